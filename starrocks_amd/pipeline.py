"""Operator-shaped pipeline driver — the §8b boundary in the reference's own
vocabulary.

Mirrors `pipeline::Operator` (reference
be/src/exec_primitive/pipeline/operator.h:48: push_chunk/pull_chunk at
:141-144, lifecycle prepare -> finishing -> finished -> close at :59-102) and
the `PipelineDriver::process` hot loop (be/src/exec/runtime/
pipeline_driver.cpp:391-500: for each adjacent operator pair with
has_output()/need_input(), pull_chunk then push_chunk, finishing propagated
down the chain once an operator is finished and drained).

This is the THIN host-side driver SURVEY.md §2 keeps: one driver <-> one HIP
stream; every pull/push pair on the GPU operators below is a kernel launch on
the session stream, and the chunks that flow between operators are device
columnar buffers that never leave HBM. The reference's 4096-row chunks
(common/config.h:1102) are sized for L2+SIMD on a CPU; ours default to 4 M
rows — the smallest chunk that still fills 256 CUs (DESIGN.md §4's grid
findings) — with the same semantics.

The driver logic itself is backend-free (CPU-tested with plain-python
operators in tests/test_pipeline_driver.py); the GPU operators
(ChunkSourceOperator -> HashJoinBuildOperator / HashJoinProbeOperator ->
AggregateSinkOperator) are parity-tested against the oracle end to end in
tests/test_gpu_parity.py.
"""

import numpy as np


class Operator:
    """Lifecycle per operator.h:59-102. Subclasses override the four data
    methods; the driver owns when they are called."""

    def __init__(self):
        self._finishing = False
        self._closed = False

    def prepare(self):
        pass

    def has_output(self) -> bool:
        raise NotImplementedError

    def need_input(self) -> bool:
        raise NotImplementedError

    def push_chunk(self, chunk) -> None:
        raise NotImplementedError

    def pull_chunk(self):
        raise NotImplementedError

    def set_finishing(self):
        """Upstream has no more chunks (hash_join_build_operator.cpp:87-96
        triggers build_ht here)."""
        self._finishing = True

    def is_finished(self) -> bool:
        """No more output will ever be produced."""
        return self._finishing and not self.has_output()

    def close(self):
        self._closed = True

    def result(self):
        """Sinks override; pass-through operators have no result."""
        return None


class PipelineDriver:
    """pipeline_driver.cpp:391-500 restated: repeatedly sweep adjacent pairs;
    move one chunk wherever curr.has_output() && next.need_input(); propagate
    set_finishing down the chain when an upstream operator is finished. A
    `process()` call runs until the whole chain is finished (no time-slice /
    blocked states — the GPU stream absorbs the asynchrony the reference's
    poller exists for)."""

    def __init__(self, operators):
        assert len(operators) >= 2
        self.ops = operators

    def process(self):
        for op in self.ops:
            op.prepare()
        self.ops[0].set_finishing()  # sources own their input
        moved = True
        while True:
            moved = False
            for curr, nxt in zip(self.ops, self.ops[1:]):
                while curr.has_output() and nxt.need_input():
                    nxt.push_chunk(curr.pull_chunk())
                    moved = True
                if curr.is_finished() and not nxt._finishing:
                    nxt.set_finishing()
                    moved = True
            if not moved:
                break
        assert all(op.is_finished() for op in self.ops), "pipeline stalled"
        for op in self.ops:
            op.close()
        return self.ops[-1].result()


# ---------------------------------------------------------------------------
# GPU operators. A chunk is {"n": rows, "cols": [DBuf, ...]} resident in HBM.
# ---------------------------------------------------------------------------

DEFAULT_CHUNK_ROWS = 4 << 20


class ChunkSourceOperator(Operator):
    """Synthetic columnar source standing in for the scan operator
    (storage is out of scope; the source yields
    BASELINE-shaped chunks straight into HBM). gen(row_start, n) -> chunk."""

    def __init__(self, gen, total_rows, chunk_rows=DEFAULT_CHUNK_ROWS):
        super().__init__()
        self._gen = gen
        self._total = total_rows
        self._chunk = chunk_rows
        self._next_row = 0

    def need_input(self):
        return False

    def has_output(self):
        return self._next_row < self._total

    def pull_chunk(self):
        n = min(self._chunk, self._total - self._next_row)
        c = self._gen(self._next_row, n)
        self._next_row += n
        return c


class HashJoinBuildOperator(Operator):
    """push_chunk appends build-side chunks (hash_joiner.cpp:221); the table
    is built once, at set_finishing (hash_join_build_operator.cpp:87-96).
    Produces no chunks — downstream probe operators reference the one
    read-only table (hash_join_probe_operator.cpp:105-111)."""

    def __init__(self, engine, build_fn):
        super().__init__()
        self._e = engine
        self._build_fn = build_fn
        self._pending = []
        self.table = None

    def need_input(self):
        return not self._finishing

    def has_output(self):
        return False

    def push_chunk(self, chunk):
        self._pending.append(chunk)

    def set_finishing(self):
        super().set_finishing()
        self.table = self._build_fn(self._e, self._pending)
        for c in self._pending:
            for b in c["cols"]:
                b.free()
        self._pending = []

    def close(self):
        # close() keeps the table: probe operators in the dependent pipeline
        # share the one read-only table after the build pipeline closes
        # (hash_join_probe_operator.cpp:105-111 clone_readable_table).
        # release() drops it once the probers are done.
        super().close()

    def release(self):
        if self.table is not None:
            self.table.destroy()
            self.table = None


class HashJoinProbeOperator(Operator):
    """Push one probe chunk, pull result chunks until drained
    (hash_join_components.cpp:65-82). The chain-walk emit is the engine's
    two-phase count/emit, so one probe chunk yields exactly one (exactly
    sized) result chunk — the reference's resumable cursor collapses into
    the count pass."""

    def __init__(self, engine, build_op, key_col=0):
        super().__init__()
        self._e = engine
        self._build_op = build_op
        self._key_col = key_col
        self._in = None
        self._out = None

    def need_input(self):
        return self._in is None and self._out is None and not self._finishing

    def has_output(self):
        if self._in is not None:
            self._probe()
        return self._out is not None

    def push_chunk(self, chunk):
        self._in = chunk

    def _probe(self):
        c, self._in = self._in, None
        t = self._build_op.table
        keys = c["cols"][self._key_col]
        cnt = self._e.join_probe_emit(t, keys, c["n"])
        op_buf = self._e.alloc(max(cnt, 1) * 4)
        ob_buf = self._e.alloc(max(cnt, 1) * 4)
        self._e.join_probe_emit(t, keys, c["n"], op_buf, ob_buf)
        # gather the probe-side payload columns through the match indices
        # (join_hash_map.hpp:284-330 _probe_output/append_selective)
        out_cols = []
        for i, col in enumerate(c["cols"]):
            if i == self._key_col:
                continue
            g = self._e.alloc(max(cnt, 1) * 4)
            if cnt > 0:
                self._e.gather_u32(col, op_buf, cnt, g)
            out_cols.append(g)
        for b in (op_buf, ob_buf, *c["cols"]):
            b.free()
        self._out = {"n": cnt, "cols": out_cols}

    def pull_chunk(self):
        if self._out is None:
            self._probe()
        c, self._out = self._out, None
        return c


class AggregateSinkOperator(Operator):
    """Blocking agg sink (aggregate_blocking_sink_operator.cpp:114-151):
    consumes every chunk into accumulator state; result() finalizes."""

    def __init__(self, engine, update_fn, result_fn):
        super().__init__()
        self._e = engine
        self._update = update_fn
        self._result = result_fn
        self._state = None

    def need_input(self):
        return not self._finishing

    def has_output(self):
        return False

    def push_chunk(self, chunk):
        self._state = self._update(self._e, self._state, chunk)
        for b in chunk["cols"]:
            b.free()

    def result(self):
        return self._result(self._e, self._state)


class StreamingAggOperator(Operator):
    """AUTO-mode streaming pre-aggregation
    (aggregate_streaming_sink_operator.cpp:224-310): per chunk, the state
    machine INIT_PREAGG -> ADJUST -> {PASS_THROUGH | PREAGG |
    SELECTIVE_PREAGG} decides between aggregating into the local table,
    streaming the chunk through to the downstream (merge) aggregate, or the
    selective mix (aggregate rows whose group already exists, stream the
    rest). Thresholds are the reference's (aggregator.h:175-178,
    aggregator.cpp:150-160): LowReduction 0.2, HighReduction 0.9,
    StableLimit 5 consecutive chunks to leave ADJUST; hit count =
    build_hash_map_with_selection's existing-group rows. At finishing the
    local table streams out as pre-aggregated partials (cnt column set), the
    merge_batch form (aggregate.h:158-168).

    Chunks: {"n", "keys": DBuf u64, "vals": DBuf i64, "cnts": DBuf|None}.
    """

    LOW_REDUCTION = 0.2     # aggregator.h:175
    HIGH_REDUCTION = 0.9    # aggregator.h:176
    STABLE_LIMIT = 5        # aggregator.h:178

    def __init__(self, engine, capacity=1 << 20, max_groups=1 << 20):
        super().__init__()
        self._e = engine
        self._at = engine.agg_table_create(capacity)
        engine.agg_table_reset(self._at)
        self._max_groups = max_groups
        self._out = []
        self._state = "INIT_PREAGG"
        self._counts = {"pass": 0, "preagg": 0, "selective": 0, "adjust": 0}
        self._drained = False

    def need_input(self):
        return not self._finishing and not self._out

    def has_output(self):
        if self._finishing and not self._drained:
            self._drain()
        return bool(self._out)

    def pull_chunk(self):
        return self._out.pop(0)

    def _preagg(self, c):
        # reference shape: try_convert_to_two_level_map runs before each
        # chunk (aggregator.cpp:1237-1241) — here the table grows + rehashes
        # so the push can never overflow (gpue_agg_table_ensure)
        self._e.agg_table_ensure(self._at, c["n"])
        self._e.hash_agg_push(self._at, c["keys"], c["vals"], c["n"], cnts=c.get("cnts"))
        for b in (c["keys"], c["vals"]):
            b.free()

    def _stream(self, c):
        self._out.append(c)

    def _selective(self, c):
        import numpy as np
        n = c["n"]
        mask = self._e.alloc(n)
        self._e.hash_agg_push(self._at, c["keys"], c["vals"], n, cnts=c.get("cnts"),
                              update_only=1, miss_mask=mask)
        m = mask.d2h(np.uint8, n)
        miss = np.flatnonzero(m).astype(np.uint32)
        mask.free()
        if len(miss):
            idx = self._e.alloc(miss.nbytes)
            idx.h2d(miss)
            ks = self._e.alloc(len(miss) * 8)
            vs = self._e.alloc(len(miss) * 8)
            self._e.gather_u64(c["keys"], idx, len(miss), ks)
            self._e.gather_u64(c["vals"], idx, len(miss), vs)
            idx.free()
            self._out.append({"n": len(miss), "keys": ks, "vals": vs, "cnts": None})
        for b in (c["keys"], c["vals"]):
            b.free()

    def push_chunk(self, c):
        n = c["n"]
        if self._state == "INIT_PREAGG":
            # first chunks always pre-aggregate; leave for ADJUST once the
            # table stops reducing (ht expansion heuristic simplified to the
            # reduction test on the next chunk)
            hits = self._e.hash_agg_probe_hits(self._at, c["keys"], n)
            if self._counts["adjust"] == 0 or hits >= self.HIGH_REDUCTION * n:
                self._counts["adjust"] += 1
                self._preagg(c)
                return
            self._state = "ADJUST"
        if self._state == "ADJUST":
            hits = self._e.hash_agg_probe_hits(self._at, c["keys"], n)
            if hits <= self.LOW_REDUCTION * n:
                self._stream(c)
                self._counts["pass"] += 1
                self._counts["preagg"] = self._counts["selective"] = 0
                if self._counts["pass"] >= self.STABLE_LIMIT:
                    self._state = "PASS_THROUGH"
            elif hits >= self.HIGH_REDUCTION * n:
                self._preagg(c)
                self._counts["preagg"] += 1
                self._counts["pass"] = self._counts["selective"] = 0
                if self._counts["preagg"] >= self.STABLE_LIMIT:
                    self._state = "PREAGG"
            else:
                self._selective(c)
                self._counts["selective"] += 1
                self._counts["pass"] = self._counts["preagg"] = 0
                if self._counts["selective"] >= self.STABLE_LIMIT:
                    self._state = "SELECTIVE_PREAGG"
        elif self._state == "PASS_THROUGH":
            self._stream(c)
        elif self._state == "SELECTIVE_PREAGG":
            self._selective(c)
        else:
            self._preagg(c)

    def _drain(self):
        """Finishing: the local table streams out as pre-agg partials."""
        self._drained = True
        ok = self._e.alloc(self._max_groups * 8)
        os_ = self._e.alloc(self._max_groups * 8)
        oc = self._e.alloc(self._max_groups * 8)
        g = self._e.hash_agg_emit(self._at, ok, os_, self._max_groups, out_counts=oc)
        if g:
            self._out.append({"n": g, "keys": ok, "vals": os_, "cnts": oc})
        else:
            for b in (ok, os_, oc):
                b.free()

    def close(self):
        self._e.agg_table_destroy(self._at)
        super().close()


class FinalAggSink(Operator):
    """The downstream (merge) aggregate: accepts raw rows AND pre-agg
    partials (cnts set) into one table — merge_batch semantics."""

    def __init__(self, engine, capacity=1 << 20, max_groups=1 << 20):
        super().__init__()
        self._e = engine
        self._at = engine.agg_table_create(capacity)
        engine.agg_table_reset(self._at)
        self._max_groups = max_groups

    def need_input(self):
        return not self._finishing

    def has_output(self):
        return False

    def push_chunk(self, c):
        self._e.hash_agg_push(self._at, c["keys"], c["vals"], c["n"], cnts=c.get("cnts"))
        for b in (c["keys"], c["vals"], c.get("cnts")):
            if b is not None:
                b.free()

    def result(self):
        import numpy as np
        ok = self._e.alloc(self._max_groups * 8)
        os_ = self._e.alloc(self._max_groups * 8)
        oc = self._e.alloc(self._max_groups * 8)
        g = self._e.hash_agg_emit(self._at, ok, os_, self._max_groups, out_counts=oc)
        keys = ok.d2h(np.uint64, g)
        sums = os_.d2h(np.int64, g)
        cnts = oc.d2h(np.int64, g)
        for b in (ok, os_, oc):
            b.free()
        self._e.agg_table_destroy(self._at)
        order = np.argsort(keys)
        return keys[order], sums[order], cnts[order]


def q1_operator_pipeline(engine, seed, total_rows, year=1993,
                         chunk_rows=DEFAULT_CHUNK_ROWS, dim_chunk_rows=1000):
    """Config-2's plan as the reference would run it — dim source -> build
    pipeline, then fact source -> probe -> agg sink over bounded chunks —
    instead of the fused q1 kernel. Bit-exact same (sum, count) as
    engine.q1_join_sum / orc.q1_pipeline; exists to exercise the push/pull
    boundary end to end, not as the bench path (the fused kernel is the fast
    form: one HBM pass, no match materialization)."""
    from starrocks_amd import gen as g

    # Dim rows pass the year predicate before the build, as the reference's
    # plan filters the dim scan below the build operator.
    datekey, dyear = g.gen_dates()
    dim_keys = datekey[dyear == year].astype(np.int32)

    def date_source(row_start, n):
        kb = engine.alloc(n * 4)
        kb.h2d(dim_keys[row_start:row_start + n])
        return {"n": n, "cols": [kb]}

    def build_dates(e, chunks):
        total = sum(c["n"] for c in chunks)
        kb = e.alloc((total + 1) * 4)  # row 0 = sentinel
        kb.h2d(np.zeros(1, np.int32))
        off = 1
        for c in chunks:
            e.dbuf_d2d(c["cols"][0], kb, c["n"] * 4, 0, off * 4)
            off += c["n"]
        t = e.join_build_range_direct(kb, total)
        kb.free()
        return t

    def gen_chunk(row_start, n):
        cols = [engine.alloc(n * 4) for _ in range(3)]
        engine.gen_lineorder_q1(seed, row_start, n, *cols)
        return {"n": n, "cols": cols}

    def agg_update(e, state, chunk):
        if state is None:
            state = e.alloc(16)
            state.h2d(np.zeros(2, np.int64))
        ep, dc = chunk["cols"]
        e.sum_prod_u32(ep, dc, chunk["n"], state)
        return state

    def agg_result(e, state):
        if state is None:
            return 0, 0
        raw = state.d2h(np.int64, 2)
        state.free()
        return int(raw[0]), int(raw[1])

    # Build pipeline first (the reference runs build and probe as two
    # pipelines with a dependency; hash_join_build_operator.cpp:87-96).
    build = HashJoinBuildOperator(engine, build_dates)
    PipelineDriver([
        ChunkSourceOperator(date_source, len(dim_keys), dim_chunk_rows),
        build,
    ]).process()

    probe_driver = PipelineDriver([
        ChunkSourceOperator(gen_chunk, total_rows, chunk_rows),
        HashJoinProbeOperator(engine, build, key_col=0),
        AggregateSinkOperator(engine, agg_update, agg_result),
    ])
    out = probe_driver.process()
    build.release()
    return out
