"""GPU mirrors of tests/test_edge_cases.py (SURVEY.md §8c edge cases):
same constructions through the C-ABI on the MI355X."""

import numpy as np
import pytest

from oracle import pyoracle as orc

pytestmark = pytest.mark.gpu

SEED = 42


def _emit(engine, t, probe, expect_pairs):
    pb = engine.alloc(max(probe.nbytes, 4))
    if len(probe):
        pb.h2d(probe)
    cnt = engine.join_probe_emit(t, pb, len(probe))
    assert cnt == len(expect_pairs)
    if cnt:
        op_buf, ob_buf = engine.alloc(cnt * 4), engine.alloc(cnt * 4)
        engine.join_probe_emit(t, pb, len(probe), op_buf, ob_buf)
        got = sorted(zip(op_buf.d2h(np.uint32, cnt).tolist(),
                         ob_buf.d2h(np.uint32, cnt).tolist()))
        assert got == sorted(expect_pairs)
        op_buf.free()
        ob_buf.free()
    pb.free()


def test_single_row_build_gpu(engine):
    keys = np.array([0, 77], np.uint32)
    kb = engine.alloc(keys.nbytes)
    kb.h2d(keys)
    t = engine.join_build_bucket_chained(kb, 1)
    _emit(engine, t, np.array([77, 78, 0], np.uint32), [(0, 1)])
    _emit(engine, t, np.empty(0, np.uint32), [])
    kb.free()
    t.destroy()


def test_all_duplicate_build_keys_chain_gpu(engine):
    n = 100_000
    keys = np.concatenate([[0], np.full(n, 12345)]).astype(np.uint32)
    kb = engine.alloc(keys.nbytes)
    kb.h2d(keys)
    t = engine.join_build_bucket_chained(kb, n)
    _emit(engine, t, np.array([12345, 999], np.uint32),
          [(0, j) for j in range(1, n + 1)])
    kb.free()
    t.destroy()


def test_extreme_key_values_gpu(engine):
    bkeys = np.concatenate([[0], [0, 1, 0x7FFFFFFF, 0x80000000, 0xFFFFFFFF]]).astype(np.uint32)
    kb = engine.alloc(bkeys.nbytes)
    kb.h2d(bkeys)
    t = engine.join_build_bucket_chained(kb, 5)
    _emit(engine, t, np.array([0xFFFFFFFF, 0x80000000, 0x7FFFFFFF, 1, 0, 2], np.uint32),
          [(0, 5), (1, 4), (2, 3), (3, 2), (4, 1)])
    kb.free()
    t.destroy()


def test_varchar_empty_string_keys_gpu(engine):
    brows = [b"", b"", b"abc", b""]
    rows = [b""] + brows
    bo = np.zeros(len(rows) + 1, np.uint32)
    np.cumsum([len(r) for r in rows], out=bo[1:])
    bb = np.frombuffer(b"".join(rows), np.uint8).copy()
    prows = [b"", b"abc", b"zzz"]
    po = np.zeros(len(prows) + 1, np.uint32)
    np.cumsum([len(r) for r in prows], out=po[1:])
    pb = np.frombuffer(b"".join(prows), np.uint8).copy()
    d_bb, d_bo = engine.alloc(max(bb.nbytes, 4)), engine.alloc(bo.nbytes)
    d_bb.h2d(bb)
    d_bo.h2d(bo)
    t = engine.join_build_varchar(d_bb, d_bo, len(brows))
    d_pb, d_po = engine.alloc(pb.nbytes), engine.alloc(po.nbytes)
    d_pb.h2d(pb)
    d_po.h2d(po)
    cnt = engine.join_probe_emit_varchar(t, d_pb, d_po, len(prows))
    assert cnt == 4
    op_buf, ob_buf = engine.alloc(cnt * 4), engine.alloc(cnt * 4)
    engine.join_probe_emit_varchar(t, d_pb, d_po, len(prows), op_buf, ob_buf)
    got = sorted(zip(op_buf.d2h(np.uint32, cnt).tolist(),
                     ob_buf.d2h(np.uint32, cnt).tolist()))
    assert got == [(0, 1), (0, 2), (0, 4), (1, 3)]
    for b in (d_bb, d_bo, d_pb, d_po, op_buf, ob_buf):
        b.free()
    t.destroy()


def test_q1_no_passing_rows_gpu(engine):
    """Payload table whose payloads are ALL zero (year matches no dim row):
    the fused kernel returns (0, 0) like the oracle."""
    n = 1_000_000
    cols = [engine.alloc(n * 4) for _ in range(3)]
    engine.gen_lineorder_q1(SEED, 0, n, *cols)
    from starrocks_amd import gen
    datekey, dyear = gen.gen_dates()
    dpay = np.zeros(len(datekey), np.uint32)
    kb = engine.alloc(datekey.nbytes)
    kb.h2d(datekey.astype(np.int32))
    pbuf = engine.alloc(dpay.nbytes)
    pbuf.h2d(dpay)
    t = engine.join_build_payload(kb, pbuf, len(datekey))
    s, cnt = engine.q1_join_sum(t, *cols, n)
    assert (s, cnt) == (0, 0)
    for b in cols + [kb, pbuf]:
        b.free()
    t.destroy()


def test_hash_agg_single_group_gpu(engine):
    n = 500_000
    keys = np.full(n, 42, np.uint64)
    vals = np.arange(n, dtype=np.int64)
    kb, vb = engine.alloc(n * 8), engine.alloc(n * 8)
    kb.h2d(keys)
    vb.h2d(vals)
    ok_b, os_b, oc_b = (engine.alloc(16 * 8) for _ in range(3))
    g = engine.hash_agg_sum_u64(kb, vb, n, ok_b, os_b, oc_b, max_out=16)
    assert g == 1
    assert ok_b.d2h(np.uint64, 1)[0] == 42
    assert os_b.d2h(np.int64, 1)[0] == vals.sum()
    assert oc_b.d2h(np.int64, 1)[0] == n
    for b in (kb, vb, ok_b, os_b, oc_b):
        b.free()


def test_hash_agg_sentinel_key_errors_gpu(engine):
    """a real group key equal to the empty-slot sentinel (~0ull)
    must raise, not silently drop the group (the reference aggregator
    accepts every key value, agg_hash_map.h:112-290)."""
    from starrocks_amd.engine import GpueError
    n = 1024
    keys = np.arange(1, n + 1, dtype=np.uint64)
    keys[100] = 0xFFFFFFFFFFFFFFFF
    vals = np.ones(n, np.int64)
    kb, vb = engine.alloc(n * 8), engine.alloc(n * 8)
    kb.h2d(keys)
    vb.h2d(vals)
    ok_b, os_b = engine.alloc(4096 * 8), engine.alloc(4096 * 8)
    with pytest.raises(GpueError, match="sentinel"):
        engine.hash_agg_sum_u64(kb, vb, n, ok_b, os_b, max_out=4096)
    # the latch clears on read: a clean call on the same session succeeds
    keys[100] = 100
    kb.h2d(keys)
    g = engine.hash_agg_sum_u64(kb, vb, n, ok_b, os_b, max_out=4096)
    assert g == n - 1  # key 100 appears twice
    for b in (kb, vb, ok_b, os_b):
        b.free()


def test_hash_agg_capacity_overflow_errors_gpu(engine):
    """capacity_hint < distinct keys must return an error, never
    spin forever (bounded probe walk)."""
    from starrocks_amd.engine import GpueError
    n = 4096
    keys = np.arange(1, n + 1, dtype=np.uint64)
    vals = np.ones(n, np.int64)
    kb, vb = engine.alloc(n * 8), engine.alloc(n * 8)
    kb.h2d(keys)
    vb.h2d(vals)
    ok_b, os_b = engine.alloc(n * 8), engine.alloc(n * 8)
    with pytest.raises(GpueError, match="full"):
        engine.hash_agg_sum_u64(kb, vb, n, ok_b, os_b, max_out=n, capacity_hint=64)
    g = engine.hash_agg_sum_u64(kb, vb, n, ok_b, os_b, max_out=n)
    assert g == n
    for b in (kb, vb, ok_b, os_b):
        b.free()


def test_pinned_ingest_parity_gpu(engine):
    """north_star: 'columnar batches pinned and streamed to HBM'. A column
    streamed in through the double-buffered pinned ingest must equal the
    same column generated on device (the generators are bit-identical by
    construction, so this pins the transport)."""
    import starrocks_amd.gen  # noqa: F401  (host-side generator helpers)
    from oracle import pyoracle as orc
    n = 8_000_000
    host_od, host_ep, host_dc = orc.gen_lineorder_q1(42, 0, n)
    dev = engine.alloc(n * 4)
    engine.gen_lineorder_q1(42, 0, n, dev, engine.alloc(n * 4), engine.alloc(n * 4))
    ing = engine.ingest_create(4 << 20)
    streamed = engine.alloc(n * 4)
    # push in uneven pieces to exercise the chunking + offsets
    engine.ingest_push(ing, host_od[:1_000_000], streamed, 0)
    engine.ingest_push(ing, host_od[1_000_000:], streamed, 1_000_000 * 4)
    engine.ingest_sync(ing)
    engine.ingest_destroy(ing)
    a = dev.d2h(np.int32, n)
    b = streamed.d2h(np.int32, n)
    assert np.array_equal(a, b)


def test_agg_table_grow_gpu(engine):
    """Growable agg table (the reference's two-level-conversion property,
    aggregator.cpp:1237-1241): start tiny, push far more distinct groups
    than the initial capacity in chunks with ensure() between — results
    must equal a one-shot aggregate and the table must have grown."""
    at = engine.agg_table_create(16)
    engine.agg_table_reset(at)
    rng = np.random.default_rng(5)
    n, chunks = 200_000, 8
    keys = rng.integers(1, 50_000, n).astype(np.uint64)
    vals = rng.integers(-1000, 1000, n).astype(np.int64)
    per = n // chunks
    for c in range(chunks):
        sl = slice(c * per, (c + 1) * per)
        kb, vb = engine.alloc(per * 8), engine.alloc(per * 8)
        kb.h2d(keys[sl])
        vb.h2d(vals[sl])
        engine.agg_table_ensure(at, per)
        engine.hash_agg_push(at, kb, vb, per)
        kb.free()
        vb.free()
    distinct = len(np.unique(keys))
    assert engine.agg_table_size(at) == distinct
    ok_b = engine.alloc(distinct * 8)
    os_b = engine.alloc(distinct * 8)
    oc_b = engine.alloc(distinct * 8)
    g = engine.hash_agg_emit(at, ok_b, os_b, distinct, oc_b)
    assert g == distinct
    got_k = ok_b.d2h(np.uint64, g)
    got_s = os_b.d2h(np.int64, g)
    got_c = oc_b.d2h(np.int64, g)
    order = np.argsort(got_k)
    import collections
    sums = collections.defaultdict(int)
    cnts = collections.defaultdict(int)
    for k, v in zip(keys.tolist(), vals.tolist()):
        sums[k] += v
        cnts[k] += 1
    ek = np.array(sorted(sums), np.uint64)
    assert np.array_equal(got_k[order], ek)
    assert np.array_equal(got_s[order], np.array([sums[k] for k in ek.tolist()]))
    assert np.array_equal(got_c[order], np.array([cnts[k] for k in ek.tolist()]))
    for b in (ok_b, os_b, oc_b):
        b.free()
    engine.agg_table_destroy(at)


def test_partition_channel_bounds_gpu(engine):
    """MAX_CH=64 channels supported (the exchange's channel cap); 65 must
    be a clean argument error."""
    from starrocks_amd.engine import GpueError
    n = 100_000
    keys = engine.alloc(n * 4)
    engine.gen_u32_mod(keys, 42, 9, 0, n, 0, 0)
    ri = engine.alloc(n * 4)
    sp = engine.partition(keys, n, 64, ri)
    assert sp[-1] == n and len(sp) == 65
    with pytest.raises(GpueError):
        engine.partition(keys, n, 65, ri)
    keys.free()
    ri.free()


def test_ingest_exact_chunk_multiple_gpu(engine):
    """ingest push where bytes is an exact multiple of the staging chunk."""
    import numpy as np
    n = (8 << 20) // 4  # exactly 2 chunks of 4 MB
    host = np.arange(n, dtype=np.int32)
    dst = engine.alloc(n * 4)
    ing = engine.ingest_create(4 << 20)
    engine.ingest_push(ing, host, dst)
    engine.ingest_sync(ing)
    engine.ingest_destroy(ing)
    assert np.array_equal(dst.d2h(np.int32, n), host)
    dst.free()


def test_hash_agg_sum_only_flag(engine):
    """GPUE_AGG_SUM_ONLY (flags bit 1): sums identical to the default path,
    the per-row count atomic skipped (counts emit as 0)."""
    rng = np.random.default_rng(31)
    n, ngroups = 2_000_000, 50_000
    keys_h = rng.integers(0, ngroups, n).astype(np.uint64)
    vals_h = rng.integers(-100, 100, n).astype(np.int64)
    keys = engine.alloc(n * 8); keys.h2d(keys_h)
    vals = engine.alloc(n * 8); vals.h2d(vals_h)
    ok = engine.alloc(2 * ngroups * 8)
    os_ = engine.alloc(2 * ngroups * 8)
    oc = engine.alloc(2 * ngroups * 8)

    def run(flags):
        at = engine.agg_table_create(2 * ngroups)
        engine.hash_agg_push(at, keys, vals, n, update_only=flags)
        got = engine.hash_agg_emit(at, ok, os_, 2 * ngroups, out_counts=oc)
        engine.agg_table_destroy(at)
        gk = ok.d2h(np.uint64, got)
        gs = os_.d2h(np.int64, got)
        gc = oc.d2h(np.int64, got)
        order = np.argsort(gk)
        return gk[order], gs[order], gc[order]

    k0, s0, c0 = run(0)
    k2, s2, c2 = run(2)
    assert np.array_equal(k0, k2)
    assert np.array_equal(s0, s2)
    assert np.all(c0 >= 1) and not np.any(c2)
    for b in (keys, vals, ok, os_, oc):
        b.free()
