"""ASOF join parity — LinearChainedAsofJoinHashMap (join_hash_map_method.h:
201-217) + AsofIndex (join_hash_table_descriptor.h:59-104, .cpp:70-134).

CPU: the oracle restatement vs an independently-derived brute force of the
ASOF semantics (LT: smallest build value > probe; LE: >=; GT: largest build
value < probe; GE: <=; ties on (key, asof) pinned to the smallest build row —
the reference's pdqsort is unstable there, so we test the deterministic
refinement both implementations pin).

GPU: gpue_asof_build_i32 / gpue_asof_probe_emit_i32 vs the oracle, covering
both segment-sort paths (LDS bitonic <= 4096 entries, single-block global
bitonic beyond), all four opcodes, INNER and LEFT_OUTER modes.
"""

import numpy as np
import pytest

from oracle import pyoracle as orc

OPCODES = {0: "LT", 1: "LE", 2: "GT", 3: "GE"}


def brute_force(bkeys, basof, pkeys, pasof, opcode):
    """Direct statement of the ASOF match, independent of the reference's
    binary-search formulation. Arrays 1-based (index 0 sentinel)."""
    out = np.zeros(len(pkeys), np.uint32)
    for i, (k, pv) in enumerate(zip(pkeys.tolist(), pasof.tolist())):
        best = None  # (asof, row)
        for row in range(1, len(bkeys)):
            if bkeys[row] != k:
                continue
            bv = int(basof[row])
            ok = (bv > pv if opcode == 0 else bv >= pv if opcode == 1
                  else bv < pv if opcode == 2 else bv <= pv)
            if not ok:
                continue
            # LT/LE want the smallest qualifying value, GT/GE the largest;
            # equal values resolve to the smallest row (pinned refinement)
            key = (bv, row) if opcode <= 1 else (-bv, row)
            if best is None or key < best:
                best = key
        out[i] = 0 if best is None else best[1]
    return out


def _case(seed, n_build, n_probe, key_lo, key_hi, val_span, tie_heavy=False):
    rng = np.random.default_rng(seed)
    bk = np.concatenate([[0], rng.integers(key_lo, key_hi + 1, n_build)]).astype(np.int32)
    pk = rng.integers(key_lo - 1, key_hi + 2, n_probe).astype(np.int32)  # incl. misses
    if tie_heavy:
        ba = np.concatenate([[0], rng.integers(0, 4, n_build)]).astype(np.int64)
        pa = rng.integers(0, 4, n_probe).astype(np.int64)
    else:
        ba = np.concatenate([[0], rng.integers(-val_span, val_span, n_build)]).astype(np.int64)
        pa = rng.integers(-val_span, val_span, n_probe).astype(np.int64)
    return bk, ba, pk, pa


@pytest.mark.parametrize("opcode", list(OPCODES))
def test_oracle_vs_bruteforce(opcode):
    for seed, tie in ((1, False), (2, True)):
        bk, ba, pk, pa = _case(seed * 10 + opcode, 60, 200, 1, 12, 50, tie)
        got = orc.asof_inner_join(bk, ba, pk, pa, opcode)
        want = brute_force(bk, ba, pk, pa, opcode)
        assert np.array_equal(got, want), OPCODES[opcode]


@pytest.mark.parametrize("opcode", list(OPCODES))
def test_oracle_edges(opcode):
    i64 = np.iinfo(np.int64)
    # single build row, extreme temporal values, probe at the extremes
    bk = np.array([0, 5], np.int32)
    ba = np.array([0, i64.max], np.int64)
    pk = np.array([5, 5, 4], np.int32)
    pa = np.array([i64.max, i64.min, 0], np.int64)
    got = orc.asof_inner_join(bk, ba, pk, pa, opcode)
    want = brute_force(bk, ba, pk, pa, opcode)
    assert np.array_equal(got, want)
    # all rows one key, all-equal asof values -> min row or nothing
    bk2 = np.array([0] + [3] * 8, np.int32)
    ba2 = np.array([0] + [7] * 8, np.int64)
    pk2 = np.array([3, 3, 3], np.int32)
    pa2 = np.array([6, 7, 8], np.int64)
    got2 = orc.asof_inner_join(bk2, ba2, pk2, pa2, opcode)
    want2 = brute_force(bk2, ba2, pk2, pa2, opcode)
    assert np.array_equal(got2, want2)


def _gpu_case(seed):
    """Build with a hot key holding ~10k entries — the > 4096-entry segment
    exercises the single-block global-bitonic sort path."""
    rng = np.random.default_rng(seed)
    n_cold, n_hot = 200_000, 10_000
    keys = rng.integers(1, 50_001, n_cold).astype(np.int32)
    keys = np.concatenate([keys, np.full(n_hot, 7, np.int32)])
    rng.shuffle(keys)
    bk = np.concatenate([[0], keys]).astype(np.int32)
    ba = np.concatenate([[0], rng.integers(-10**12, 10**12, len(keys))]).astype(np.int64)
    n_probe = 2_000_000
    pk = rng.integers(1, 50_001, n_probe).astype(np.int32)
    pa = rng.integers(-10**12, 10**12, n_probe).astype(np.int64)
    return bk, ba, pk, pa


@pytest.mark.gpu
@pytest.mark.parametrize("opcode", list(OPCODES))
def test_gpu_parity(engine, opcode):
    bk, ba, pk, pa = _gpu_case(100 + opcode)
    want = orc.asof_inner_join(bk, ba, pk, pa, opcode)

    kb = engine.alloc(bk.nbytes); kb.h2d(bk)
    ab = engine.alloc(ba.nbytes); ab.h2d(ba)
    t = engine.asof_build(kb, ab, len(bk) - 1, opcode)
    pkb = engine.alloc(pk.nbytes); pkb.h2d(pk)
    pab = engine.alloc(pa.nbytes); pab.h2d(pa)

    # INNER: matched probe rows only, ordered by probe row
    cnt = engine.asof_probe_emit(t, pkb, pab, len(pk), 0)
    exp_rows = np.flatnonzero(want).astype(np.uint32)
    assert cnt == len(exp_rows)
    op = engine.alloc(max(cnt, 1) * 4)
    ob = engine.alloc(max(cnt, 1) * 4)
    engine.asof_probe_emit(t, pkb, pab, len(pk), 0, op, ob)
    assert np.array_equal(op.d2h(np.uint32, cnt), exp_rows)
    assert np.array_equal(ob.d2h(np.uint32, cnt), want[exp_rows])

    # LEFT_OUTER: every probe row, misses carry build row 0
    cnt3 = engine.asof_probe_emit(t, pkb, pab, len(pk), 3)
    assert cnt3 == len(pk)
    op3 = engine.alloc(cnt3 * 4)
    ob3 = engine.alloc(cnt3 * 4)
    engine.asof_probe_emit(t, pkb, pab, len(pk), 3, op3, ob3)
    assert np.array_equal(op3.d2h(np.uint32, cnt3), np.arange(len(pk), dtype=np.uint32))
    assert np.array_equal(ob3.d2h(np.uint32, cnt3), want)

    for b in (op, ob, op3, ob3, pkb, pab, kb, ab):
        b.free()
    t.destroy()


@pytest.mark.gpu
def test_gpu_ties_and_small_segments(engine):
    # tie-heavy small case across all opcodes: duplicate (key, asof) pairs
    # must resolve to the smallest build row on both sides
    for opcode in OPCODES:
        # negative keys included: build/probe must agree through the u32 cast
        bk, ba, pk, pa = _case(500 + opcode, 5_000, 100_000, -15, 40, 3, tie_heavy=True)
        want = orc.asof_inner_join(bk, ba, pk, pa, opcode)
        kb = engine.alloc(bk.nbytes); kb.h2d(bk)
        ab = engine.alloc(ba.nbytes); ab.h2d(ba)
        t = engine.asof_build(kb, ab, len(bk) - 1, opcode)
        pkb = engine.alloc(pk.nbytes); pkb.h2d(pk)
        pab = engine.alloc(pa.nbytes); pab.h2d(pa)
        cnt = engine.asof_probe_emit(t, pkb, pab, len(pk), 3)
        assert cnt == len(pk)
        op = engine.alloc(cnt * 4)
        ob = engine.alloc(cnt * 4)
        engine.asof_probe_emit(t, pkb, pab, len(pk), 3, op, ob)
        assert np.array_equal(ob.d2h(np.uint32, cnt), want), OPCODES[opcode]
        for b in (op, ob, pkb, pab, kb, ab):
            b.free()
        t.destroy()


def test_oracle_randomized_sweep():
    """Wider randomized sweep: many seeds x opcodes x shapes against the
    brute force, including single-key builds, wide key spaces (mostly
    misses), and negative/positive value mixes."""
    shapes = [
        (30, 80, 1, 1, 10),      # one key only
        (50, 120, 1, 200, 40),   # sparse keys, mostly misses
        (120, 150, 1, 6, 5),     # dense keys, heavy duplicates
        (80, 150, -20, 20, 30),  # NEGATIVE keys (the u32-cast slot path)
    ]
    for opcode in OPCODES:
        for si, (nb, np_, klo, khi, span) in enumerate(shapes):
            for seed in range(3):
                bk, ba, pk, pa = _case(7000 + opcode * 100 + si * 10 + seed,
                                       nb, np_, klo, khi, span)
                got = orc.asof_inner_join(bk, ba, pk, pa, opcode)
                want = brute_force(bk, ba, pk, pa, opcode)
                assert np.array_equal(got, want), (OPCODES[opcode], si, seed)


def test_oracle_nulls_vs_bruteforce():
    """Nulls variant: flagged build rows skipped, null probe rows unmatched
    (join_hash_table_descriptor.h:447-456) — checked by masking inside the
    brute force's input instead."""
    rng = np.random.default_rng(99)
    for opcode in OPCODES:
        bk, ba, pk, pa = _case(900 + opcode, 80, 250, 1, 10, 30)
        bn = np.concatenate([[0], rng.integers(0, 2, len(bk) - 1)]).astype(np.uint8)
        pn = rng.integers(0, 2, len(pk)).astype(np.uint8)
        got = orc.asof_inner_join_nulls(bk, ba, bn, pk, pa, pn, opcode)
        # brute force: null build rows get a key no probe carries; null
        # probe rows zeroed after the fact
        bk_m = bk.copy()
        bk_m[bn != 0] = -(10**9)
        want = brute_force(bk_m, ba, pk, pa, opcode)
        want[pn != 0] = 0
        assert np.array_equal(got, want), OPCODES[opcode]
        # None masks == the plain entry point
        assert np.array_equal(orc.asof_inner_join_nulls(bk, ba, None, pk, pa, None, opcode),
                              orc.asof_inner_join(bk, ba, pk, pa, opcode))


@pytest.mark.gpu
def test_gpu_nulls_parity(engine):
    rng = np.random.default_rng(321)
    for opcode in (0, 3):  # one ascending, one descending
        n_build, n_probe = 100_000, 1_000_000
        bk = np.concatenate([[0], rng.integers(1, 20_001, n_build)]).astype(np.int32)
        ba = np.concatenate([[0], rng.integers(-10**9, 10**9, n_build)]).astype(np.int64)
        bn = np.concatenate([[0], (rng.random(n_build) < 0.3)]).astype(np.uint8)
        pk = rng.integers(1, 20_001, n_probe).astype(np.int32)
        pa = rng.integers(-10**9, 10**9, n_probe).astype(np.int64)
        pn = (rng.random(n_probe) < 0.2).astype(np.uint8)
        want = orc.asof_inner_join_nulls(bk, ba, bn, pk, pa, pn, opcode)

        kb = engine.alloc(bk.nbytes); kb.h2d(bk)
        ab = engine.alloc(ba.nbytes); ab.h2d(ba)
        nb = engine.alloc(bn.nbytes); nb.h2d(bn)
        t = engine.asof_build_nulls(kb, ab, nb, n_build, opcode)
        pkb = engine.alloc(pk.nbytes); pkb.h2d(pk)
        pab = engine.alloc(pa.nbytes); pab.h2d(pa)
        pnb = engine.alloc(pn.nbytes); pnb.h2d(pn)
        cnt = engine.asof_probe_emit_nulls(t, pkb, pab, pnb, n_probe, 3)
        assert cnt == n_probe
        op = engine.alloc(cnt * 4)
        ob = engine.alloc(cnt * 4)
        engine.asof_probe_emit_nulls(t, pkb, pab, pnb, n_probe, 3, op, ob)
        assert np.array_equal(ob.d2h(np.uint32, cnt), want), OPCODES[opcode]
        # INNER count = non-null matched rows
        cnt0 = engine.asof_probe_emit_nulls(t, pkb, pab, pnb, n_probe, 0)
        assert cnt0 == int(np.count_nonzero(want))
        for b in (op, ob, pkb, pab, pnb, kb, ab, nb):
            b.free()
        t.destroy()


@pytest.mark.gpu
def test_gpu_asof_agg_chain(engine):
    """Composability: ASOF join -> payload gather by build row -> generic CAS
    hash aggregate, vs the same chain through the oracle. The 'latest value
    as of each event, summed per key' shape ASOF joins exist for."""
    rng = np.random.default_rng(777)
    n_build, n_probe = 50_000, 500_000
    bk = np.concatenate([[0], rng.integers(1, 2_001, n_build)]).astype(np.int32)
    ba = np.concatenate([[0], rng.integers(0, 10**9, n_build)]).astype(np.int64)
    bval = np.concatenate([[0], rng.integers(1, 1000, n_build)]).astype(np.uint32)
    pk = rng.integers(1, 2_001, n_probe).astype(np.int32)
    pa = rng.integers(0, 10**9, n_probe).astype(np.int64)

    # oracle chain: per-probe matched build row -> gather bval -> sum by key
    want_match = orc.asof_inner_join(bk, ba, pk, pa, 3)  # GE: latest <= probe
    hit = want_match != 0
    expect = np.zeros(2001, np.int64)
    np.add.at(expect, pk[hit], bval[want_match[hit]].astype(np.int64))

    # GPU chain through the generic operators
    kb = engine.alloc(bk.nbytes); kb.h2d(bk)
    ab = engine.alloc(ba.nbytes); ab.h2d(ba)
    t = engine.asof_build(kb, ab, n_build, 3)
    pkb = engine.alloc(pk.nbytes); pkb.h2d(pk)
    pab = engine.alloc(pa.nbytes); pab.h2d(pa)
    cnt = engine.asof_probe_emit(t, pkb, pab, n_probe, 0)
    assert cnt == int(np.count_nonzero(hit))
    op = engine.alloc(max(cnt, 1) * 4)
    ob = engine.alloc(max(cnt, 1) * 4)
    engine.asof_probe_emit(t, pkb, pab, n_probe, 0, op, ob)
    # gather: group key by probe row, value by matched build row
    bval_d = engine.alloc(bval.nbytes); bval_d.h2d(bval)
    gkey = engine.alloc(max(cnt, 1) * 4)
    gval = engine.alloc(max(cnt, 1) * 4)
    engine.gather_u32(pkb, op, cnt, gkey)
    engine.gather_u32(bval_d, ob, cnt, gval)
    # widen to the agg's u64 key / i64 value layout
    keys64 = engine.alloc(max(cnt, 1) * 8)
    keys64.h2d(gkey.d2h(np.uint32, cnt).astype(np.uint64))
    vals64 = engine.alloc(max(cnt, 1) * 8)
    vals64.h2d(gval.d2h(np.uint32, cnt).astype(np.int64))
    ok_b = engine.alloc(4096 * 8)
    os_b = engine.alloc(4096 * 8)
    ng = engine.hash_agg_sum_u64(keys64, vals64, cnt, ok_b, os_b, max_out=4096)
    got = np.zeros(2001, np.int64)
    gk = ok_b.d2h(np.uint64, ng)
    gs = os_b.d2h(np.int64, ng)
    got[gk.astype(np.int64)] = gs
    assert np.array_equal(got, expect)
    for b in (kb, ab, pkb, pab, op, ob, bval_d, gkey, gval, keys64, vals64, ok_b, os_b):
        b.free()
    t.destroy()


@pytest.mark.gpu
def test_gpu_edge_all_null_build_and_empty_probe(engine):
    """Degenerate shapes: every build row null (no groups exist at all) and a
    zero-row probe must both work, not crash or hang."""
    n_build = 1_000
    bk = np.arange(n_build + 1, dtype=np.int32)
    ba = np.arange(n_build + 1, dtype=np.int64)
    bn = np.ones(n_build + 1, np.uint8)
    kb = engine.alloc(bk.nbytes); kb.h2d(bk)
    ab = engine.alloc(ba.nbytes); ab.h2d(ba)
    nb = engine.alloc(bn.nbytes); nb.h2d(bn)
    t = engine.asof_build_nulls(kb, ab, nb, n_build, 0)
    pk = np.arange(1, 101, dtype=np.int32)
    pa = np.zeros(100, np.int64)
    pkb = engine.alloc(pk.nbytes); pkb.h2d(pk)
    pab = engine.alloc(pa.nbytes); pab.h2d(pa)
    assert engine.asof_probe_emit(t, pkb, pab, 100, 0) == 0   # all miss
    assert engine.asof_probe_emit(t, pkb, pab, 100, 3) == 100  # outer: all rows
    assert engine.asof_probe_emit(t, pkb, pab, 0, 0) == 0      # empty probe
    ob = engine.alloc(100 * 4)
    op = engine.alloc(100 * 4)
    engine.asof_probe_emit(t, pkb, pab, 100, 3, op, ob)
    assert not np.any(ob.d2h(np.uint32, 100))
    for b in (kb, ab, nb, pkb, pab, op, ob):
        b.free()
    t.destroy()
