"""GPU↔oracle parity (the §8c gate). Every test here calls through the C-ABI
(libgpue.so) and compares against the CPU oracle on identical seeded inputs:
bit-exact for all integer work; join pair multisets compared sorted
(match ORDER is the only nondeterminism — SURVEY.md §7 hard part (a)).

Runs on a real MI355X via gpurun; marked gpu.
"""

import numpy as np
import pytest

from oracle import pyoracle as orc
from starrocks_amd import gen

pytestmark = pytest.mark.gpu

SEED = 42


def test_native_lib_is_intree(engine):
    """The loaded engine must be the in-tree HIP library, not a fallback."""
    from starrocks_amd.engine import lib_path
    import os
    assert os.path.exists(lib_path())
    assert "starrocks_amd" in lib_path()


def test_gen_u32_matches_oracle(engine):
    n = 1_000_000
    buf = engine.alloc(n * 4)
    engine.gen_u32_mod(buf, SEED, 77, 123, n, 1000, 5)
    got = buf.d2h(np.uint32, n)
    idx = np.arange(123, 123 + n, dtype=np.uint64)
    expect = (gen.gen_u64(SEED, 77, idx) % np.uint64(1000)).astype(np.uint32) + 5
    assert np.array_equal(got, expect)
    buf.free()


def test_gen_lineorder_q1_matches_oracle(engine):
    n = 2_000_000
    od, ep, dc = engine.alloc(n * 4), engine.alloc(n * 4), engine.alloc(n * 4)
    engine.gen_lineorder_q1(SEED, 500, n, od, ep, dc)
    od_c, ep_c, dc_c = orc.gen_lineorder_q1(SEED, 500, n)
    assert np.array_equal(od.d2h(np.int32, n), od_c)
    assert np.array_equal(ep.d2h(np.int32, n), ep_c)
    assert np.array_equal(dc.d2h(np.int32, n), dc_c)
    for b in (od, ep, dc):
        b.free()


@pytest.mark.parametrize("n,sel", [(1_000_000, 0.1), (10_000_000, 0.5), (999_999, 0.01), (4096, 1.0), (1000, 0.0)])
def test_filter_parity_bit_exact_ordered(engine, n, sel):
    inp = engine.alloc(n * 8)
    engine.gen_i64(inp, SEED, 3, 0, n)
    data = inp.d2h(np.int64, n)
    # theta at the requested selectivity over the uniform u64-as-i64 domain
    theta = int(np.quantile(data.astype(np.float64), sel)) if 0 < sel < 1 else (
        np.iinfo(np.int64).max if sel >= 1 else np.iinfo(np.int64).min)
    out = engine.alloc(n * 8)
    cnt = engine.scan_filter_i64_lt(inp, n, theta, out)
    expect = orc.filter_i64_lt(data, theta)
    assert cnt == len(expect)
    got = out.d2h(np.int64, cnt) if cnt else np.empty(0, np.int64)
    assert np.array_equal(got, expect)  # bit-exact INCLUDING order
    inp.free()
    out.free()


@pytest.mark.parametrize("n,sel", [(1_000_000, 0.1), (10_000_000, 0.5), (999_999, 0.01),
                                   (50_000_000, 0.3), (4096, 1.0), (1000, 0.0)])
def test_filter_singlepass_parity(engine, n, sel):
    """Single-pass decoupled-lookback compaction: bit-exact INCLUDING order
    vs the oracle (and therefore vs the two-pass kernel)."""
    inp = engine.alloc(n * 8)
    engine.gen_i64(inp, SEED, 3, 0, n)
    data = inp.d2h(np.int64, n)
    theta = int(np.quantile(data.astype(np.float64), sel)) if 0 < sel < 1 else (
        np.iinfo(np.int64).max if sel >= 1 else np.iinfo(np.int64).min)
    out = engine.alloc(n * 8)
    cnt = engine.scan_filter_i64_lt_sp(inp, n, theta, out)
    expect = orc.filter_i64_lt(data, theta)
    assert cnt == len(expect)
    got = out.d2h(np.int64, cnt) if cnt else np.empty(0, np.int64)
    assert np.array_equal(got, expect)
    inp.free()
    out.free()


def test_join_build_probe_emit_parity(engine):
    """Range-direct build + probe emit vs oracle: multiset of (probe,build)."""
    rng = np.random.default_rng(5)
    build_keys = np.concatenate([[0], rng.integers(100, 5000, 20_000)]).astype(np.int32)
    probe_keys = rng.integers(0, 6000, 300_000).astype(np.int32)

    kb = engine.alloc(build_keys.nbytes)
    kb.h2d(build_keys)
    t = engine.join_build_range_direct(kb, len(build_keys) - 1)
    mn, mx = t.minmax
    assert mn == int(build_keys[1:].min()) and mx == int(build_keys[1:].max())

    pb = engine.alloc(probe_keys.nbytes)
    pb.h2d(probe_keys)
    cnt = engine.join_probe_emit(t, pb, len(probe_keys))
    op_buf = engine.alloc(max(cnt, 1) * 4)
    ob_buf = engine.alloc(max(cnt, 1) * 4)
    cnt2 = engine.join_probe_emit(t, pb, len(probe_keys), op_buf, ob_buf)
    assert cnt2 == cnt
    gop = op_buf.d2h(np.uint32, cnt)
    gob = ob_buf.d2h(np.uint32, cnt)

    first, nxt = orc.range_direct_build(build_keys, mn, mx)
    heads = orc.range_direct_lookup(probe_keys, mn, mx, first)
    eop, eob = orc.probe_emit(build_keys.view(np.uint32), nxt, probe_keys.view(np.uint32), heads)
    assert len(eop) == cnt
    got = np.sort(gop.astype(np.uint64) << np.uint64(32) | gob.astype(np.uint64))
    exp = np.sort(eop.astype(np.uint64) << np.uint64(32) | eob.astype(np.uint64))
    assert np.array_equal(got, exp)
    for b in (kb, pb, op_buf, ob_buf):
        b.free()
    t.destroy()


def test_bucket_chained_join_parity(engine):
    """Generic (non-dense-key) join: GPU bucket-chained build+probe vs the
    oracle's restatement of the reference fallback method — match multisets
    equal; the hash/bucket assignment itself is pinned by the oracle KATs."""
    rng = np.random.default_rng(23)
    build_keys = np.concatenate([[0], rng.integers(0, 2**31, 100_000)]).astype(np.uint32)
    probe_keys = rng.choice(np.concatenate([build_keys[1:], rng.integers(0, 2**31, 50_000).astype(np.uint32)]),
                            400_000).astype(np.uint32)
    kb = engine.alloc(build_keys.nbytes)
    kb.h2d(build_keys.view(np.int32))
    t = engine.join_build_bucket_chained(kb, len(build_keys) - 1)
    pb = engine.alloc(probe_keys.nbytes)
    pb.h2d(probe_keys.view(np.int32))
    cnt = engine.join_probe_emit(t, pb, len(probe_keys))
    op_buf = engine.alloc(max(cnt, 1) * 4)
    ob_buf = engine.alloc(max(cnt, 1) * 4)
    engine.join_probe_emit(t, pb, len(probe_keys), op_buf, ob_buf)
    gop = op_buf.d2h(np.uint32, cnt)
    gob = ob_buf.d2h(np.uint32, cnt)

    first, nxt, bs, log = orc.bucket_chained_build(build_keys)
    heads = orc.bucket_chained_lookup(probe_keys, first, bs, log)
    eop, eob = orc.probe_emit(build_keys, nxt, probe_keys, heads)
    assert cnt == len(eop)
    got = np.sort(gop.astype(np.uint64) << np.uint64(32) | gob.astype(np.uint64))
    exp = np.sort(eop.astype(np.uint64) << np.uint64(32) | eob.astype(np.uint64))
    assert np.array_equal(got, exp)
    for b in (kb, pb, op_buf, ob_buf):
        b.free()
    t.destroy()


def test_linear_chained_join_parity(engine):
    """LINEAR_CHAINED (fp-packed, linear probing) vs the oracle restatement:
    slot placement under concurrent collisions may differ from sequential
    insertion, but the probe follows the same (fp,key) search — the match
    multiset is identical."""
    rng = np.random.default_rng(31)
    build_keys = np.concatenate([[0], rng.integers(0, 2**30, 80_000)]).astype(np.uint32)
    probe_keys = np.concatenate([build_keys[1:],
                                 rng.integers(0, 2**30, 40_000).astype(np.uint32)])
    rng.shuffle(probe_keys)
    kb = engine.alloc(build_keys.nbytes)
    kb.h2d(build_keys.view(np.int32))
    t = engine.join_build_linear_chained(kb, len(build_keys) - 1)
    pb = engine.alloc(probe_keys.nbytes)
    pb.h2d(probe_keys.view(np.int32))
    cnt = engine.join_probe_emit(t, pb, len(probe_keys))
    op_buf = engine.alloc(max(cnt, 1) * 4)
    ob_buf = engine.alloc(max(cnt, 1) * 4)
    engine.join_probe_emit(t, pb, len(probe_keys), op_buf, ob_buf)
    gop = op_buf.d2h(np.uint32, cnt)
    gob = ob_buf.d2h(np.uint32, cnt)

    lf, ln, ls, ll = orc.linear_chained_build(build_keys)
    heads = orc.linear_chained_lookup(build_keys, probe_keys, lf, ls, ll)
    eop, eob = orc.probe_emit(build_keys, ln, probe_keys, heads)
    assert cnt == len(eop)
    got = np.sort(gop.astype(np.uint64) << np.uint64(32) | gob.astype(np.uint64))
    exp = np.sort(eop.astype(np.uint64) << np.uint64(32) | eob.astype(np.uint64))
    assert np.array_equal(got, exp)
    for b in (kb, pb, op_buf, ob_buf):
        b.free()
    t.destroy()


def _build_date_table(engine, year_filter):
    datekey, dyear = gen.gen_dates()
    mn, mx, first = gen.build_date_dim_payload(year_filter)
    keys = engine.alloc(datekey.nbytes)
    keys.h2d(datekey.astype(np.int32))
    payload = np.where((dyear == year_filter) if year_filter else np.ones(len(dyear), bool),
                       dyear - 1992 + 1, 0).astype(np.uint32)
    pay = engine.alloc(payload.nbytes)
    pay.h2d(payload)
    t = engine.join_build_payload(keys, pay, len(datekey))
    # the device payload table must equal the host-computed one
    assert np.array_equal(t.first_d2h(mx - mn + 1), first)
    keys.free()
    pay.free()
    return t


@pytest.mark.parametrize("n_rows", [1_000_000, 10_000_000])
def test_q1_join_sum_parity(engine, n_rows):
    year = 1993
    od, ep, dc = (engine.alloc(n_rows * 4) for _ in range(3))
    engine.gen_lineorder_q1(SEED, 0, n_rows, od, ep, dc)
    dates = _build_date_table(engine, year)
    s, cnt = engine.q1_join_sum(dates, od, ep, dc, n_rows)
    es, ecnt = orc.q1_pipeline(SEED, 0, n_rows, year)
    assert (s, cnt) == (es, ecnt)  # bit-exact int64 sum
    for b in (od, ep, dc):
        b.free()
    dates.destroy()


def test_q21_star_agg_parity(engine):
    n_rows, cat, reg = 5_000_000, 12, 2
    pk, sk, od, rv = (engine.alloc(n_rows * 4) for _ in range(4))
    engine.gen_lineorder_q21(SEED, 0, n_rows, pk, sk, od, rv)

    pfirst = gen.build_part_dim_payload(SEED, gen.N_PARTS_SF100, cat)
    sfirst = gen.build_supp_dim_payload(SEED, gen.N_SUPPS_SF100, reg)

    pkeys = engine.alloc(gen.N_PARTS_SF100 * 4)
    pkeys.h2d(np.arange(1, gen.N_PARTS_SF100 + 1, dtype=np.int32))
    ppay = engine.alloc(pfirst.nbytes)
    ppay.h2d(pfirst)
    parts = engine.join_build_payload(pkeys, ppay, gen.N_PARTS_SF100)

    skeys = engine.alloc(gen.N_SUPPS_SF100 * 4)
    skeys.h2d(np.arange(1, gen.N_SUPPS_SF100 + 1, dtype=np.int32))
    spay = engine.alloc(sfirst.nbytes)
    spay.h2d(sfirst)
    supps = engine.join_build_payload(skeys, spay, gen.N_SUPPS_SF100)

    dates = _build_date_table(engine, None)

    got = engine.q21_star_agg(parts, supps, dates, pk, sk, od, rv, n_rows)
    expect = orc.q21_pipeline(SEED, 0, n_rows, cat, reg)
    assert np.array_equal(got, expect)  # bit-exact per-group int64 sums
    for b in (pk, sk, od, rv, pkeys, ppay, skeys, spay):
        b.free()
    for t in (parts, supps, dates):
        t.destroy()


def _build_q43_tables(engine):
    R, N, C = 1, 7, 12
    tabs = {}
    for name, keys, pay in [
        ("custs", np.arange(1, gen.N_CUSTS_SF100 + 1, dtype=np.int32),
         gen.build_cust_dim_q43(SEED, gen.N_CUSTS_SF100, R)),
        ("supps", np.arange(1, gen.N_SUPPS_SF100 + 1, dtype=np.int32),
         gen.build_supp_dim_q43(SEED, gen.N_SUPPS_SF100, N)),
        ("parts", np.arange(1, gen.N_PARTS_SF100 + 1, dtype=np.int32),
         gen.build_part_dim_q43(SEED, gen.N_PARTS_SF100, C)),
    ]:
        kb = engine.alloc(keys.nbytes)
        kb.h2d(keys)
        pb = engine.alloc(pay.nbytes)
        pb.h2d(pay)
        tabs[name] = engine.join_build_payload(kb, pb, len(keys))
        kb.free()
        pb.free()
    datekey, dyear = gen.gen_dates()
    dpay = np.where(dyear == 1997, 1, np.where(dyear == 1998, 2, 0)).astype(np.uint32)
    kb = engine.alloc(datekey.nbytes)
    kb.h2d(datekey.astype(np.int32))
    pb = engine.alloc(dpay.nbytes)
    pb.h2d(dpay)
    tabs["dates"] = engine.join_build_payload(kb, pb, len(datekey))
    kb.free()
    pb.free()
    return tabs


def test_q43_star_agg_parity(engine):
    from starrocks_amd.engine import c_u64, c_vp, _ck
    import ctypes
    n = 8_000_000
    cols = [engine.alloc(n * 4) for _ in range(6)]
    engine.gen_lineorder_q43(SEED, 0, n, *cols)
    tabs = _build_q43_tables(engine)
    acc = engine.alloc(800 * 8)
    engine.q43_star_agg_async(tabs["custs"], tabs["supps"], tabs["parts"],
                              tabs["dates"], *cols, n, acc)
    got = acc.d2h(np.int64, 800)
    from oracle import pyoracle as orc
    expect = orc.q43_pipeline(SEED, 0, n, 1, 7, 12)
    assert np.array_equal(got, expect)  # bit-exact int64 group sums
    for b in cols:
        b.free()
    acc.free()
    for t in tabs.values():
        t.destroy()


def test_q43_partition_gather_roundtrip(engine):
    """Single-GPU simulation of the N>1 partitioned pipeline (bench.py q43):
    partition rows by lo_custkey, gather all 6 columns into channel order,
    run the agg per channel slice against the channel's customer shard, merge
    — must equal the unpartitioned result bit-exactly. This is the same math
    the RCCL all-to-all path executes across 8 GPUs."""
    n, world = 4_000_000, 4
    cols = [engine.alloc(n * 4) for _ in range(6)]
    engine.gen_lineorder_q43(SEED, 0, n, *cols)
    tabs = _build_q43_tables(engine)

    ridx = engine.alloc(n * 4)
    sp = engine.partition(cols[0], n, world, ridx)
    send = [engine.alloc(n * 4) for _ in range(6)]
    for c, s in zip(cols, send):
        engine.gather_u32(c, ridx, n, s)

    ckeys = np.arange(1, gen.N_CUSTS_SF100 + 1, dtype=np.uint32)
    cust_owner = gen.partition_channels(ckeys, world)
    cpay_full = gen.build_cust_dim_q43(SEED, gen.N_CUSTS_SF100, 1)
    merged = np.zeros(800, np.int64)
    acc = engine.alloc(800 * 8)
    for r in range(world):
        lo, hi = int(sp[r]), int(sp[r + 1])
        if hi == lo:
            continue
        cpay_r = np.where(cust_owner == r, cpay_full, 0).astype(np.uint32)
        kb = engine.alloc(ckeys.nbytes)
        kb.h2d(ckeys.view(np.int32))
        pb = engine.alloc(cpay_r.nbytes)
        pb.h2d(cpay_r)
        custs_r = engine.join_build_payload(kb, pb, len(ckeys))
        kb.free()
        pb.free()
        # per-channel slice: wrap offsets into the gathered buffers
        slice_bufs = [engine.wrap_ptr_offset(s, lo * 4, (hi - lo) * 4) for s in send]
        engine.q43_star_agg_async(custs_r, tabs["supps"], tabs["parts"], tabs["dates"],
                                  *slice_bufs, hi - lo, acc)
        merged += acc.d2h(np.int64, 800)
        custs_r.destroy()

    from oracle import pyoracle as orc
    expect = orc.q43_pipeline(SEED, 0, n, 1, 7, 12)
    assert np.array_equal(merged, expect)
    for b in cols + send + [ridx, acc]:
        b.free()
    for t in tabs.values():
        t.destroy()


@pytest.mark.parametrize("nkeys,card", [(2_000_000, 100), (2_000_000, 500_000), (1_000_000, 2**62)])
def test_hash_agg_parity(engine, nkeys, card):
    """Generic hash aggregate vs oracle: key-sorted (keys, sums, counts)
    must match bit-exactly (int64, order-independent)."""
    rng = np.random.default_rng(17)
    keys = (rng.integers(0, card, nkeys).astype(np.uint64)
            if card <= 2**32 else
            rng.integers(0, 2**62, nkeys, dtype=np.int64).astype(np.uint64))
    vals = rng.integers(-10**9, 10**9, nkeys).astype(np.int64)
    kb = engine.alloc(keys.nbytes)
    kb.h2d(keys)
    vb = engine.alloc(vals.nbytes)
    vb.h2d(vals)
    max_out = min(nkeys, 1_200_000)
    ok = engine.alloc(max_out * 8)
    os_ = engine.alloc(max_out * 8)
    oc = engine.alloc(max_out * 8)
    g = engine.hash_agg_sum_u64(kb, vb, nkeys, ok, os_, oc, max_out)
    gk = ok.d2h(np.uint64, g)
    gs = os_.d2h(np.int64, g)
    gc = oc.d2h(np.int64, g)
    ek, es, ec = orc.hash_agg_sum(keys, vals)
    assert g == len(ek)
    order = np.argsort(gk)
    eorder = np.argsort(ek)
    assert np.array_equal(gk[order], ek[eorder])
    assert np.array_equal(gs[order], es[eorder])
    assert np.array_equal(gc[order], ec[eorder])
    for b in (kb, vb, ok, os_, oc):
        b.free()


def test_q3_pipeline_parity(engine):
    """Config 5 (TPC-H Q3-shaped) on device vs oracle: string-dict filter,
    orders semi-join bitset, high-cardinality GROUP BY — key-sorted (keys,
    sums) bit-exact."""
    n, n_orders, n_custs = 3_000_000, 200_000, 40_000
    mkt = engine.alloc(n_custs * 16)
    engine.gen_cust_mkt16(SEED, n_custs, mkt)
    cbits = engine.alloc((n_custs + 31) // 32 * 4)
    engine.bits_str16_eq(mkt, n_custs, orc.mkt_literal(1), cbits)
    oc, od = engine.alloc(n_orders * 4), engine.alloc(n_orders * 4)
    engine.gen_orders_q3(SEED, n_orders, n_custs, oc, od)
    obits = engine.alloc((n_orders + 31) // 32 * 4)
    engine.q3_order_bits(oc, od, n_orders, cbits, 19950315, obits)
    lk, ext, disc = (engine.alloc(n * 8) for _ in range(3))
    ship = engine.alloc(n * 4)
    engine.gen_lineitem_q3(SEED, 0, n, n_orders, lk, ext, disc, ship)
    max_out = n
    ok_b, os_b = engine.alloc(max_out * 8), engine.alloc(max_out * 8)
    g = engine.q3_probe_agg(lk, ext, disc, ship, n, obits, 19950315, ok_b, os_b, max_out)
    gk = ok_b.d2h(np.uint64, g)
    gs = os_b.d2h(np.int64, g)
    order = np.argsort(gk)
    ek, es = orc.q3_pipeline(SEED, 0, n, n_orders, n_custs)
    assert g == len(ek)
    assert np.array_equal(gk[order], ek)
    assert np.array_equal(gs[order], es)
    for b in (mkt, cbits, oc, od, obits, lk, ext, disc, ship, ok_b, os_b):
        b.free()


def test_hash_agg_stats_parity(engine):
    """SUM/COUNT/MIN/MAX states + AVG finalize vs oracle — int64 bit-exact;
    AVG is the double division of the exact sum/count pair (identical IEEE
    operands => bit-identical; stated tolerance 1e-12 regardless)."""
    rng = np.random.default_rng(41)
    n = 1_000_000
    keys = rng.integers(0, 10_000, n).astype(np.uint64)
    vals = rng.integers(-10**12, 10**12, n).astype(np.int64)
    kb = engine.alloc(keys.nbytes)
    kb.h2d(keys)
    vb = engine.alloc(vals.nbytes)
    vb.h2d(vals)
    outs = [engine.alloc(20_000 * 8) for _ in range(5)]
    g = engine.hash_agg_stats_u64(kb, vb, n, *outs, 20_000)
    gk = outs[0].d2h(np.uint64, g)
    gs, gc, gmn, gmx = (o.d2h(np.int64, g) for o in outs[1:])
    ek, es, ec, emn, emx = orc.hash_agg_stats(keys, vals)
    assert g == len(ek)
    o1, o2 = np.argsort(gk), np.argsort(ek)
    assert np.array_equal(gk[o1], ek[o2])
    assert np.array_equal(gs[o1], es[o2])
    assert np.array_equal(gc[o1], ec[o2])
    assert np.array_equal(gmn[o1], emn[o2])
    assert np.array_equal(gmx[o1], emx[o2])
    # AVG finalize (float path): double division of exact int64 pairs
    gavg = gs[o1].astype(np.float64) / gc[o1]
    eavg = es[o2].astype(np.float64) / ec[o2]
    assert np.allclose(gavg, eavg, rtol=1e-12, atol=0)
    for b in [kb, vb] + outs:
        b.free()


def test_hash_agg_sum128_parity(engine):
    """Decimal128 SUM (int128 lo/hi with carry) vs oracle __int128 — exact."""
    rng = np.random.default_rng(43)
    n = 2_000_000
    keys = rng.integers(0, 500, n).astype(np.uint64)
    vals = rng.integers(-2**62, 2**62, n).astype(np.int64)
    kb = engine.alloc(keys.nbytes)
    kb.h2d(keys)
    vb = engine.alloc(vals.nbytes)
    vb.h2d(vals)
    ok_b, lo_b, hi_b = (engine.alloc(1000 * 8) for _ in range(3))
    g = engine.hash_agg_sum128_u64(kb, vb, n, ok_b, lo_b, hi_b, 1000)
    gk = ok_b.d2h(np.uint64, g)
    glo = lo_b.d2h(np.uint64, g)
    ghi = hi_b.d2h(np.int64, g)
    ek, elo, ehi = orc.hash_agg_sum128(keys, vals)
    assert g == len(ek)
    o1, o2 = np.argsort(gk), np.argsort(ek)
    assert np.array_equal(gk[o1], ek[o2])
    assert np.array_equal(glo[o1], elo[o2])
    assert np.array_equal(ghi[o1], ehi[o2])
    for b in (kb, vb, ok_b, lo_b, hi_b):
        b.free()


def test_partition_i64_and_gather_u64_parity(engine):
    """BIGINT-key partition (fnv64 + ReduceOp) + u64 gather — the q3 N>1
    exchange pieces: channel sizes and per-channel row sets bit-identical to
    the oracle; gathered i64 columns match a numpy take."""
    n, nch = 2_000_000, 8
    keys = engine.alloc(n * 8)
    engine.gen_i64(keys, SEED, 15, 0, n)
    host_keys = keys.d2h(np.int64, n)
    ri = engine.alloc(n * 4)
    sp = engine.partition_i64(keys, n, nch, ri)
    ch = orc.partition_channels_u64(host_keys.view(np.uint64), nch)
    esp, _ = orc.partition_counting_sort(ch, nch)
    assert np.array_equal(sp, esp)
    got_ri = ri.d2h(np.uint32, n)
    for c in range(nch):
        rows = got_ri[int(sp[c]):int(sp[c + 1])]
        assert (ch[rows] == c).all()
        assert len(np.unique(rows)) == len(rows)
    out = engine.alloc(n * 8)
    engine.gather_u64(keys, ri, n, out)
    assert np.array_equal(out.d2h(np.int64, n), host_keys[got_ri])
    for b in (keys, ri, out):
        b.free()


def test_topk_parity(engine):
    """TopN (ORDER BY value DESC LIMIT k): deterministic (value,key)
    lexicographic order vs numpy, including value ties."""
    rng = np.random.default_rng(47)
    n = 3_000_000
    keys = rng.integers(1, 2**40, n).astype(np.uint64)
    vals = rng.integers(0, 50_000, n).astype(np.int64)  # many ties
    kb = engine.alloc(keys.nbytes)
    kb.h2d(keys)
    vb = engine.alloc(vals.nbytes)
    vb.h2d(vals)
    for k in (1, 10, 16):
        gk, gv = engine.topk_i64(kb, vb, n, k)
        order = np.lexsort((keys, vals))[::-1][:k]  # value desc, key desc tie-break
        assert np.array_equal(gv, vals[order])
        assert np.array_equal(gk, keys[order])
    kb.free()
    vb.free()


def test_q3_with_topn(engine):
    """Q3's full result shape: join + high-cardinality agg + ORDER BY revenue
    DESC LIMIT 10 — top-10 of the group sums vs numpy over the oracle's."""
    n, n_orders, n_custs = 2_000_000, 150_000, 30_000
    mkt = engine.alloc(n_custs * 16)
    engine.gen_cust_mkt16(SEED, n_custs, mkt)
    cbits = engine.alloc((n_custs + 31) // 32 * 4)
    engine.bits_str16_eq(mkt, n_custs, orc.mkt_literal(1), cbits)
    oc, od = engine.alloc(n_orders * 4), engine.alloc(n_orders * 4)
    engine.gen_orders_q3(SEED, n_orders, n_custs, oc, od)
    obits = engine.alloc((n_orders + 31) // 32 * 4)
    engine.q3_order_bits(oc, od, n_orders, cbits, 19950315, obits)
    lk, ext, disc = (engine.alloc(n * 8) for _ in range(3))
    ship = engine.alloc(n * 4)
    engine.gen_lineitem_q3(SEED, 0, n, n_orders, lk, ext, disc, ship)
    ok_b, os_b = engine.alloc(n * 8), engine.alloc(n * 8)
    g = engine.q3_probe_agg(lk, ext, disc, ship, n, obits, 19950315, ok_b, os_b, n)
    tk, tv = engine.topk_i64(ok_b, os_b, g, 10)
    ek, es = orc.q3_pipeline(SEED, 0, n, n_orders, n_custs)
    order = np.lexsort((ek, es))[::-1][:10]
    assert np.array_equal(tv, es[order])
    assert np.array_equal(tk, ek[order])
    for b in (mkt, cbits, oc, od, obits, lk, ext, disc, ship, ok_b, os_b):
        b.free()


def test_page_decode_parity(engine):
    """Storage ingress: GPU bitshuffle+LZ4 page decode == oracle decode ==
    original values, for compressible and incompressible pages."""
    rng = np.random.default_rng(53)
    for vals in [rng.integers(0, 50, 2048 * 200).astype(np.int32),
                 rng.integers(0, 2**31, 2048 * 37 + 8).astype(np.int32),
                 np.arange(2048 * 10, dtype=np.int32)]:
        page = orc.bshuf_lz4_encode_i32(vals)
        pbuf = engine.alloc(len(page))
        pbuf.h2d(page)
        out = engine.alloc(len(vals) * 4)
        engine.page_decode_bshuf_lz4_i32(pbuf, len(vals), out)
        got = out.d2h(np.int32, len(vals))
        assert np.array_equal(got, vals)
        pbuf.free()
        out.free()


def test_page_decode_feeds_join(engine):
    """Ingress composed with the hot path: lo_orderdate arrives as a
    bitshuffle+LZ4 page, is decoded on device, and feeds the q1 join+SUM —
    result bit-exact vs the all-generated run."""
    n = 4_000_000
    od_vals, _, _ = orc.gen_lineorder_q1(SEED, 0, n)
    page = orc.bshuf_lz4_encode_i32(od_vals)
    pbuf = engine.alloc(len(page))
    pbuf.h2d(page)
    od = engine.alloc(n * 4)
    engine.page_decode_bshuf_lz4_i32(pbuf, n, od)
    ep, dc = engine.alloc(n * 4), engine.alloc(n * 4)
    od2 = engine.alloc(n * 4)
    engine.gen_lineorder_q1(SEED, 0, n, od2, ep, dc)
    dates = _build_date_table(engine, 1993)
    s1, c1 = engine.q1_join_sum(dates, od, ep, dc, n)
    s2, c2 = engine.q1_join_sum(dates, od2, ep, dc, n)
    es, ec = orc.q1_pipeline(SEED, 0, n, 1993)
    assert (s1, c1) == (s2, c2) == (es, ec)
    for b in (pbuf, od, od2, ep, dc):
        b.free()
    dates.destroy()


def test_join_modes_parity(engine):
    """LEFT SEMI/ANTI/OUTER + RIGHT SEMI/ANTI probe variants vs oracle, over
    both RANGE_DIRECT and BUCKET_CHAINED tables."""
    rng = np.random.default_rng(61)
    build_keys = np.concatenate([[0], rng.integers(100, 4000, 30_000)]).astype(np.uint32)
    probe_keys = rng.integers(0, 5000, 200_000).astype(np.uint32)
    kb = engine.alloc(build_keys.nbytes)
    kb.h2d(build_keys.view(np.int32))
    pb = engine.alloc(probe_keys.nbytes)
    pb.h2d(probe_keys.view(np.int32))
    mn, mx = int(build_keys[1:].min()), int(build_keys[1:].max())

    for kind in ("rd", "bc"):
        if kind == "rd":
            t = engine.join_build_range_direct(kb, len(build_keys) - 1)
            efirst, enxt = orc.range_direct_build(build_keys.view(np.int32), mn, mx)
            heads = orc.range_direct_lookup(probe_keys.view(np.int32), mn, mx, efirst)
        else:
            t = engine.join_build_bucket_chained(kb, len(build_keys) - 1)
            efirst, enxt, bs, log = orc.bucket_chained_build(build_keys)
            heads = orc.bucket_chained_lookup(probe_keys, efirst, bs, log)
        for mode in (0, 1, 2, 3):
            cnt = engine.join_probe_emit_mode(t, pb, len(probe_keys), mode)
            op_b = engine.alloc(max(cnt, 1) * 4)
            ob_b = engine.alloc(max(cnt, 1) * 4)
            engine.join_probe_emit_mode(t, pb, len(probe_keys), mode, op_b, ob_b)
            gop = op_b.d2h(np.uint32, cnt)
            gob = ob_b.d2h(np.uint32, cnt)
            eop, eob = orc.probe_emit_mode(build_keys, enxt, probe_keys, heads, mode)
            assert cnt == len(eop), (kind, mode, cnt, len(eop))
            if mode == 1:
                # SEMI's emitted build row is chain-order-dependent: the probe
                # row set + key equality are the contract
                assert sorted(gop.tolist()) == sorted(eop.tolist())
                assert (build_keys[gob] == probe_keys[gop]).all()
            else:
                got = np.sort(gop.astype(np.uint64) << np.uint64(32) | gob.astype(np.uint64))
                exp = np.sort(eop.astype(np.uint64) << np.uint64(32) | eob.astype(np.uint64))
                assert np.array_equal(got, exp), (kind, mode)
            op_b.free()
            ob_b.free()
        for anti in (0, 1):
            cnt = engine.join_probe_right(t, pb, len(probe_keys), anti)
            ob_b = engine.alloc(max(cnt, 1) * 4)
            engine.join_probe_right(t, pb, len(probe_keys), anti, ob_b)
            got = np.sort(ob_b.d2h(np.uint32, cnt))
            exp = np.sort(orc.probe_right(build_keys, enxt, probe_keys, heads, anti))
            assert np.array_equal(got, exp), (kind, anti)
            ob_b.free()
        t.destroy()
    kb.free()
    pb.free()


def test_nullable_join_parity(engine):
    """Nullable build + probe (JoinBuildProbeFuncNullable semantics,
    reference join_hash_map_test.cpp:1190-1240): null build rows never
    match; null probe rows match nothing (ANTI/OUTER emit them unmatched)."""
    rng = np.random.default_rng(67)
    nbuild, nprobe = 20_000, 150_000
    build_keys = np.concatenate([[0], rng.integers(0, 2000, nbuild)]).astype(np.uint32)
    build_nulls = np.concatenate([[0], (rng.random(nbuild) < 0.3)]).astype(np.uint8)
    probe_keys = rng.integers(0, 2500, nprobe).astype(np.uint32)
    probe_nulls = (rng.random(nprobe) < 0.2).astype(np.uint8)

    kb = engine.alloc(build_keys.nbytes)
    kb.h2d(build_keys.view(np.int32))
    nb = engine.alloc(build_nulls.nbytes)
    nb.h2d(build_nulls)
    t = engine.join_build_bucket_chained_nulls(kb, nb, nbuild)
    pb = engine.alloc(probe_keys.nbytes)
    pb.h2d(probe_keys.view(np.int32))
    pn = engine.alloc(probe_nulls.nbytes)
    pn.h2d(probe_nulls)

    first, nxt, bs, log = orc.bucket_chained_build_nulls(build_keys, build_nulls)
    heads = orc.bucket_chained_lookup_nulls(probe_keys, probe_nulls, first, bs, log)
    for mode in (0, 1, 2, 3):
        cnt = engine.join_probe_emit_nulls(t, pb, pn, nprobe, mode)
        op_b = engine.alloc(max(cnt, 1) * 4)
        ob_b = engine.alloc(max(cnt, 1) * 4)
        engine.join_probe_emit_nulls(t, pb, pn, nprobe, mode, op_b, ob_b)
        gop = op_b.d2h(np.uint32, cnt)
        gob = ob_b.d2h(np.uint32, cnt)
        eop, eob = orc.probe_emit_mode(build_keys, nxt, probe_keys, heads, mode)
        assert cnt == len(eop), (mode, cnt, len(eop))
        if mode == 1:
            assert sorted(gop.tolist()) == sorted(eop.tolist())
            assert (build_keys[gob] == probe_keys[gop]).all()
        else:
            got = np.sort(gop.astype(np.uint64) << np.uint64(32) | gob.astype(np.uint64))
            exp = np.sort(eop.astype(np.uint64) << np.uint64(32) | eob.astype(np.uint64))
            assert np.array_equal(got, exp), mode
        op_b.free()
        ob_b.free()
    for b in (kb, nb, pb, pn):
        b.free()
    t.destroy()


def test_packed_multikey_groupby(engine):
    """Multi-column GROUP BY via SERIALIZED_FIXED_SIZE packing: pack two i32
    key columns to u64 on device, aggregate, compare with a numpy groupby
    over the column PAIR."""
    rng = np.random.default_rng(71)
    n = 2_000_000
    a = rng.integers(0, 50, n).astype(np.int32)
    b = rng.integers(0, 40, n).astype(np.int32)
    v = rng.integers(-10**9, 10**9, n).astype(np.int64)
    ab = engine.alloc(n * 4)
    ab.h2d(a)
    bb = engine.alloc(n * 4)
    bb.h2d(b)
    keys = engine.alloc(n * 8)
    engine.pack_keys_2xi32(ab, bb, n, keys)
    # device packing == oracle packing
    assert np.array_equal(keys.d2h(np.uint64, n), orc.pack_keys_2xi32(a, b))
    vb = engine.alloc(n * 8)
    vb.h2d(v)
    outs = [engine.alloc(4000 * 8) for _ in range(2)]
    g = engine.hash_agg_sum_u64(keys, vb, n, outs[0], outs[1], None, 4000)
    gk = outs[0].d2h(np.uint64, g)
    gs = outs[1].d2h(np.int64, g)
    packed = a.astype(np.uint64) | (b.astype(np.uint64) << np.uint64(32))
    uk, inv = np.unique(packed, return_inverse=True)
    es = np.zeros(len(uk), np.int64)
    np.add.at(es, inv, v)
    order = np.argsort(gk)
    assert np.array_equal(gk[order], uk)
    assert np.array_equal(gs[order], es)
    for x in [ab, bb, keys, vb] + outs:
        x.free()


def test_partition_parity(engine):
    n, nch = 3_000_000, 8
    keys = engine.alloc(n * 4)
    engine.gen_u32_mod(keys, SEED, 9, 0, n, 0, 0)
    host_keys = keys.d2h(np.uint32, n)
    ri = engine.alloc(n * 4)
    sp = engine.partition(keys, n, nch, ri)
    ch = orc.partition_channels(host_keys, nch)
    esp, _ = orc.partition_counting_sort(ch, nch)
    assert np.array_equal(sp, esp)  # channel sizes bit-identical
    got_ri = ri.d2h(np.uint32, n)
    for c in range(nch):
        rows = got_ri[int(sp[c]):int(sp[c + 1])]
        # every row landed on its reference channel; per-channel row SET pinned
        assert (ch[rows] == c).all()
        assert len(np.unique(rows)) == len(rows)
    keys.free()
    ri.free()


def _varchar_cols(rng, n, card, one_based):
    """BinaryColumn-shaped (bytes, offsets) with duplicate-heavy strings."""
    pool = [f"city_{i:04d}".encode() + b"x" * int(rng.integers(0, 9)) for i in range(card)]
    rows = [pool[int(i)] for i in rng.integers(0, card, n)]
    if one_based:
        rows = [b""] + rows  # row 0 = empty sentinel
    offsets = np.zeros(len(rows) + 1, np.uint32)
    np.cumsum([len(r) for r in rows], out=offsets[1:])
    return np.frombuffer(b"".join(rows), np.uint8).copy(), offsets, rows


def test_varchar_join_parity(engine):
    """SERIALIZED_VARCHAR / Slice keys (join_hash_map.cpp:269-281): GPU
    crc-hash chained build + byte-verify probe vs the oracle restatement —
    match-pair multisets equal; the Slice hash itself is pinned by the
    ref-shim crc cross-check in test_oracle_golden."""
    rng = np.random.default_rng(77)
    n_build, n_probe, card = 40_000, 200_000, 700
    bb, bo, brows = _varchar_cols(rng, n_build, card, one_based=True)
    pb_, po, prows = _varchar_cols(rng, n_probe, 2 * card, one_based=False)

    eop, eob = orc.slice_join(bb, bo, n_build, pb_, po, n_probe, 64_000_000)

    d_bb, d_bo = engine.alloc(bb.nbytes), engine.alloc(bo.nbytes)
    d_bb.h2d(bb)
    d_bo.h2d(bo)
    t = engine.join_build_varchar(d_bb, d_bo, n_build)
    d_pb, d_po = engine.alloc(pb_.nbytes), engine.alloc(po.nbytes)
    d_pb.h2d(pb_)
    d_po.h2d(po)
    cnt = engine.join_probe_emit_varchar(t, d_pb, d_po, n_probe)
    assert cnt == len(eop)
    op_buf, ob_buf = engine.alloc(max(cnt, 1) * 4), engine.alloc(max(cnt, 1) * 4)
    cnt2 = engine.join_probe_emit_varchar(t, d_pb, d_po, n_probe, op_buf, ob_buf)
    assert cnt2 == cnt
    gop = op_buf.d2h(np.uint32, cnt)
    gob = ob_buf.d2h(np.uint32, cnt)
    # every emitted pair joins equal strings
    for i in rng.integers(0, cnt, 200):
        assert prows[gop[i]] == brows[gob[i]]
    got = np.sort(gop.astype(np.uint64) << np.uint64(32) | gob.astype(np.uint64))
    exp = np.sort(eop.astype(np.uint64) << np.uint64(32) | eob.astype(np.uint64))
    assert np.array_equal(got, exp)
    for b in (d_bb, d_bo, d_pb, d_po, op_buf, ob_buf):
        b.free()
    t.destroy()


def test_q1_operator_pipeline_parity(engine):
    """The chunked operator plan (source -> build -> probe+gather -> agg sink,
    starrocks_amd/pipeline.py mirroring operator.h:141-144 push/pull) produces
    the SAME bit-exact (sum, count) as the fused q1 kernel and the oracle —
    the §8b boundary exercised end to end over multiple bounded chunks."""
    from starrocks_amd.pipeline import q1_operator_pipeline
    n = 3_000_000
    got = q1_operator_pipeline(engine, SEED, n, chunk_rows=700_000)
    expect = orc.q1_pipeline(SEED, 0, n, 1993)
    assert got == expect


def test_graph_replay_q1(engine):
    """hipGraph capture/replay of the q1 step (include/gpue.h graph API):
    two replays produce the same bit-exact (sum, count) as the direct call
    and the oracle."""
    n = 2_000_000
    cols = [engine.alloc(n * 4) for _ in range(3)]
    engine.gen_lineorder_q1(SEED, 0, n, *cols)
    from starrocks_amd import gen
    datekey, dyear = gen.gen_dates()
    dpay = np.where(dyear == 1993, dyear - 1992 + 1, 0).astype(np.uint32)
    kb = engine.alloc(datekey.nbytes)
    kb.h2d(datekey.astype(np.int32))
    pb = engine.alloc(dpay.nbytes)
    pb.h2d(dpay)
    t = engine.join_build_payload(kb, pb, len(datekey))
    acc = engine.alloc(16)
    g = engine.graph_capture(
        lambda: engine.q1_join_sum_async(t, cols[0], cols[1], cols[2], n, acc))
    expect = orc.q1_pipeline(SEED, 0, n, 1993)
    for _ in range(2):
        engine.graph_launch(g)
        got = acc.d2h(np.int64, 2)
        assert (int(got[0]), int(got[1])) == expect
    engine.graph_destroy(g)
    direct = engine.q1_join_sum(t, *cols, n)
    assert direct == expect
    for b in cols + [kb, pb, acc]:
        b.free()
    t.destroy()


def test_varchar_join_modes_parity(engine):
    """Slice-key SEMI/ANTI/OUTER probes vs the oracle mode restatement."""
    rng = np.random.default_rng(13)
    n_build, n_probe, card = 20_000, 120_000, 500
    bb, bo, brows = _varchar_cols(rng, n_build, card, one_based=True)
    pb_, po, prows = _varchar_cols(rng, n_probe, 2 * card, one_based=False)
    d_bb, d_bo = engine.alloc(bb.nbytes), engine.alloc(bo.nbytes)
    d_bb.h2d(bb)
    d_bo.h2d(bo)
    t = engine.join_build_varchar(d_bb, d_bo, n_build)
    d_pb, d_po = engine.alloc(pb_.nbytes), engine.alloc(po.nbytes)
    d_pb.h2d(pb_)
    d_po.h2d(po)
    for mode in (1, 2, 3):
        eop, eob = orc.slice_join_mode(bb, bo, n_build, pb_, po, n_probe, mode, 64_000_000)
        cnt = engine.join_probe_emit_varchar_mode(t, d_pb, d_po, n_probe, mode)
        assert cnt == len(eop), mode
        op_buf, ob_buf = engine.alloc(max(cnt, 1) * 4), engine.alloc(max(cnt, 1) * 4)
        engine.join_probe_emit_varchar_mode(t, d_pb, d_po, n_probe, mode, op_buf, ob_buf)
        gop = op_buf.d2h(np.uint32, cnt)
        gob = ob_buf.d2h(np.uint32, cnt)
        if mode == 1:
            # SEMI: one emit per matching probe row; duplicate choice is
            # chain-order dependent — compare probe sets + key equality
            assert np.array_equal(np.sort(gop), np.sort(eop))
            for i in rng.integers(0, cnt, 100):
                assert prows[gop[i]] == brows[gob[i]]
        else:
            got = np.sort(gop.astype(np.uint64) << np.uint64(32) | gob.astype(np.uint64))
            exp = np.sort(eop.astype(np.uint64) << np.uint64(32) | eob.astype(np.uint64))
            assert np.array_equal(got, exp), mode
        op_buf.free()
        ob_buf.free()
    for b in (d_bb, d_bo, d_pb, d_po):
        b.free()
    t.destroy()


def test_cabi_cpp_demo(engine):
    """The standalone C++ caller (examples/cabi_q1.cpp, linked against
    libgpue.so with no Python in the loop) produces the oracle's exact
    (sum, count) for the config-2 plan."""
    import json
    import os
    import subprocess
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    exe = os.path.join(repo, "examples", "cabi_q1")
    if not os.path.exists(exe):
        subprocess.run(["g++", "-O2", f"-I{os.path.join(repo, 'include')}",
                        os.path.join(repo, "examples", "cabi_q1.cpp"),
                        f"-L{os.path.join(repo, 'starrocks_amd')}", "-lgpue",
                        "-Wl,-rpath,$ORIGIN/../starrocks_amd", "-o", exe], check=True)
    n = 1_500_000
    out = subprocess.run([exe, str(n)], capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr
    got = json.loads(out.stdout.strip())
    es, ecnt = orc.q1_pipeline(42, 0, n, 1993)
    assert (got["sum"], got["count"]) == (es, ecnt)


def test_sbf_runtime_filter_parity(engine):
    """SimdBlockFilter (split-block bloom runtime filter, runtime_filter.h:
    79-232): the GPU atomicOr build produces a directory BIT-IDENTICAL to the
    oracle's serial build (lane ORs are order-independent), membership tests
    match per row, and probing it before a join (the reference's scan-side
    early prune) never changes the join result."""
    rng = np.random.default_rng(19)
    n_build, n_probe = 200_000, 1_000_000
    bkeys = rng.choice(20_000_000, n_build, replace=False).astype(np.int32)
    pkeys = rng.integers(0, 40_000_000, n_probe).astype(np.int32)

    directory, log = orc.sbf_build(bkeys)
    kb = engine.alloc(bkeys.nbytes)
    kb.h2d(bkeys)
    d_dir = engine.alloc(directory.nbytes)
    engine.sbf_build(kb, n_build, log, d_dir)
    assert np.array_equal(d_dir.d2h(np.uint32, len(directory)), directory)

    pb = engine.alloc(pkeys.nbytes)
    pb.h2d(pkeys)
    d_out = engine.alloc(n_probe)
    engine.sbf_test(pb, n_probe, d_dir, log, d_out)
    got = d_out.d2h(np.uint8, n_probe)
    assert np.array_equal(got, orc.sbf_test(pkeys, directory, log))

    # no false negatives: every build key tests positive
    d_out2 = engine.alloc(n_build)
    engine.sbf_test(kb, n_build, d_dir, log, d_out2)
    assert d_out2.d2h(np.uint8, n_build).all()

    # integration: join with SBF pre-prune == join without (result unchanged)
    bkeys1 = np.concatenate([[0], bkeys]).astype(np.uint32)
    kb1 = engine.alloc(bkeys1.nbytes)
    kb1.h2d(bkeys1)
    t = engine.join_build_bucket_chained(kb1, n_build)
    full = engine.join_probe_emit(t, pb, n_probe)
    kept = pkeys[got.astype(bool)]
    pb2 = engine.alloc(max(kept.nbytes, 4))
    pb2.h2d(kept)
    pruned = engine.join_probe_emit(t, pb2, len(kept))
    assert pruned == full  # the prune dropped only non-matching rows
    assert len(kept) < n_probe  # and it did drop rows
    for b in (kb, kb1, d_dir, pb, pb2, d_out, d_out2):
        b.free()
    t.destroy()


def test_partition_2xi32_parity(engine):
    """Two-column chained-FNV partition: channel sizes + per-channel row SETS
    match the oracle (within-channel order is block-atomic nondeterministic,
    as for the single-column kernels)."""
    rng = np.random.default_rng(29)
    n, nch = 2_000_000, 8
    a = rng.integers(-2**31, 2**31, n).astype(np.int32)
    b = rng.integers(-2**31, 2**31, n).astype(np.int32)
    da, db = engine.alloc(a.nbytes), engine.alloc(b.nbytes)
    da.h2d(a)
    db.h2d(b)
    ridx = engine.alloc(n * 4)
    sp = engine.partition_2xi32(da, db, n, nch, ridx)
    ch = gen.partition_channels_2xi32(a, b, nch)
    expect_counts = np.bincount(ch, minlength=nch)
    assert np.array_equal(np.diff(sp.astype(np.int64)), expect_counts)
    got = ridx.d2h(np.uint32, n)
    for c in range(nch):
        seg = got[int(sp[c]):int(sp[c + 1])]
        assert np.array_equal(np.sort(seg), np.flatnonzero(ch == c).astype(np.uint32))
    for buf in (da, db, ridx):
        buf.free()


def test_varchar_join_nulls_parity(engine):
    """Nullable Slice-key join vs the oracle restatement across modes."""
    rng = np.random.default_rng(43)
    n_build, n_probe, card = 15_000, 80_000, 400
    bb, bo, brows = _varchar_cols(rng, n_build, card, one_based=True)
    pb_, po, prows = _varchar_cols(rng, n_probe, 2 * card, one_based=False)
    bnulls = np.concatenate([[0], rng.integers(0, 2, n_build)]).astype(np.uint8)
    pnulls = rng.integers(0, 2, n_probe).astype(np.uint8)
    d_bb, d_bo = engine.alloc(bb.nbytes), engine.alloc(bo.nbytes)
    d_bb.h2d(bb)
    d_bo.h2d(bo)
    d_bn = engine.alloc(bnulls.nbytes)
    d_bn.h2d(bnulls)
    t = engine.join_build_varchar_nulls(d_bb, d_bo, d_bn, n_build)
    d_pb, d_po = engine.alloc(pb_.nbytes), engine.alloc(po.nbytes)
    d_pb.h2d(pb_)
    d_po.h2d(po)
    d_pn = engine.alloc(pnulls.nbytes)
    d_pn.h2d(pnulls)
    for mode in (0, 2, 3):
        eop, eob = orc.slice_join_nulls(bb, bo, bnulls, n_build, pb_, po, pnulls,
                                        n_probe, mode, 64_000_000)
        cnt = engine.join_probe_emit_varchar_nulls(t, d_pb, d_po, d_pn, n_probe, mode)
        assert cnt == len(eop), mode
        op_buf, ob_buf = engine.alloc(max(cnt, 1) * 4), engine.alloc(max(cnt, 1) * 4)
        engine.join_probe_emit_varchar_nulls(t, d_pb, d_po, d_pn, n_probe, mode,
                                             op_buf, ob_buf)
        gop = op_buf.d2h(np.uint32, cnt)
        gob = ob_buf.d2h(np.uint32, cnt)
        got = np.sort(gop.astype(np.uint64) << np.uint64(32) | gob.astype(np.uint64))
        exp = np.sort(eop.astype(np.uint64) << np.uint64(32) | eob.astype(np.uint64))
        assert np.array_equal(got, exp), mode
        op_buf.free()
        ob_buf.free()
    for b in (d_bb, d_bo, d_bn, d_pb, d_po, d_pn):
        b.free()
    t.destroy()


def test_dict_decode_and_varchar_join_chain(engine):
    """Storage ingress chain for string columns: bitshuffle+LZ4 codes page
    decode -> dict decode (binary_dict_page.cpp:229-280) -> BinaryColumn ->
    varchar join, each stage bit-exact vs the oracle."""
    rng = np.random.default_rng(53)
    words = [f"w{i:03d}".encode() + b"q" * int(rng.integers(0, 5)) for i in range(200)]
    doff = np.zeros(201, np.uint32)
    np.cumsum([len(w) for w in words], out=doff[1:])
    dbytes = np.frombuffer(b"".join(words), np.uint8).copy()
    n = 100_000
    codes = rng.integers(0, 200, n).astype(np.int32)
    # encode the codes page with the oracle encoder, decode it on GPU
    page = orc.bshuf_lz4_encode_i32(codes)
    d_page = engine.alloc(page.nbytes)
    d_page.h2d(page)
    d_codes = engine.alloc(n * 4)
    engine.page_decode_bshuf_lz4_i32(d_page, n, d_codes)
    assert np.array_equal(d_codes.d2h(np.int32, n), codes)
    # dict decode on GPU vs oracle
    d_db, d_do = engine.alloc(dbytes.nbytes), engine.alloc(doff.nbytes)
    d_db.h2d(dbytes)
    d_do.h2d(doff)
    total = engine.dict_decode_binary(d_db, d_do, d_codes, n)
    eb, eo = orc.dict_decode_binary(dbytes, doff, codes)
    assert total == len(eb)
    d_ob, d_oo = engine.alloc(total), engine.alloc((n + 1) * 4)
    engine.dict_decode_binary(d_db, d_do, d_codes, n, d_ob, d_oo)
    assert np.array_equal(d_ob.d2h(np.uint8, total), eb)
    assert np.array_equal(d_oo.d2h(np.uint32, n + 1), eo)
    # the decoded BinaryColumn probes a varchar build table (subset of words)
    brows = [b""] + [words[i] for i in range(0, 200, 2)]
    bo = np.zeros(len(brows) + 1, np.uint32)
    np.cumsum([len(r) for r in brows], out=bo[1:])
    bb = np.frombuffer(b"".join(brows), np.uint8).copy()
    d_bb, d_bo = engine.alloc(bb.nbytes), engine.alloc(bo.nbytes)
    d_bb.h2d(bb)
    d_bo.h2d(bo)
    t = engine.join_build_varchar(d_bb, d_bo, len(brows) - 1)
    cnt = engine.join_probe_emit_varchar(t, d_ob, d_oo, n)
    expect_cnt = int(np.sum(codes % 2 == 0))  # even codes are in the build set
    assert cnt == expect_cnt
    for b in (d_page, d_codes, d_db, d_do, d_ob, d_oo, d_bb, d_bo):
        b.free()
    t.destroy()


def test_streaming_preagg_auto_policy(engine):
    """AUTO streaming pre-agg (aggregate_streaming_sink_operator.cpp:224-310
    restated in StreamingAggOperator): a low-cardinality phase drives the
    operator to PREAGG, a high-cardinality phase drives it to PASS_THROUGH,
    and the final merged (keys, sums, counts) equal the oracle's one-shot
    aggregate over the whole input regardless of the route each chunk took."""
    from starrocks_amd.pipeline import (StreamingAggOperator, FinalAggSink,
                                        PipelineDriver, ChunkSourceOperator)
    rng = np.random.default_rng(61)
    chunks = []
    # phase 1: 8 chunks over 100 groups (reduction ~1.0 -> PREAGG)
    for _ in range(8):
        chunks.append((rng.integers(1, 101, 40_000).astype(np.uint64),
                       rng.integers(0, 1000, 40_000).astype(np.int64)))
    # phase 2: 10 chunks of mostly-unique keys (reduction ~0 -> PASS_THROUGH)
    for i in range(10):
        chunks.append((rng.integers(10_000 + i * 10**7, 10_000 + (i + 1) * 10**7,
                                    40_000).astype(np.uint64),
                       rng.integers(0, 1000, 40_000).astype(np.int64)))

    all_keys = np.concatenate([k for k, _ in chunks])
    all_vals = np.concatenate([v for _, v in chunks])
    ek, es, ec = orc.hash_agg_sum(all_keys, all_vals)
    order = np.argsort(ek)
    ek, es, ec = ek[order], es[order], ec[order]

    def source(row_start, n):
        k, v = chunks[row_start]
        kb = engine.alloc(k.nbytes)
        kb.h2d(k)
        vb = engine.alloc(v.nbytes)
        vb.h2d(v)
        return {"n": len(k), "keys": kb, "vals": vb, "cnts": None}

    sagg = StreamingAggOperator(engine)
    sink = FinalAggSink(engine)
    gk, gs, gc = PipelineDriver(
        [ChunkSourceOperator(source, len(chunks), 1), sagg, sink]).process()
    assert sagg._state in ("PASS_THROUGH", "ADJUST")
    assert sagg._counts["pass"] >= 1  # phase 2 streamed
    assert np.array_equal(gk, ek)
    assert np.array_equal(gs, es)
    assert np.array_equal(gc, ec)


def test_bucket_chained_u64_join_parity(engine):
    """BIGINT-key chained join: GPU build/probe vs the oracle — match
    multisets equal (chain order nondeterministic), modes covered."""
    rng = np.random.default_rng(71)
    n_build, n_probe = 120_000, 600_000
    bkeys = np.concatenate([[0], rng.integers(0, 2**62, n_build, dtype=np.uint64) |
                            np.uint64(1) << np.uint64(40)]).astype(np.uint64)
    bkeys[1:n_build // 2] = rng.integers(0, 5000, n_build // 2 - 1)  # duplicate-heavy half
    probe = np.concatenate([rng.integers(0, 5000, n_probe - 1000, dtype=np.uint64),
                            bkeys[1:1001]]).astype(np.uint64)
    eop, eob = orc.bucket_chained_join_u64(bkeys, probe, 64_000_000)
    kb = engine.alloc(bkeys.nbytes)
    kb.h2d(bkeys)
    t = engine.join_build_bucket_chained_u64(kb, n_build)
    pb = engine.alloc(probe.nbytes)
    pb.h2d(probe)
    cnt = engine.join_probe_emit_u64(t, pb, len(probe))
    assert cnt == len(eop)
    op_buf, ob_buf = engine.alloc(max(cnt, 1) * 4), engine.alloc(max(cnt, 1) * 4)
    engine.join_probe_emit_u64(t, pb, len(probe), 0, op_buf, ob_buf)
    got = np.sort(op_buf.d2h(np.uint32, cnt).astype(np.uint64) << np.uint64(32) |
                  ob_buf.d2h(np.uint32, cnt).astype(np.uint64))
    exp = np.sort(eop.astype(np.uint64) << np.uint64(32) | eob.astype(np.uint64))
    assert np.array_equal(got, exp)
    # SEMI: distinct matching probe rows
    semi = engine.join_probe_emit_u64(t, pb, len(probe), 1)
    assert semi == len(np.unique(eop))
    for b in (kb, pb, op_buf, ob_buf):
        b.free()
    t.destroy()


def test_distinct_count_via_agg_claims(engine):
    """COUNT(DISTINCT col): agg-table insert where distinct = rows - existing
    hits (each first-insertion claims exactly once under concurrent CAS —
    the agg_hash_set semantics of the reference's distinct aggregate)."""
    rng = np.random.default_rng(73)
    n = 2_000_000
    keys = rng.integers(1, 300_000, n).astype(np.uint64)
    expect = len(np.unique(keys))
    at = engine.agg_table_create(1 << 20)
    engine.agg_table_reset(at)
    kb = engine.alloc(keys.nbytes)
    kb.h2d(keys)
    hits = engine.hash_agg_push(at, kb, None, n, want_hits=True)
    assert n - hits == expect
    # and the emitted group count agrees
    ok, os_ = engine.alloc(expect * 8), engine.alloc(expect * 8)
    g = engine.hash_agg_emit(at, ok, os_, expect)
    assert g == expect
    for b in (kb, ok, os_):
        b.free()
    engine.agg_table_destroy(at)


def test_varchar_probe_right_parity(engine):
    """RIGHT SEMI/ANTI over Slice keys: matched/unmatched build-row sets
    equal the oracle's."""
    rng = np.random.default_rng(79)
    n_build, n_probe, card = 30_000, 90_000, 800
    bb, bo, brows = _varchar_cols(rng, n_build, card, one_based=True)
    pb_, po, prows = _varchar_cols(rng, n_probe, card // 2, one_based=False)
    d_bb, d_bo = engine.alloc(bb.nbytes), engine.alloc(bo.nbytes)
    d_bb.h2d(bb)
    d_bo.h2d(bo)
    t = engine.join_build_varchar(d_bb, d_bo, n_build)
    d_pb, d_po = engine.alloc(pb_.nbytes), engine.alloc(po.nbytes)
    d_pb.h2d(pb_)
    d_po.h2d(po)
    for anti in (0, 1):
        expect = orc.slice_probe_right(bb, bo, n_build, pb_, po, n_probe, anti)
        ob = engine.alloc(n_build * 4)
        cnt = engine.join_probe_right_varchar(t, d_pb, d_po, n_probe, anti, ob)
        assert cnt == len(expect), anti
        got = np.sort(ob.d2h(np.uint32, cnt))
        assert np.array_equal(got, expect), anti
        ob.free()
    for b in (d_bb, d_bo, d_pb, d_po):
        b.free()
    t.destroy()


def test_eval_conjuncts_parity(engine):
    """GPU eager-prune conjunct evaluation vs the oracle: same survivors,
    same stable order, across prune-triggering and non-triggering
    selectivities (chunk_predicate_evaluator.cpp:31-80)."""
    rng = np.random.default_rng(89)
    n = 3_000_000
    od = rng.integers(19920101, 19990101, n).astype(np.int32)
    dc = rng.integers(0, 11, n).astype(np.int32)
    qt = rng.integers(1, 51, n).astype(np.int32)
    for preds in ([(0, 2, 19930101, 19931231), (1, 2, 1, 3), (2, 1, 0, 25)],  # sparse -> prunes
                  [(2, 1, 0, 45), (1, 2, 0, 9)],                               # dense -> no prune
                  [(1, 0, 99, 0)],                                             # all-false
                  [(2, 1, 0, 100)]):                                           # all-true
        cols_np = [od.copy(), dc.copy(), qt.copy()]
        em = orc.eval_conjuncts(cols_np, preds)
        dcols = []
        for a in (od, dc, qt):
            b = engine.alloc(a.nbytes)
            b.h2d(a)
            dcols.append(b)
        gm = engine.eval_conjuncts(dcols, n, preds)
        assert gm == em, preds
        for b, exp in zip(dcols, cols_np):
            if gm:
                assert np.array_equal(b.d2h(np.int32, gm), exp[:gm])
            b.free()


def test_q3_partitioned_probe_parity(engine):
    """Range-partitioned q3 probe (pass A partition + pass B L2-resident
    slice probes) produces the SAME groups/sums as the fused kernel and the
    oracle."""
    n, n_orders, n_custs = 4_000_000, 2_000_000, 200_000
    mkt = engine.alloc(n_custs * 16)
    engine.gen_cust_mkt16(SEED, n_custs, mkt)
    cbits = engine.alloc((n_custs + 31) // 32 * 4)
    engine.bits_str16_eq(mkt, n_custs, orc.mkt_literal(1), cbits)
    oc, od = engine.alloc(n_orders * 4), engine.alloc(n_orders * 4)
    engine.gen_orders_q3(SEED, n_orders, n_custs, oc, od)
    obits = engine.alloc((n_orders + 31) // 32 * 4)
    engine.q3_order_bits(oc, od, n_orders, cbits, 19950315, obits)
    lk, ext, disc = (engine.alloc(n * 8) for _ in range(3))
    ship = engine.alloc(n * 4)
    engine.gen_lineitem_q3(SEED, 0, n, n_orders, lk, ext, disc, ship)
    at = engine.agg_table_create(1 << 22)
    ks, vs = engine.alloc(n * 4), engine.alloc(n * 8)
    max_out = 2_000_000
    ok_b, os_b = engine.alloc(max_out * 8), engine.alloc(max_out * 8)
    g = engine.q3_probe_agg_part(lk, ext, disc, ship, n, n_orders, obits, 19950315,
                                 at, ks, vs, 32, ok_b, os_b, max_out)
    gk = ok_b.d2h(np.uint64, g)
    gs = os_b.d2h(np.int64, g)
    order = np.argsort(gk)
    ek, es = orc.q3_pipeline(SEED, 0, n, n_orders, n_custs)
    assert g == len(ek)
    assert np.array_equal(gk[order], ek)
    assert np.array_equal(gs[order], es)
    for b in (mkt, cbits, oc, od, obits, lk, ext, disc, ship, ks, vs, ok_b, os_b):
        b.free()
    engine.agg_table_destroy(at)


def test_eval_conjuncts_i64_parity(engine):
    """i64-column conjunct evaluation (the q3 lineitem column type) vs the
    oracle: survivors and stable order equal."""
    rng = np.random.default_rng(101)
    n = 2_000_000
    a = rng.integers(0, 10**12, n).astype(np.int64)
    b = rng.integers(0, 100, n).astype(np.int64)
    preds = [(0, 2, 10**11, 5 * 10**11), (1, 1, 0, 30)]
    cols_np = [a.copy(), b.copy()]
    em = orc.eval_conjuncts_i64(cols_np, preds)
    dcols = []
    for arr in (a, b):
        buf = engine.alloc(arr.nbytes)
        buf.h2d(arr)
        dcols.append(buf)
    gm = engine.eval_conjuncts_i64(dcols, n, preds)
    assert gm == em
    for buf, exp in zip(dcols, cols_np):
        assert np.array_equal(buf.d2h(np.int64, gm), exp[:gm])
        buf.free()


def test_q1_accum_step_soak(engine):
    """The accumulating q1 step (persistent acc + host diff, bench default at
    N=1): 300 consecutive steps each produce the exact oracle (sum, count)
    as a difference of readbacks — no drift, no race."""
    n = 1_000_000
    cols = [engine.alloc(n * 4) for _ in range(3)]
    engine.gen_lineorder_q1(SEED, 0, n, *cols)
    from starrocks_amd import gen
    datekey, dyear = gen.gen_dates()
    dpay = np.where(dyear == 1993, dyear - 1992 + 1, 0).astype(np.uint32)
    kb = engine.alloc(datekey.nbytes)
    kb.h2d(datekey.astype(np.int32))
    pb = engine.alloc(dpay.nbytes)
    pb.h2d(dpay)
    t = engine.join_build_payload(kb, pb, len(datekey))
    acc = engine.alloc(16)
    acc.h2d(np.zeros(2, np.int64))
    expect = orc.q1_pipeline(SEED, 0, n, 1993)
    prev = np.zeros(2, np.int64)
    for step in range(300):
        engine.q1_join_sum_accum(t, cols[0], cols[1], cols[2], n, acc)
        cur = acc.d2h(np.int64, 2)
        d = cur - prev
        prev = cur
        assert (int(d[0]), int(d[1])) == expect, step
    for b in cols + [kb, pb, acc]:
        b.free()
    t.destroy()


def test_null_aware_left_anti_join_parity(engine):
    """NULL_AWARE_LEFT_ANTI (the NOT IN lowering, mode 6): identical to
    LEFT_ANTI except null probe rows are EXCLUDED from the output — NULL
    NOT IN (...) is never true (join_hash_map.hpp:1225-1240; hash_joiner.cpp
    :97 maps NOT-IN anti joins to this type). Compared against a direct
    numpy statement of that rule and cross-checked against plain LEFT_ANTI."""
    rng = np.random.default_rng(91)
    nbuild, nprobe = 10_000, 120_000
    build_keys = np.concatenate([[0], rng.integers(0, 1500, nbuild)]).astype(np.uint32)
    build_nulls = np.concatenate([[0], (rng.random(nbuild) < 0.2)]).astype(np.uint8)
    probe_keys = rng.integers(0, 2000, nprobe).astype(np.uint32)
    probe_nulls = (rng.random(nprobe) < 0.25).astype(np.uint8)

    kb = engine.alloc(build_keys.nbytes)
    kb.h2d(build_keys.view(np.int32))
    nb = engine.alloc(build_nulls.nbytes)
    nb.h2d(build_nulls)
    t = engine.join_build_bucket_chained_nulls(kb, nb, nbuild)
    pb = engine.alloc(probe_keys.nbytes)
    pb.h2d(probe_keys.view(np.int32))
    pn = engine.alloc(probe_nulls.nbytes)
    pn.h2d(probe_nulls)

    # expected: non-null probe rows whose key is absent from the NON-NULL
    # build rows
    present = set(build_keys[1:][build_nulls[1:] == 0].tolist())
    exp = np.flatnonzero((probe_nulls == 0) &
                         ~np.isin(probe_keys, np.fromiter(present, np.uint32)))

    cnt = engine.join_probe_emit_nulls(t, pb, pn, nprobe, 6)
    assert cnt == len(exp)
    op_b = engine.alloc(max(cnt, 1) * 4)
    ob_b = engine.alloc(max(cnt, 1) * 4)
    engine.join_probe_emit_nulls(t, pb, pn, nprobe, 6, op_b, ob_b)
    gop = np.sort(op_b.d2h(np.uint32, cnt))
    assert np.array_equal(gop, exp.astype(np.uint32))
    assert (ob_b.d2h(np.uint32, cnt) == 0).all()  # anti: build idx 0 sentinel

    # plain LEFT_ANTI (mode 2) additionally emits every null probe row
    cnt2 = engine.join_probe_emit_nulls(t, pb, pn, nprobe, 2)
    assert cnt2 == cnt + int(probe_nulls.sum())
    for b in (kb, nb, pb, pn, op_b, ob_b):
        b.free()
    t.destroy()


def test_null_aware_left_anti_varchar_parity(engine):
    """mode 6 over Slice keys (the serialized-varchar constructor path)."""
    rng = np.random.default_rng(92)
    bb, boff, brows = _varchar_cols(rng, 3000, 40, one_based=True)
    pb_, poff, prows = _varchar_cols(rng, 20_000, 55, one_based=False)
    pnulls = (rng.random(20_000) < 0.3).astype(np.uint8)
    b_bytes = engine.alloc(max(bb.nbytes, 1))
    b_bytes.h2d(bb)
    b_off = engine.alloc(boff.nbytes)
    b_off.h2d(boff)
    t = engine.join_build_varchar(b_bytes, b_off, 3000)
    p_bytes = engine.alloc(max(pb_.nbytes, 1))
    p_bytes.h2d(pb_)
    p_off = engine.alloc(poff.nbytes)
    p_off.h2d(poff)
    p_n = engine.alloc(pnulls.nbytes)
    p_n.h2d(pnulls)
    present = set(brows[1:])
    exp = np.array([i for i in range(20_000)
                    if not pnulls[i] and prows[i] not in present], np.uint32)
    cnt = engine.join_probe_emit_varchar_nulls(t, p_bytes, p_off, p_n, 20_000, 6)
    assert cnt == len(exp)
    op_b = engine.alloc(max(cnt, 1) * 4)
    ob_b = engine.alloc(max(cnt, 1) * 4)
    engine.join_probe_emit_varchar_nulls(t, p_bytes, p_off, p_n, 20_000, 6, op_b, ob_b)
    assert np.array_equal(np.sort(op_b.d2h(np.uint32, cnt)), exp)
    for b in (b_bytes, b_off, p_bytes, p_off, p_n, op_b, ob_b):
        b.free()
    t.destroy()


def test_dense_range_direct_join_parity(engine):
    """DENSE_RANGE_DIRECT (rank/select 2-bit map, join_hash_map_method.h:378,
    .hpp:781-940): sparse-wide build keys with duplicates; every probe mode
    must produce the same multiset as the bucket-chained build of the same
    rows, and the selector's DENSE decision (10 M rows / 100 M interval in
    the CPU-threshold scenario) is what routes here via join_build_auto."""
    rng = np.random.default_rng(77)
    nbuild, nprobe = 120_000, 400_000
    # interval ~3M >> nbuild: the dense regime; ~25% duplicate keys
    base = rng.integers(1, 3_000_000, nbuild - nbuild // 4)
    keys = np.concatenate([[0], base, rng.choice(base, nbuild // 4)]).astype(np.int32)
    kb = engine.alloc(keys.nbytes)
    kb.h2d(keys)
    td = engine.join_build_dense_range_direct(kb, nbuild)
    tb = engine.join_build_bucket_chained(kb, nbuild)
    probe = rng.integers(-10, 3_100_000, nprobe).astype(np.int32)
    pb = engine.alloc(probe.nbytes)
    pb.h2d(probe)
    for mode in (0, 1, 2, 3):
        cd = engine.join_probe_emit_mode(td, pb, nprobe, mode)
        cb = engine.join_probe_emit_mode(tb, pb, nprobe, mode)
        assert cd == cb, mode
        opd, obd = engine.alloc(max(cd, 1) * 4), engine.alloc(max(cd, 1) * 4)
        opb, obb = engine.alloc(max(cb, 1) * 4), engine.alloc(max(cb, 1) * 4)
        engine.join_probe_emit_mode(td, pb, nprobe, mode, opd, obd)
        engine.join_probe_emit_mode(tb, pb, nprobe, mode, opb, obb)
        gop, gob = opd.d2h(np.uint32, cd), obd.d2h(np.uint32, cd)
        eop, eob = opb.d2h(np.uint32, cb), obb.d2h(np.uint32, cb)
        if mode == 1:
            # SEMI emits ONE representative build row per matched probe row;
            # chain order differs between structures, so compare the probe
            # set + the key-match property (as the reference's own semi
            # semantics define it)
            assert np.array_equal(np.sort(gop), np.sort(eop))
            assert (keys[gob] == probe[gop]).all()
        else:
            pack = lambda a, b: np.sort(a.astype(np.uint64) << np.uint64(32) | b)
            assert np.array_equal(pack(gop, gob), pack(eop, eob)), mode
        for x in (opd, obd, opb, obb):
            x.free()
    td.destroy()
    tb.destroy()
    kb.free()
    pb.free()


def test_join_build_auto_dense_route(engine):
    """join_build_auto must route a DENSE decision to the dense build and the
    table must answer probes (CPU-cache thresholds force the DENSE branch)."""
    rng = np.random.default_rng(78)
    n = 400_000
    # interval 2.5M: > bucket 524k, > the 64 KB l2 below, and the dense
    # inequality interval/4 + rc*4 <= (b+b/10)*4 holds (2.23M <= 2.31M)
    keys = np.concatenate([[0], rng.integers(1, 2_500_000, n)]).astype(np.int32)
    kb = engine.alloc(keys.nbytes)
    kb.h2d(keys)
    # CPU-threshold scenario forcing the DENSE branch
    t, method = engine.join_build_auto(kb, n, l2_size=64 << 10, l3_size=32 << 20)
    assert engine.JM_NAMES[method] == "DENSE_RANGE_DIRECT"
    probe = keys[1:5001]
    pb = engine.alloc(probe.nbytes)
    pb.h2d(probe)
    cnt = engine.join_probe_emit_mode(t, pb, 5000, 1)  # LEFT_SEMI
    assert cnt == 5000
    kb.free()
    pb.free()
    t.destroy()


def test_dense_right_semi_anti_parity(engine):
    """RIGHT SEMI/ANTI over a DENSE_RANGE_DIRECT table (the probe pass marks
    matched BUILD rows) must match the bucket-chained build's sets."""
    rng = np.random.default_rng(81)
    nbuild, nprobe = 60_000, 200_000
    keys = np.concatenate([[0], rng.integers(1, 1_500_000, nbuild)]).astype(np.int32)
    kb = engine.alloc(keys.nbytes)
    kb.h2d(keys)
    td = engine.join_build_dense_range_direct(kb, nbuild)
    tb = engine.join_build_bucket_chained(kb, nbuild)
    probe = rng.integers(1, 1_600_000, nprobe).astype(np.int32)
    pb = engine.alloc(probe.nbytes)
    pb.h2d(probe)
    for anti in (0, 1):
        cd = engine.join_probe_right(td, pb, nprobe, anti)
        cb = engine.join_probe_right(tb, pb, nprobe, anti)
        assert cd == cb, anti
        od = engine.alloc(max(cd, 1) * 4)
        ob2 = engine.alloc(max(cb, 1) * 4)
        engine.join_probe_right(td, pb, nprobe, anti, od)
        engine.join_probe_right(tb, pb, nprobe, anti, ob2)
        assert np.array_equal(np.sort(od.d2h(np.uint32, cd)),
                              np.sort(ob2.d2h(np.uint32, cb))), anti
        od.free()
        ob2.free()
    td.destroy()
    tb.destroy()
    kb.free()
    pb.free()


def test_partition_async_matches_sync(engine):
    """The async partition (device-side scan) must produce the same
    per-channel ROW SETS at the same split boundaries as the sync form (the
    routing contract; order within a (block,channel) segment is emission
    order — atomic-cursor nondeterministic even between two sync runs)."""
    n, nch = 2_000_000, 8
    keys = engine.alloc(n * 4)
    engine.gen_u32_mod(keys, 42, 9, 0, n, 0, 0)
    host_keys = keys.d2h(np.uint32, n)
    ri_sync = engine.alloc(n * 4)
    sp = engine.partition(keys, n, nch, ri_sync)
    ri_async = engine.alloc(n * 4)
    scr = engine.alloc(engine.partition_scratch_bytes(n, nch))
    engine.partition_async(keys, n, nch, ri_async, scr)
    engine.sync()
    ch = orc.partition_channels(host_keys, nch)
    got = ri_async.d2h(np.uint32, n)
    for c in range(nch):
        rows = got[int(sp[c]):int(sp[c + 1])]
        assert (ch[rows] == c).all()
        assert len(np.unique(rows)) == len(rows)
    # i64 form
    k64 = engine.alloc(n * 8)
    engine.gen_i64(k64, 42, 3, 0, n)
    host64 = k64.d2h(np.uint64, n)
    r1 = engine.alloc(n * 4)
    sp64 = engine.partition_i64(k64, n, nch, r1)
    r2 = engine.alloc(n * 4)
    engine.partition_i64_async(k64, n, nch, r2, scr)
    engine.sync()
    ch64 = orc.partition_channels_u64(host64, nch)
    got64 = r2.d2h(np.uint32, n)
    for c in range(nch):
        rows = got64[int(sp64[c]):int(sp64[c + 1])]
        assert (ch64[rows] == c).all()
        assert len(np.unique(rows)) == len(rows)
    for b in (keys, ri_sync, ri_async, scr, k64, r1, r2):
        b.free()


@pytest.mark.gpu
def test_filter_singlepass_layout_ab(engine):
    """Both single-pass layouts — the wave-coalesced default and the
    per-thread-contiguous baseline (GPUE_FILT_WL=0) — must be bit-exact
    including order on the same input."""
    import os
    n = 7_345_677  # non-multiple of every tile size
    inp = engine.alloc(n * 8)
    engine.gen_i64(inp, SEED, 9, 0, n)
    data = inp.d2h(np.int64, n)
    theta = int(np.quantile(data.astype(np.float64), 0.13))
    out = engine.alloc(n * 8)
    cnt = engine.scan_filter_i64_lt_sp(inp, n, theta, out)
    got_wl = out.d2h(np.int64, cnt)
    os.environ["GPUE_FILT_WL"] = "0"
    try:
        cnt2 = engine.scan_filter_i64_lt_sp(inp, n, theta, out)
        got_base = out.d2h(np.int64, cnt2)
    finally:
        del os.environ["GPUE_FILT_WL"]
    assert cnt == cnt2
    assert np.array_equal(got_wl, got_base)
    inp.free()
    out.free()
