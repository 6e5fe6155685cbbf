"""Config 5 (TPC-H Q3-shaped) — oracle correctness on CPU."""

import numpy as np

from oracle import pyoracle as orc

SEED = 42
N_ORDERS = 100_000
N_CUSTS = 20_000


def brute_q3(n):
    mkt = orc.gen_cust_mkt16(SEED, N_CUSTS).reshape(N_CUSTS, 16)
    lit = np.frombuffer(orc.mkt_literal(1), np.uint8)
    cpass = (mkt == lit).all(axis=1)
    oc, od = orc.gen_orders_q3(SEED, N_ORDERS, N_CUSTS)
    opass = (od < 19950315) & cpass[oc - 1]
    lk, ext, disc, ship = orc.gen_lineitem_q3(SEED, 0, n, N_ORDERS)
    lpass = (ship > 19950315) & opass[lk - 1]
    keys = lk[lpass].astype(np.uint64)
    vals = ext[lpass] * (100 - disc[lpass])
    uk, inv = np.unique(keys, return_inverse=True)
    sums = np.zeros(len(uk), np.int64)
    np.add.at(sums, inv, vals)
    return uk, sums


def test_q3_oracle_vs_numpy():
    n = 500_000
    gk, gs = orc.q3_pipeline(SEED, 0, n, N_ORDERS, N_CUSTS)
    ek, es = brute_q3(n)
    assert np.array_equal(gk, ek)
    assert np.array_equal(gs, es)
    assert len(gk) > 0


def test_q3_mkt_dictionary():
    mkt = orc.gen_cust_mkt16(SEED, 1000).reshape(1000, 16)
    segs = {bytes(row) for row in mkt}
    assert len(segs) == 5
    assert orc.mkt_literal(1) == b"BUILDING        "
    for s in segs:
        assert len(s) == 16


def test_q3_partitioned_execution_math():
    """Simulate bench.py's N>1 hash-partitioned q3 exactly: rows route by
    fnv64(l_orderkey) channels; each simulated rank aggregates only its
    received rows; the group spaces are DISJOINT across ranks and their
    union equals the unpartitioned result bit-exactly."""
    from starrocks_amd import gen
    n, world = 400_000, 4
    lk, ext, disc, ship = orc.gen_lineitem_q3(SEED, 0, n, N_ORDERS)
    row_channel = gen.partition_channels_u64(lk.astype(np.uint64), world)
    # reference-pinned channel function: oracle vs numpy restatement
    assert np.array_equal(row_channel, orc.partition_channels_u64(lk.astype(np.uint64), world))

    mkt = orc.gen_cust_mkt16(SEED, N_CUSTS)
    cbits = np.zeros((N_CUSTS + 7) // 8, np.uint8)
    orc.load().orc_q3_build_cust_bits(orc._p(mkt), N_CUSTS, orc.mkt_literal(1), orc._p(cbits))
    oc, od = orc.gen_orders_q3(SEED, N_ORDERS, N_CUSTS)
    obits = np.zeros((N_ORDERS + 7) // 8, np.uint8)
    orc.load().orc_q3_build_order_bits(orc._p(oc), orc._p(od), N_ORDERS, orc._p(cbits),
                                       19950315, orc._p(obits))
    merged = {}
    seen_keys_per_rank = []
    for r in range(world):
        sel = row_channel == r
        cap = max(int(sel.sum()), 16)
        ok = np.empty(cap, np.uint64)
        os_ = np.empty(cap, np.int64)
        g = orc.load().orc_q3_probe_agg(
            orc._p(np.ascontiguousarray(lk[sel])), orc._p(np.ascontiguousarray(ext[sel])),
            orc._p(np.ascontiguousarray(disc[sel])), orc._p(np.ascontiguousarray(ship[sel])),
            int(sel.sum()), orc._p(obits), 19950315, orc._p(ok), orc._p(os_), cap)
        assert g != 2**64 - 1
        seen_keys_per_rank.append(set(ok[:g].tolist()))
        for k, s in zip(ok[:g].tolist(), os_[:g].tolist()):
            merged[k] = merged.get(k, 0) + s
    # disjointness: each group key appears on exactly one rank
    for a in range(world):
        for b in range(a + 1, world):
            assert not (seen_keys_per_rank[a] & seen_keys_per_rank[b])
    wk, ws = orc.q3_pipeline(SEED, 0, n, N_ORDERS, N_CUSTS)
    mk = np.array(sorted(merged), np.uint64)
    ms = np.array([merged[k] for k in sorted(merged)], np.int64)
    assert np.array_equal(mk, wk)
    assert np.array_equal(ms, ws)


def test_q3_sharded_merge():
    """Lineitem row-sharding merges bit-exactly (group-keyed sum merge)."""
    n = 400_000
    wk, ws = orc.q3_pipeline(SEED, 0, n, N_ORDERS, N_CUSTS)
    ak, as_ = orc.q3_pipeline(SEED, 0, n // 2, N_ORDERS, N_CUSTS)
    bk, bs = orc.q3_pipeline(SEED, n // 2, n - n // 2, N_ORDERS, N_CUSTS)
    merged = {}
    for k, s in zip(ak.tolist(), as_.tolist()):
        merged[k] = merged.get(k, 0) + s
    for k, s in zip(bk.tolist(), bs.tolist()):
        merged[k] = merged.get(k, 0) + s
    mk = np.array(sorted(merged), np.uint64)
    ms = np.array([merged[k] for k in sorted(merged)], np.int64)
    assert np.array_equal(mk, wk)
    assert np.array_equal(ms, ws)
