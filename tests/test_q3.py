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
