"""Generic hash aggregate (high-cardinality GROUP BY) — oracle vs numpy."""

import numpy as np

from oracle import pyoracle as orc


def numpy_groupby(keys, vals):
    uk, inv = np.unique(keys, return_inverse=True)
    sums = np.zeros(len(uk), np.int64)
    counts = np.zeros(len(uk), np.int64)
    np.add.at(sums, inv, vals)
    np.add.at(counts, inv, 1)
    return uk, sums, counts


def _check(keys, vals):
    gk, gs, gc = orc.hash_agg_sum(keys, vals)
    ek, es, ec = numpy_groupby(keys, vals)
    order = np.argsort(gk)
    assert np.array_equal(gk[order], ek)
    assert np.array_equal(gs[order], es)
    assert np.array_equal(gc[order], ec)


def test_small_dense():
    rng = np.random.default_rng(2)
    keys = rng.integers(0, 100, 10_000).astype(np.uint64)
    vals = rng.integers(-1000, 1000, 10_000).astype(np.int64)
    _check(keys, vals)


def test_high_cardinality():
    rng = np.random.default_rng(3)
    keys = rng.integers(0, 2**62, 200_000).astype(np.uint64)
    vals = rng.integers(-10**9, 10**9, 200_000).astype(np.int64)
    _check(keys, vals)


def test_stats_vs_numpy():
    rng = np.random.default_rng(5)
    keys = rng.integers(0, 1000, 50_000).astype(np.uint64)
    vals = rng.integers(-10**12, 10**12, 50_000).astype(np.int64)
    gk, gs, gc, gmn, gmx = orc.hash_agg_stats(keys, vals)
    order = np.argsort(gk)
    uk = np.unique(keys)
    assert np.array_equal(gk[order], uk)
    for i, k in enumerate(uk):
        sel = vals[keys == k]
        j = order[i]
        assert gs[j] == sel.sum()
        assert gc[j] == len(sel)
        assert gmn[j] == sel.min()
        assert gmx[j] == sel.max()


def test_sum128_vs_python_bigints():
    rng = np.random.default_rng(6)
    keys = rng.integers(0, 50, 20_000).astype(np.uint64)
    # values near int64 extremes so group sums overflow 64 bits
    vals = rng.integers(-2**62, 2**62, 20_000).astype(np.int64)
    gk, lo, hi = orc.hash_agg_sum128(keys, vals)
    order = np.argsort(gk)
    for i, k in enumerate(np.unique(keys)):
        expect = sum(int(v) for v in vals[keys == k])  # exact python bigint
        j = order[i]
        got = (int(hi[j]) << 64) | int(lo[j])
        # interpret as signed 128-bit
        if got >= 2**127:
            got -= 2**128
        assert got == expect


def test_single_group_and_edges():
    keys = np.zeros(1000, np.uint64)
    vals = np.full(1000, 7, np.int64)
    gk, gs, gc = orc.hash_agg_sum(keys, vals)
    assert len(gk) == 1 and gs[0] == 7000 and gc[0] == 1000
    # empty input
    gk, gs, gc = orc.hash_agg_sum(np.empty(0, np.uint64), np.empty(0, np.int64))
    assert len(gk) == 0
