"""The three generator implementations (oracle C, numpy, HIP) must be
bit-identical; CPU↔CPU here, CPU↔GPU in test_gpu_parity.py."""

import numpy as np

from oracle import pyoracle as orc
from starrocks_amd import gen


def test_gen_u64_scalar():
    lib = orc.load()
    rng = np.random.default_rng(1)
    for _ in range(200):
        seed = int(rng.integers(0, 2**63))
        tag = int(rng.integers(0, 16))
        i = int(rng.integers(0, 2**40))
        a = lib.orc_gen_u64(seed, tag, i)
        b = int(gen.gen_u64(seed, tag, np.array([i], np.uint64))[0])
        assert a == b


def test_dates():
    datekey, dyear = gen.gen_dates()
    dk = np.zeros(gen.N_DAYS, np.int32)
    dy = np.zeros(gen.N_DAYS, np.int32)
    orc.load().orc_gen_dates(gen.N_DAYS, orc._p(dk), orc._p(dy))
    assert np.array_equal(datekey, dk)
    assert np.array_equal(dyear, dy)
    assert datekey[0] == 19920101
    assert dyear[-1] == 1998
    # leap day present in 1992 and 1996
    assert 19920229 in datekey and 19960229 in datekey


def test_lineorder_q1_columns():
    od_c, ep_c, dc_c = orc.gen_lineorder_q1(42, 1000, 50_000)
    od_n, ep_n, dc_n = gen.gen_lineorder_q1(42, 1000, 50_000)
    assert np.array_equal(od_c, od_n)
    assert np.array_equal(ep_c, ep_n)
    assert np.array_equal(dc_c, dc_n)
    assert ep_c.min() >= 1 and ep_c.max() <= 100000
    assert dc_c.min() >= 0 and dc_c.max() <= 10


def test_lineorder_q21_columns():
    a = orc.gen_lineorder_q21(42, 0, 50_000)
    b = gen.gen_lineorder_q21(42, 0, 50_000)
    for x, y in zip(a, b):
        assert np.array_equal(x, y)
    pk = a[0]
    assert pk.min() >= 1 and pk.max() <= gen.N_PARTS_SF100


def test_partition_channels_2xi32_oracle_vs_numpy():
    """Chained 2-column FNV partition (exchange_sink_operator.cpp:611-617):
    oracle C vs numpy restatement, bit-exact."""
    import numpy as np
    from oracle import pyoracle as orc
    from starrocks_amd import gen
    rng = np.random.default_rng(21)
    a = rng.integers(-2**31, 2**31, 30_000).astype(np.int32)
    b = rng.integers(-2**31, 2**31, 30_000).astype(np.int32)
    for nch in (2, 5, 8, 64):
        out = np.zeros(len(a), np.uint32)
        orc.load().orc_partition_channel_2xi32(orc._p(a), orc._p(b), len(a), nch, orc._p(out))
        assert np.array_equal(out, gen.partition_channels_2xi32(a, b, nch))
        # chain property: equals fnv(b, seed=fnv(a, FNV_SEED)) reduced
        h = gen.fnv_u32_seeded(b, gen.fnv_u32(a))
        assert np.array_equal(out, ((h.astype(np.uint64) * nch) >> np.uint64(32)).astype(np.uint32))


def test_mkt_segments_match_oracle():
    """gen.MKT_SEGMENTS (product path) must equal the oracle's restatement of
    the TPC-H c_mktsegment dictionary (oracle.c MKT_SEGMENTS) — the product
    path may not import oracle/ (DESIGN.md §2), so the constant lives twice."""
    for i in range(5):
        assert gen.mkt_literal(i) == orc.mkt_literal(i)
        assert len(gen.mkt_literal(i)) == 16


def test_bench_script_compiles():
    """bench.py is never imported by the CPU suite, so a syntax error would
    only surface on the GPU box — compile it here."""
    import ast
    import os
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    for fname in ("bench.py", "__graft_entry__.py"):
        with open(os.path.join(root, fname)) as f:
            ast.parse(f.read(), filename=fname)
