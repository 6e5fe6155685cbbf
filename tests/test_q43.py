"""Config 4 (SSB Q4.3) — oracle correctness + the partitioned-execution math.

The partitioned multi-GPU path (bench.py --workload q43, N>1) is validated
here on CPU by simulating the exact per-rank computation: partition rows by
the reference's FNV+ReduceOp channel function, give each simulated rank only
its owned customer-dim shard, and check that the merged per-rank group sums
equal the unpartitioned result bit-exactly.
"""

import numpy as np

from oracle import pyoracle as orc
from starrocks_amd import gen

SEED = 42
REGION, NATION, CATEGORY = 1, 7, 12


def brute_q43(n):
    ck, sk, pk, od, rv, sc = gen.gen_lineorder_q43(SEED, 0, n)
    cfirst = gen.build_cust_dim_q43(SEED, gen.N_CUSTS_SF100, REGION)
    sfirst = gen.build_supp_dim_q43(SEED, gen.N_SUPPS_SF100, NATION)
    pfirst = gen.build_part_dim_q43(SEED, gen.N_PARTS_SF100, CATEGORY)
    mn, _, dfirst = gen.build_date_dim_q43()
    ppay = pfirst[pk - 1]
    spay = sfirst[sk - 1]
    dpay = dfirst[od - mn]
    ok = (ppay != 0) & (spay != 0) & (cfirst[ck - 1] != 0) & (dpay != 0)
    gid = ((dpay[ok].astype(np.int64) - 1) * 400 + (spay[ok].astype(np.int64) - 1) * 40
           + (ppay[ok].astype(np.int64) - 1))
    out = np.zeros(800, np.int64)
    np.add.at(out, gid, rv[ok].astype(np.int64) - sc[ok])
    return out


def test_q43_oracle_vs_numpy():
    n = 300_000
    gs = orc.q43_pipeline(SEED, 0, n, REGION, NATION, CATEGORY)
    assert np.array_equal(gs, brute_q43(n))
    assert gs.sum() != 0  # selectivity ~1/875: 300k rows -> ~340 matches


def test_q43_gen_consistency():
    a = orc.gen_lineorder_q43(SEED, 123, 50_000)
    b = gen.gen_lineorder_q43(SEED, 123, 50_000)
    for x, y in zip(a, b):
        assert np.array_equal(x, y)


def test_q43_sharded_rows_merge():
    """Row-sharding (weak scaling) merge == whole, bit-exact."""
    n = 400_000
    whole = orc.q43_pipeline(SEED, 0, n, REGION, NATION, CATEGORY)
    a = orc.q43_pipeline(SEED, 0, n // 2, REGION, NATION, CATEGORY)
    b = orc.q43_pipeline(SEED, n // 2, n - n // 2, REGION, NATION, CATEGORY)
    assert np.array_equal(a + b, whole)


def test_q43_partitioned_execution_math():
    """Simulate bench.py's N>1 hash-partitioned mode exactly:
    rows route by channel(lo_custkey); rank r owns only customers with
    channel==r. Merged result must equal the unpartitioned one bit-exactly."""
    n, world = 500_000, 4
    ck, sk, pk, od, rv, sc = gen.gen_lineorder_q43(SEED, 0, n)
    cfirst_full = gen.build_cust_dim_q43(SEED, gen.N_CUSTS_SF100, REGION)
    sfirst = gen.build_supp_dim_q43(SEED, gen.N_SUPPS_SF100, NATION)
    pfirst = gen.build_part_dim_q43(SEED, gen.N_PARTS_SF100, CATEGORY)
    mn, _, dfirst = gen.build_date_dim_q43()

    ckeys = np.arange(1, gen.N_CUSTS_SF100 + 1, dtype=np.uint32)
    cust_owner = gen.partition_channels(ckeys, world)
    row_channel = gen.partition_channels(ck.view(np.uint32), world)
    merged = np.zeros(800, np.int64)
    for r in range(world):
        # rank r's received rows = every row whose key routes to r
        sel = row_channel == r
        cfirst_r = np.where(cust_owner == r, cfirst_full, 0).astype(np.uint32)
        merged += orc.q43_kernel(
            np.ascontiguousarray(ck[sel]), np.ascontiguousarray(sk[sel]),
            np.ascontiguousarray(pk[sel]), np.ascontiguousarray(od[sel]),
            np.ascontiguousarray(rv[sel]), np.ascontiguousarray(sc[sel]),
            cfirst_r, sfirst, pfirst, dfirst, mn)
    whole = orc.q43_pipeline(SEED, 0, n, REGION, NATION, CATEGORY)
    assert np.array_equal(merged, whole)
    # sanity: every row routed somewhere and keys agree with their owner rank
    assert (np.bincount(row_channel, minlength=world).sum()) == n
    assert (cust_owner[ck[row_channel == 1] - 1] == 1).all()
