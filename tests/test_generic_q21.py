"""SSB Q2.1 through the GENERIC operator chain — no fused kernel.

The fused k_q21_star_agg_pfq is the plan-selected fast path; this test
computes the same query the way a plan of generic operators would
(the reference's own operator decomposition, hash_join_node.cpp:193-346 +
aggregator.cpp): selector-checked RANGE_DIRECT builds over the FILTERED dim
keys, probe-emit + gathers per join, SERIALIZED_FIXED_SIZE key packing for
the 2-column GROUP BY, and the generic CAS hash aggregate — and must match
both the fused kernel and the oracle bit-exactly. This pins that the engine
is a composable operator set, not a family of bench-specific kernels.
"""

import numpy as np
import pytest

from oracle import pyoracle as orc
from starrocks_amd import gen

pytestmark = pytest.mark.gpu

SEED, CAT, REG = 42, 12, 2
N = 5_000_000


def _filtered_range_direct(engine, keys_1based_i32):
    """Build a RANGE_DIRECT table over the (already filtered) dim keys,
    consulting the selector first — the JoinHashTable::build flow."""
    rc = len(keys_1based_i32) - 1
    kb = engine.alloc(keys_1based_i32.nbytes)
    kb.h2d(keys_1based_i32)
    t, method = engine.join_build_auto(kb, rc)
    kb.free()
    return t, method


def test_q21_generic_operator_chain(engine):
    # --- scan: generate the probe-side columns on device ---
    cols = [engine.alloc(N * 4) for _ in range(4)]  # pk, sk, od, rv
    engine.gen_lineorder_q21(SEED, 0, N, *cols)
    pk, sk, od, rv = cols

    # --- dim scans with their predicates applied (the reference pushes the
    # category/region filters to the dim scans), then selector-driven builds
    pfirst = gen.build_part_dim_payload(SEED, gen.N_PARTS_SF100, CAT)
    passing_parts = np.flatnonzero(pfirst) + 1  # part keys in the category
    brand1_by_bidx = np.concatenate(  # build rows are 1-based
        [[0], pfirst[passing_parts - 1]]).astype(np.uint32)
    pkeys = np.concatenate([[0], passing_parts]).astype(np.int32)
    parts_t, m1 = _filtered_range_direct(engine, pkeys)

    sfirst = gen.build_supp_dim_payload(SEED, gen.N_SUPPS_SF100, REG)
    skeys = np.concatenate([[0], np.flatnonzero(sfirst) + 1]).astype(np.int32)
    supps_t, m2 = _filtered_range_direct(engine, skeys)

    datekey, dyear = gen.gen_dates()
    year1_by_bidx = np.concatenate([[0], dyear - 1992 + 1]).astype(np.uint32)
    dkeys = np.concatenate([[0], datekey]).astype(np.int32)
    dates_t, m3 = _filtered_range_direct(engine, dkeys)
    for m in (m1, m2, m3):
        assert engine.JM_NAMES[m].startswith("RANGE_DIRECT") or \
            engine.JM_NAMES[m] == "DENSE_RANGE_DIRECT", engine.JM_NAMES[m]

    # --- join 1: lineorder ⋈ parts (INNER emit + payload gather) ---
    c1 = engine.join_probe_emit(parts_t, pk, N)
    rows1 = engine.alloc(max(c1, 1) * 4)
    bidx1 = engine.alloc(max(c1, 1) * 4)
    engine.join_probe_emit(parts_t, pk, N, rows1, bidx1)
    # brand payload gathered by build index (the _build_output gather)
    brand_arr = engine.alloc(brand1_by_bidx.nbytes)
    brand_arr.h2d(brand1_by_bidx)
    brand1 = engine.alloc(max(c1, 1) * 4)
    engine.gather_u32(brand_arr, bidx1, c1, brand1)
    # probe-side columns gathered by probe index (_probe_output)
    sk1 = engine.alloc(max(c1, 1) * 4)
    od1 = engine.alloc(max(c1, 1) * 4)
    rv1 = engine.alloc(max(c1, 1) * 4)
    for src, dst in ((sk, sk1), (od, od1), (rv, rv1)):
        engine.gather_u32(src, rows1, c1, dst)

    # --- join 2: ⋈ suppliers (LEFT SEMI — only the filter effect) ---
    c2 = engine.join_probe_emit_mode(supps_t, sk1, c1, 1)
    rows2 = engine.alloc(max(c2, 1) * 4)
    b2 = engine.alloc(max(c2, 1) * 4)
    engine.join_probe_emit_mode(supps_t, sk1, c1, 1, rows2, b2)
    od2 = engine.alloc(max(c2, 1) * 4)
    rv2 = engine.alloc(max(c2, 1) * 4)
    brand2 = engine.alloc(max(c2, 1) * 4)
    for src, dst in ((od1, od2), (rv1, rv2), (brand1, brand2)):
        engine.gather_u32(src, rows2, c2, dst)

    # --- join 3: ⋈ dates (INNER; every datekey matches, payload = year) ---
    c3 = engine.join_probe_emit(dates_t, od2, c2)
    assert c3 == c2  # dates are dense over the full key range
    rows3 = engine.alloc(max(c3, 1) * 4)
    bidx3 = engine.alloc(max(c3, 1) * 4)
    engine.join_probe_emit(dates_t, od2, c2, rows3, bidx3)
    year_arr = engine.alloc(year1_by_bidx.nbytes)
    year_arr.h2d(year1_by_bidx)
    year3 = engine.alloc(max(c3, 1) * 4)
    engine.gather_u32(year_arr, bidx3, c3, year3)
    rv3 = engine.alloc(max(c3, 1) * 4)
    brand3 = engine.alloc(max(c3, 1) * 4)
    engine.gather_u32(rv2, rows3, c3, rv3)
    engine.gather_u32(brand2, rows3, c3, brand3)

    # --- GROUP BY (year, brand): SERIALIZED_FIXED_SIZE 2xi32 packing +
    # generic CAS hash aggregate (the aggregator's build_hash_map path) ---
    keys64 = engine.alloc(max(c3, 1) * 8)
    engine.pack_keys_2xi32(year3, brand3, c3, keys64)
    vals = engine.alloc(max(c3, 1) * 8)
    rv_host = rv3.d2h(np.uint32, c3).astype(np.int64)  # widen rv to i64 vals
    vals.h2d(rv_host)
    ok_b = engine.alloc(8192 * 8)
    os_b = engine.alloc(8192 * 8)
    ng = engine.hash_agg_sum_u64(keys64, vals, c3, ok_b, os_b, max_out=8192)

    # --- expectations: the oracle one-shot q21 AND the fused kernel ---
    expect = orc.q21_pipeline(SEED, 0, N, CAT, REG)  # 7000 dense group sums
    got = np.zeros(7000, np.int64)
    gk = ok_b.d2h(np.uint64, ng)
    gs = os_b.d2h(np.int64, ng)
    for k, s in zip(gk.tolist(), gs.tolist()):
        year1 = k & 0xFFFFFFFF
        brand1_v = k >> 32
        got[(year1 - 1) * 1000 + (brand1_v - 1)] = s
    assert np.array_equal(got, expect)

    fused = engine.q21_star_agg(parts_payload_tables(engine)[0],
                                parts_payload_tables(engine)[1],
                                parts_payload_tables(engine)[2],
                                pk, sk, od, rv, N)
    assert np.array_equal(np.asarray(fused, np.int64), expect)


_cached_tables = None


def parts_payload_tables(engine):
    """The fused path's payload tables (what the bench builds)."""
    global _cached_tables
    if _cached_tables is None:
        def pay(keys, payload):
            k = engine.alloc(keys.nbytes)
            k.h2d(keys)
            p = engine.alloc(payload.nbytes)
            p.h2d(payload)
            t = engine.join_build_payload(k, p, len(keys))
            k.free()
            p.free()
            return t
        pf = gen.build_part_dim_payload(SEED, gen.N_PARTS_SF100, CAT)
        sf = gen.build_supp_dim_payload(SEED, gen.N_SUPPS_SF100, REG)
        datekey, dyear = gen.gen_dates()
        _cached_tables = (
            pay(np.arange(1, gen.N_PARTS_SF100 + 1, dtype=np.int32), pf),
            pay(np.arange(1, gen.N_SUPPS_SF100 + 1, dtype=np.int32), sf),
            pay(datekey.astype(np.int32), (dyear - 1992 + 1).astype(np.uint32)))
    return _cached_tables
