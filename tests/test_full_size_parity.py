"""Parity at BASELINE.json's FULL configured sizes (§8c: small sizes compare
against the oracle directly; the full sizes are checked here, also directly —
the oracle's OpenMP restatement finishes in seconds on the GPU box's host
cores, so no weaker size-independent property is needed). All integer results
compare bit-exact.

Runs on the MI355X box (the oracle .so travels with the snapshot); marked gpu.
"""

import numpy as np
import pytest

from oracle import pyoracle as orc
from starrocks_amd import gen

pytestmark = pytest.mark.gpu

SEED = 42


def _payload_table(engine, keys_i32, payload_u32):
    kb = engine.alloc(keys_i32.nbytes)
    kb.h2d(keys_i32)
    pb = engine.alloc(payload_u32.nbytes)
    pb.h2d(payload_u32)
    t = engine.join_build_payload(kb, pb, len(keys_i32))
    kb.free()
    pb.free()
    return t


def test_q1_full_sf10(engine):
    n = gen.SF10_LINEORDER_ROWS  # 59,986,052 — §8d config 2 exactly
    cols = [engine.alloc(n * 4) for _ in range(3)]
    engine.gen_lineorder_q1(SEED, 0, n, *cols)
    datekey, dyear = gen.gen_dates()
    dpay = np.where(dyear == 1993, dyear - 1992 + 1, 0).astype(np.uint32)
    dates = _payload_table(engine, datekey.astype(np.int32), dpay)
    s, cnt = engine.q1_join_sum(dates, *cols, n)
    es, ecnt = orc.q1_pipeline(SEED, 0, n, 1993)
    assert (s, cnt) == (es, ecnt)
    for b in cols:
        b.free()
    dates.destroy()


def test_q21_full_sf100(engine):
    n = gen.SF100_LINEORDER_ROWS  # 600,000,000 — §8d config 3 exactly
    cols = [engine.alloc(n * 4) for _ in range(4)]
    engine.gen_lineorder_q21(SEED, 0, n, *cols)
    parts = _payload_table(engine, np.arange(1, gen.N_PARTS_SF100 + 1, dtype=np.int32),
                           gen.build_part_dim_payload(SEED, gen.N_PARTS_SF100, 12))
    supps = _payload_table(engine, np.arange(1, gen.N_SUPPS_SF100 + 1, dtype=np.int32),
                           gen.build_supp_dim_payload(SEED, gen.N_SUPPS_SF100, 2))
    datekey, dyear = gen.gen_dates()
    dates = _payload_table(engine, datekey.astype(np.int32),
                           (dyear - 1992 + 1).astype(np.uint32))
    got = engine.q21_star_agg(parts, supps, dates, *cols, n)
    expect = orc.q21_pipeline(SEED, 0, n, 12, 2)
    assert np.array_equal(got, expect)
    for b in cols:
        b.free()
    for t in (parts, supps, dates):
        t.destroy()


def test_q43_full_sf100(engine):
    n = gen.SF100_LINEORDER_ROWS  # §8d config 4 exactly (single-GPU leg)
    cols = [engine.alloc(n * 4) for _ in range(6)]
    engine.gen_lineorder_q43(SEED, 0, n, *cols)
    custs = _payload_table(engine, np.arange(1, gen.N_CUSTS_SF100 + 1, dtype=np.int32),
                           gen.build_cust_dim_q43(SEED, gen.N_CUSTS_SF100, 1))
    supps = _payload_table(engine, np.arange(1, gen.N_SUPPS_SF100 + 1, dtype=np.int32),
                           gen.build_supp_dim_q43(SEED, gen.N_SUPPS_SF100, 7))
    parts = _payload_table(engine, np.arange(1, gen.N_PARTS_SF100 + 1, dtype=np.int32),
                           gen.build_part_dim_q43(SEED, gen.N_PARTS_SF100, 12))
    datekey, dyear = gen.gen_dates()
    dpay = np.where(dyear == 1997, 1, np.where(dyear == 1998, 2, 0)).astype(np.uint32)
    dates = _payload_table(engine, datekey.astype(np.int32), dpay)
    acc = engine.alloc(800 * 8)
    engine.q43_star_agg_async(custs, supps, parts, dates, *cols, n, acc)
    got = acc.d2h(np.int64, 800)
    expect = orc.q43_pipeline(SEED, 0, n, 1, 7, 12)
    assert np.array_equal(got, expect)
    for b in cols + [acc]:
        b.free()
    for t in (custs, supps, parts, dates):
        t.destroy()


def test_q3_full_sf300_shard(engine):
    """Config 5 at the full dim sizes (450 M orders, 45 M customers) and the
    per-GPU lineitem shard size the bench runs (225 M rows)."""
    n, n_orders, n_custs = 225_000_000, 450_000_000, 45_000_000
    mkt = engine.alloc(n_custs * 16)
    engine.gen_cust_mkt16(SEED, n_custs, mkt)
    cbits = engine.alloc((n_custs + 31) // 32 * 4)
    engine.bits_str16_eq(mkt, n_custs, orc.mkt_literal(1), cbits)
    mkt.free()
    oc, od = engine.alloc(n_orders * 4), engine.alloc(n_orders * 4)
    engine.gen_orders_q3(SEED, n_orders, n_custs, oc, od)
    obits = engine.alloc((n_orders + 31) // 32 * 4)
    engine.q3_order_bits(oc, od, n_orders, cbits, 19950315, obits)
    lk, ext, disc = (engine.alloc(n * 8) for _ in range(3))
    ship = engine.alloc(n * 4)
    engine.gen_lineitem_q3(SEED, 0, n, n_orders, lk, ext, disc, ship)
    max_out = 64_000_000
    ok_b, os_b = engine.alloc(max_out * 8), engine.alloc(max_out * 8)
    g = engine.q3_probe_agg(lk, ext, disc, ship, n, obits, 19950315, ok_b, os_b,
                            max_out, capacity_hint=64_000_000)
    gk = ok_b.d2h(np.uint64, g)
    gs = os_b.d2h(np.int64, g)
    order = np.argsort(gk)
    ek, es = orc.q3_pipeline(SEED, 0, n, n_orders, n_custs)
    assert g == len(ek)
    assert np.array_equal(gk[order], ek)
    assert np.array_equal(gs[order], es)
    # and the ORDER BY revenue DESC LIMIT 10 on top
    tk, tv = engine.topk_i64(ok_b, os_b, g, 10)
    eorder = np.lexsort((ek, es))[::-1][:10]
    assert np.array_equal(tv, es[eorder])
    assert np.array_equal(tk, ek[eorder])
    for b in (cbits, oc, od, obits, lk, ext, disc, ship, ok_b, os_b):
        b.free()
