"""JoinHashMapSelector restatement tests (VERDICT r01 missing #1 / next #4).

The reference picks the key constructor and hash-map method at build time
(`be/src/exec/join/join_hash_table.cpp:164-344`). The reference's own test
suite has no direct selector unit test, so the scenarios below are derived
from the cited decision rules themselves, with every threshold computed the
way the reference computes it (calc_bucket_size, interval gates, cache-size
comparisons). Both restatements (oracle C and the product C-ABI in
libgpue.so — pure host functions, loadable without a GPU) must agree with
the rules and with each other.
"""

import ctypes
import os

import numpy as np
import pytest

from oracle import pyoracle as orc
from starrocks_amd.engine import lib_path

# constructor classes
ONE_KEY, ONE_KEY_VARCHAR = 0, 1
FIXED_INT, FIXED_BIGINT, FIXED_LARGEINT, SERIALIZED_VARCHAR = 2, 3, 4, 5
# methods
DIRECT, RANGE_DIRECT, RANGE_DIRECT_SET, DENSE_RANGE_DIRECT = 0, 1, 2, 3
LINEAR_CHAINED, LINEAR_CHAINED_SET, BUCKET_CHAINED = 4, 5, 6
# lt classes
LT_TINY, LT_INT, LT_BIGINT, LT_OTHER, LT_VARCHAR = 0, 1, 2, 3, 4
# join modes
INNER, LEFT_SEMI, LEFT_ANTI, LEFT_OUTER = 0, 1, 2, 3

L2 = 1 << 20   # a CPU-typical L2, so the scenarios exercise the same
L3 = 32 << 20  # comparisons the reference's CpuInfo values would


def _gpue_lib():
    lib = ctypes.CDLL(lib_path())
    lib.gpue_join_select_key_constructor.restype = ctypes.c_int32
    lib.gpue_join_select_key_constructor.argtypes = [
        ctypes.c_int32, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int32,
        ctypes.c_void_p]
    lib.gpue_join_select_varchar_constructor.restype = ctypes.c_int32
    lib.gpue_join_select_varchar_constructor.argtypes = [ctypes.c_int32, ctypes.c_int32]
    lib.gpue_join_select_method.restype = ctypes.c_int32
    lib.gpue_join_select_method.argtypes = [
        ctypes.c_int32, ctypes.c_int32, ctypes.c_uint64, ctypes.c_int64,
        ctypes.c_int64, ctypes.c_int32, ctypes.c_int32, ctypes.c_int32,
        ctypes.c_int32, ctypes.c_uint64, ctypes.c_uint64]
    return lib


# ---- key constructor (_determine_key_constructor, :164-229) ----

KC_CASES = [
    # (fixed_sizes, null_safe, expect_constructor, expect_packed)
    ([4], None, ONE_KEY, 4),            # single int32 key
    ([8], None, ONE_KEY, 8),            # single int64 key
    ([4, 4], None, FIXED_BIGINT, 8),    # 2xi32 packs to 8 B (q43-style)
    ([4, 8], None, FIXED_LARGEINT, 12),
    ([4, 4, 4, 4], None, FIXED_LARGEINT, 16),
    ([4, 4, 4, 4, 4], None, SERIALIZED_VARCHAR, 0),  # 20 B > 16 -> serialize
    ([2, 2], None, FIXED_INT, 4),
    # null-safe equal keeps a null byte per key (:211)
    ([4, 4], [0, 1], FIXED_LARGEINT, 9),
    # single null-safe key takes the multi-key packing branch (:175 gate)
    ([4], [1], FIXED_BIGINT, 5),
    # multi-key with a fixable varchar (max_size 6) packs; unfixable forces
    # full serialization
    ([6, 4], None, FIXED_LARGEINT, 10),
    ([0, 4], None, SERIALIZED_VARCHAR, 0),
]


def test_key_constructor_scenarios():
    for sizes, ns, want_kc, want_pb in KC_CASES:
        kc, pb = orc.join_select_key_constructor(sizes, ns)
        assert (kc, pb) == (want_kc, want_pb), (sizes, ns, orc.KEYCON_NAMES[kc])


def test_varchar_constructor_boundaries():
    """:178-194 — single varchar key packs fixed when max string length <= 16."""
    for ms, want in [(1, FIXED_INT), (4, FIXED_INT), (5, FIXED_BIGINT),
                     (8, FIXED_BIGINT), (9, FIXED_LARGEINT), (16, FIXED_LARGEINT),
                     (17, ONE_KEY_VARCHAR), (0, ONE_KEY_VARCHAR)]:
        assert orc.join_select_varchar_constructor(ms) == want, ms
    # session flag off (enable_hash_join_serialize_fixed_size_string)
    assert orc.join_select_varchar_constructor(8, 0) == ONE_KEY_VARCHAR


# ---- hash-map method (_determine_hash_map_method + helpers, :231-344) ----

def bucket_size(rc):
    return orc.load().orc_calc_bucket_size(min(rc + 1, 0xFFFFFFFF))


METHOD_CASES = [
    # (lt, rc, mn, mx, mode, conj, expect, why)
    (LT_TINY, 100, 0, 0, INNER, 0, DIRECT, "bool/tiny/smallint -> DIRECT (:239)"),
    # SSB date dim: interval 69 K > bucket 4096 but <= L2 -> RANGE_DIRECT (:307)
    (LT_INT, 2556, 19920101, 19981230, INNER, 0, RANGE_DIRECT, "dates"),
    # SSB part dim: interval 1.4 M <= bucket 2 M -> RANGE_DIRECT
    (LT_INT, 1_400_000, 1, 1_400_000, INNER, 0, RANGE_DIRECT, "parts"),
    # semi-join without other conjunct, small interval -> 1-bit SET (:301-305)
    (LT_INT, 1000, 1, 60_000, LEFT_SEMI, 0, RANGE_DIRECT_SET, "semi set"),
    (LT_INT, 1000, 1, 60_000, LEFT_ANTI, 0, RANGE_DIRECT_SET, "anti set"),
    # ... but WITH another conjunct the set probe can't answer, so the
    # non-set branch runs; interval 60 K <= L2 -> RANGE_DIRECT
    (LT_INT, 1000, 1, 60_000, LEFT_SEMI, 1, RANGE_DIRECT, "semi w/ conjunct"),
    # semi-join, huge interval: 1e9/8 B > bucket*64 and > L3/2 -> linear set
    (LT_INT, 1000, 0, 10**9 - 1, LEFT_SEMI, 0, LINEAR_CHAINED_SET, "sparse semi"),
    # inner, 10 M rows, interval 100 M: > bucket 2^24, > L2, but
    # interval/4 + rc*4 <= (b + b/10)*4 -> 2-bit DENSE (:310-317)
    (LT_INT, 10_000_000, 1, 100_000_000, INNER, 0, DENSE_RANGE_DIRECT, "dense"),
    # same rows, interval 500 M: dense inequality fails; bucket 2^24 exceeds
    # the 24-bit fp cap -> BUCKET_CHAINED (:335, join_hash_map_method.h:135)
    (LT_INT, 10_000_000, 1, 500_000_000, INNER, 0, BUCKET_CHAINED, "fallback"),
    # small table, interval >= 2^32 -> range-direct gate closes (:288),
    # linear fits -> LINEAR_CHAINED
    (LT_BIGINT, 100_000, 0, 2**33, INNER, 0, LINEAR_CHAINED, "sparse 64-bit"),
    # int64 min/max overflow guard (:283)
    (LT_BIGINT, 100, -2**63, 2**63 - 1, INNER, 0, LINEAR_CHAINED, "overflow guard"),
    # non-int single key (e.g. largeint/date) skips range-direct entirely
    (LT_OTHER, 2556, 0, 0, INNER, 0, LINEAR_CHAINED, "non-int one-key"),
    (LT_VARCHAR, 2556, 0, 0, INNER, 0, LINEAR_CHAINED, "slice key"),
    # big build side: bucket > 16 M buckets -> bucket-chained
    (LT_VARCHAR, 20_000_000, 0, 0, INNER, 0, BUCKET_CHAINED, "big slice"),
    (LT_INT, 0, 0, 0, INNER, 0, LINEAR_CHAINED, "empty build"),
]


def test_method_scenarios():
    for lt, rc, mn, mx, mode, conj, want, why in METHOD_CASES:
        got = orc.join_select_method(ONE_KEY, lt, rc, mn, mx, mode, conj,
                                     l2_size=L2, l3_size=L3)
        assert got == want, f"{why}: got {orc.JM_NAMES[got]} want {orc.JM_NAMES[want]}"
    # sanity on the derived thresholds the cases rely on
    assert bucket_size(2556) == 4096
    assert bucket_size(10_000_000) == 1 << 24


def test_method_flag_gates():
    """Session flags enable_hash_join_{range_direct_mapping,linear_chained}_opt
    (default true, SessionVariable.java:2060-2064) gate each tier."""
    args = (ONE_KEY, LT_INT, 2556, 19920101, 19981230, INNER, 0)
    assert orc.join_select_method(*args, 1, 1, L2, L3) == RANGE_DIRECT
    assert orc.join_select_method(*args, 0, 1, L2, L3) == LINEAR_CHAINED
    assert orc.join_select_method(*args, 0, 0, L2, L3) == BUCKET_CHAINED
    # multi-key constructors never take the range-direct path (:242 gate)
    assert orc.join_select_method(FIXED_BIGINT, LT_INT, 2556, 19920101, 19981230,
                                  INNER, 0, 1, 1, L2, L3) == LINEAR_CHAINED


def test_oracle_vs_gpue_selector_cross_impl():
    """The product C-ABI restatement must agree with the oracle's on a
    randomized scenario sweep (pure host functions — no GPU needed)."""
    lib = _gpue_lib()
    rng = np.random.default_rng(7)
    for _ in range(2000):
        lt = int(rng.integers(0, 5))
        kc = int(rng.integers(0, 6))
        rc = int(rng.choice([0, 1, 100, 4096, 10**5, 10**7, 2 * 10**7]))
        mn = int(rng.choice([0, 1, -100, 19920101, -2**62]))
        span = int(rng.choice([0, 10, 4000, 70000, 10**6, 10**8, 2**33]))
        mode = int(rng.integers(0, 4))
        conj = int(rng.integers(0, 2))
        erd = int(rng.integers(0, 2))
        elc = int(rng.integers(0, 2))
        a = orc.join_select_method(kc, lt, rc, mn, mn + span, mode, conj, erd, elc,
                                   L2, L3)
        b = lib.gpue_join_select_method(kc, lt, rc, mn, mn + span, mode, conj,
                                        erd, elc, L2, L3)
        assert a == b, (kc, lt, rc, mn, span, mode, conj, erd, elc,
                        orc.JM_NAMES[a], orc.JM_NAMES[b])
    for sizes, ns, _, _ in KC_CASES:
        n = len(sizes)
        fs = np.asarray(sizes, np.int32)
        nsa = np.asarray(ns if ns else [0] * n, np.uint8)
        pb = ctypes.c_int32()
        g = lib.gpue_join_select_key_constructor(
            n, fs.ctypes.data_as(ctypes.c_void_p), nsa.ctypes.data_as(ctypes.c_void_p),
            1, ctypes.byref(pb))
        o, opb = orc.join_select_key_constructor(sizes, ns)
        assert (g, pb.value) == (o, opb)


# ---- GPU: selector-driven build end to end ----

@pytest.mark.gpu
def test_join_build_auto_dense_dim(engine):
    """Dense dim keys (the SSB shape) must auto-select RANGE_DIRECT and the
    built table must probe identically to the explicit range-direct build."""
    from starrocks_amd import gen
    datekey, _ = gen.gen_dates()
    keys = np.concatenate([[0], datekey]).astype(np.int32)  # 1-based, row 0 sentinel
    kb = engine.alloc(keys.nbytes)
    kb.h2d(keys)
    t, method = engine.join_build_auto(kb, len(datekey))
    assert engine.JM_NAMES[method] == "RANGE_DIRECT"
    probe = datekey[::7].astype(np.int32)
    pb = engine.alloc(probe.nbytes)
    pb.h2d(probe)
    cnt = engine.join_probe_emit(t, pb, len(probe))
    assert cnt == len(probe)  # unique dense keys: every probe matches once
    t.destroy()
    # sparse 64-bit-range... not expressible with i32 keys; instead force the
    # linear tier with the flagged decision: a 100 K-row table whose interval
    # (~2^31) exceeds bucket and the MI355X L2 default
    rng = np.random.default_rng(3)
    sk = np.concatenate([[0], rng.integers(1, 2**31 - 1, 100_000)]).astype(np.int32)
    skb = engine.alloc(sk.nbytes)
    skb.h2d(sk)
    t2, m2 = engine.join_build_auto(skb, 100_000)
    assert engine.JM_NAMES[m2] == "LINEAR_CHAINED"
    # probe half present / half absent keys; compare against bucket-chained
    probe2 = np.concatenate([sk[1:2000], (rng.integers(1, 2**31 - 1, 2000))]).astype(np.int32)
    pb2 = engine.alloc(probe2.nbytes)
    pb2.h2d(probe2)
    c_auto = engine.join_probe_emit(t2, pb2, len(probe2))
    t3 = engine.join_build_bucket_chained(skb, 100_000)
    c_ref = engine.join_probe_emit(t3, pb2, len(probe2))
    assert c_auto == c_ref
    for b in (kb, pb, skb, pb2):
        b.free()
    t2.destroy()
    t3.destroy()


@pytest.mark.gpu
def test_join_build_auto_semi_set(engine):
    """LEFT SEMI without other conjuncts on a small interval picks the 1-bit
    SET method; the GPU maps it onto the direct-mapped table and the semi
    probe answers identically."""
    keys = np.concatenate([[0], np.arange(1, 5001)]).astype(np.int32)
    kb = engine.alloc(keys.nbytes)
    kb.h2d(keys)
    t, method = engine.join_build_auto(kb, 5000, mode=1)
    assert engine.JM_NAMES[method] in ("RANGE_DIRECT_SET", "RANGE_DIRECT")
    probe = np.arange(4000, 7000, dtype=np.int32)
    pb = engine.alloc(probe.nbytes)
    pb.h2d(probe)
    cnt = engine.join_probe_emit_mode(t, pb, len(probe), 1)  # LEFT_SEMI
    assert cnt == 1001  # 4000..5000 present
    kb.free()
    pb.free()
    t.destroy()


def test_bench_dims_select_range_direct_on_mi355x_defaults():
    """The four SSB dim builds in bench.py route through the selector with
    MI355X cache defaults (4 MiB XCD L2) — all must land in the
    direct-mapped family or the fused payload-table path would refuse."""
    L2M, L3M = 4 << 20, 256 << 20
    dims = [(2556, 19920101, 19981230), (1_400_000, 1, 1_400_000),
            (200_000, 1, 200_000), (3_000_000, 1, 3_000_000)]
    for rc, mn, mx in dims:
        m = orc.join_select_method(ONE_KEY, LT_INT, rc, mn, mx, INNER, 0,
                                   l2_size=L2M, l3_size=L3M)
        assert m == RANGE_DIRECT, (rc, orc.JM_NAMES[m])


@pytest.mark.gpu
def test_join_build_auto_u64(engine):
    """8-byte-key auto build: decision reported (sparse 64-bit interval >=
    2^32 closes the range-direct gate -> LINEAR decision; physical tier is
    the u64 bucket table), probes answer correctly."""
    rng = np.random.default_rng(61)
    n = 50_000
    keys = np.concatenate([[0], rng.integers(1, 2**40, n)]).astype(np.uint64)
    kb = engine.alloc(keys.nbytes)
    kb.h2d(keys)
    t, method = engine.join_build_auto_u64(kb, n)
    assert orc.JM_NAMES[method] == "LINEAR_CHAINED"  # interval >= 2^32
    probe = np.concatenate([keys[1:1001], rng.integers(1, 2**40, 1000)]).astype(np.uint64)
    pb = engine.alloc(probe.nbytes)
    pb.h2d(probe)
    cnt = engine.join_probe_emit_u64(t, pb, len(probe), mode=1)
    present = set(keys[1:].tolist())
    assert cnt == sum(p in present for p in probe.tolist())
    t.destroy()
    kb.free()
    pb.free()
