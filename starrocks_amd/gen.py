"""Deterministic synthetic SSB/TPC-H columns — numpy restatement.

This is the SAME counter-based generator (splitmix64 finalizer) implemented in
oracle/oracle.c (orc_gen_*) and starrocks_amd/csrc/gpue.hip (k_gen_*): all
three produce bit-identical columns for a given (seed, tag, row index), so CPU
oracle and GPU engine operate on identical data with no transfer.

Schema follows the reference's SSB DDL (reference
test/common/sql/ssb/create.sql) at the granularity BASELINE.json's configs
need; SURVEY.md §8d fixes the column value ranges.
"""

import numpy as np

_GOLD = np.uint64(0x9E3779B97F4A7C15)
_M1 = np.uint64(0xBF58476D1CE4E5B9)
_M2 = np.uint64(0x94D049BB133111EB)

# tags shared with oracle/oracle.c and csrc/gpue.hip
TAG_ORDERDATE = 1
TAG_EXTPRICE = 2
TAG_DISCOUNT = 3
TAG_PARTKEY = 4
TAG_SUPPKEY = 5
TAG_REVENUE = 6
TAG_PCAT = 7
TAG_PBRD = 8
TAG_SREG = 9
TAG_CUSTKEY = 10
TAG_SUPPCOST = 11
TAG_CREG = 12
TAG_SNAT = 13
TAG_SCITY = 14

N_DAYS = 2556
N_PARTS_SF100 = 1_400_000
N_SUPPS_SF100 = 200_000
N_CUSTS_SF100 = 3_000_000
SF10_LINEORDER_ROWS = 59_986_052
SF100_LINEORDER_ROWS = 600_000_000


def sm64(x: np.ndarray) -> np.ndarray:
    with np.errstate(over="ignore"):
        x = x.astype(np.uint64) + _GOLD
        x = (x ^ (x >> np.uint64(30))) * _M1
        x = (x ^ (x >> np.uint64(27))) * _M2
        return x ^ (x >> np.uint64(31))


def gen_u64(seed: int, tag: int, idx: np.ndarray) -> np.ndarray:
    with np.errstate(over="ignore"):
        base = np.uint64(seed) + np.uint64(tag) * _GOLD
        return sm64(base + idx.astype(np.uint64))


def gen_dates(n_days: int = N_DAYS):
    """datekey/d_year arrays from 1992-01-01 (identical to orc_gen_dates)."""
    mdays = [31, 28, 31, 30, 31, 30, 31, 31, 30, 31, 30, 31]
    datekey = np.empty(n_days, dtype=np.int32)
    dyear = np.empty(n_days, dtype=np.int32)
    y, m, d = 1992, 1, 1
    for i in range(n_days):
        datekey[i] = y * 10000 + m * 100 + d
        dyear[i] = y
        leap = (y % 4 == 0 and y % 100 != 0) or y % 400 == 0
        md = mdays[m - 1] + (1 if (m == 2 and leap) else 0)
        d += 1
        if d > md:
            d = 1
            m += 1
            if m > 12:
                m, y = 1, y + 1
    return datekey, dyear


def gen_lineorder_q1(seed: int, row_start: int, n: int):
    idx = np.arange(row_start, row_start + n, dtype=np.uint64)
    datekey, _ = gen_dates()
    od = datekey[(gen_u64(seed, TAG_ORDERDATE, idx) % np.uint64(N_DAYS)).astype(np.int64)]
    ep = (gen_u64(seed, TAG_EXTPRICE, idx) % np.uint64(100000)).astype(np.int32) + 1
    dc = (gen_u64(seed, TAG_DISCOUNT, idx) % np.uint64(11)).astype(np.int32)
    return od.astype(np.int32), ep, dc


def gen_lineorder_q21(seed: int, row_start: int, n: int):
    idx = np.arange(row_start, row_start + n, dtype=np.uint64)
    datekey, _ = gen_dates()
    pk = (gen_u64(seed, TAG_PARTKEY, idx) % np.uint64(N_PARTS_SF100)).astype(np.int32) + 1
    sk = (gen_u64(seed, TAG_SUPPKEY, idx) % np.uint64(N_SUPPS_SF100)).astype(np.int32) + 1
    od = datekey[(gen_u64(seed, TAG_ORDERDATE, idx) % np.uint64(N_DAYS)).astype(np.int64)]
    rv = (gen_u64(seed, TAG_REVENUE, idx) % np.uint64(10000000)).astype(np.int32)
    return pk, sk, od.astype(np.int32), rv


def gen_lineorder_q43(seed: int, row_start: int, n: int):
    idx = np.arange(row_start, row_start + n, dtype=np.uint64)
    datekey, _ = gen_dates()
    ck = (gen_u64(seed, TAG_CUSTKEY, idx) % np.uint64(N_CUSTS_SF100)).astype(np.int32) + 1
    sk = (gen_u64(seed, TAG_SUPPKEY, idx) % np.uint64(N_SUPPS_SF100)).astype(np.int32) + 1
    pk = (gen_u64(seed, TAG_PARTKEY, idx) % np.uint64(N_PARTS_SF100)).astype(np.int32) + 1
    od = datekey[(gen_u64(seed, TAG_ORDERDATE, idx) % np.uint64(N_DAYS)).astype(np.int64)]
    rv = (gen_u64(seed, TAG_REVENUE, idx) % np.uint64(10000000)).astype(np.int32)
    sc = (gen_u64(seed, TAG_SUPPCOST, idx) % np.uint64(100000)).astype(np.int32) + 1
    return ck, sk, pk, od.astype(np.int32), rv, sc


def build_cust_dim_q43(seed: int, n_custs: int, region: int) -> np.ndarray:
    keys = np.arange(1, n_custs + 1, dtype=np.uint64)
    reg = (gen_u64(seed, TAG_CREG, keys) % np.uint64(5)).astype(np.uint32)
    return (reg == region).astype(np.uint32)


def build_supp_dim_q43(seed: int, n_supps: int, nation: int) -> np.ndarray:
    keys = np.arange(1, n_supps + 1, dtype=np.uint64)
    nat = (gen_u64(seed, TAG_SNAT, keys) % np.uint64(25)).astype(np.uint32)
    city = (gen_u64(seed, TAG_SCITY, keys) % np.uint64(10)).astype(np.uint32)
    return np.where(nat == nation, city + 1, 0).astype(np.uint32)


def build_part_dim_q43(seed: int, n_parts: int, category: int) -> np.ndarray:
    keys = np.arange(1, n_parts + 1, dtype=np.uint64)
    cat = part_category(seed, keys)
    brand_in_cat = (gen_u64(seed, TAG_PBRD, keys) % np.uint64(40)).astype(np.uint32)
    return np.where(cat == category, brand_in_cat + 1, 0).astype(np.uint32)


def build_date_dim_q43() -> tuple:
    datekey, dyear = gen_dates()
    mn, mx = int(datekey[0]), int(datekey[-1])
    first = np.zeros(mx - mn + 1, dtype=np.uint32)
    first[datekey[dyear == 1997] - mn] = 1
    first[datekey[dyear == 1998] - mn] = 2
    return mn, mx, first


def fnv_u32(keys: np.ndarray) -> np.ndarray:
    """HashUtil::fnv_hash over the 4 LE bytes of each uint32 key (reference
    hash_util.hpp:133-143), vectorized; identical to oracle orc_fnv_hash and
    the HIP fnv_u32."""
    with np.errstate(over="ignore"):
        h = np.full(keys.shape, 0x811C9DC5, np.uint32)
        k = keys.astype(np.uint32)
        for b in range(4):
            byte = ((k >> np.uint32(8 * b)) & np.uint32(0xFF))
            h = (byte ^ h) * np.uint32(16777619)
        return h


def fnv_u64(keys: np.ndarray) -> np.ndarray:
    """FNV over the 8 LE bytes of each uint64 key (BIGINT columns)."""
    with np.errstate(over="ignore"):
        h = np.full(keys.shape, 0x811C9DC5, np.uint32)
        k = keys.astype(np.uint64)
        for b in range(8):
            byte = ((k >> np.uint64(8 * b)) & np.uint64(0xFF)).astype(np.uint32)
            h = (byte ^ h) * np.uint32(16777619)
        return h


def partition_channels_u64(keys: np.ndarray, num_channels: int) -> np.ndarray:
    return ((fnv_u64(keys).astype(np.uint64) * np.uint64(num_channels))
            >> np.uint64(32)).astype(np.uint32)


def fnv_u32_seeded(keys: np.ndarray, seeds: np.ndarray) -> np.ndarray:
    """fnv over 4 LE bytes with a per-row running seed (the exchange sink's
    chained multi-column hash, exchange_sink_operator.cpp:611-617)."""
    with np.errstate(over="ignore"):
        h = seeds.astype(np.uint32).copy()
        k = keys.astype(np.uint32)
        for b in range(4):
            byte = ((k >> np.uint32(8 * b)) & np.uint32(0xFF))
            h = (byte ^ h) * np.uint32(16777619)
        return h


def partition_channels_2xi32(a: np.ndarray, b: np.ndarray, num_channels: int) -> np.ndarray:
    h = fnv_u32_seeded(b, fnv_u32(a))
    return ((h.astype(np.uint64) * np.uint64(num_channels)) >> np.uint64(32)).astype(np.uint32)


def partition_channels(keys: np.ndarray, num_channels: int) -> np.ndarray:
    """ReduceOp channel assignment (shuffler.h:71-86): (fnv(key)*n)>>32."""
    return ((fnv_u32(keys).astype(np.uint64) * np.uint64(num_channels))
            >> np.uint64(32)).astype(np.uint32)


def part_category(seed: int, partkeys: np.ndarray) -> np.ndarray:
    return (gen_u64(seed, TAG_PCAT, partkeys) % np.uint64(25)).astype(np.uint32)


def part_brand(seed: int, partkeys: np.ndarray) -> np.ndarray:
    cat = part_category(seed, partkeys)
    return cat * 40 + (gen_u64(seed, TAG_PBRD, partkeys) % np.uint64(40)).astype(np.uint32)


def supp_region(seed: int, suppkeys: np.ndarray) -> np.ndarray:
    return (gen_u64(seed, TAG_SREG, suppkeys) % np.uint64(5)).astype(np.uint32)


def build_date_dim_payload(year_filter: int | None):
    """Direct-mapped date payload array: value (d_year-1992)+1
    where the dim row passes the year filter, else 0. Returns (min_key,
    max_key, first)."""
    datekey, dyear = gen_dates()
    mn, mx = int(datekey[0]), int(datekey[-1])
    first = np.zeros(mx - mn + 1, dtype=np.uint32)
    keep = np.ones(len(datekey), bool) if year_filter is None else (dyear == year_filter)
    first[datekey[keep] - mn] = (dyear[keep] - 1992 + 1).astype(np.uint32)
    return mn, mx, first


def build_part_dim_payload(seed: int, n_parts: int, category: int) -> np.ndarray:
    """first[p-1] = brand+1 if p_category==category else 0."""
    pkeys = np.arange(1, n_parts + 1, dtype=np.uint64)
    cat = part_category(seed, pkeys)
    brand = part_brand(seed, pkeys)
    return np.where(cat == category, brand + 1, 0).astype(np.uint32)


def build_supp_dim_payload(seed: int, n_supps: int, region: int) -> np.ndarray:
    skeys = np.arange(1, n_supps + 1, dtype=np.uint64)
    return (supp_region(seed, skeys) == region).astype(np.uint32)


# TPC-H c_mktsegment 5-value dictionary, 16-byte space-padded fixed strings
# (the SERIALIZED_FIXED_SIZE ≤16 B key packing of join_key_constructor.h:40-153
# applied to the Q3 string literal). Product-path copy of the constant the
# oracle also restates (oracle.c MKT_SEGMENTS) — kept here so the product path
# has zero oracle imports (oracle/ is test infrastructure only);
# tests/test_gen_consistency.py asserts the two stay identical.
MKT_SEGMENTS = [b"AUTOMOBILE      ", b"BUILDING        ", b"FURNITURE       ",
                b"MACHINERY       ", b"HOUSEHOLD       "]


def mkt_literal(idx: int = 1) -> bytes:
    """16-byte mktsegment literal (Q3 filters c_mktsegment = 'BUILDING')."""
    return MKT_SEGMENTS[idx]
