"""ctypes host bindings over the C-ABI (include/gpue.h).

This is the host side of the drop-in boundary: the same calls a thin C++
Operator wrapper inside the reference BE would make from the pipeline driver
loop (INTEGRATION.md shows that wrapper). There is NO CPU fallback: a missing
libgpue.so or absent GPU raises GpueError.
"""

import ctypes
import os

import numpy as np

_PKG_DIR = os.path.dirname(os.path.abspath(__file__))


def lib_path() -> str:
    p = os.environ.get("GPUE_LIB")
    if p:
        return p
    return os.path.join(_PKG_DIR, "libgpue.so")


_lib = None


def _load():
    global _lib
    if _lib is not None:
        return _lib
    p = lib_path()
    if not os.path.exists(p):
        raise GpueError(
            f"HIP engine library not found at {p}; build it with "
            "`python -c 'import __graft_entry__; __graft_entry__.build()'` — "
            "the product path has no CPU fallback."
        )
    _lib = ctypes.CDLL(p)
    _declare(_lib)
    return _lib


class GpueError(RuntimeError):
    pass


c_u64 = ctypes.c_uint64
c_i64 = ctypes.c_int64
c_u32 = ctypes.c_uint32
c_i32 = ctypes.c_int32
c_vp = ctypes.c_void_p


def _declare(lib):
    lib.gpue_last_error.restype = ctypes.c_char_p
    sigs = {
        "gpue_session_create": (c_i32, [c_i32, ctypes.POINTER(c_vp)]),
        "gpue_session_destroy": (None, [c_vp]),
        "gpue_device_count": (c_i32, [ctypes.POINTER(c_i32)]),
        "gpue_sync": (c_i32, [c_vp]),
        "gpue_dbuf_alloc": (c_i32, [c_vp, c_u64, ctypes.POINTER(c_vp)]),
        "gpue_dbuf_free": (None, [c_vp]),
        "gpue_dbuf_h2d": (c_i32, [c_vp, c_vp, c_u64, c_u64]),
        "gpue_dbuf_d2h": (c_i32, [c_vp, c_vp, c_u64, c_u64]),
        "gpue_dbuf_memset": (c_i32, [c_vp, c_i32, c_u64]),
        "gpue_join_select_key_constructor": (c_i32, [c_i32, c_vp, c_vp, c_i32,
                                                     ctypes.POINTER(c_i32)]),
        "gpue_join_select_varchar_constructor": (c_i32, [c_i32, c_i32]),
        "gpue_join_select_method": (c_i32, [c_i32, c_i32, c_u64, c_i64, c_i64, c_i32,
                                            c_i32, c_i32, c_i32, c_u64, c_u64]),
        "gpue_join_build_auto_i32": (c_i32, [c_vp, c_vp, c_u64, c_i32, c_i32, c_u64,
                                             c_u64, ctypes.POINTER(c_vp),
                                             ctypes.POINTER(c_i32)]),
        "gpue_join_build_auto_u64": (c_i32, [c_vp, c_vp, c_u64, c_i32, c_i32, c_u64,
                                             c_u64, ctypes.POINTER(c_vp),
                                             ctypes.POINTER(c_i32)]),
        "gpue_gen_u32_mod": (c_i32, [c_vp, c_vp, c_u64, c_u64, c_u64, c_u64, c_u32, c_u32]),
        "gpue_gen_i64": (c_i32, [c_vp, c_vp, c_u64, c_u64, c_u64, c_u64]),
        "gpue_gen_lineorder_q1": (c_i32, [c_vp, c_u64, c_u64, c_u64, c_vp, c_vp, c_vp]),
        "gpue_gen_lineorder_q21": (c_i32, [c_vp, c_u64, c_u64, c_u64, c_vp, c_vp, c_vp, c_vp]),
        "gpue_gen_lineorder_q43": (c_i32, [c_vp, c_u64, c_u64, c_u64] + [c_vp] * 6),
        "gpue_eval_conjuncts_i32": (c_i32, [c_vp, ctypes.POINTER(c_vp), c_i32, c_u64,
                                            c_vp, c_vp, c_vp, c_vp, c_i32,
                                            ctypes.POINTER(c_u64)]),
        "gpue_eval_conjuncts_i64": (c_i32, [c_vp, ctypes.POINTER(c_vp), c_i32, c_u64,
                                            c_vp, c_vp, c_vp, c_vp, c_i32,
                                            ctypes.POINTER(c_u64)]),
        "gpue_scan_filter_i64_lt": (c_i32, [c_vp, c_vp, c_u64, c_i64, c_vp, ctypes.POINTER(c_u64)]),
        "gpue_scan_filter_i64_lt_sp": (c_i32, [c_vp, c_vp, c_u64, c_i64, c_vp, ctypes.POINTER(c_u64)]),
        "gpue_join_build_payload_i32": (c_i32, [c_vp, c_vp, c_vp, c_u64, ctypes.POINTER(c_vp)]),
        "gpue_join_build_range_direct_i32": (c_i32, [c_vp, c_vp, c_u64, ctypes.POINTER(c_vp)]),
        "gpue_join_build_dense_range_direct_i32": (c_i32, [c_vp, c_vp, c_u64,
                                                           ctypes.POINTER(c_vp)]),
        "gpue_join_build_bucket_chained_u32": (c_i32, [c_vp, c_vp, c_u64, ctypes.POINTER(c_vp)]),
        "gpue_join_build_linear_chained_u32": (c_i32, [c_vp, c_vp, c_u64, ctypes.POINTER(c_vp)]),
        "gpue_join_build_bucket_chained_u64": (c_i32, [c_vp, c_vp, c_u64, ctypes.POINTER(c_vp)]),
        "gpue_join_build_bucket_chained_u128": (c_i32, [c_vp, c_vp, c_u64,
                                                        ctypes.POINTER(c_vp)]),
        "gpue_join_probe_emit_mode_u128": (c_i32, [c_vp, c_vp, c_vp, c_u64, c_i32, c_vp,
                                                   c_vp, ctypes.POINTER(c_u64)]),
        "gpue_join_probe_emit_mode_u64": (c_i32, [c_vp, c_vp, c_vp, c_u64, c_i32, c_vp, c_vp,
                                                  ctypes.POINTER(c_u64)]),
        "gpue_join_table_destroy": (None, [c_vp]),
        "gpue_join_table_minmax": (c_i32, [c_vp, ctypes.POINTER(c_i64), ctypes.POINTER(c_i64)]),
        "gpue_join_table_first_d2h": (c_i32, [c_vp, c_vp, c_u64]),
        "gpue_join_probe_emit_i32": (c_i32, [c_vp, c_vp, c_vp, c_u64, c_vp, c_vp, ctypes.POINTER(c_u64)]),
        "gpue_join_probe_emit_mode_i32": (c_i32, [c_vp, c_vp, c_vp, c_u64, c_i32, c_vp, c_vp, ctypes.POINTER(c_u64)]),
        "gpue_join_probe_right_varchar": (c_i32, [c_vp, c_vp, c_vp, c_vp, c_u64, c_i32,
                                                  c_vp, ctypes.POINTER(c_u64)]),
        "gpue_join_probe_right_i32": (c_i32, [c_vp, c_vp, c_vp, c_u64, c_i32, c_vp, ctypes.POINTER(c_u64)]),
        "gpue_join_build_bucket_chained_nulls_u32": (c_i32, [c_vp, c_vp, c_vp, c_u64, ctypes.POINTER(c_vp)]),
        "gpue_join_probe_emit_nulls_i32": (c_i32, [c_vp, c_vp, c_vp, c_vp, c_u64, c_i32, c_vp, c_vp, ctypes.POINTER(c_u64)]),
        "gpue_pack_keys_2xi32": (c_i32, [c_vp, c_vp, c_vp, c_u64, c_vp]),
        "gpue_pack_keys_2xi64": (c_i32, [c_vp, c_vp, c_vp, c_u64, c_vp]),
        "gpue_q1_join_sum": (c_i32, [c_vp, c_vp, c_vp, c_vp, c_vp, c_u64,
                                     ctypes.POINTER(c_i64), ctypes.POINTER(c_u64)]),
        "gpue_q21_star_agg": (c_i32, [c_vp, c_vp, c_vp, c_vp, c_vp, c_vp, c_vp, c_vp, c_u64, c_vp]),
        "gpue_q1_join_sum_async": (c_i32, [c_vp, c_vp, c_vp, c_vp, c_vp, c_u64, c_vp]),
        "gpue_q1_join_sum_accum": (c_i32, [c_vp, c_vp, c_vp, c_vp, c_vp, c_u64, c_vp]),
        "gpue_dbuf_wrap": (c_i32, [c_vp, c_vp, c_u64, ctypes.POINTER(c_vp)]),
        "gpue_dbuf_ptr": (c_i32, [c_vp, ctypes.POINTER(c_vp)]),
        "gpue_q43_star_agg_async": (c_i32, [c_vp] * 11 + [c_u64, c_vp]),
        "gpue_q21_star_agg_async": (c_i32, [c_vp, c_vp, c_vp, c_vp, c_vp, c_vp, c_vp, c_vp, c_u64, c_vp]),
        "gpue_q21_star_agg_pipe": (c_i32, [c_vp] * 8 + [c_u64, c_vp, c_vp, c_i32]),
        "gpue_partition_i32": (c_i32, [c_vp, c_vp, c_u64, c_u32, c_vp, c_vp]),
        "gpue_partition_i32_async": (c_i32, [c_vp, c_vp, c_u64, c_u32, c_vp, c_vp]),
        "gpue_partition_i64_async": (c_i32, [c_vp, c_vp, c_u64, c_u32, c_vp, c_vp]),
        "gpue_session_stream": (c_vp, [c_vp]),
        "gpue_ingest_create": (c_i32, [c_vp, c_u64, ctypes.POINTER(c_vp)]),
        "gpue_pinned_alloc": (c_i32, [c_vp, c_u64, ctypes.POINTER(c_vp)]),
        "gpue_pinned_free": (None, [c_vp]),
        "gpue_ingest_push": (c_i32, [c_vp, c_vp, c_u64, c_vp, c_u64]),
        "gpue_ingest_sync": (c_i32, [c_vp]),
        "gpue_ingest_destroy": (None, [c_vp]),
        "gpue_q43_star_agg_accum_async": (c_i32, [c_vp] * 5 + [c_vp] * 6 + [c_u64, c_vp]),
        "gpue_q3_probe_accum": (c_i32, [c_vp] + [c_vp] * 4 + [c_u64, c_vp, c_i32, c_vp]),
        "gpue_partition_xxh3_i32": (c_i32, [c_vp, c_vp, c_u64, c_u32, c_vp, c_vp]),
        "gpue_partition_crc_i32": (c_i32, [c_vp, c_vp, c_u64, c_u32, c_vp, c_vp]),
        "gpue_partition_varchar": (c_i32, [c_vp, c_vp, c_vp, c_u64, c_u32, c_vp, c_vp]),
        "gpue_gather_u32": (c_i32, [c_vp, c_vp, c_vp, c_u64, c_vp]),
        "gpue_gather_u64": (c_i32, [c_vp, c_vp, c_vp, c_u64, c_vp]),
        "gpue_partition_i64": (c_i32, [c_vp, c_vp, c_u64, c_u32, c_vp, c_vp]),
        "gpue_partition_2xi32": (c_i32, [c_vp, c_vp, c_vp, c_u64, c_u32, c_vp, c_vp]),
        "gpue_gen_lineitem_q3": (c_i32, [c_vp, c_u64, c_u64, c_u64, c_u64] + [c_vp] * 4),
        "gpue_gen_orders_q3": (c_i32, [c_vp, c_u64, c_u64, c_u32, c_vp, c_vp]),
        "gpue_gen_cust_mkt16": (c_i32, [c_vp, c_u64, c_u32, c_vp]),
        "gpue_bits_str16_eq": (c_i32, [c_vp, c_vp, c_u64, c_vp, c_vp]),
        "gpue_q3_order_bits": (c_i32, [c_vp, c_vp, c_vp, c_u64, c_vp, c_i32, c_vp]),
        "gpue_agg_table_create": (c_i32, [c_vp, c_u64, ctypes.POINTER(c_vp)]),
        "gpue_agg_table_destroy": (None, [c_vp]),
        "gpue_agg_table_reset": (c_i32, [c_vp]),
        "gpue_agg_table_size": (c_i32, [c_vp, c_vp, ctypes.POINTER(c_u64)]),
        "gpue_agg_table_ensure": (c_i32, [c_vp, c_vp, c_u64]),
        "gpue_q3_probe_agg_part": (c_i32, [c_vp] + [c_vp] * 4 + [c_u64, c_u64, c_vp,
                                           c_i32, c_vp, c_vp, c_vp, c_u32, c_vp, c_vp,
                                           c_u64, ctypes.POINTER(c_u64)]),
        "gpue_q3_probe_agg_t": (c_i32, [c_vp] + [c_vp] * 4 + [c_u64, c_vp, c_i32, c_vp,
                                       c_vp, c_vp, c_u64, ctypes.POINTER(c_u64)]),
        "gpue_q3_probe_agg": (c_i32, [c_vp] + [c_vp] * 4 + [c_u64, c_vp, c_i32, c_u64,
                                     c_vp, c_vp, c_u64, ctypes.POINTER(c_u64)]),
        "gpue_hash_agg_stats_u64": (c_i32, [c_vp, c_vp, c_vp, c_u64, c_u64] + [c_vp] * 5 +
                                    [c_u64, ctypes.POINTER(c_u64)]),
        "gpue_hash_agg_sum128_u64": (c_i32, [c_vp, c_vp, c_vp, c_u64, c_u64] + [c_vp] * 3 +
                                     [c_u64, ctypes.POINTER(c_u64)]),
        "gpue_hash_agg_sum_u64": (c_i32, [c_vp, c_vp, c_vp, c_u64, c_u64, c_vp, c_vp, c_vp,
                                          c_u64, ctypes.POINTER(c_u64)]),
        "gpue_dbuf_d2d": (c_i32, [c_vp, c_vp, c_u64, c_u64, c_u64]),
        "gpue_q3_decomp": (c_i32, [c_vp] + [c_vp] * 4 + [c_u64, c_vp, c_i32, c_i32,
                                   c_vp, c_vp]),
        "gpue_sum_prod_u32": (c_i32, [c_vp, c_vp, c_vp, c_u64, c_vp]),
        "gpue_graph_begin": (c_i32, [c_vp]),
        "gpue_graph_end": (c_i32, [c_vp, ctypes.POINTER(c_vp)]),
        "gpue_graph_launch": (c_i32, [c_vp, c_vp]),
        "gpue_graph_destroy": (None, [c_vp]),
        "gpue_join_build_varchar": (c_i32, [c_vp, c_vp, c_vp, c_u64, ctypes.POINTER(c_vp)]),
        "gpue_join_probe_emit_varchar": (c_i32, [c_vp, c_vp, c_vp, c_vp, c_u64, c_vp, c_vp,
                                                 ctypes.POINTER(c_u64)]),
        "gpue_join_probe_emit_varchar_mode": (c_i32, [c_vp, c_vp, c_vp, c_vp, c_u64, c_i32,
                                                      c_vp, c_vp, ctypes.POINTER(c_u64)]),
        "gpue_join_build_varchar_nulls": (c_i32, [c_vp, c_vp, c_vp, c_vp, c_u64,
                                                  ctypes.POINTER(c_vp)]),
        "gpue_join_probe_emit_varchar_nulls": (c_i32, [c_vp, c_vp, c_vp, c_vp, c_vp, c_u64,
                                                       c_i32, c_vp, c_vp,
                                                       ctypes.POINTER(c_u64)]),
        "gpue_dict_decode_binary": (c_i32, [c_vp, c_vp, c_vp, c_vp, c_u64, c_vp, c_vp,
                                            ctypes.POINTER(c_u64)]),
        "gpue_hash_agg_push_u64": (c_i32, [c_vp, c_vp, c_vp, c_vp, c_vp, c_u64, c_i32,
                                           c_vp, ctypes.POINTER(c_u64)]),
        "gpue_hash_agg_probe_hits_u64": (c_i32, [c_vp, c_vp, c_vp, c_u64,
                                                 ctypes.POINTER(c_u64)]),
        "gpue_hash_agg_emit_u64": (c_i32, [c_vp, c_vp, c_vp, c_vp, c_vp, c_u64,
                                           ctypes.POINTER(c_u64)]),
        "gpue_sbf_build_i32": (c_i32, [c_vp, c_vp, c_u64, c_i32, c_vp]),
        "gpue_sbf_test_i32": (c_i32, [c_vp, c_vp, c_u64, c_vp, c_i32, c_vp]),
        "gpue_topk_i64": (c_i32, [c_vp, c_vp, c_vp, c_u64, c_i32, c_vp, c_vp]),
        "gpue_page_decode_rle_i32": (c_i32, [c_vp, c_vp, c_u64, c_vp]),
        "gpue_page_decode_plain_i32": (c_i32, [c_vp, c_vp, c_u64, c_vp]),
        "gpue_page_decode_rle_bool": (c_i32, [c_vp, c_vp, c_u64, c_vp]),
        "gpue_page_decode_for_i32": (c_i32, [c_vp, c_vp, c_u64, c_vp]),
        "gpue_page_decode_binary_plain": (c_i32, [c_vp, c_vp, c_u64, c_vp, c_vp]),
        "gpue_page_decode_binary_prefix": (c_i32, [c_vp, c_vp, c_u64, c_vp, c_vp]),
        "gpue_ubench_bitgather": (c_i32, [c_vp, c_vp, c_u64, c_vp, c_u64, c_i32,
                                          ctypes.POINTER(ctypes.c_float)]),
        "gpue_page_decode_bshuf_lz4_i32": (c_i32, [c_vp, c_vp, c_u32, c_vp]),
        "gpue_asof_build_i32": (c_i32, [c_vp, c_vp, c_vp, c_u64, c_i32, c_vp]),
        "gpue_asof_build_nulls_i32": (c_i32, [c_vp, c_vp, c_vp, c_vp, c_u64, c_i32,
                                              c_vp]),
        "gpue_asof_probe_emit_nulls_i32": (c_i32, [c_vp, c_vp, c_vp, c_vp, c_vp, c_u64,
                                                   c_i32, c_vp, c_vp,
                                                   ctypes.POINTER(c_u64)]),
        "gpue_asof_probe_emit_i32": (c_i32, [c_vp, c_vp, c_vp, c_vp, c_u64, c_i32,
                                             c_vp, c_vp, ctypes.POINTER(c_u64)]),
        "gpue_asof_table_destroy": (c_i32, [c_vp]),
        "gpue_timer_start": (c_i32, [c_vp]),
        "gpue_timer_stop": (c_i32, [c_vp, ctypes.POINTER(ctypes.c_float)]),
    }
    for name, (res, args) in sigs.items():
        fn = getattr(lib, name)
        fn.restype = res
        fn.argtypes = args


def _ck(lib, status):
    if status != 0:
        raise GpueError(f"gpue status {status}: {lib.gpue_last_error().decode()}")


class DBuf:
    """Device HBM buffer (columnar container analog)."""

    def __init__(self, eng: "Engine", nbytes: int):
        self._lib = eng._lib
        self.nbytes = nbytes
        h = c_vp()
        _ck(self._lib, self._lib.gpue_dbuf_alloc(eng._h, nbytes, ctypes.byref(h)))
        self._h = h

    def h2d(self, arr: np.ndarray, dst_off: int = 0):
        arr = np.ascontiguousarray(arr)
        _ck(self._lib, self._lib.gpue_dbuf_h2d(self._h, arr.ctypes.data_as(c_vp),
                                               arr.nbytes, dst_off))
        return self

    def d2h(self, dtype, count: int, src_off: int = 0) -> np.ndarray:
        out = np.empty(count, dtype=dtype)
        _ck(self._lib, self._lib.gpue_dbuf_d2h(self._h, out.ctypes.data_as(c_vp),
                                               out.nbytes, src_off))
        return out

    def free(self):
        if self._h:
            self._lib.gpue_dbuf_free(self._h)
            self._h = None


class JoinTable:
    def __init__(self, eng: "Engine", handle):
        self._lib = eng._lib
        self._h = handle

    @property
    def minmax(self):
        mn, mx = c_i64(), c_i64()
        _ck(self._lib, self._lib.gpue_join_table_minmax(self._h, ctypes.byref(mn), ctypes.byref(mx)))
        return mn.value, mx.value

    def first_d2h(self, n_entries: int) -> np.ndarray:
        out = np.empty(n_entries, dtype=np.uint32)
        _ck(self._lib, self._lib.gpue_join_table_first_d2h(self._h, out.ctypes.data_as(c_vp), n_entries))
        return out

    def destroy(self):
        if self._h:
            self._lib.gpue_join_table_destroy(self._h)
            self._h = None


class AsofTable:
    """ASOF join table (LinearChainedAsofJoinHashMap, join_hash_map_method.h:
    201-217): equi-key slots + per-key temporal index sorted per opcode."""

    def __init__(self, eng: "Engine", handle):
        self._lib = eng._lib
        self._h = handle

    def destroy(self):
        if self._h:
            self._lib.gpue_asof_table_destroy(self._h)
            self._h = None


class Engine:
    """One session == one device + one HIP stream (one pipeline driver)."""

    def __init__(self, device: int = 0):
        self._lib = _load()
        h = c_vp()
        _ck(self._lib, self._lib.gpue_session_create(device, ctypes.byref(h)))
        self._h = h
        self.device = device

    @staticmethod
    def device_count() -> int:
        lib = _load()
        n = c_i32()
        lib.gpue_device_count(ctypes.byref(n))
        return n.value

    def close(self):
        if self._h:
            self._lib.gpue_session_destroy(self._h)
            self._h = None

    def sync(self):
        _ck(self._lib, self._lib.gpue_sync(self._h))

    def alloc(self, nbytes: int) -> DBuf:
        return DBuf(self, nbytes)

    # ---- synthetic chunk source (scan replacement) ----
    def gen_u32_mod(self, out: DBuf, seed, tag, row_start, n, mod, add=0):
        _ck(self._lib, self._lib.gpue_gen_u32_mod(self._h, out._h, seed, tag, row_start, n, mod, add))

    def gen_i64(self, out: DBuf, seed, tag, row_start, n):
        _ck(self._lib, self._lib.gpue_gen_i64(self._h, out._h, seed, tag, row_start, n))

    def gen_lineorder_q1(self, seed, row_start, n, od: DBuf, ep: DBuf, dc: DBuf):
        _ck(self._lib, self._lib.gpue_gen_lineorder_q1(self._h, seed, row_start, n, od._h, ep._h, dc._h))

    def gen_lineorder_q21(self, seed, row_start, n, pk, sk, od, rv):
        _ck(self._lib, self._lib.gpue_gen_lineorder_q21(self._h, seed, row_start, n,
                                                        pk._h, sk._h, od._h, rv._h))

    def gen_lineorder_q43(self, seed, row_start, n, ck, sk, pk, od, rv, sc):
        _ck(self._lib, self._lib.gpue_gen_lineorder_q43(self._h, seed, row_start, n,
                                                        ck._h, sk._h, pk._h, od._h,
                                                        rv._h, sc._h))

    # ---- operators ----
    def scan_filter_i64_lt(self, inp: DBuf, n, theta, out: DBuf) -> int:
        cnt = c_u64()
        _ck(self._lib, self._lib.gpue_scan_filter_i64_lt(self._h, inp._h, n, theta, out._h,
                                                         ctypes.byref(cnt)))
        return cnt.value

    def scan_filter_i64_lt_sp(self, inp: DBuf, n, theta, out: DBuf) -> int:
        cnt = c_u64()
        _ck(self._lib, self._lib.gpue_scan_filter_i64_lt_sp(self._h, inp._h, n, theta,
                                                            out._h, ctypes.byref(cnt)))
        return cnt.value

    def eval_conjuncts(self, cols, n_rows, preds) -> int:
        """preds: list of (col_index, op, lo, hi); op 0 EQ / 1 LT / 2 BETWEEN.
        Compacts cols stably in place; returns surviving rows."""
        arr = (c_vp * len(cols))(*[c._h for c in cols])
        pc = np.array([p[0] for p in preds], np.int32)
        po = np.array([p[1] for p in preds], np.int32)
        pl = np.array([p[2] for p in preds], np.int32)
        ph = np.array([p[3] for p in preds], np.int32)
        out = c_u64()
        _ck(self._lib, self._lib.gpue_eval_conjuncts_i32(
            self._h, arr, len(cols), n_rows, pc.ctypes.data_as(c_vp),
            po.ctypes.data_as(c_vp), pl.ctypes.data_as(c_vp), ph.ctypes.data_as(c_vp),
            len(preds), ctypes.byref(out)))
        return out.value

    def eval_conjuncts_i64(self, cols, n_rows, preds) -> int:
        arr = (c_vp * len(cols))(*[c._h for c in cols])
        pc = np.array([p[0] for p in preds], np.int32)
        po = np.array([p[1] for p in preds], np.int32)
        pl = np.array([p[2] for p in preds], np.int64)
        ph = np.array([p[3] for p in preds], np.int64)
        out = c_u64()
        _ck(self._lib, self._lib.gpue_eval_conjuncts_i64(
            self._h, arr, len(cols), n_rows, pc.ctypes.data_as(c_vp),
            po.ctypes.data_as(c_vp), pl.ctypes.data_as(c_vp), ph.ctypes.data_as(c_vp),
            len(preds), ctypes.byref(out)))
        return out.value

    # --- JoinHashMapSelector restatement (join_hash_table.cpp:164-344) ---
    JM_NAMES = {0: "DIRECT", 1: "RANGE_DIRECT", 2: "RANGE_DIRECT_SET",
                3: "DENSE_RANGE_DIRECT", 4: "LINEAR_CHAINED",
                5: "LINEAR_CHAINED_SET", 6: "BUCKET_CHAINED"}
    KEYCON_NAMES = {0: "ONE_KEY", 1: "ONE_KEY_VARCHAR", 2: "FIXED_INT",
                    3: "FIXED_BIGINT", 4: "FIXED_LARGEINT", 5: "SERIALIZED_VARCHAR"}

    def join_select_key_constructor(self, fixed_sizes, null_safe=None,
                                    enable_fixed_size_string=1):
        """_determine_key_constructor (:164-229). fixed_sizes: bytes per key
        column (varchar: its max_size when fixable, else 0). Returns
        (constructor, packed_bytes)."""
        n = len(fixed_sizes)
        fs = (c_i32 * n)(*fixed_sizes)
        ns = (ctypes.c_uint8 * n)(*(null_safe or [0] * n))
        pb = c_i32()
        kc = self._lib.gpue_join_select_key_constructor(
            n, fs, ns, enable_fixed_size_string, ctypes.byref(pb))
        return kc, pb.value

    def join_select_varchar_constructor(self, max_size, enable_fixed_size_string=1):
        return self._lib.gpue_join_select_varchar_constructor(
            max_size, enable_fixed_size_string)

    def join_select_method(self, key_constructor, lt_class, row_count, min_value,
                           max_value, mode=0, with_other_conjunct=0,
                           enable_range_direct=1, enable_linear_chained=1,
                           l2_size=4 << 20, l3_size=256 << 20):
        """_determine_hash_map_method (:231-344). Pure decision — returns a
        GPUE_JM_* value (JM_NAMES). Cache-size defaults are the MI355X
        residency tiers (4 MiB XCD L2 / 256 MiB Infinity Cache), matching
        gpue_join_build_auto_i32; the reference reads CpuInfo here."""
        return self._lib.gpue_join_select_method(
            key_constructor, lt_class, row_count, min_value, max_value, mode,
            with_other_conjunct, enable_range_direct, enable_linear_chained,
            l2_size, l3_size)

    def join_build_auto(self, keys: DBuf, row_count, mode=0, with_other_conjunct=0,
                        l2_size=0, l3_size=0):
        """Selector-driven build (JoinHashTable::build :350-380): min/max on
        device -> reference decision -> matching GPU table. Returns
        (JoinTable, chosen GPUE_JM_* method)."""
        h = c_vp()
        m = c_i32()
        _ck(self._lib, self._lib.gpue_join_build_auto_i32(
            self._h, keys._h, row_count, mode, with_other_conjunct, l2_size, l3_size,
            ctypes.byref(h), ctypes.byref(m)))
        return JoinTable(self, h), m.value

    def join_build_auto_u64(self, keys: DBuf, row_count, mode=0, with_other_conjunct=0,
                            l2_size=0, l3_size=0):
        """Selector-driven build for 8-byte keys (decision reported; every
        physical tier maps onto the u64 bucket-chained table)."""
        h = c_vp()
        m = c_i32()
        _ck(self._lib, self._lib.gpue_join_build_auto_u64(
            self._h, keys._h, row_count, mode, with_other_conjunct, l2_size, l3_size,
            ctypes.byref(h), ctypes.byref(m)))
        return JoinTable(self, h), m.value

    def join_build_payload(self, keys: DBuf, payloads: DBuf, n_rows) -> JoinTable:
        h = c_vp()
        _ck(self._lib, self._lib.gpue_join_build_payload_i32(self._h, keys._h, payloads._h,
                                                             n_rows, ctypes.byref(h)))
        return JoinTable(self, h)

    def join_build_range_direct(self, keys: DBuf, row_count) -> JoinTable:
        h = c_vp()
        _ck(self._lib, self._lib.gpue_join_build_range_direct_i32(self._h, keys._h, row_count,
                                                                  ctypes.byref(h)))
        return JoinTable(self, h)

    def join_build_dense_range_direct(self, keys: DBuf, row_count) -> JoinTable:
        """DENSE_RANGE_DIRECT_MAPPING (rank/select-compressed direct map,
        join_hash_map_method.h:378)."""
        h = c_vp()
        _ck(self._lib, self._lib.gpue_join_build_dense_range_direct_i32(
            self._h, keys._h, row_count, ctypes.byref(h)))
        return JoinTable(self, h)

    def join_build_bucket_chained(self, keys: DBuf, row_count) -> JoinTable:
        h = c_vp()
        _ck(self._lib, self._lib.gpue_join_build_bucket_chained_u32(
            self._h, keys._h, row_count, ctypes.byref(h)))
        return JoinTable(self, h)

    def join_build_bucket_chained_u128(self, keys: DBuf, row_count) -> JoinTable:
        """16-byte key (LARGEINT / packed ≤16 B) bucket-chained build
        (JoinKeyHash<T,16>: crc32 over the key bytes)."""
        h = c_vp()
        _ck(self._lib, self._lib.gpue_join_build_bucket_chained_u128(
            self._h, keys._h, row_count, ctypes.byref(h)))
        return JoinTable(self, h)

    def join_probe_emit_mode_u128(self, table, probe_keys, n_rows, mode,
                                  out_probe=None, out_build=None) -> int:
        cnt = c_u64()
        op = out_probe._h if out_probe else None
        ob = out_build._h if out_build else None
        _ck(self._lib, self._lib.gpue_join_probe_emit_mode_u128(
            self._h, table._h, probe_keys._h, n_rows, mode, op, ob, ctypes.byref(cnt)))
        return cnt.value

    def join_build_bucket_chained_u64(self, keys: DBuf, row_count) -> JoinTable:
        h = c_vp()
        _ck(self._lib, self._lib.gpue_join_build_bucket_chained_u64(
            self._h, keys._h, row_count, ctypes.byref(h)))
        return JoinTable(self, h)

    def join_probe_emit_u64(self, table, probe_keys, n_rows, mode=0,
                            out_probe=None, out_build=None) -> int:
        cnt = c_u64()
        op = out_probe._h if out_probe else None
        ob = out_build._h if out_build else None
        _ck(self._lib, self._lib.gpue_join_probe_emit_mode_u64(
            self._h, table._h, probe_keys._h, n_rows, mode, op, ob, ctypes.byref(cnt)))
        return cnt.value

    def join_build_linear_chained(self, keys: DBuf, row_count) -> JoinTable:
        h = c_vp()
        _ck(self._lib, self._lib.gpue_join_build_linear_chained_u32(
            self._h, keys._h, row_count, ctypes.byref(h)))
        return JoinTable(self, h)

    def join_probe_emit(self, table: JoinTable, probe_keys: DBuf, n_rows,
                        out_probe: DBuf = None, out_build: DBuf = None) -> int:
        cnt = c_u64()
        op = out_probe._h if out_probe else None
        ob = out_build._h if out_build else None
        _ck(self._lib, self._lib.gpue_join_probe_emit_i32(self._h, table._h, probe_keys._h,
                                                          n_rows, op, ob, ctypes.byref(cnt)))
        return cnt.value

    def join_probe_emit_mode(self, table, probe_keys, n_rows, mode,
                             out_probe=None, out_build=None) -> int:
        cnt = c_u64()
        op = out_probe._h if out_probe else None
        ob = out_build._h if out_build else None
        _ck(self._lib, self._lib.gpue_join_probe_emit_mode_i32(
            self._h, table._h, probe_keys._h, n_rows, mode, op, ob, ctypes.byref(cnt)))
        return cnt.value

    # --- ASOF join (join_hash_map_method.h:201-217 + AsofIndex,
    # join_hash_table_descriptor.h:59-104 / .cpp:70-134) ---
    ASOF_NAMES = {0: "LT", 1: "LE", 2: "GT", 3: "GE"}

    def asof_build(self, keys: DBuf, asof: DBuf, row_count, opcode) -> AsofTable:
        """keys: (row_count+1) int32 equi keys, asof: (row_count+1) int64
        temporal values, both 1-based (row 0 sentinel). opcode: 0 LT / 1 LE /
        2 GT / 3 GE (the probe-vs-build temporal condition)."""
        h = c_vp()
        _ck(self._lib, self._lib.gpue_asof_build_i32(
            self._h, keys._h, asof._h, row_count, opcode, ctypes.byref(h)))
        return AsofTable(self, h)

    def asof_probe_emit(self, table: AsofTable, probe_keys: DBuf, probe_asof: DBuf,
                        n_rows, mode=0, out_probe=None, out_build=None) -> int:
        """Two-call contract like join_probe_emit: without outputs, returns the
        match count; with outputs, fills (probe row, 1-based build row) pairs
        ordered by probe row. mode 0 = ASOF inner, 3 = ASOF left outer (miss
        emits build row 0)."""
        cnt = c_u64()
        op = out_probe._h if out_probe else None
        ob = out_build._h if out_build else None
        _ck(self._lib, self._lib.gpue_asof_probe_emit_i32(
            self._h, table._h, probe_keys._h, probe_asof._h, n_rows, mode, op, ob,
            ctypes.byref(cnt)))
        return cnt.value

    def asof_build_nulls(self, keys: DBuf, asof: DBuf, nulls: DBuf, row_count,
                         opcode) -> AsofTable:
        """Nullable build: nulls is the (row_count+1) u8 OR of the equi-key
        and temporal null masks; flagged rows are skipped
        (join_hash_table_descriptor.h:447-456)."""
        h = c_vp()
        _ck(self._lib, self._lib.gpue_asof_build_nulls_i32(
            self._h, keys._h, asof._h, nulls._h, row_count, opcode, ctypes.byref(h)))
        return AsofTable(self, h)

    def asof_probe_emit_nulls(self, table: AsofTable, probe_keys: DBuf,
                              probe_asof: DBuf, probe_nulls: DBuf, n_rows, mode=0,
                              out_probe=None, out_build=None) -> int:
        cnt = c_u64()
        op = out_probe._h if out_probe else None
        ob = out_build._h if out_build else None
        _ck(self._lib, self._lib.gpue_asof_probe_emit_nulls_i32(
            self._h, table._h, probe_keys._h, probe_asof._h, probe_nulls._h, n_rows,
            mode, op, ob, ctypes.byref(cnt)))
        return cnt.value

    def join_build_bucket_chained_nulls(self, keys: DBuf, nulls: DBuf, row_count):
        h = c_vp()
        _ck(self._lib, self._lib.gpue_join_build_bucket_chained_nulls_u32(
            self._h, keys._h, nulls._h, row_count, ctypes.byref(h)))
        return JoinTable(self, h)

    def join_probe_emit_nulls(self, table, probe_keys, probe_nulls, n_rows, mode,
                              out_probe=None, out_build=None) -> int:
        cnt = c_u64()
        op = out_probe._h if out_probe else None
        ob = out_build._h if out_build else None
        _ck(self._lib, self._lib.gpue_join_probe_emit_nulls_i32(
            self._h, table._h, probe_keys._h, probe_nulls._h, n_rows, mode, op, ob,
            ctypes.byref(cnt)))
        return cnt.value

    def join_build_varchar(self, bytes_: DBuf, offsets: DBuf, row_count) -> JoinTable:
        h = c_vp()
        _ck(self._lib, self._lib.gpue_join_build_varchar(
            self._h, bytes_._h, offsets._h, row_count, ctypes.byref(h)))
        return JoinTable(self, h)

    def join_probe_emit_varchar(self, table, pbytes: DBuf, poffsets: DBuf, n_rows,
                                out_probe=None, out_build=None) -> int:
        cnt = c_u64()
        op = out_probe._h if out_probe else None
        ob = out_build._h if out_build else None
        _ck(self._lib, self._lib.gpue_join_probe_emit_varchar(
            self._h, table._h, pbytes._h, poffsets._h, n_rows, op, ob, ctypes.byref(cnt)))
        return cnt.value

    def join_build_varchar_nulls(self, bytes_: DBuf, offsets: DBuf, nulls: DBuf,
                                 row_count) -> JoinTable:
        h = c_vp()
        _ck(self._lib, self._lib.gpue_join_build_varchar_nulls(
            self._h, bytes_._h, offsets._h, nulls._h, row_count, ctypes.byref(h)))
        return JoinTable(self, h)

    def join_probe_emit_varchar_nulls(self, table, pbytes, poffsets, pnulls, n_rows, mode,
                                      out_probe=None, out_build=None) -> int:
        cnt = c_u64()
        op = out_probe._h if out_probe else None
        ob = out_build._h if out_build else None
        _ck(self._lib, self._lib.gpue_join_probe_emit_varchar_nulls(
            self._h, table._h, pbytes._h, poffsets._h, pnulls._h, n_rows, mode, op, ob,
            ctypes.byref(cnt)))
        return cnt.value

    def join_probe_emit_varchar_mode(self, table, pbytes: DBuf, poffsets: DBuf, n_rows,
                                     mode, out_probe=None, out_build=None) -> int:
        cnt = c_u64()
        op = out_probe._h if out_probe else None
        ob = out_build._h if out_build else None
        _ck(self._lib, self._lib.gpue_join_probe_emit_varchar_mode(
            self._h, table._h, pbytes._h, poffsets._h, n_rows, mode, op, ob,
            ctypes.byref(cnt)))
        return cnt.value

    def pack_keys_2xi64(self, a: DBuf, b: DBuf, n, out: DBuf):
        """Two i64 key columns packed into 16-byte keys
        (SERIALIZED_FIXED_SIZE_LARGEINT)."""
        _ck(self._lib, self._lib.gpue_pack_keys_2xi64(self._h, a._h, b._h, n, out._h))

    def pack_keys_2xi32(self, a: DBuf, b: DBuf, n, out: DBuf):
        _ck(self._lib, self._lib.gpue_pack_keys_2xi32(self._h, a._h, b._h, n, out._h))

    def join_probe_right_varchar(self, table, pbytes, poffsets, n_rows, anti,
                                 out_build=None) -> int:
        cnt = c_u64()
        ob = out_build._h if out_build else None
        _ck(self._lib, self._lib.gpue_join_probe_right_varchar(
            self._h, table._h, pbytes._h, poffsets._h, n_rows, anti, ob, ctypes.byref(cnt)))
        return cnt.value

    def join_probe_right(self, table, probe_keys, n_rows, anti, out_build=None) -> int:
        cnt = c_u64()
        ob = out_build._h if out_build else None
        _ck(self._lib, self._lib.gpue_join_probe_right_i32(
            self._h, table._h, probe_keys._h, n_rows, anti, ob, ctypes.byref(cnt)))
        return cnt.value

    def q1_join_sum(self, dates: JoinTable, od: DBuf, ep: DBuf, dc: DBuf, n):
        s, c = c_i64(), c_u64()
        _ck(self._lib, self._lib.gpue_q1_join_sum(self._h, dates._h, od._h, ep._h, dc._h, n,
                                                  ctypes.byref(s), ctypes.byref(c)))
        return s.value, c.value

    def dbuf_ptr(self, b: DBuf) -> int:
        p = c_vp()
        _ck(self._lib, self._lib.gpue_dbuf_ptr(b._h, ctypes.byref(p)))
        return p.value

    def wrap_ptr_offset(self, b: DBuf, offset: int, nbytes: int) -> DBuf:
        """Sub-buffer view (per-channel slice of a partitioned column)."""
        return self.wrap_ptr(self.dbuf_ptr(b) + offset, nbytes)

    def wrap_ptr(self, device_ptr: int, nbytes: int) -> DBuf:
        """Wrap external device memory (e.g. torch_tensor.data_ptr())."""
        b = DBuf.__new__(DBuf)
        b._lib = self._lib
        b.nbytes = nbytes
        h = c_vp()
        _ck(self._lib, self._lib.gpue_dbuf_wrap(self._h, c_vp(device_ptr), nbytes,
                                                ctypes.byref(h)))
        b._h = h
        return b

    def q43_star_agg_async(self, custs, supps, parts, dates, ck, sk, pk, od, rv, sc,
                           n, group_sums: DBuf):
        _ck(self._lib, self._lib.gpue_q43_star_agg_async(
            self._h, custs._h, supps._h, parts._h, dates._h, ck._h, sk._h, pk._h,
            od._h, rv._h, sc._h, n, group_sums._h))

    def q1_join_sum_accum(self, dates: JoinTable, od: DBuf, ep: DBuf, dc: DBuf, n, acc: DBuf):
        _ck(self._lib, self._lib.gpue_q1_join_sum_accum(self._h, dates._h, od._h, ep._h,
                                                        dc._h, n, acc._h))

    def q1_join_sum_async(self, dates: JoinTable, od: DBuf, ep: DBuf, dc: DBuf, n, acc: DBuf):
        _ck(self._lib, self._lib.gpue_q1_join_sum_async(self._h, dates._h, od._h, ep._h,
                                                        dc._h, n, acc._h))

    def q21_star_agg_pipe(self, parts, supps, dates, pk, sk, od, rv, n,
                          brand_scratch: DBuf, group_sums: DBuf, n_chunks=8):
        _ck(self._lib, self._lib.gpue_q21_star_agg_pipe(
            self._h, parts._h, supps._h, dates._h, pk._h, sk._h, od._h, rv._h, n,
            brand_scratch._h, group_sums._h, n_chunks))

    def q21_star_agg_async(self, parts, supps, dates, pk, sk, od, rv, n, group_sums: DBuf):
        _ck(self._lib, self._lib.gpue_q21_star_agg_async(self._h, parts._h, supps._h, dates._h,
                                                         pk._h, sk._h, od._h, rv._h, n,
                                                         group_sums._h))

    def q21_star_agg(self, parts: JoinTable, supps: JoinTable, dates: JoinTable,
                     pk: DBuf, sk: DBuf, od: DBuf, rv: DBuf, n) -> np.ndarray:
        out = np.zeros(7000, dtype=np.int64)
        _ck(self._lib, self._lib.gpue_q21_star_agg(self._h, parts._h, supps._h, dates._h,
                                                   pk._h, sk._h, od._h, rv._h, n,
                                                   out.ctypes.data_as(c_vp)))
        return out

    def gen_lineitem_q3(self, seed, row_start, n, n_orders, lk, ext, disc, ship):
        _ck(self._lib, self._lib.gpue_gen_lineitem_q3(self._h, seed, row_start, n, n_orders,
                                                      lk._h, ext._h, disc._h, ship._h))

    def gen_orders_q3(self, seed, n_orders, n_custs, ocust, odate):
        _ck(self._lib, self._lib.gpue_gen_orders_q3(self._h, seed, n_orders, n_custs,
                                                    ocust._h, odate._h))

    def gen_cust_mkt16(self, seed, n_custs, out):
        _ck(self._lib, self._lib.gpue_gen_cust_mkt16(self._h, seed, n_custs, out._h))

    def bits_str16_eq(self, col16, n, lit16: bytes, bits):
        assert len(lit16) == 16
        _ck(self._lib, self._lib.gpue_bits_str16_eq(self._h, col16._h, n, lit16, bits._h))

    def q3_order_bits(self, ocust, odate, n_orders, cust_bits, cutoff, order_bits):
        _ck(self._lib, self._lib.gpue_q3_order_bits(self._h, ocust._h, odate._h, n_orders,
                                                    cust_bits._h, cutoff, order_bits._h))

    def agg_table_create(self, capacity):
        h = c_vp()
        _ck(self._lib, self._lib.gpue_agg_table_create(self._h, capacity, ctypes.byref(h)))
        return h

    def agg_table_reset(self, h):
        _ck(self._lib, self._lib.gpue_agg_table_reset(h))

    def agg_table_size(self, h) -> int:
        """Claimed group count (device counter over the claim sites)."""
        g = c_u64()
        _ck(self._lib, self._lib.gpue_agg_table_size(self._h, h, ctypes.byref(g)))
        return g.value

    def agg_table_ensure(self, h, additional_rows):
        """Grow + rehash so a push of additional_rows can never overflow —
        the reference's try_convert_to_two_level_map check before each chunk
        (aggregator.cpp:1237-1241)."""
        _ck(self._lib, self._lib.gpue_agg_table_ensure(self._h, h, additional_rows))

    def agg_table_destroy(self, h):
        self._lib.gpue_agg_table_destroy(h)

    def q3_probe_agg_t(self, lk, ext, disc, ship, n, order_bits, ship_cutoff, at,
                       out_keys, out_sums, max_out):
        g = c_u64()
        _ck(self._lib, self._lib.gpue_q3_probe_agg_t(
            self._h, lk._h, ext._h, disc._h, ship._h, n, order_bits._h, ship_cutoff,
            at, out_keys._h, out_sums._h, max_out, ctypes.byref(g)))
        return g.value

    def q3_probe_agg(self, lk, ext, disc, ship, n, order_bits, ship_cutoff,
                     out_keys, out_sums, max_out, capacity_hint=0):
        g = c_u64()
        _ck(self._lib, self._lib.gpue_q3_probe_agg(
            self._h, lk._h, ext._h, disc._h, ship._h, n, order_bits._h, ship_cutoff,
            capacity_hint, out_keys._h, out_sums._h, max_out, ctypes.byref(g)))
        return g.value

    def q3_decomp(self, lk, ext, disc, ship, n, order_bits, ship_cutoff, legs, at, sink):
        _ck(self._lib, self._lib.gpue_q3_decomp(
            self._h, lk._h, ext._h, disc._h, ship._h, n, order_bits._h, ship_cutoff, legs,
            at, sink._h))

    def q3_probe_agg_part(self, lk, ext, disc, ship, n, n_orders, order_bits,
                          ship_cutoff, at, keys_scratch, vals_scratch, nparts,
                          out_keys, out_sums, max_out) -> int:
        g = c_u64()
        _ck(self._lib, self._lib.gpue_q3_probe_agg_part(
            self._h, lk._h, ext._h, disc._h, ship._h, n, n_orders, order_bits._h,
            ship_cutoff, at, keys_scratch._h, vals_scratch._h, nparts, out_keys._h,
            out_sums._h, max_out, ctypes.byref(g)))
        return g.value

    def hash_agg_sum_u64(self, keys: DBuf, vals: DBuf, n, out_keys: DBuf, out_sums: DBuf,
                         out_counts: DBuf = None, max_out=0, capacity_hint=0):
        g = c_u64()
        oc = out_counts._h if out_counts else None
        _ck(self._lib, self._lib.gpue_hash_agg_sum_u64(
            self._h, keys._h, vals._h, n, capacity_hint, out_keys._h, out_sums._h, oc,
            max_out or out_keys.nbytes // 8, ctypes.byref(g)))
        return g.value

    def hash_agg_stats_u64(self, keys, vals, n, out_keys, out_sums, out_counts,
                           out_mins, out_maxs, max_out, capacity_hint=0):
        g = c_u64()
        _ck(self._lib, self._lib.gpue_hash_agg_stats_u64(
            self._h, keys._h, vals._h, n, capacity_hint, out_keys._h, out_sums._h,
            out_counts._h, out_mins._h, out_maxs._h, max_out, ctypes.byref(g)))
        return g.value

    def hash_agg_sum128_u64(self, keys, vals, n, out_keys, out_lo, out_hi, max_out,
                            capacity_hint=0):
        g = c_u64()
        _ck(self._lib, self._lib.gpue_hash_agg_sum128_u64(
            self._h, keys._h, vals._h, n, capacity_hint, out_keys._h, out_lo._h,
            out_hi._h, max_out, ctypes.byref(g)))
        return g.value

    def graph_capture(self, fn):
        """Capture fn()'s async launches as a hipGraph; returns the exec
        handle for graph_launch. fn must only enqueue stream-async work."""
        _ck(self._lib, self._lib.gpue_graph_begin(self._h))
        try:
            fn()
        except BaseException:
            h = c_vp()
            self._lib.gpue_graph_end(self._h, ctypes.byref(h))  # abort capture
            if h:
                self._lib.gpue_graph_destroy(h)
            raise
        h = c_vp()
        _ck(self._lib, self._lib.gpue_graph_end(self._h, ctypes.byref(h)))
        return h

    def graph_launch(self, g):
        _ck(self._lib, self._lib.gpue_graph_launch(self._h, g))

    def graph_destroy(self, g):
        self._lib.gpue_graph_destroy(g)

    def dbuf_d2d(self, src: DBuf, dst: DBuf, nbytes, src_off=0, dst_off=0):
        _ck(self._lib, self._lib.gpue_dbuf_d2d(src._h, dst._h, nbytes, src_off, dst_off))

    def sum_prod_u32(self, a: DBuf, b: DBuf, n, acc: DBuf):
        _ck(self._lib, self._lib.gpue_sum_prod_u32(self._h, a._h, b._h, n, acc._h))

    def gather_u32(self, inp: DBuf, idx: DBuf, n, out: DBuf):
        _ck(self._lib, self._lib.gpue_gather_u32(self._h, inp._h, idx._h, n, out._h))

    def gather_u64(self, inp: DBuf, idx: DBuf, n, out: DBuf):
        _ck(self._lib, self._lib.gpue_gather_u64(self._h, inp._h, idx._h, n, out._h))

    def partition_i64(self, keys: DBuf, n, num_channels, row_indexes: DBuf) -> np.ndarray:
        sp = np.zeros(num_channels + 1, dtype=np.uint64)
        _ck(self._lib, self._lib.gpue_partition_i64(self._h, keys._h, n, num_channels,
                                                    sp.ctypes.data_as(c_vp), row_indexes._h))
        return sp

    def partition(self, keys: DBuf, n, num_channels, row_indexes: DBuf) -> np.ndarray:
        sp = np.zeros(num_channels + 1, dtype=np.uint64)
        _ck(self._lib, self._lib.gpue_partition_i32(self._h, keys._h, n, num_channels,
                                                    sp.ctypes.data_as(c_vp), row_indexes._h))
        return sp

    def stream_ptr(self) -> int:
        """The session's HIP stream (for torch.cuda.ExternalStream wrapping —
        event-orders RCCL collectives against engine kernels)."""
        return int(self._lib.gpue_session_stream(self._h) or 0)

    def q43_star_agg_accum_async(self, custs, supps, parts, dates, ck, sk, pk, od,
                                 rv, sc, n, group_sums: DBuf):
        _ck(self._lib, self._lib.gpue_q43_star_agg_accum_async(
            self._h, custs._h, supps._h, parts._h, dates._h, ck._h, sk._h, pk._h,
            od._h, rv._h, sc._h, n, group_sums._h))

    def q3_probe_accum(self, lk, ext, disc, ship, n, order_bits, ship_cutoff, at):
        _ck(self._lib, self._lib.gpue_q3_probe_accum(
            self._h, lk._h, ext._h, disc._h, ship._h, n, order_bits._h,
            ship_cutoff, at))

    def pinned_alloc(self, nbytes) -> int:
        """Page-locked host allocation (returns raw host pointer)."""
        h = c_vp()
        _ck(self._lib, self._lib.gpue_pinned_alloc(self._h, nbytes, ctypes.byref(h)))
        return h.value

    def pinned_free(self, host_ptr):
        self._lib.gpue_pinned_free(c_vp(host_ptr))

    def ingest_create(self, chunk_bytes=32 << 20):
        """Pinned double-buffered H2D ingest (scan_operator.h:40 morsel
        shape); push host arrays with ingest_push, then ingest_sync."""
        h = c_vp()
        _ck(self._lib, self._lib.gpue_ingest_create(self._h, chunk_bytes,
                                                    ctypes.byref(h)))
        return h

    def ingest_push(self, ing, arr: np.ndarray, dst: DBuf, dst_off=0):
        _ck(self._lib, self._lib.gpue_ingest_push(
            ing, arr.ctypes.data_as(c_vp), arr.nbytes, dst._h, dst_off))

    def ingest_sync(self, ing):
        _ck(self._lib, self._lib.gpue_ingest_sync(ing))

    def ingest_destroy(self, ing):
        self._lib.gpue_ingest_destroy(ing)

    def partition_scratch_bytes(self, n, num_channels):
        """Scratch size for the async partition forms (per-(block,channel)
        hist + offsets)."""
        nb = min((n + 255) // 256, 4096)  # conservative upper bound on grid
        return int(nb * num_channels * 12 + 4096)

    def partition_async(self, keys: DBuf, n, num_channels, row_indexes: DBuf,
                        scratch: DBuf):
        """Steady-state partition: hist -> device scan -> emit, fully async
        (no host readback; splits must already be known — they are static
        per shard)."""
        _ck(self._lib, self._lib.gpue_partition_i32_async(
            self._h, keys._h, n, num_channels, row_indexes._h, scratch._h))

    def partition_i64_async(self, keys: DBuf, n, num_channels, row_indexes: DBuf,
                            scratch: DBuf):
        _ck(self._lib, self._lib.gpue_partition_i64_async(
            self._h, keys._h, n, num_channels, row_indexes._h, scratch._h))

    def partition_crc(self, keys: DBuf, n, num_channels, row_indexes: DBuf) -> np.ndarray:
        """Bucket-shuffle hash path (zlib crc32 seed 0,
        exchange_sink_operator.cpp:617-622)."""
        sp = np.zeros(num_channels + 1, dtype=np.uint64)
        _ck(self._lib, self._lib.gpue_partition_crc_i32(
            self._h, keys._h, n, num_channels, sp.ctypes.data_as(c_vp), row_indexes._h))
        return sp

    def partition_varchar(self, bytes_: DBuf, offsets: DBuf, n, num_channels,
                          row_indexes: DBuf) -> np.ndarray:
        """Varchar partition key (FNV over the slice bytes)."""
        sp = np.zeros(num_channels + 1, dtype=np.uint64)
        _ck(self._lib, self._lib.gpue_partition_varchar(
            self._h, bytes_._h, offsets._h, n, num_channels,
            sp.ctypes.data_as(c_vp), row_indexes._h))
        return sp

    def partition_xxh3(self, keys: DBuf, n, num_channels, row_indexes: DBuf) -> np.ndarray:
        """Version-1 exchange hash (xxh3) partition
        (exchange_sink_operator.cpp:604-610; FNV stays the default)."""
        sp = np.zeros(num_channels + 1, dtype=np.uint64)
        _ck(self._lib, self._lib.gpue_partition_xxh3_i32(
            self._h, keys._h, n, num_channels, sp.ctypes.data_as(c_vp), row_indexes._h))
        return sp

    def partition_2xi32(self, a: DBuf, b: DBuf, n, num_channels,
                        row_indexes: DBuf) -> np.ndarray:
        sp = np.zeros(num_channels + 1, dtype=np.uint64)
        _ck(self._lib, self._lib.gpue_partition_2xi32(
            self._h, a._h, b._h, n, num_channels, sp.ctypes.data_as(c_vp), row_indexes._h))
        return sp

    def ubench_bitgather(self, idx: DBuf, n, bits: DBuf, nbits_pow2, reps=5) -> float:
        ms = ctypes.c_float()
        _ck(self._lib, self._lib.gpue_ubench_bitgather(
            self._h, idx._h, n, bits._h, nbits_pow2, reps, ctypes.byref(ms)))
        return ms.value

    def page_decode_binary_prefix(self, page: DBuf, n_values, out_bytes: DBuf,
                                  out_offsets: DBuf):
        """BinaryPrefixPage (front coding, restart every 16) -> BinaryColumn."""
        _ck(self._lib, self._lib.gpue_page_decode_binary_prefix(
            self._h, page._h, n_values, out_bytes._h, out_offsets._h))

    def page_decode_binary_plain(self, page: DBuf, n_values, out_bytes: DBuf,
                                 out_offsets: DBuf):
        """BinaryPlainPage -> BinaryColumn (bytes + offsets[n+1])."""
        _ck(self._lib, self._lib.gpue_page_decode_binary_plain(
            self._h, page._h, n_values, out_bytes._h, out_offsets._h))

    def page_decode_for_i32(self, page: DBuf, n_values, out: DBuf):
        """Frame-of-reference page decode (FOR_ENCODING)."""
        _ck(self._lib, self._lib.gpue_page_decode_for_i32(self._h, page._h,
                                                          n_values, out._h))

    def page_decode_rle_bool(self, page: DBuf, n_values, out: DBuf):
        """RLE bool page decode (bit_width 1; u8 output)."""
        _ck(self._lib, self._lib.gpue_page_decode_rle_bool(self._h, page._h,
                                                           n_values, out._h))

    def page_decode_plain_i32(self, page: DBuf, n_values, out: DBuf):
        """PlainPage numeric decode (plain_page.h:51,83-102,148-158)."""
        _ck(self._lib, self._lib.gpue_page_decode_plain_i32(
            self._h, page._h, n_values, out._h))

    def page_decode_rle_i32(self, page: DBuf, n_values, out: DBuf):
        """RLE page decode (rle_page.h + rle_encoding.h at bit_width 32)."""
        _ck(self._lib, self._lib.gpue_page_decode_rle_i32(self._h, page._h,
                                                          n_values, out._h))

    def page_decode_bshuf_lz4_i32(self, page: DBuf, n_values, out: DBuf):
        _ck(self._lib, self._lib.gpue_page_decode_bshuf_lz4_i32(self._h, page._h,
                                                                n_values, out._h))

    def hash_agg_push(self, at, keys: DBuf, vals: DBuf, n, cnts: DBuf = None,
                      update_only=0, miss_mask: DBuf = None, want_hits=False):
        hv = c_u64()
        _ck(self._lib, self._lib.gpue_hash_agg_push_u64(
            self._h, at, keys._h, vals._h if vals else None, cnts._h if cnts else None, n,
            update_only, miss_mask._h if miss_mask else None,
            ctypes.byref(hv) if want_hits else None))
        return hv.value if want_hits else None

    def hash_agg_probe_hits(self, at, keys: DBuf, n) -> int:
        hv = c_u64()
        _ck(self._lib, self._lib.gpue_hash_agg_probe_hits_u64(self._h, at, keys._h, n,
                                                              ctypes.byref(hv)))
        return hv.value

    def hash_agg_emit(self, at, out_keys: DBuf, out_sums: DBuf, max_out,
                      out_counts: DBuf = None) -> int:
        g = c_u64()
        _ck(self._lib, self._lib.gpue_hash_agg_emit_u64(
            self._h, at, out_keys._h, out_sums._h, out_counts._h if out_counts else None,
            max_out, ctypes.byref(g)))
        return g.value

    def dict_decode_binary(self, dict_bytes: DBuf, dict_offsets: DBuf, codes: DBuf, n,
                           out_bytes: DBuf = None, out_offsets: DBuf = None) -> int:
        t = c_u64()
        ob = out_bytes._h if out_bytes else None
        oo = out_offsets._h if out_offsets else None
        _ck(self._lib, self._lib.gpue_dict_decode_binary(
            self._h, dict_bytes._h, dict_offsets._h, codes._h, n, ob, oo, ctypes.byref(t)))
        return t.value

    def sbf_build(self, keys: DBuf, n, log_num_buckets, directory: DBuf):
        _ck(self._lib, self._lib.gpue_sbf_build_i32(self._h, keys._h, n, log_num_buckets,
                                                    directory._h))

    def sbf_test(self, keys: DBuf, n, directory: DBuf, log_num_buckets, out: DBuf):
        _ck(self._lib, self._lib.gpue_sbf_test_i32(self._h, keys._h, n, directory._h,
                                                   log_num_buckets, out._h))

    def topk_i64(self, keys: DBuf, vals: DBuf, n, k):
        ok = np.zeros(k, np.uint64)
        ov = np.zeros(k, np.int64)
        _ck(self._lib, self._lib.gpue_topk_i64(self._h, keys._h, vals._h, n, k,
                                               ok.ctypes.data_as(c_vp),
                                               ov.ctypes.data_as(c_vp)))
        return ok, ov

    # ---- stream event timing (bench) ----
    def timer_start(self):
        _ck(self._lib, self._lib.gpue_timer_start(self._h))

    def timer_stop(self) -> float:
        ms = ctypes.c_float()
        _ck(self._lib, self._lib.gpue_timer_stop(self._h, ctypes.byref(ms)))
        return ms.value
