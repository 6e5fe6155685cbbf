/* oracle.h — CPU restatement of the StarRocks BE hot-path algorithms.
 *
 * TEST INFRASTRUCTURE ONLY. Per DESIGN.md §2, only tests/, __graft_entry__'s
 * smoke() and bench.py's cpu_baseline leg may load this library. The product
 * path (starrocks_amd/ + libgpue.so) must never call into it.
 *
 * Every function cites the reference source (path:line relative to
 * /root/reference) whose algorithm it restates. Parity is pinned by the
 * reference's own known-answer tests, ported in tests/test_oracle_golden.py.
 */
#ifndef GPUE_ORACLE_H
#define GPUE_ORACLE_H
#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- hashes (be/src/base/hash/hash.h:96-132, exec/join/join_hash_map_helper.h:23-77) ---- */
uint32_t orc_crc_hash_32(const void* data, int32_t bytes, uint32_t seed);
uint32_t orc_fnv_hash(const void* data, int32_t bytes, uint32_t seed);
uint32_t orc_xorshift32(uint32_t x);
uint32_t orc_join_hash_u32(uint32_t v, uint32_t num_log_buckets);
uint32_t orc_join_hash_u64(uint64_t v, uint32_t num_log_buckets);
uint32_t orc_join_hash_slice(const void* p, int32_t n, uint32_t num_buckets);
uint32_t orc_calc_bucket_size(uint32_t size);

/* JoinHashMapSelector restatement (join_hash_table.cpp:164-344); enums and
 * parameter semantics documented at the definitions in oracle.c */
int orc_join_select_key_constructor(int num_keys, const int32_t* fixed_sizes,
                                    const uint8_t* null_safe,
                                    int enable_fixed_size_string,
                                    int32_t* packed_bytes_out);
int orc_join_select_varchar_constructor(int32_t max_size, int enable_fixed_size_string);
int orc_join_select_method(int key_constructor, int lt_class, uint64_t row_count,
                           int64_t min_value, int64_t max_value, int mode,
                           int with_other_conjunct, int enable_range_direct,
                           int enable_linear_chained, uint64_t l2_size,
                           uint64_t l3_size);

/* ---- deterministic synthetic data (stateless splitmix64 finalizer) ---- */
uint64_t orc_gen_u64(uint64_t seed, uint64_t tag, uint64_t i);
void orc_gen_u32_mod(uint64_t seed, uint64_t tag, uint64_t start, uint64_t n,
                     uint32_t mod, uint32_t add, uint32_t* out);
void orc_gen_i64(uint64_t seed, uint64_t tag, uint64_t start, uint64_t n, int64_t* out);

/* SSB date dimension: n_days entries from 1992-01-01, d_datekey=y*10000+m*100+d */
void orc_gen_dates(int32_t n_days, int32_t* datekey, int32_t* dyear);

/* ---- join hash maps ---- */
/* BucketChained build: exec/join/join_hash_map_method.hpp:37-85. keys is
 * 1-based (keys[0] = sentinel default row, join_hash_table.cpp:590-596);
 * first[bucket_size] and next[row_count+1] are zero-initialised by callee. */
void orc_bucket_chained_build_u32(const uint32_t* keys, uint32_t row_count,
                                  uint32_t* first, uint32_t* next,
                                  uint32_t bucket_size, uint32_t log_bucket_size);
/* lookup_init: join_hash_map_method.hpp:88-120 — chain head per probe row */
void orc_bucket_chained_lookup_u32(const uint32_t* probe_keys, uint32_t probe_rows,
                                   const uint32_t* first, uint32_t bucket_size,
                                   uint32_t log_bucket_size, uint32_t* heads);

void orc_bucket_chained_build_u64(const uint64_t* keys, uint32_t row_count,
                                  uint32_t* first, uint32_t* next,
                                  uint32_t bucket_size, uint32_t log_bucket_size);
void orc_bucket_chained_lookup_u64(const uint64_t* probe_keys, uint32_t probe_rows,
                                   const uint32_t* first, uint32_t bucket_size,
                                   uint32_t log_bucket_size, uint32_t* heads);
uint64_t orc_probe_emit_u64(const uint64_t* build_keys, const uint32_t* next,
                            const uint64_t* probe_keys, const uint32_t* heads,
                            uint32_t probe_rows, int collision_free,
                            uint32_t* out_probe_idx, uint32_t* out_build_idx);
void orc_slice_build_u32(const uint8_t* bytes, const uint32_t* offsets, uint32_t row_count,
                         uint32_t* first, uint32_t* next, uint32_t bucket_size,
                         uint32_t log_bucket_size);
uint64_t orc_phmap_mix8(uint64_t a);
int32_t orc_sbf_log_num_buckets(uint64_t nums);
void orc_sbf_insert_hash(uint32_t* directory, int32_t log_num_buckets, uint64_t h);
int orc_sbf_test_hash(const uint32_t* directory, int32_t log_num_buckets, uint64_t h);
void orc_sbf_build_i32(const int32_t* keys, uint64_t n, uint32_t* directory,
                       int32_t log_num_buckets);
void orc_sbf_test_i32(const int32_t* keys, uint64_t n, const uint32_t* directory,
                      int32_t log_num_buckets, uint8_t* out);
void orc_slice_build_nulls_u32(const uint8_t* bytes, const uint32_t* offsets,
                               const uint8_t* is_nulls, uint32_t row_count, uint32_t* first,
                               uint32_t* next, uint32_t bucket_size,
                               uint32_t log_bucket_size);
uint64_t orc_slice_probe_emit_nulls(const uint8_t* bbytes, const uint32_t* boffsets,
                                    const uint32_t* next, uint32_t bucket_size,
                                    const uint32_t* first, const uint8_t* pbytes,
                                    const uint32_t* poffsets, const uint8_t* probe_nulls,
                                    uint32_t probe_rows, int mode, uint32_t* out_probe_idx,
                                    uint32_t* out_build_idx);
uint64_t orc_slice_probe_right(const uint8_t* bbytes, const uint32_t* boffsets,
                               const uint32_t* next, uint32_t bucket_size,
                               const uint32_t* first, uint32_t build_rows,
                               const uint8_t* pbytes, const uint32_t* poffsets,
                               uint32_t probe_rows, int anti, uint32_t* out_build_idx);
uint64_t orc_eval_conjuncts_i32(int32_t** cols, int n_cols, uint64_t n_rows,
                                const int32_t* pred_col, const int32_t* pred_op,
                                const int32_t* pred_lo, const int32_t* pred_hi,
                                int n_preds);
uint64_t orc_eval_conjuncts_i64(int64_t** cols, int n_cols, uint64_t n_rows,
                                const int32_t* pred_col, const int32_t* pred_op,
                                const int64_t* pred_lo, const int64_t* pred_hi,
                                int n_preds);
uint64_t orc_dict_decode_binary(const uint8_t* dict_bytes, const uint32_t* dict_offsets,
                                const int32_t* codes, uint64_t n, uint8_t* out_bytes,
                                uint32_t* out_offsets);
uint64_t orc_slice_probe_emit_mode(const uint8_t* bbytes, const uint32_t* boffsets,
                                   const uint32_t* next, uint32_t bucket_size,
                                   const uint32_t* first, const uint8_t* pbytes,
                                   const uint32_t* poffsets, uint32_t probe_rows, int mode,
                                   uint32_t* out_probe_idx, uint32_t* out_build_idx);
uint64_t orc_slice_probe_emit(const uint8_t* bbytes, const uint32_t* boffsets,
                              const uint32_t* next, uint32_t bucket_size,
                              const uint32_t* first, const uint8_t* pbytes,
                              const uint32_t* poffsets, uint32_t probe_rows,
                              uint32_t* out_probe_idx, uint32_t* out_build_idx);
void orc_bucket_chained_build_nulls_u32(const uint32_t* keys, const uint8_t* is_nulls,
                                        uint32_t row_count, uint32_t* first, uint32_t* next,
                                        uint32_t bucket_size, uint32_t log_bucket_size);
void orc_bucket_chained_lookup_nulls_u32(const uint32_t* probe_keys, const uint8_t* is_nulls,
                                         uint32_t probe_rows, const uint32_t* first,
                                         uint32_t bucket_size, uint32_t log_bucket_size,
                                         uint32_t* heads);
void orc_pack_keys_2xi32(const int32_t* a, const int32_t* b, uint64_t n, uint64_t* out);

/* TLinearChained (fp-packed) build+lookup: join_hash_map_method.hpp:125-368,
 * join_hash_map_method.h:118-150 (FP_BITS=8, fp|24-bit index packed in first) */
void orc_linear_chained_build_u32(const uint32_t* keys, uint32_t row_count,
                                  uint32_t* first, uint32_t* next,
                                  uint32_t bucket_size, uint32_t log_bucket_size);
void orc_linear_chained_lookup_u32(const uint32_t* build_keys, const uint32_t* probe_keys,
                                   uint32_t probe_rows, const uint32_t* first,
                                   uint32_t bucket_size, uint32_t log_bucket_size,
                                   uint32_t* heads);

/* RangeDirectMapping: join_hash_map_method.hpp:625-707 */
void orc_range_direct_build_i32(const int32_t* keys, uint32_t row_count,
                                int64_t min_value, uint32_t* first, uint32_t* next);
void orc_range_direct_lookup_i32(const int32_t* probe_keys, uint64_t probe_rows,
                                 int64_t min_value, int64_t max_value,
                                 const uint32_t* first, uint32_t* heads);

/* Probe chain-walk, emit ALL (probe_idx, build_idx) pairs:
 * join_hash_map.hpp:717-795 (_probe_from_ht); the chunk_size-resumable cursor
 * is an iteration detail — the emitted multiset is the result. Returns number
 * of pairs. build_keys/next are 1-based; heads from a lookup_init above.
 * collision_free: skip key compare (is_collision_free_and_unique fast path,
 * join_hash_map.hpp:752-761). */
uint64_t orc_probe_emit_u32(const uint32_t* build_keys, const uint32_t* next,
                            const uint32_t* probe_keys, const uint32_t* heads,
                            uint32_t probe_rows, int collision_free,
                            uint32_t* out_probe_idx, uint32_t* out_build_idx);

/* per-join-type probe emits (join_hash_map.h:228-333): mode 0 INNER,
 * 1 LEFT_SEMI, 2 LEFT_ANTI, 3 LEFT_OUTER */
uint64_t orc_probe_emit_mode_u32(const uint32_t* build_keys, const uint32_t* next,
                                 const uint32_t* probe_keys, const uint32_t* heads,
                                 uint32_t probe_rows, int mode,
                                 uint32_t* out_probe_idx, uint32_t* out_build_idx);
uint64_t orc_probe_right_u32(const uint32_t* build_keys, const uint32_t* next,
                             uint32_t build_rows, const uint32_t* probe_keys,
                             const uint32_t* heads, uint32_t probe_rows, int anti,
                             uint32_t* out_build_idx);

/* ---- predicate filter (base/simd/filter.h:26-38 + chunk_predicate_evaluator.cpp:31-80) ----
 * stable stream compaction of int64 values where v < theta; returns count */
uint64_t orc_filter_i64_lt(const int64_t* in, uint64_t n, int64_t theta, int64_t* out);
/* OpenMP-parallel variant used as the timed CPU baseline */
uint64_t orc_filter_i64_lt_mt(const int64_t* in, uint64_t n, int64_t theta, int64_t* out);

/* ---- exchange partition (exchange_sink_operator.cpp:611-660, shuffler.h:71-102,
 * hash_util.hpp:120-262) ----
 * hashes: per-row FNV over the 4-byte key with running seed FNV_SEED, then
 * channel = ReduceOp(hash, num_channels) = (hash * n) >> 32 (HASH_PARTITIONED). */
void orc_partition_channel_2xi32(const int32_t* a, const int32_t* b, uint64_t n,
                                 uint32_t num_channels, uint32_t* out);
void orc_partition_channel_u32(const uint32_t* keys, uint64_t n, uint32_t num_channels,
                               uint32_t* channel_ids);
/* counting-sort row layout: start_points[ch+1] sizes then reverse emit
 * (exchange_sink_operator.cpp:629-660). row_indexes gets source row per slot. */
void orc_partition_channel_u64(const uint64_t* keys, uint64_t n, uint32_t num_channels,
                               uint32_t* channel_ids);
void orc_partition_counting_sort(const uint32_t* channel_ids, uint64_t n,
                                 uint32_t num_channels, uint64_t* start_points,
                                 uint32_t* row_indexes);

/* ---- fused pipelines (the CPU baseline legs; OpenMP) ---- */
/* Config 2 (SSB SF10 Q1-like): lineorder(lo_orderdate,lo_extendedprice,
 * lo_discount) ⋈ date filtered d_year==year, SUM(extendedprice*discount).
 * Columns generated internally from (seed, n_rows, rank offset). Returns sum;
 * *match_count gets matched row count. */
int64_t orc_q1_pipeline(uint64_t seed, uint64_t row_start, uint64_t n_rows,
                        int32_t year, int threads, uint64_t* match_count);
/* Config 3 (SSB SF100 Q2.1): 3-way star probe + GROUP BY (d_year,p_brand).
 * group_sums must hold 7*1000 int64 (index (year-1992)*1000+brand), zeroed. */
void orc_q21_pipeline(uint64_t seed, uint64_t row_start, uint64_t n_rows,
                      int32_t category, int32_t region, int threads,
                      int64_t* group_sums);

/* column generators for the pipelines, exposed so tests can cross-check the
 * GPU generator bit-for-bit */
void orc_gen_lineorder_q1(uint64_t seed, uint64_t row_start, uint64_t n,
                          int32_t* lo_orderdate, int32_t* lo_extendedprice,
                          int32_t* lo_discount);
void orc_gen_lineorder_q21(uint64_t seed, uint64_t row_start, uint64_t n,
                           int32_t* lo_partkey, int32_t* lo_suppkey,
                           int32_t* lo_orderdate, int32_t* lo_revenue);
/* dim payload arrays as the GPU builds them:
 * date:  first[datekey-min] = (d_year-1992)+1 if in range else 0
 * part:  first[p-1] = brand+1 if p_category==category else 0
 * supp:  first[s-1] = 1 if s_region==region else 0 */
void orc_build_date_dim(int32_t n_days, int32_t year_filter /* -1: no filter */,
                        int32_t* min_key, int32_t* max_key, uint32_t** first_out,
                        uint32_t* size_out);
void orc_build_part_dim(uint64_t seed, uint32_t n_parts, int32_t category, uint32_t* first);
void orc_build_supp_dim(uint64_t seed, uint32_t n_supps, int32_t region, uint32_t* first);

/* p_category / p_brand / s_region generators (shared with GPU) */
uint32_t orc_part_category(uint64_t seed, uint32_t partkey);
uint32_t orc_part_brand(uint64_t seed, uint32_t partkey);
uint32_t orc_supp_region(uint64_t seed, uint32_t suppkey);

/* ---- SSB Q4.3 (config 4): 4-way star join + 2-key GROUP BY ---- */
void orc_gen_lineorder_q43(uint64_t seed, uint64_t row_start, uint64_t n,
                           int32_t* lo_custkey, int32_t* lo_suppkey,
                           int32_t* lo_partkey, int32_t* lo_orderdate,
                           int32_t* lo_revenue, int32_t* lo_supplycost);
uint32_t orc_cust_region(uint64_t seed, uint32_t custkey);
uint32_t orc_supp_nation(uint64_t seed, uint32_t suppkey);
uint32_t orc_supp_city_in_nation(uint64_t seed, uint32_t suppkey);
uint32_t orc_part_brand_in_category(uint64_t seed, uint32_t partkey);
void orc_build_cust_dim_q43(uint64_t seed, uint32_t n_custs, int32_t region, uint32_t* first);
void orc_build_supp_dim_q43(uint64_t seed, uint32_t n_supps, int32_t nation, uint32_t* first);
void orc_build_part_dim_q43(uint64_t seed, uint32_t n_parts, int32_t category, uint32_t* first);
void orc_q43_kernel(const int32_t* ck, const int32_t* sk, const int32_t* pk,
                    const int32_t* od, const int32_t* rv, const int32_t* sc,
                    uint64_t n_rows, const uint32_t* cfirst, const uint32_t* sfirst,
                    const uint32_t* pfirst, const uint32_t* dfirst, int64_t dmin,
                    int threads, int64_t* group_sums /*800*/);
void orc_q43_pipeline(uint64_t seed, uint64_t row_start, uint64_t n_rows,
                      int32_t region, int32_t nation, int32_t category,
                      int threads, int64_t* group_sums /*800*/);

/* ---- TPC-H Q3 (config 5) ---- */
void orc_gen_lineitem_q3(uint64_t seed, uint64_t row_start, uint64_t n,
                         uint64_t n_orders, int64_t* l_orderkey,
                         int64_t* l_extendedprice, int64_t* l_discount,
                         int32_t* l_shipdate);
void orc_gen_orders_q3(uint64_t seed, uint64_t n_orders, uint32_t n_custs,
                       int32_t* o_custkey, int32_t* o_orderdate);
void orc_gen_cust_mkt16(uint64_t seed, uint32_t n_custs, uint8_t* out);
const char* orc_mkt_segment_literal(int idx);
void orc_q3_build_cust_bits(const uint8_t* mkt16, uint32_t n_custs,
                            const char* lit16, uint8_t* bits);
void orc_q3_build_order_bits(const int32_t* o_custkey, const int32_t* o_orderdate,
                             uint64_t n_orders, const uint8_t* cust_bits,
                             int32_t date_cutoff, uint8_t* bits);
uint64_t orc_q3_probe_agg(const int64_t* lk, const int64_t* ext, const int64_t* disc,
                          const int32_t* ship, uint64_t n, const uint8_t* order_bits,
                          int32_t ship_cutoff, uint64_t* out_keys, int64_t* out_sums,
                          uint64_t max_out);

/* generic hash aggregate (agg_hash_map.h:112-290 restatement); returns group
 * count or UINT64_MAX if max_out exceeded */
uint64_t orc_hash_agg_sum_u64(const uint64_t* keys, const int64_t* vals, uint64_t n,
                              uint64_t* out_keys, int64_t* out_sums, int64_t* out_counts,
                              uint64_t max_out);

uint64_t orc_hash_agg_stats_u64(const uint64_t* keys, const int64_t* vals, uint64_t n,
                                uint64_t* out_keys, int64_t* out_sums, int64_t* out_counts,
                                int64_t* out_mins, int64_t* out_maxs, uint64_t max_out);
uint64_t orc_hash_agg_sum128_u64(const uint64_t* keys, const int64_t* vals, uint64_t n,
                                 uint64_t* out_keys, uint64_t* out_lo, int64_t* out_hi,
                                 uint64_t max_out);

/* compute-only legs timed by bench.py's cpu_baseline (columns pre-generated) */
int64_t orc_q1_kernel(const int32_t* od, const int32_t* ep, const int32_t* dc,
                      uint64_t n_rows, const uint32_t* dfirst, int64_t mn, int64_t mx,
                      int threads, uint64_t* match_count);
void orc_q21_kernel(const int32_t* pk, const int32_t* sk, const int32_t* od,
                    const int32_t* rv, uint64_t n_rows, const uint32_t* pfirst,
                    const uint32_t* sfirst, const uint32_t* dfirst, int64_t dmin,
                    int threads, int64_t* group_sums);

/* storage ingress: bitshuffle(0.5.1 published algorithm)+LZ4(block spec)
 * page body encode/decode for int32 */
uint64_t orc_bshuf_lz4_encode_i32(const int32_t* values, uint32_t n, uint8_t* out,
                                  uint32_t* block_starts);
uint64_t orc_bshuf_lz4_decode_i32(const uint8_t* in, uint32_t n, int32_t* values);

void orc_free(void* p);

#ifdef __cplusplus
}
#endif
#endif

/* XXH3-64 exchange hash version 1 (exchange_sink_operator.cpp:604-610;
 * restated from the published XXH3 spec, pinned to python-xxhash vectors) */
uint64_t orc_xxh3_64_4to8(const void* data, int32_t len, uint64_t seed);
void orc_xxh3_hash_i32(const int32_t* col, uint64_t n, uint32_t* hashes);
void orc_xxh3_hash_i64(const int64_t* col, uint64_t n, uint32_t* hashes);
void orc_partition_channel_xxh3_u32(const uint32_t* keys, uint64_t n,
                                    uint32_t num_channels, uint32_t* channel_ids);
/* set the OMP team size for the parameterless kernels (q3 legs) */
void orc_set_threads(int n);

/* RLE page codec for int32 (rle_page.h + base/bit/rle_encoding.h at
 * bit_width 32; Parquet-style RLE/bit-pack hybrid, byte-aligned) */
uint64_t orc_rle_page_encode_i32(const int32_t* values, uint32_t n, uint8_t* out);
uint64_t orc_rle_page_decode_i32(const uint8_t* page, int32_t* values);
uint64_t orc_rle_page_encode_bool(const uint8_t* values, uint32_t n, uint8_t* out);
uint64_t orc_rle_page_decode_bool(const uint8_t* page, uint8_t* values);

/* frame-of-reference page codec for int32 (FOR_ENCODING,
 * frame_of_reference_coding.{h,cpp}; decoder-authoritative layout) */
uint64_t orc_for_page_encode_i32(const int32_t* values, uint32_t n, uint8_t* out);
uint64_t orc_for_page_decode_i32(const uint8_t* page, uint64_t page_bytes, int32_t* values);
/* BinaryPlainPage codec (PLAIN_ENCODING, binary_plain_page.h:28-46) */
uint64_t orc_binary_plain_encode(const uint8_t* bytes, const uint32_t* offsets,
                                 uint32_t n, uint8_t* out);
uint64_t orc_binary_plain_decode(const uint8_t* page, uint64_t page_bytes,
                                 uint8_t* out_bytes, uint32_t* out_offsets);
/* zlib CRC32 + the exchange's crc (bucket-shuffle) and varchar-key paths */
uint32_t orc_zlib_crc32(const void* data, int32_t n, uint32_t seed);
void orc_partition_channel_crc_u32(const uint32_t* keys, uint64_t n,
                                   uint32_t num_channels, uint32_t* channel_ids);
void orc_partition_channel_fnv_slice(const uint8_t* bytes, const uint32_t* offsets,
                                     uint64_t n, uint32_t num_channels,
                                     uint32_t* channel_ids);
/* raw LZ4 block codec (pinned against pyarrow's bundled lz4_raw in tests) */
uint64_t orc_lz4_compress_block(const uint8_t* src, uint64_t n, uint8_t* dst);
uint64_t orc_lz4_decompress_block(const uint8_t* src, uint64_t comp_n, uint8_t* dst,
                                  uint64_t cap);
/* bit-plane transpose stage (cross-checked vs an independent numpy
 * restatement of published bitshuffle 0.5.1 in tests) */
uint64_t orc_bshuf_transpose_i32(const int32_t* in, uint32_t elems, uint8_t* out);
/* BinaryPrefixPage codec (PREFIX_ENCODING, binary_prefix_page.{h,cpp}) */
uint64_t orc_binary_prefix_encode(const uint8_t* bytes, const uint32_t* offsets,
                                  uint32_t n, uint8_t* out);
uint64_t orc_binary_prefix_decode(const uint8_t* page, uint64_t page_bytes,
                                  uint8_t* out_bytes, uint32_t* out_offsets);
/* ASOF inner/left-outer join restatement (LinearChainedAsofJoinHashMap,
 * join_hash_map_method.h:201-217 + AsofIndex, join_hash_table_descriptor.h:
 * 59-104 / .cpp:70-134). opcode 0 LT / 1 LE / 2 GT / 3 GE. Build arrays are
 * 1-based (row 0 sentinel). out_build[i] = 1-based matched build row for
 * probe row i, 0 = no match. Tie refinement: duplicate (key, asof) pairs
 * resolve to the smallest build row (the reference's pdqsort is unstable
 * there; gpue pins the same refinement). */
void orc_asof_inner_join(const int32_t* build_keys, const int64_t* build_asof,
                         uint32_t build_rows, const int32_t* probe_keys,
                         const int64_t* probe_asof, uint64_t n, int opcode,
                         uint32_t* out_build);
/* nulls variant: 1-based build_nulls (equi+temporal masks ORed) skips build
 * rows (join_hash_table_descriptor.h:447-456); null probe rows never match.
 * Either mask may be NULL. */
void orc_asof_inner_join_nulls(const int32_t* build_keys, const int64_t* build_asof,
                               const uint8_t* build_nulls, uint32_t build_rows,
                               const int32_t* probe_keys, const int64_t* probe_asof,
                               const uint8_t* probe_nulls, uint64_t n, int opcode,
                               uint32_t* out_build);
/* PlainPage numeric codec (plain_page.h:51,83-102,148-158) */
uint64_t orc_plain_page_encode_i32(const int32_t* values, uint32_t n, uint8_t* out);
uint64_t orc_plain_page_decode_i32(const uint8_t* page, uint64_t page_bytes,
                                   int32_t* values);
