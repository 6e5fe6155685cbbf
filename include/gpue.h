/* gpue.h — C-ABI boundary of the MI355X-native execution engine for the
 * StarRocks BE hot path.
 *
 * These entry points are what thin C++ Operator wrappers inside the reference
 * BE would call from the pipeline driver loop
 * (reference be/src/exec/runtime/pipeline_driver.cpp:391-500); the mapping of
 * each entry to the reference interface it replaces is given per declaration.
 * INTEGRATION.md shows the Operator-side binding a maintainer would add.
 *
 * Conventions: extern "C"; plain pointers + sizes; opaque handles; int status
 * returns (0 == GPUE_OK, matching Status::ok() of reference
 * be/src/base/status.h); no torch types. The implementation (HIP/C++,
 * starrocks_amd/csrc/gpue.hip) FAILS at session creation when no AMD GPU is
 * present — there is no CPU fallback in the product path.
 */
#ifndef GPUE_H
#define GPUE_H
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

#define GPUE_OK 0
#define GPUE_ERR_HIP 1        /* HIP runtime failure (details: gpue_last_error) */
#define GPUE_ERR_ARG 2        /* bad argument */
#define GPUE_ERR_NO_GPU 3     /* no AMD GPU visible */

/* Human-readable detail of the last non-OK status on this thread. */
const char* gpue_last_error(void);

/* ---- session ----
 * Engine lifetime per device; analog of FragmentExecutor::prepare/close
 * (reference be/src/orchestration/fragment_executor.h:40-42). One session ==
 * one HIP device + one HIP stream (one pipeline driver <-> one stream,
 * SURVEY.md §2 pipeline row). */
typedef struct gpue_session gpue_session;
int gpue_session_create(int device_index, gpue_session** out);
void gpue_session_destroy(gpue_session* s);
int gpue_device_count(int* out);
int gpue_sync(gpue_session* s);

/* ---- device buffers ----
 * Raw HBM allocations; the engine-side analog of Chunk column containers
 * (reference be/src/column/fixed_length_column_base.h:283) held resident on
 * device. Chunked Operator pushes append into these (gpue_dbuf_h2d with
 * dst_off). */
typedef struct gpue_dbuf gpue_dbuf;
int gpue_dbuf_alloc(gpue_session* s, uint64_t bytes, gpue_dbuf** out);
void gpue_dbuf_free(gpue_dbuf* b);
int gpue_dbuf_h2d(gpue_dbuf* b, const void* src, uint64_t bytes, uint64_t dst_off);
int gpue_dbuf_d2h(gpue_dbuf* b, void* dst, uint64_t bytes, uint64_t src_off);
int gpue_dbuf_memset(gpue_dbuf* b, int value, uint64_t bytes);
int gpue_dbuf_d2d(gpue_dbuf* src, gpue_dbuf* dst, uint64_t bytes, uint64_t src_off,
                  uint64_t dst_off);
/* SUM(a[i]*b[i]) + row count accumulated into acc (i64[2]) — the agg sink
 * update for the chunked (unfused) config-2 plan (Aggregator::update_batch
 * with sum<int64> states, reference be/src/exprs/agg/sum.h:45-181). */
int gpue_sum_prod_u32(gpue_session* s, gpue_dbuf* a, gpue_dbuf* b, uint64_t n,
                      gpue_dbuf* acc);

/* ---- hipGraph step replay ----
 * Capture the async launch sequence of one step between begin/end, then
 * replay it with a single hipGraphLaunch per step. Removes the per-launch
 * host cost that dominates small (SF10-sized) steps; the CDNA-native analog
 * of the reference driver's time-slice batching. Only stream-async entries
 * (the *_async pipelines, memsets) may sit inside a capture. */
typedef struct gpue_graph gpue_graph;
int gpue_graph_begin(gpue_session* s);
int gpue_graph_end(gpue_session* s, gpue_graph** out);
int gpue_graph_launch(gpue_session* s, gpue_graph* g);
void gpue_graph_destroy(gpue_graph* g);
/* Wrap external device memory (e.g. a torch tensor's data_ptr) so kernels
 * operate in place and torch.distributed (RCCL) moves the same buffers —
 * the all-to-all leg of configs 4-5. Caller keeps ownership. */
int gpue_dbuf_wrap(gpue_session* s, void* device_ptr, uint64_t bytes, gpue_dbuf** out);
/* Expose the device pointer (torch interop / sub-buffer views). */
int gpue_dbuf_ptr(gpue_dbuf* b, void** out);

/* ---- synthetic chunk source ----
 * Replaces the scan operator with a seeded on-device generator (SURVEY.md §2
 * scan row: storage engine OUT OF SCOPE, synthetic source instead). The
 * generator (splitmix64 finalizer on row index) is bit-identical to
 * oracle/oracle.c orc_gen_* and tests' numpy restatement. */
int gpue_gen_u32_mod(gpue_session* s, gpue_dbuf* out, uint64_t seed, uint64_t tag,
                     uint64_t row_start, uint64_t n, uint32_t mod, uint32_t add);
int gpue_gen_i64(gpue_session* s, gpue_dbuf* out, uint64_t seed, uint64_t tag,
                 uint64_t row_start, uint64_t n);
/* SSB lineorder column sets (schema: reference test/common/sql/ssb/create.sql) */
int gpue_gen_lineorder_q1(gpue_session* s, uint64_t seed, uint64_t row_start, uint64_t n,
                          gpue_dbuf* lo_orderdate, gpue_dbuf* lo_extendedprice,
                          gpue_dbuf* lo_discount);
int gpue_gen_lineorder_q21(gpue_session* s, uint64_t seed, uint64_t row_start, uint64_t n,
                           gpue_dbuf* lo_partkey, gpue_dbuf* lo_suppkey,
                           gpue_dbuf* lo_orderdate, gpue_dbuf* lo_revenue);
int gpue_gen_lineorder_q43(gpue_session* s, uint64_t seed, uint64_t row_start, uint64_t n,
                           gpue_dbuf* lo_custkey, gpue_dbuf* lo_suppkey,
                           gpue_dbuf* lo_partkey, gpue_dbuf* lo_orderdate,
                           gpue_dbuf* lo_revenue, gpue_dbuf* lo_supplycost);

/* ---- scan + predicate filter ----
 * Replaces ChunkPredicateEvaluator::eval_conjuncts + Column::filter_range
 * (reference be/src/exprs/chunk_predicate_evaluator.cpp:31-80,
 * be/src/base/simd/filter.h:26-38). Ordered (stable) compaction of int64
 * values v < theta; output order bit-identical to the reference loop. */
int gpue_scan_filter_i64_lt(gpue_session* s, gpue_dbuf* in, uint64_t n, int64_t theta,
                            gpue_dbuf* out, uint64_t* out_count);
/* Single-pass variant (decoupled lookback): one coalesced read + one
 * coalesced write per element — the reference's algorithmic byte count. */
int gpue_scan_filter_i64_lt_sp(gpue_session* s, gpue_dbuf* in, uint64_t n, int64_t theta,
                               gpue_dbuf* out, uint64_t* out_count);

/* Multi-conjunct predicate evaluation with the reference's eager-prune
 * strategy (chunk_predicate_evaluator.cpp:31-80: AND-merge per conjunct,
 * all-true skip, all-false short-circuit, compact all columns when zeros
 * exceed max(0.8*rows, 1024)). preds are (col_index, op, lo, hi) arrays;
 * op: 0 EQ(lo), 1 LT(hi), 2 BETWEEN[lo,hi]. Columns compact stably in
 * place; out_rows = survivors. */
int gpue_eval_conjuncts_i32(gpue_session* s, gpue_dbuf** cols, int n_cols, uint64_t n_rows,
                            const int32_t* pred_col, const int32_t* pred_op,
                            const int32_t* pred_lo, const int32_t* pred_hi, int n_preds,
                            uint64_t* out_rows);
int gpue_eval_conjuncts_i64(gpue_session* s, gpue_dbuf** cols, int n_cols, uint64_t n_rows,
                            const int32_t* pred_col, const int32_t* pred_op,
                            const int64_t* pred_lo, const int64_t* pred_hi, int n_preds,
                            uint64_t* out_rows);

/* ---- hash-join build ----
 * Replaces JoinHashTable::build with the RANGE_DIRECT_MAPPING method the
 * selector takes for dense int keys (reference
 * be/src/exec/join/join_hash_table.cpp:263-321,
 * join_hash_map_method.hpp:625-707). Build rows are 1-based; row 0 is the
 * chain-end sentinel (join_hash_table.cpp:590-596).
 *
 * Payload variant (the fused fast path): first[key-min] holds
 * a caller-provided uint32 payload (0 = dim row filtered out / absent). */
typedef struct gpue_join_table gpue_join_table;
int gpue_join_build_payload_i32(gpue_session* s, gpue_dbuf* keys /*i32*/,
                                gpue_dbuf* payloads /*u32*/, uint64_t n_rows,
                                gpue_join_table** out);
/* BUCKET_CHAINED variant — the selector's generic fallback for non-dense
 * keys (join_hash_map_method.hpp:37-120): multiplicative-hash buckets +
 * first/next chains; probe compares build keys along the chain. */
int gpue_join_build_bucket_chained_u32(gpue_session* s, gpue_dbuf* keys /*u32, 1-based*/,
                                       uint64_t row_count, gpue_join_table** out);
/* LINEAR_CHAINED variant — the selector's preferred method under its
 * 16M-bucket cap (join_hash_map_method.h:118-150): 8-bit fingerprint packed
 * in first[], linear probing, same-key chains via next[]. */
int gpue_join_build_linear_chained_u32(gpue_session* s, gpue_dbuf* keys /*u32, 1-based*/,
                                       uint64_t row_count, gpue_join_table** out);
/* 8-byte (BIGINT) key bucket-chained variant — JoinKeyHash<8>
 * (join_hash_map_helper.h:46-55), u64 build-key compares along chains. */
int gpue_join_build_bucket_chained_u64(gpue_session* s, gpue_dbuf* keys /*u64, 1-based*/,
                                       uint64_t row_count, gpue_join_table** out);
/* 16-byte keys (TYPE_LARGEINT / SERIALIZED_FIXED_SIZE_LARGEINT packing):
 * the generic JoinKeyHash<T,16> — crc32 over the key bytes, CRC_SEED,
 * masked by bucket_size-1 (join_hash_map_helper.h:23-30). */
int gpue_join_build_bucket_chained_u128(gpue_session* s, gpue_dbuf* keys /*16 B, 1-based*/,
                                        uint64_t row_count, gpue_join_table** out);
int gpue_join_probe_emit_mode_u128(gpue_session* s, gpue_join_table* t,
                                   gpue_dbuf* probe_keys, uint64_t n_rows, int mode,
                                   gpue_dbuf* out_probe_idx, gpue_dbuf* out_build_idx,
                                   uint64_t* match_count);
int gpue_join_probe_emit_mode_u64(gpue_session* s, gpue_join_table* t, gpue_dbuf* probe_keys,
                                  uint64_t n_rows, int mode, gpue_dbuf* out_probe_idx,
                                  gpue_dbuf* out_build_idx, uint64_t* match_count);
/* Row-index variant: first/next chain structure exactly as the reference
 * builds it (chain order under duplicate keys is scatter-order, which on GPU
 * is nondeterministic — the emitted match multiset is identical). */
int gpue_join_build_range_direct_i32(gpue_session* s, gpue_dbuf* keys /*i32, 1-based row 0 sentinel*/,
                                     uint64_t row_count, gpue_join_table** out);
/* DENSE_RANGE_DIRECT_MAPPING (join_hash_map_method.h:378, .hpp:781-940):
 * rank/select-compressed direct map — per 32-key group a {start_index,
 * bitset} pair (2 bits/interval position amortized), first[] sized by the
 * PRESENT keys; probe = bit test + popcount + one first[] load; chains hold
 * identical keys (no compare). The selector's choice when the interval
 * exceeds bucket/L2 but 2b/pos + 4 B/row beats bucket-chained. */
int gpue_join_build_dense_range_direct_i32(gpue_session* s, gpue_dbuf* keys,
                                           uint64_t row_count, gpue_join_table** out);

/* ---- JoinHashMapSelector (join_hash_table.cpp:164-344) --------------------
 * The reference's automatic key-constructor + map-method decision, restated
 * as pure host functions. Probe-mode values double as the join_type input.
 */
#define GPUE_JOIN_INNER 0
#define GPUE_JOIN_LEFT_SEMI 1
#define GPUE_JOIN_LEFT_ANTI 2
#define GPUE_JOIN_LEFT_OUTER 3
/* key-constructor classes (JoinKeyConstructorUnaryType, :164-229) */
#define GPUE_KEYCON_ONE_KEY 0
#define GPUE_KEYCON_ONE_KEY_VARCHAR 1
#define GPUE_KEYCON_FIXED_INT 2      /* SERIALIZED_FIXED_SIZE_INT, packed <= 4 B */
#define GPUE_KEYCON_FIXED_BIGINT 3   /* packed <= 8 B */
#define GPUE_KEYCON_FIXED_LARGEINT 4 /* packed <= 16 B */
#define GPUE_KEYCON_SERIALIZED_VARCHAR 5
/* map methods (JoinHashMapMethodType, join_hash_map_method.h) */
#define GPUE_JM_DIRECT 0
#define GPUE_JM_RANGE_DIRECT 1
#define GPUE_JM_RANGE_DIRECT_SET 2
#define GPUE_JM_DENSE_RANGE_DIRECT 3
#define GPUE_JM_LINEAR_CHAINED 4
#define GPUE_JM_LINEAR_CHAINED_SET 5
#define GPUE_JM_BUCKET_CHAINED 6
/* logical-type classes for the method decision */
#define GPUE_LT_TINY 0    /* BOOLEAN/TINYINT/SMALLINT -> DIRECT_MAPPING (:239) */
#define GPUE_LT_INT 1
#define GPUE_LT_BIGINT 2
#define GPUE_LT_OTHER 3   /* other fixed-width (largeint/decimal/date...) */
#define GPUE_LT_VARCHAR 4

/* _determine_key_constructor (:164-229). fixed_sizes[i]: key column i's
 * fixed byte width; for varchar keys pass _get_binary_column_max_size's
 * result (1..16 when the fixed-size-string opt applies, else 0).
 * packed_bytes_out: total packed width (0 when not fixed-packed). */
int gpue_join_select_key_constructor(int num_keys, const int32_t* fixed_sizes,
                                     const uint8_t* null_safe,
                                     int enable_fixed_size_string,
                                     int32_t* packed_bytes_out);
/* single-VARCHAR refinement (:178-194) */
int gpue_join_select_varchar_constructor(int32_t max_size, int enable_fixed_size_string);
/* _determine_hash_map_method (:231-256) + range-direct (:270-321) + linear
 * (:323-344) gates. min/max_value: the single int key's bounds over build
 * rows (ignored unless the range-direct gate applies). l2_size/l3_size:
 * reference reads CpuInfo L2 and halves L3 itself (:295-296). Returns a
 * GPUE_JM_* value. */
int gpue_join_select_method(int key_constructor, int lt_class, uint64_t row_count,
                            int64_t min_value, int64_t max_value, int mode,
                            int with_other_conjunct, int enable_range_direct,
                            int enable_linear_chained, uint64_t l2_size,
                            uint64_t l3_size);
/* Auto build for a single i32 key column (keys 1-based, row 0 sentinel):
 * computes the build keys' min/max on device, runs the selector with the
 * reference's default-on session flags, and builds the matching GPU table.
 * l2_size/l3_size 0 -> MI355X defaults (4 MiB XCD L2 / 256 MiB Infinity
 * Cache). method_out (nullable) receives the GPUE_JM_* decision. Mapping to
 * physical layouts: DIRECT/RANGE_DIRECT/RANGE_DIRECT_SET/DENSE map onto the
 * u32 first[] direct-mapped table (the 1-bit SET and 2-bit DENSE packings
 * are CPU-cache idioms; with interval < 2^32 the u32 array is HBM-resident
 * and probes in ONE load — DESIGN.md (S)3), LINEAR_CHAINED(+SET) onto the
 * fp-packed linear-probed table, BUCKET_CHAINED onto first/next chains. */
int gpue_join_build_auto_i32(gpue_session* s, gpue_dbuf* keys, uint64_t row_count,
                             int mode, int with_other_conjunct, uint64_t l2_size,
                             uint64_t l3_size, gpue_join_table** out, int* method_out);
/* 8-byte-key auto build: selector decision with LT_BIGINT reported via
 * method_out; every physical tier maps onto the u64 bucket-chained table
 * (the u64 domain has no range-direct/linear specialization here). */
int gpue_join_build_auto_u64(gpue_session* s, gpue_dbuf* keys, uint64_t row_count,
                             int mode, int with_other_conjunct, uint64_t l2_size,
                             uint64_t l3_size, gpue_join_table** out, int* method_out);
void gpue_join_table_destroy(gpue_join_table* t);
int gpue_join_table_minmax(gpue_join_table* t, int64_t* min_out, int64_t* max_out);
/* d2h the first[] array (tests) */
int gpue_join_table_first_d2h(gpue_join_table* t, uint32_t* dst, uint64_t n_entries);

/* ---- hash-join probe ----
 * Replaces lookup_init + _probe_from_ht chain-walk emit
 * (reference be/src/exec/join/join_hash_map_method.hpp:88-120,
 * join_hash_map.hpp:717-795). Emits all (probe_idx, build_idx) match pairs;
 * out buffers must be sized for the match count (call with out_* NULL to get
 * the count first — the CPU's chunk_size-resumable cursor maps to this
 * two-phase count/emit, SURVEY.md §7 hard part (c)). */
int gpue_join_probe_emit_i32(gpue_session* s, gpue_join_table* t, gpue_dbuf* probe_keys,
                             uint64_t n_rows, gpue_dbuf* out_probe_idx,
                             gpue_dbuf* out_build_idx, uint64_t* match_count);
/* Per-join-type probe (join_hash_map.h:228-333): mode 0 INNER, 1 LEFT_SEMI,
 * 2 LEFT_ANTI (unmatched rows, build index 0 = NULL sentinel), 3 LEFT_OUTER. */
int gpue_join_probe_emit_mode_i32(gpue_session* s, gpue_join_table* t, gpue_dbuf* probe_keys,
                                  uint64_t n_rows, int mode, gpue_dbuf* out_probe_idx,
                                  gpue_dbuf* out_build_idx, uint64_t* match_count);
/* Nullable variants (NullableColumn is_nulls path,
 * join_hash_map_method.hpp:56-120): null build rows never enter a chain;
 * null probe rows match nothing (ANTI/OUTER emit them as unmatched).
 * mode 6 = NULL_AWARE_LEFT_ANTI (the NOT IN lowering, hash_joiner.cpp:97,
 * join_hash_map.hpp:1225-1240): as LEFT_ANTI but null probe rows are
 * EXCLUDED from the output — NULL NOT IN (...) is never true. Also accepted
 * by the varchar nulls probe. */
int gpue_join_build_bucket_chained_nulls_u32(gpue_session* s, gpue_dbuf* keys,
                                             gpue_dbuf* is_nulls /*u8, 1-based*/,
                                             uint64_t row_count, gpue_join_table** out);
int gpue_join_probe_emit_nulls_i32(gpue_session* s, gpue_join_table* t, gpue_dbuf* probe_keys,
                                   gpue_dbuf* probe_nulls /*u8*/, uint64_t n_rows, int mode,
                                   gpue_dbuf* out_probe_idx, gpue_dbuf* out_build_idx,
                                   uint64_t* match_count);
/* Multi-column key packing (SERIALIZED_FIXED_SIZE_BIGINT,
 * join_hash_map_helper.h:112-136): two int32 key columns -> one 8-byte key. */
int gpue_pack_keys_2xi32(gpue_session* s, gpue_dbuf* a, gpue_dbuf* b, uint64_t n,
                         gpue_dbuf* out);
/* two i64 key columns packed into one 16-byte key
 * (SERIALIZED_FIXED_SIZE_LARGEINT, join_key_constructor.h:40-153) */
int gpue_pack_keys_2xi64(gpue_session* s, gpue_dbuf* a, gpue_dbuf* b, uint64_t n,
                         gpue_dbuf* out);
/* SERIALIZED_VARCHAR / Slice keys (the selector's remaining constructor
 * branch, join_hash_map.cpp:269-281 -> JoinKeyHash<Slice>,
 * join_hash_map_helper.h:57-64: crc_hash_32(bytes,len,CRC_HASH_SEED1) masked
 * to the bucket count; probe verifies with a byte compare along the chain).
 * Columns are BinaryColumn-shaped: a bytes buffer + uint32 offsets
 * (binary_column.h:458-459). Build rows are 1-based with row 0 the empty
 * sentinel, so offsets has row_count+2 entries; probe rows are 0-based with
 * n_rows+1 entries. */
int gpue_join_build_varchar(gpue_session* s, gpue_dbuf* bytes, gpue_dbuf* offsets,
                            uint64_t row_count, gpue_join_table** out);
int gpue_join_probe_emit_varchar(gpue_session* s, gpue_join_table* t, gpue_dbuf* pbytes,
                                 gpue_dbuf* poffsets, uint64_t n_rows,
                                 gpue_dbuf* out_probe_idx, gpue_dbuf* out_build_idx,
                                 uint64_t* match_count);
/* Per-join-type Slice probe: mode 0 INNER, 1 LEFT_SEMI, 2 LEFT_ANTI,
 * 3 LEFT_OUTER (join_hash_map.h:228-333 semantics as for the i32 paths). */
int gpue_join_probe_emit_varchar_mode(gpue_session* s, gpue_join_table* t, gpue_dbuf* pbytes,
                                      gpue_dbuf* poffsets, uint64_t n_rows, int mode,
                                      gpue_dbuf* out_probe_idx, gpue_dbuf* out_build_idx,
                                      uint64_t* match_count);
/* Nullable Slice keys (is_nulls semantics as the fixed-size paths:
 * join_hash_map_method.hpp:56-120 — null build rows never chain, null probe
 * rows match nothing; ANTI/OUTER emit them unmatched). is_nulls u8, build
 * side 1-based. */
int gpue_join_build_varchar_nulls(gpue_session* s, gpue_dbuf* bytes, gpue_dbuf* offsets,
                                  gpue_dbuf* is_nulls, uint64_t row_count,
                                  gpue_join_table** out);
int gpue_join_probe_emit_varchar_nulls(gpue_session* s, gpue_join_table* t, gpue_dbuf* pbytes,
                                       gpue_dbuf* poffsets, gpue_dbuf* probe_nulls,
                                       uint64_t n_rows, int mode, gpue_dbuf* out_probe_idx,
                                       gpue_dbuf* out_build_idx, uint64_t* match_count);
/* Accumulating config-2 step: NO per-step accumulator reset — the caller
 * keeps a persistent i64[2] acc and differences successive readbacks.
 * Removes one launch from the 2-op step (SF10 is launch-cost-sensitive). */
int gpue_q1_join_sum_accum(gpue_session* s, gpue_join_table* dates, gpue_dbuf* od,
                           gpue_dbuf* ep, gpue_dbuf* dc, uint64_t n, gpue_dbuf* acc);

typedef struct gpue_agg_table gpue_agg_table; /* defined with the agg section below */

/* ---- streaming pre-aggregation building blocks ----
 * The AUTO-mode streaming agg (aggregate_streaming_sink_operator.cpp:224-310;
 * thresholds aggregator.h:175-178: LowReduction 0.2, HighReduction 0.9,
 * StableLimit 5) over the persistent gpue_agg_table. push = build_hash_map +
 * compute_batch_agg_states (cnts NULL: each row counts 1; non-NULL: rows are
 * pre-aggregated partials, merge_batch semantics aggregate.h:158-168;
 * update_only=1 aggregates only rows whose group exists and marks the rest
 * in miss_mask — the selective pre-agg form); probe_hits =
 * build_hash_map_with_selection's hit count; emit = convert_hash_map_to_chunk.
 * update_only is a FLAGS word: bit 0 = update-only (above); bit 1 =
 * GPUE_AGG_SUM_ONLY — skip the per-row group-count atomic for SUM-only
 * aggregations (the insert leg is atomic-rate bound,
 * profiles/r02_agg_card_ubench.json; a query with no COUNT/AVG function
 * needs no counts and the fused kernels already omit them). Under
 * SUM_ONLY, emit's out_counts for groups touched only by such pushes
 * read 0. */
#define GPUE_AGG_SUM_ONLY 2
int gpue_hash_agg_push_u64(gpue_session* s, gpue_agg_table* at, gpue_dbuf* keys,
                           gpue_dbuf* vals, gpue_dbuf* cnts, uint64_t n, int update_only,
                           gpue_dbuf* miss_mask, uint64_t* hits_out);
int gpue_hash_agg_probe_hits_u64(gpue_session* s, gpue_agg_table* at, gpue_dbuf* keys,
                                 uint64_t n, uint64_t* hits_out);
int gpue_hash_agg_emit_u64(gpue_session* s, gpue_agg_table* at, gpue_dbuf* out_keys,
                           gpue_dbuf* out_sums, gpue_dbuf* out_counts, uint64_t max_out,
                           uint64_t* n_groups);

/* Dictionary-encoded binary page decode (binary_dict_page.cpp:229-280): the
 * data page's int32 codewords (bitshuffle layer via
 * gpue_page_decode_bshuf_lz4_i32) index the dict page's distinct strings
 * (binary_plain_page.h string_at_index); output is a BinaryColumn (bytes +
 * uint32 offsets, 0-based rows). Call with out_* NULL for the byte count. */
int gpue_dict_decode_binary(gpue_session* s, gpue_dbuf* dict_bytes, gpue_dbuf* dict_offsets,
                            gpue_dbuf* codes, uint64_t n, gpue_dbuf* out_bytes,
                            gpue_dbuf* out_offsets, uint64_t* total_bytes);

/* ---- SimdBlockFilter runtime filter ----
 * The reference's split-block bloom (runtime_filter.h:79-232, upstream
 * fastfilter_cpp): 32-byte buckets of 8 uint32 lanes, one bit per lane from
 * (key*SALT[i])>>27; inserted hash = phmap_mix<8>(value) for integer keys
 * (runtime_filter.h:1271-1276). directory must hold 32<<log_num_buckets
 * bytes, where log_num_buckets = max(1, ceil(log2(n))-5)
 * (runtime_filter.cpp:26-36). Build is atomicOr — the directory is
 * bit-identical to the reference's serial build. test writes one u8 per row
 * (1 = maybe-member): the scan-side early prune pushed to probe operators
 * (operator.h:188-199). */
int gpue_sbf_build_i32(gpue_session* s, gpue_dbuf* keys, uint64_t n,
                       int32_t log_num_buckets, gpue_dbuf* directory);
int gpue_sbf_test_i32(gpue_session* s, gpue_dbuf* keys, uint64_t n, gpue_dbuf* directory,
                      int32_t log_num_buckets, gpue_dbuf* out);
/* RIGHT SEMI (anti=0) / RIGHT ANTI (anti=1) over Slice keys. */
int gpue_join_probe_right_varchar(gpue_session* s, gpue_join_table* t, gpue_dbuf* pbytes,
                                  gpue_dbuf* poffsets, uint64_t n_rows, int anti,
                                  gpue_dbuf* out_build_idx, uint64_t* count);
/* RIGHT SEMI (anti=0) / RIGHT ANTI (anti=1): matched/unmatched BUILD rows. */
int gpue_join_probe_right_i32(gpue_session* s, gpue_join_table* t, gpue_dbuf* probe_keys,
                              uint64_t n_rows, int anti, gpue_dbuf* out_build_idx,
                              uint64_t* count);

/* ---- fused probe + aggregate pipelines ----
 * Replace the probe -> output gather -> Aggregator::update_batch chain
 * (reference be/src/exec/join/join_hash_map.hpp:284-330,
 * be/src/exec/aggregator.cpp:937-959, be/src/exprs/agg/sum.h:45-181) with one
 * fused kernel per BASELINE config. Sums are int64, order-independent =>
 * bit-exact vs the oracle. */
/* Config 2 (SSB SF10 Q1-shaped): SUM(lo_extendedprice*lo_discount) over rows
 * whose lo_orderdate matches the (pre-filtered) date dim payload table. */
int gpue_q1_join_sum(gpue_session* s, gpue_join_table* dates, gpue_dbuf* lo_orderdate,
                     gpue_dbuf* lo_extendedprice, gpue_dbuf* lo_discount, uint64_t n_rows,
                     int64_t* sum_out, uint64_t* match_count_out);
/* Async bench variant: zero + launch into a caller-owned 16-byte accumulator
 * ({int64 sum, uint64 count}); no sync, no transient allocation. */
int gpue_q1_join_sum_async(gpue_session* s, gpue_join_table* dates, gpue_dbuf* lo_orderdate,
                           gpue_dbuf* lo_extendedprice, gpue_dbuf* lo_discount,
                           uint64_t n_rows, gpue_dbuf* acc);
/* Config 3 (SSB SF100 Q2.1-shaped): 3-way star probe + GROUP BY
 * (d_year,p_brand) -> dense group id (year_idx*1000+brand), SUM(lo_revenue).
 * group_sums_out must hold 7000 int64. */
int gpue_q21_star_agg(gpue_session* s, gpue_join_table* parts, gpue_join_table* supps,
                      gpue_join_table* dates, gpue_dbuf* lo_partkey, gpue_dbuf* lo_suppkey,
                      gpue_dbuf* lo_orderdate, gpue_dbuf* lo_revenue, uint64_t n_rows,
                      int64_t* group_sums_out);

/* Two-stream pipelined variant: K_A (part runtime-filter probe + brand
 * payload -> u16 per row, brand_scratch) overlaps K_B (remaining streams +
 * supplier/date probes + group agg) across n_chunks chunks — the two legs
 * measured fully additive inside one kernel. */
int gpue_q21_star_agg_pipe(gpue_session* s, gpue_join_table* parts, gpue_join_table* supps,
                           gpue_join_table* dates, gpue_dbuf* lo_partkey, gpue_dbuf* lo_suppkey,
                           gpue_dbuf* lo_orderdate, gpue_dbuf* lo_revenue, uint64_t n_rows,
                           gpue_dbuf* brand_scratch /* n_rows × u16 */, gpue_dbuf* group_sums,
                           int n_chunks);
/* Async bench variant of the star aggregate (7000 × int64 device buffer). */
int gpue_q21_star_agg_async(gpue_session* s, gpue_join_table* parts, gpue_join_table* supps,
                            gpue_join_table* dates, gpue_dbuf* lo_partkey, gpue_dbuf* lo_suppkey,
                            gpue_dbuf* lo_orderdate, gpue_dbuf* lo_revenue, uint64_t n_rows,
                            gpue_dbuf* group_sums);

/* Config 4 (SSB Q4.3): 4-way star join + GROUP BY (d_year,s_city,p_brand) as
 * compact filtered indexes -> 800 groups; SUM(lo_revenue - lo_supplycost).
 * group_sums holds 800 x int64 on device. */
int gpue_q43_star_agg_async(gpue_session* s, gpue_join_table* custs, gpue_join_table* supps,
                            gpue_join_table* parts, gpue_join_table* dates,
                            gpue_dbuf* lo_custkey, gpue_dbuf* lo_suppkey,
                            gpue_dbuf* lo_partkey, gpue_dbuf* lo_orderdate,
                            gpue_dbuf* lo_revenue, gpue_dbuf* lo_supplycost,
                            uint64_t n_rows, gpue_dbuf* group_sums);

/* ---- exchange partition (shuffle groundwork for configs 4-5) ----
 * Replaces the ExchangeSinkOperator partition stage
 * (reference be/src/exec/pipeline/exchange/exchange_sink_operator.cpp:611-660,
 * shuffler.h:71-86): per-row FNV hash -> ReduceOp channel -> counting-sort
 * row layout, bit-identical to the reference's (stable: each channel's rows
 * ascend by source row). start_points has num_channels+1 entries. */
/* Version-1 exchange hash (xxh3) partition — the
 * `_exchange_hash_function_version == 1` branch of
 * exchange_sink_operator.cpp:604-610 (XXH3_64bits_withSeed per key value,
 * seed XXH3_SEED_32, truncated u32; restated from the published XXH3 spec
 * and pinned to python-xxhash vectors in tests/golden/xxh3_kats.json).
 * FNV (gpue_partition_i32) stays the reference's backward-compatible
 * default. */
int gpue_partition_xxh3_i32(gpue_session* s, gpue_dbuf* keys, uint64_t n,
                            uint32_t num_channels, uint64_t* start_points_out,
                            gpue_dbuf* row_indexes_out);
/* The bucket-shuffle hash path (zlib crc32, seed 0 — the third of the
 * exchange's three hash functions, exchange_sink_operator.cpp:617-622;
 * pinned live against python zlib). */
int gpue_partition_crc_i32(gpue_session* s, gpue_dbuf* keys, uint64_t n,
                           uint32_t num_channels, uint64_t* start_points_out,
                           gpue_dbuf* row_indexes_out);
/* Varchar (BinaryColumn) partition key: FNV over the slice bytes. */
int gpue_partition_varchar(gpue_session* s, gpue_dbuf* bytes, gpue_dbuf* offsets,
                           uint64_t n, uint32_t num_channels,
                           uint64_t* start_points_out, gpue_dbuf* row_indexes_out);
/* Steady-state async partition: hist -> DEVICE-side scan -> emit with no
 * host round-trip (the sync forms host-scan the histogram to return split
 * sizes; in the chunked exchange the splits are static per shard, so the
 * timed step only needs row_indexes). scratch holds the per-(block,channel)
 * histogram + offsets (>= grid*channels*12 bytes). */
int gpue_partition_i32_async(gpue_session* s, gpue_dbuf* keys, uint64_t n,
                             uint32_t num_channels, gpue_dbuf* row_indexes_out,
                             gpue_dbuf* scratch);
int gpue_partition_i64_async(gpue_session* s, gpue_dbuf* keys, uint64_t n,
                             uint32_t num_channels, gpue_dbuf* row_indexes_out,
                             gpue_dbuf* scratch);
int gpue_partition_i32(gpue_session* s, gpue_dbuf* keys, uint64_t n, uint32_t num_channels,
                       uint64_t* start_points_out, gpue_dbuf* row_indexes_out);
/* Multi-column partition key: the sink seeds FNV_SEED then CHAINS fnv_hash
 * per partition column, each column seeding with the running hash
 * (exchange_sink_operator.cpp:611-617). Two-int32-column variant. */
int gpue_partition_2xi32(gpue_session* s, gpue_dbuf* a, gpue_dbuf* b, uint64_t n,
                         uint32_t num_channels, uint64_t* start_points_out,
                         gpue_dbuf* row_indexes_out);

/* ---- config 5: TPC-H Q3-shaped ----
 * lineitem ⋈ orders ⋈ customer; c_mktsegment = 16-byte dictionary string
 * (SERIALIZED_FIXED_SIZE_LARGEINT packing, join_hash_table.cpp:185-192);
 * revenue = extendedprice(cents) × (100−discount), scale-4 decimal int64;
 * GROUP BY l_orderkey (high cardinality) via the hash aggregate below. */
int gpue_gen_lineitem_q3(gpue_session* s, uint64_t seed, uint64_t row_start, uint64_t n,
                         uint64_t n_orders, gpue_dbuf* l_orderkey /*i64*/,
                         gpue_dbuf* l_extendedprice /*i64*/, gpue_dbuf* l_discount /*i64*/,
                         gpue_dbuf* l_shipdate /*i32*/);
int gpue_gen_orders_q3(gpue_session* s, uint64_t seed, uint64_t n_orders, uint32_t n_custs,
                       gpue_dbuf* o_custkey, gpue_dbuf* o_orderdate);
int gpue_gen_cust_mkt16(gpue_session* s, uint64_t seed, uint32_t n_custs, gpue_dbuf* out);
/* 16-byte fixed-string equality -> membership bitset (two u64 compares) */
int gpue_bits_str16_eq(gpue_session* s, gpue_dbuf* col16, uint64_t n, const void* lit16,
                       gpue_dbuf* bits);
/* orders pass bitset: o_orderdate < cutoff AND customer bit set */
int gpue_q3_order_bits(gpue_session* s, gpue_dbuf* o_custkey, gpue_dbuf* o_orderdate,
                       uint64_t n_orders, gpue_dbuf* cust_bits, int32_t cutoff,
                       gpue_dbuf* order_bits);
/* persistent aggregate table (per-query hash map analog, reused across
 * passes so repeated executions pay reset, not allocation) */
int gpue_agg_table_create(gpue_session* s, uint64_t capacity, gpue_agg_table** out);
void gpue_agg_table_destroy(gpue_agg_table* t);
/* fused lineitem filter + orders semi-probe + hash aggregate */
int gpue_q3_probe_agg(gpue_session* s, gpue_dbuf* lk, gpue_dbuf* ext, gpue_dbuf* disc,
                      gpue_dbuf* ship, uint64_t n, gpue_dbuf* order_bits, int32_t ship_cutoff,
                      uint64_t capacity_hint, gpue_dbuf* out_keys, gpue_dbuf* out_sums,
                      uint64_t max_out, uint64_t* n_groups);
/* variant over a persistent gpue_agg_table (reset + probe + emit) */
int gpue_q3_probe_agg_t(gpue_session* s, gpue_dbuf* lk, gpue_dbuf* ext, gpue_dbuf* disc,
                        gpue_dbuf* ship, uint64_t n, gpue_dbuf* order_bits,
                        int32_t ship_cutoff, gpue_agg_table* at, gpue_dbuf* out_keys,
                        gpue_dbuf* out_sums, uint64_t max_out, uint64_t* n_groups);

/* ---- generic hash aggregate ----
 * Replaces AggHashMapWithKey::compute_agg_states + update_batch +
 * convert_hash_map_to_chunk (reference be/src/exec/agg_hash_map.h:112-290,
 * aggregator.cpp:937-959,1742-1816) for HIGH-cardinality GROUP BY (group-key
 * packed <= 8 B, as the SERIALIZED_FIXED_SIZE key constructors pack). SUM +
 * COUNT states; emission order is table order (results are a set). Key
 * sentinel ~0ull must not occur in keys. */
int gpue_hash_agg_sum_u64(gpue_session* s, gpue_dbuf* keys /*u64*/, gpue_dbuf* vals /*i64*/,
                          uint64_t n, uint64_t capacity_hint, gpue_dbuf* out_keys,
                          gpue_dbuf* out_sums, gpue_dbuf* out_counts /*nullable*/,
                          uint64_t max_out, uint64_t* n_groups);

/* Full aggregate-function states: SUM + COUNT + MIN + MAX (AVG = SUM/COUNT
 * at finalize, reference exprs/agg/aggregate.h:136-269). */
int gpue_hash_agg_stats_u64(gpue_session* s, gpue_dbuf* keys, gpue_dbuf* vals, uint64_t n,
                            uint64_t capacity_hint, gpue_dbuf* out_keys, gpue_dbuf* out_sums,
                            gpue_dbuf* out_counts, gpue_dbuf* out_mins, gpue_dbuf* out_maxs,
                            uint64_t max_out, uint64_t* n_groups);
/* Decimal SUM widened to int128 (exprs/agg/sum.h:181): lo/hi pair with an
 * explicit carry, exact mod 2^128, order-independent. */
int gpue_hash_agg_sum128_u64(gpue_session* s, gpue_dbuf* keys, gpue_dbuf* vals, uint64_t n,
                             uint64_t capacity_hint, gpue_dbuf* out_keys, gpue_dbuf* out_lo,
                             gpue_dbuf* out_hi, uint64_t max_out, uint64_t* n_groups);

/* Row gather by index — the exchange sink's add_rows_selective analog
 * (exchange_sink_operator.cpp:670): materializes per-channel row slices at
 * the counting-sorted indexes before the RCCL all-to-all. */
int gpue_gather_u32(gpue_session* s, gpue_dbuf* in, gpue_dbuf* idx, uint64_t n,
                    gpue_dbuf* out);
int gpue_gather_u64(gpue_session* s, gpue_dbuf* in, gpue_dbuf* idx, uint64_t n,
                    gpue_dbuf* out);
/* BIGINT-key partition (FNV over the 8 LE bytes of an int64 key column) */
int gpue_partition_i64(gpue_session* s, gpue_dbuf* keys, uint64_t n, uint32_t num_channels,
                       uint64_t* start_points_out, gpue_dbuf* row_indexes_out);

/* ---- storage ingress (the step upstream of the scan) ----
 * Decode the reference's numeric page body (bitshuffle+LZ4,
 * be/src/storage/rowset/bitshuffle_page.h framing after the 16-byte header;
 * bitshuffle 0.5.1 published algorithm + LZ4 block format restated in
 * oracle/oracle.c): n_values int32 (multiple of 8) from the device-resident
 * page bytes. */
int gpue_page_decode_bshuf_lz4_i32(gpue_session* s, gpue_dbuf* page, uint32_t n_values,
                                   gpue_dbuf* out);

/* ---- TopN ----
 * ORDER BY value DESC LIMIT k (reference exec/chunks_sorter_topn.cpp):
 * deterministic (value, key) lexicographic descending; k <= 16. */
int gpue_topk_i64(gpue_session* s, gpue_dbuf* keys, gpue_dbuf* vals, uint64_t n,
                  int k, uint64_t* out_keys, int64_t* out_vals);

/* ---- chunked-exchange overlap support (SURVEY.md §7 hard part (d)) ------
 * The session's HIP stream as an opaque pointer (wrap as a torch
 * ExternalStream to event-order RCCL collectives against engine kernels),
 * plus accumulate-only forms of the probe+agg steps so a received row-block
 * can be probed while the next block's all-to-all is in flight — the analog
 * of the reference's overlapped SinkBuffer (sink_buffer.cpp:533-536). */
void* gpue_session_stream(gpue_session* s);
int gpue_q43_star_agg_accum_async(gpue_session* s, gpue_join_table* custs,
                                  gpue_join_table* supps, gpue_join_table* parts,
                                  gpue_join_table* dates, gpue_dbuf* ck, gpue_dbuf* sk,
                                  gpue_dbuf* pk, gpue_dbuf* od, gpue_dbuf* rv,
                                  gpue_dbuf* sc, uint64_t n, gpue_dbuf* group_sums);
int gpue_q3_probe_accum(gpue_session* s, gpue_dbuf* lk, gpue_dbuf* ext, gpue_dbuf* disc,
                        gpue_dbuf* ship, uint64_t n, gpue_dbuf* order_bits,
                        int32_t ship_cutoff, gpue_agg_table* at);

/* ---- growable agg table ----
 * The MI355X analog of the reference's two-level conversion
 * (Aggregator::try_convert_to_two_level_map, aggregator.cpp:1237-1241;
 * agg_hash_variant.cpp:318): size() reads the claimed-group counter;
 * ensure() doubles + rehashes on device until a push of additional_rows can
 * never overflow (load factor kept <= 5/8). Call before each chunk push,
 * as the reference checks before each chunk. */
int gpue_agg_table_size(gpue_session* s, gpue_agg_table* t, uint64_t* n_groups);
int gpue_agg_table_ensure(gpue_session* s, gpue_agg_table* t, uint64_t additional_rows);

/* ---- pinned double-buffered H2D ingest ----
 * north_star's "columnar batches pinned and streamed to HBM": two
 * hipHostMalloc staging buffers on the session's second stream; while chunk
 * k DMAs pinned->HBM, the host fills the other buffer — the morsel-driven
 * async-io shape of the reference's scan operator
 * (be/src/exec/pipeline/scan/scan_operator.h:40). */
typedef struct gpue_ingest gpue_ingest;
int gpue_ingest_create(gpue_session* s, uint64_t chunk_bytes, gpue_ingest** out);
/* page-locked host memory for producers that fill batches in place (zero
 * staging copy; DMA at full PCIe rate) */
int gpue_pinned_alloc(gpue_session* s, uint64_t bytes, void** host_ptr);
void gpue_pinned_free(void* host_ptr);
int gpue_ingest_push(gpue_ingest* g, const void* host, uint64_t bytes, gpue_dbuf* dst,
                     uint64_t dst_off);
int gpue_ingest_sync(gpue_ingest* g);
void gpue_ingest_destroy(gpue_ingest* g);

/* RLE page decode for int32 (storage ingress):
 * storage/rowset/rle_page.h header + base/bit/rle_encoding.h's
 * Parquet-style RLE/bit-pack hybrid at bit_width 32 (byte-aligned runs).
 * Two-phase device decode: run-table scan + binary-search parallel fill. */
int gpue_page_decode_rle_i32(gpue_session* s, gpue_dbuf* page, uint64_t n_values,
                             gpue_dbuf* out);
/* BOOL variant (bit_width 1, bit-packed literal groups; u8 output) */
int gpue_page_decode_rle_bool(gpue_session* s, gpue_dbuf* page, uint64_t n_values,
                              gpue_dbuf* out);
/* Frame-of-reference page decode (FOR_ENCODING,
 * frame_of_reference_coding.{h,cpp}): independent 128-value frames
 * (LE min + MSB-first bit-packed deltas; formats 0 min-delta / 1 ascending
 * prefix / 2 raw), decoded one block per frame with an LDS scan for the
 * ascending format. */
int gpue_page_decode_for_i32(gpue_session* s, gpue_dbuf* page, uint64_t n_values,
                             gpue_dbuf* out);
/* PlainPage numeric decode (plain_page.h:51,83-102,148-158): u32 LE count
 * + raw LE values; the fallback numeric encoding (encoding_info.cpp). */
int gpue_page_decode_plain_i32(gpue_session* s, gpue_dbuf* page, uint64_t n_values,
                               gpue_dbuf* out);
/* BinaryPlainPage -> BinaryColumn (binary_plain_page.h:28-46: string body +
 * u32 absolute-offset trailer + count). The dict page's dictionary format. */
int gpue_page_decode_binary_plain(gpue_session* s, gpue_dbuf* page, uint64_t n_values,
                                  gpue_dbuf* out_bytes, gpue_dbuf* out_offsets);
/* BinaryPrefixPage -> BinaryColumn (PREFIX_ENCODING,
 * binary_prefix_page.{h,cpp}: front coding, restart every 16 entries;
 * decode parallelizes per restart group). */
int gpue_page_decode_binary_prefix(gpue_session* s, gpue_dbuf* page, uint64_t n_values,
                                   gpue_dbuf* out_bytes, gpue_dbuf* out_offsets);

/* ---- ASOF join (reference LinearChainedAsofJoinHashMap,
 * join_hash_map_method.h:201-217 + AsofIndex, join_hash_table_descriptor.h:
 * 59-104 / .cpp:70-134). The selector routes every ASOF join here
 * (_get_fallback_method, join_hash_table.cpp:257-263). Equi-key lookup plus a
 * per-key temporal index sorted ascending (LT/LE) or descending (GT/GE),
 * probed with the reference's branchless lower-bound search; at most one
 * build match per probe row. GPU layout: open-addressing key slots + one
 * contiguous (asof_value, row) segment per key — binary search wants
 * contiguous sorted runs, not the CPU's pointer chains. Tie order among
 * duplicate (key, asof) pairs is unspecified in the reference (pdqsort is
 * unstable, comparator reads only asof_value); we pin the deterministic
 * refinement "smallest build row wins the boundary slot". */
#define GPUE_ASOF_LT 0 /* probe <  build: match smallest build value >  probe */
#define GPUE_ASOF_LE 1 /* probe <= build: match smallest build value >= probe */
#define GPUE_ASOF_GT 2 /* probe >  build: match largest  build value <  probe */
#define GPUE_ASOF_GE 3 /* probe >= build: match largest  build value <= probe */
typedef struct gpue_asof_table gpue_asof_table;
/* keys: (row_count+1) int32 equi keys, asof: (row_count+1) int64 temporal
 * values; row 0 is the sentinel "no match" row (never matched), as in the
 * reference's 1-based build rows. opcode: GPUE_ASOF_*. */
int gpue_asof_build_i32(gpue_session* s, gpue_dbuf* keys, gpue_dbuf* asof,
                        uint64_t row_count, int opcode, gpue_asof_table** out);
/* mode: GPUE_JOIN_INNER (matched rows only) or GPUE_JOIN_LEFT_OUTER (miss
 * emits build index 0), the reference's ASOF_INNER / ASOF_LEFT_OUTER
 * (join_hash_table.cpp:744). Two-call contract like gpue_join_probe_emit:
 * null outputs -> count only. Output ordered by probe row. */
int gpue_asof_probe_emit_i32(gpue_session* s, gpue_asof_table* t, gpue_dbuf* probe_keys,
                             gpue_dbuf* probe_asof, uint64_t n_rows, int mode,
                             gpue_dbuf* out_probe_idx, gpue_dbuf* out_build_idx,
                             uint64_t* match_count);
/* Nullable variants: is_nulls is a (row_count+1) u8 mask — the caller ORs the
 * equi-key and temporal null masks, and flagged build rows are skipped
 * exactly as the reference's is_null_row does
 * (join_hash_table_descriptor.h:447-456); null probe rows never match
 * (INNER drops them, LEFT_OUTER emits build row 0). */
int gpue_asof_build_nulls_i32(gpue_session* s, gpue_dbuf* keys, gpue_dbuf* asof,
                              gpue_dbuf* is_nulls, uint64_t row_count, int opcode,
                              gpue_asof_table** out);
int gpue_asof_probe_emit_nulls_i32(gpue_session* s, gpue_asof_table* t,
                                   gpue_dbuf* probe_keys, gpue_dbuf* probe_asof,
                                   gpue_dbuf* probe_nulls, uint64_t n_rows, int mode,
                                   gpue_dbuf* out_probe_idx, gpue_dbuf* out_build_idx,
                                   uint64_t* match_count);
int gpue_asof_table_destroy(gpue_asof_table* t);

/* ---- event timing on the session stream (bench roofline evidence) ---- */
int gpue_timer_start(gpue_session* s);
int gpue_timer_stop(gpue_session* s, float* ms_out);

#ifdef __cplusplus
}
#endif
#endif
