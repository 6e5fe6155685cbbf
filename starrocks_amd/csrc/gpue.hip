// gpue.hip — MI355X-native (gfx950/CDNA4) execution engine for the StarRocks
// BE hot path. Implements the C-ABI declared in include/gpue.h (see that
// header and DESIGN.md §1 for the reference interfaces each entry replaces).
//
// Design notes: every kernel on this path is HBM-bound —
// grid-stride loops over 256-thread blocks with the grid capped near 2048
// blocks (≫256 CUs across 8 XCDs), coalesced loads, wave64 ballot/prefix-sum
// compaction, one atomic per wave/block for reductions. No MFMA: there is no
// dense contraction on this path. No CUDA shims, no CPU fallback.
//
// Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 -shared -fPIC \
//          -I../../include gpue.hip -o ../libgpue.so
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstring>
#include <cstdlib>
#include <thread>
#include "gpue.h"

// ---------------------------------------------------------------------------
// status / error plumbing
// ---------------------------------------------------------------------------
static thread_local char g_err[512] = {0};

const char* gpue_last_error(void) { return g_err; }

#define HIP_CHECK(expr)                                                      \
    do {                                                                     \
        hipError_t _e = (expr);                                              \
        if (_e != hipSuccess) {                                              \
            snprintf(g_err, sizeof(g_err), "%s failed: %s (%s:%d)", #expr,   \
                     hipGetErrorString(_e), __FILE__, __LINE__);             \
            return GPUE_ERR_HIP;                                             \
        }                                                                    \
    } while (0)

#define ARG_CHECK(cond)                                                      \
    do {                                                                     \
        if (!(cond)) {                                                       \
            snprintf(g_err, sizeof(g_err), "bad argument: %s (%s:%d)",       \
                     #cond, __FILE__, __LINE__);                             \
            return GPUE_ERR_ARG;                                             \
        }                                                                    \
    } while (0)

static constexpr int BLOCK = 256;
static constexpr int WAVE = 64;
static constexpr uint64_t MAX_GRID = 2048; // Guideline 11: grid-stride past this

static uint32_t env_cap(const char* name, uint32_t dflt) {
    const char* e = getenv(name);
    int v = e ? atoi(e) : 0;
    return v > 0 ? (uint32_t)v : dflt;
}

static inline uint32_t grid_capped(uint64_t n, uint64_t cap) {
    uint64_t b = (n + BLOCK - 1) / BLOCK;
    if (b > cap) b = cap;
    if (b == 0) b = 1;
    return (uint32_t)b;
}

// latency-bound kernels (chain walks, hash inserts): many blocks for wave
// occupancy (GPUE_GRID env overrides for experiments)
static inline uint32_t grid_for(uint64_t n) {
    static uint32_t cap = 0;
    if (!cap) cap = env_cap("GPUE_GRID", MAX_GRID);
    return grid_capped(n, cap);
}

// streaming-bound kernels: measured FASTEST at ~1 block/CU (256 blocks) —
// the q1 grid sweep (profiles/r01 q1_grid_sweep) runs 5205 GB/s @SF10 /
// 6342 GB/s @600 M rows at 256 blocks vs 2819/4895 at the 2048-block cap:
// fewer, longer-lived blocks stream long contiguous ranges per wave
static inline uint32_t grid_stream(uint64_t n) {
    static uint32_t cap = 0;
    if (!cap) cap = env_cap("GPUE_GRID_STREAM", 256);
    return grid_capped(n, cap);
}

// per-join-type count transform (reference join_hash_map.h:228-333):
// mode 0 INNER, 1 LEFT_SEMI, 2 LEFT_ANTI, 3 LEFT_OUTER
__device__ static inline uint32_t join_mode_count(uint32_t c, int mode) {
    switch (mode) {
    case 1: return c ? 1u : 0u;
    case 2: return c ? 0u : 1u;
    case 3: return c ? c : 1u;
    default: return c;
    }
}

// ---------------------------------------------------------------------------
// session / buffers
// ---------------------------------------------------------------------------
struct gpue_session {
    int device;
    hipStream_t stream;
    hipEvent_t ev_start, ev_stop;
    // cached SSB date dimension (datekey per day index), built lazily
    int32_t* d_datekey = nullptr; // device, N_DAYS entries
    // pinned bounce buffer: small result reads (the per-step accumulator
    // pull) pay ~5 µs instead of a pageable-copy ~20 µs
    void* pinned = nullptr;
    static constexpr uint64_t PINNED_BYTES = 65536;
    // second stream + event pool for pipelined two-kernel operators
    hipStream_t stream2 = nullptr;
    static constexpr int N_CHUNK_EVENTS = 32;
    hipEvent_t chunk_ev[N_CHUNK_EVENTS] = {};
    // device error latch for hash-aggregate kernels: bit 0 =
    // sentinel key (~0ull) seen in input — the reference aggregator accepts
    // every key value, our open-addressing table reserves ~0ull, so a real
    // ~0ull key must fail loudly, never silently drop its group; bit 1 =
    // probe walk exhausted the table (capacity_hint < distinct keys) —
    // bounded walk + error instead of an infinite-spin GPU hang
    unsigned int* d_agg_err = nullptr;
};
#define AGG_ERR_SENTINEL 1u
#define AGG_ERR_FULL 2u

struct gpue_dbuf {
    gpue_session* s;
    void* ptr;
    uint64_t bytes;
    bool owned = true; // false: wraps external device memory (e.g. a torch tensor)
};

struct gpue_join_table {
    gpue_session* s = nullptr;
    int64_t min_key = 0, max_key = 0;
    uint32_t* first = nullptr;   // device, (max-min+1) entries: payload or head row idx
    uint32_t* next = nullptr;    // device, (row_count+1) entries, or nullptr (payload variant)
    // Derived probe structures (payload variant only):
    uint32_t* bitset = nullptr;  // 1 bit per key in [set_min,set_max] — the
                                 // reference's RANGE_DIRECT_MAPPING_SET
                                 // (join_hash_table.cpp:297-303) for semi-join
                                 // probes, clamped to the PASSING keys' bounding
                                 // range (data-dependent specialization, like the
                                 // selector's interval checks) so it stages in LDS
    int64_t set_min = 0, set_max = -1; // bounding range of keys with payload != 0
    uint16_t* first16 = nullptr; // 16-bit payload copy when all payloads < 65536:
                                 // halves the random-gather footprint (L2 per XCD is 4 MiB)
    uint32_t* prefilter = nullptr; // 2^19-bit (64 KB) fold of `bitset` for
                                   // LDS-resident prefiltering (k_q21_star_agg_pf)
    uint32_t* prefilter2w = nullptr; // WIDE split fold (2 x 2^19 bits,
                                     // 128 KB): q43's 6.4 KB group array
                                     // leaves LDS room for full-size folds
                                     // (fpr ~1% vs the single fold's 6.1%)
    uint32_t* prefilter2 = nullptr; // split two-probe fold (2 x 2^18 bits):
                                    // words [0,8192) = i & MASK18,
                                    // [8192,16384) = (i*2654435761)>>14 —
                                    // the bloom-k=2 experiment (mode 6)
    uint2* dense_groups = nullptr; // DENSE_RANGE_DIRECT: per-32-key group
                                   // {start_index, bitset} (rank/select slot map,
                                   // join_hash_map_method.h:378, .hpp:781-904)
    uint64_t bucket_size = 0;
    uint64_t row_count = 0;
    // method discriminator — the GPU analog of JoinHashMapSelector's choice
    // (reference join_hash_table.cpp:164-344)
    enum Kind { PAYLOAD = 0, RANGE_DIRECT = 1, BUCKET_CHAINED = 2,
                DENSE_RANGE_DIRECT = 7, BUCKET_CHAINED128 = 8,
                LINEAR_CHAINED = 3, VARCHAR = 4, BUCKET_CHAINED64 = 5 } kind = PAYLOAD;
    uint32_t log_bucket_size = 0;
    uint32_t* build_keys = nullptr; // chained methods keep the build keys for the
                                    // probe-side equality check (1-based, row 0 sentinel)
    uint8_t* key_bytes = nullptr;   // VARCHAR: BinaryColumn bytes + uint32 offsets
    uint32_t* key_offsets = nullptr;
    uint8_t* key_nulls = nullptr;   // VARCHAR nullable: is_nulls (1-based), or null
    uint64_t* build_keys64 = nullptr; // BUCKET_CHAINED64: 8-byte build keys
    ulonglong2* build_keys128 = nullptr; // BUCKET_CHAINED128: 16-byte build keys
};

int gpue_device_count(int* out) {
    int n = 0;
    hipError_t e = hipGetDeviceCount(&n);
    if (e != hipSuccess) n = 0;
    *out = n;
    return GPUE_OK;
}

int gpue_session_create(int device_index, gpue_session** out) {
    ARG_CHECK(out != nullptr);
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess || n == 0) {
        snprintf(g_err, sizeof(g_err), "no AMD GPU visible (hipGetDeviceCount)");
        return GPUE_ERR_NO_GPU;
    }
    ARG_CHECK(device_index >= 0 && device_index < n);
    HIP_CHECK(hipSetDevice(device_index));
    gpue_session* s = new gpue_session();
    s->device = device_index;
    HIP_CHECK(hipStreamCreate(&s->stream));
    HIP_CHECK(hipEventCreate(&s->ev_start));
    HIP_CHECK(hipEventCreate(&s->ev_stop));
    HIP_CHECK(hipHostMalloc(&s->pinned, gpue_session::PINNED_BYTES));
    HIP_CHECK(hipStreamCreate(&s->stream2));
    for (int i = 0; i < gpue_session::N_CHUNK_EVENTS; i++)
        HIP_CHECK(hipEventCreateWithFlags(&s->chunk_ev[i], hipEventDisableTiming));
    HIP_CHECK(hipMalloc(&s->d_agg_err, sizeof(unsigned int)));
    HIP_CHECK(hipMemset(s->d_agg_err, 0, sizeof(unsigned int)));
    *out = s;
    return GPUE_OK;
}

// Read + clear the session's aggregate-kernel error latch. Call sites sit
// right after a stream sync that already waits for the aggregating kernel.
static int agg_err_check(gpue_session* s, const char* what) {
    unsigned int e = 0;
    HIP_CHECK(hipMemcpy(&e, s->d_agg_err, sizeof(e), hipMemcpyDeviceToHost));
    if (e == 0) return GPUE_OK;
    HIP_CHECK(hipMemset(s->d_agg_err, 0, sizeof(e)));
    snprintf(g_err, sizeof(g_err), "%s: %s%s%s", what,
             (e & AGG_ERR_SENTINEL) ? "group key 0xFFFFFFFFFFFFFFFF collides with the "
                                      "empty-slot sentinel" : "",
             (e & (AGG_ERR_SENTINEL | AGG_ERR_FULL)) == (AGG_ERR_SENTINEL | AGG_ERR_FULL)
                 ? "; " : "",
             (e & AGG_ERR_FULL) ? "hash table full (capacity < distinct keys)" : "");
    return GPUE_ERR_ARG;
}

void gpue_session_destroy(gpue_session* s) {
    if (!s) return;
    if (s->d_agg_err) (void)hipFree(s->d_agg_err);
    if (s->pinned) (void)hipHostFree(s->pinned);
    for (int i = 0; i < gpue_session::N_CHUNK_EVENTS; i++)
        if (s->chunk_ev[i]) (void)hipEventDestroy(s->chunk_ev[i]);
    if (s->stream2) (void)hipStreamDestroy(s->stream2);
    if (s->d_datekey) (void)hipFree(s->d_datekey);
    (void)hipEventDestroy(s->ev_start);
    (void)hipEventDestroy(s->ev_stop);
    (void)hipStreamDestroy(s->stream);
    delete s;
}

int gpue_sync(gpue_session* s) {
    ARG_CHECK(s);
    HIP_CHECK(hipStreamSynchronize(s->stream));
    return GPUE_OK;
}

// The session's HIP stream as an opaque pointer — lets the host wrap it as
// a torch ExternalStream so RCCL collectives on a second stream can be
// event-ordered against engine kernels without full-device syncs (the
// exchange/compute overlap of SURVEY.md (S)7 hard part (d); the reference
// overlaps its sink the same way, sink_buffer.cpp:533-536).
extern "C" void* gpue_session_stream(gpue_session* s);
void* gpue_session_stream(gpue_session* s) { return s ? (void*)s->stream : nullptr; }

int gpue_dbuf_alloc(gpue_session* s, uint64_t bytes, gpue_dbuf** out) {
    ARG_CHECK(s && out && bytes > 0);
    HIP_CHECK(hipSetDevice(s->device));
    void* p = nullptr;
    HIP_CHECK(hipMalloc(&p, bytes));
    gpue_dbuf* b = new gpue_dbuf{s, p, bytes, true};
    *out = b;
    return GPUE_OK;
}

void gpue_dbuf_free(gpue_dbuf* b) {
    if (!b) return;
    if (b->owned) (void)hipFree(b->ptr);
    delete b;
}

// Wrap external device memory (a torch tensor's data_ptr) as a gpue_dbuf so
// kernels operate in place and torch.distributed (RCCL) moves the same
// buffers — the all-to-all leg of configs 4-5.
int gpue_dbuf_wrap(gpue_session* s, void* device_ptr, uint64_t bytes, gpue_dbuf** out) {
    ARG_CHECK(s && device_ptr && out && bytes > 0);
    gpue_dbuf* b = new gpue_dbuf{s, device_ptr, bytes, false};
    *out = b;
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// Pinned double-buffered H2D ingest — north_star's "columnar batches pinned
// and streamed to HBM"; the morsel-driven async-io shape of the reference's
// scan operator (be/src/exec/pipeline/scan/scan_operator.h:40: each morsel
// is staged and shipped while the next is prepared). Two hipHostMalloc
// staging buffers on the session's second stream: while chunk k DMAs
// pinned->HBM, the host memcpys chunk k+1 into the other buffer. Pageable
// single-shot h2d measured 56.5 GB/s on this path; pinned
// DMA is the PCIe-rate fix.
// ---------------------------------------------------------------------------
struct gpue_ingest {
    gpue_session* s;
    void* pin[2];
    uint64_t chunk_bytes;
    hipEvent_t done[2]; // DMA completion per staging buffer
    int cur;
};

extern "C" {
int gpue_ingest_create(gpue_session* s, uint64_t chunk_bytes, gpue_ingest** out);
int gpue_pinned_alloc(gpue_session* s, uint64_t bytes, void** host_ptr);
void gpue_pinned_free(void* host_ptr);
int gpue_ingest_push(gpue_ingest* g, const void* host, uint64_t bytes, gpue_dbuf* dst,
                     uint64_t dst_off);
int gpue_ingest_sync(gpue_ingest* g);
void gpue_ingest_destroy(gpue_ingest* g);
}

// Page-locked host memory for zero-copy-staging producers: a caller that
// writes its batches directly into pinned memory (the scan io-task model)
// DMAs at the full PCIe rate with no staging memcpy at all.
int gpue_pinned_alloc(gpue_session* s, uint64_t bytes, void** host_ptr) {
    ARG_CHECK(s && host_ptr && bytes > 0);
    HIP_CHECK(hipHostMalloc(host_ptr, bytes));
    return GPUE_OK;
}

void gpue_pinned_free(void* host_ptr) {
    if (host_ptr) (void)hipHostFree(host_ptr);
}

int gpue_ingest_create(gpue_session* s, uint64_t chunk_bytes, gpue_ingest** out) {
    ARG_CHECK(s && out && chunk_bytes >= 4096);
    gpue_ingest* g = new gpue_ingest{s, {nullptr, nullptr}, chunk_bytes, {}, 0};
    for (int i = 0; i < 2; i++) {
        HIP_CHECK(hipHostMalloc(&g->pin[i], chunk_bytes));
        HIP_CHECK(hipEventCreateWithFlags(&g->done[i], hipEventDisableTiming));
        HIP_CHECK(hipEventRecord(g->done[i], s->stream2)); // initially signalled
    }
    *out = g;
    return GPUE_OK;
}

int gpue_ingest_push(gpue_ingest* g, const void* host, uint64_t bytes, gpue_dbuf* dst,
                     uint64_t dst_off) {
    ARG_CHECK(g && host && dst && dst_off + bytes <= dst->bytes);
    const uint8_t* src = (const uint8_t*)host;
    uint64_t off = 0;
    while (off < bytes) {
        uint64_t sz = bytes - off < g->chunk_bytes ? bytes - off : g->chunk_bytes;
        int c = g->cur;
        // wait for this staging buffer's previous DMA before refilling it —
        // the OTHER buffer's DMA proceeds concurrently with this memcpy
        HIP_CHECK(hipEventSynchronize(g->done[c]));
        // staging copy: a single-threaded memcpy into pinned memory measured
        // ~32 GB/s and throttled the whole pipeline below the pageable path;
        // split big chunks across 4 threads
        if (sz >= (8u << 20)) {
            const int T = 4;
            uint64_t part = (sz + T - 1) / T;
            std::thread th[T];
            for (int t = 0; t < T; t++) {
                uint64_t lo = t * part, hi = lo + part < sz ? lo + part : sz;
                th[t] = std::thread([&, lo, hi, c]() {
                    if (lo < hi) memcpy((uint8_t*)g->pin[c] + lo, src + off + lo, hi - lo);
                });
            }
            for (auto& t : th) t.join();
        } else {
            memcpy(g->pin[c], src + off, sz);
        }
        HIP_CHECK(hipMemcpyAsync((uint8_t*)dst->ptr + dst_off + off, g->pin[c], sz,
                                 hipMemcpyHostToDevice, g->s->stream2));
        HIP_CHECK(hipEventRecord(g->done[c], g->s->stream2));
        g->cur ^= 1;
        off += sz;
    }
    return GPUE_OK;
}

int gpue_ingest_sync(gpue_ingest* g) {
    ARG_CHECK(g);
    HIP_CHECK(hipStreamSynchronize(g->s->stream2));
    return GPUE_OK;
}

void gpue_ingest_destroy(gpue_ingest* g) {
    if (!g) return;
    for (int i = 0; i < 2; i++) {
        if (g->pin[i]) (void)hipHostFree(g->pin[i]);
        if (g->done[i]) (void)hipEventDestroy(g->done[i]);
    }
    delete g;
}

// expose the device pointer (for torch interop / sub-buffer views)
extern "C" int gpue_dbuf_ptr(gpue_dbuf* b, void** out);
int gpue_dbuf_ptr(gpue_dbuf* b, void** out) {
    ARG_CHECK(b && out);
    *out = b->ptr;
    return GPUE_OK;
}

int gpue_dbuf_h2d(gpue_dbuf* b, const void* src, uint64_t bytes, uint64_t dst_off) {
    ARG_CHECK(b && src && dst_off + bytes <= b->bytes);
    HIP_CHECK(hipMemcpyAsync((char*)b->ptr + dst_off, src, bytes,
                             hipMemcpyHostToDevice, b->s->stream));
    HIP_CHECK(hipStreamSynchronize(b->s->stream));
    return GPUE_OK;
}

int gpue_dbuf_d2h(gpue_dbuf* b, void* dst, uint64_t bytes, uint64_t src_off) {
    ARG_CHECK(b && dst && src_off + bytes <= b->bytes);
    if (bytes <= gpue_session::PINNED_BYTES && b->s->pinned) {
        HIP_CHECK(hipMemcpyAsync(b->s->pinned, (char*)b->ptr + src_off, bytes,
                                 hipMemcpyDeviceToHost, b->s->stream));
        HIP_CHECK(hipStreamSynchronize(b->s->stream));
        memcpy(dst, b->s->pinned, bytes);
        return GPUE_OK;
    }
    HIP_CHECK(hipMemcpyAsync(dst, (char*)b->ptr + src_off, bytes,
                             hipMemcpyDeviceToHost, b->s->stream));
    HIP_CHECK(hipStreamSynchronize(b->s->stream));
    return GPUE_OK;
}

int gpue_dbuf_memset(gpue_dbuf* b, int value, uint64_t bytes) {
    ARG_CHECK(b && bytes <= b->bytes);
    HIP_CHECK(hipMemsetAsync(b->ptr, value, bytes, b->s->stream));
    return GPUE_OK;
}

extern "C" int gpue_dbuf_d2d(gpue_dbuf* src, gpue_dbuf* dst, uint64_t bytes,
                             uint64_t src_off, uint64_t dst_off);
int gpue_dbuf_d2d(gpue_dbuf* src, gpue_dbuf* dst, uint64_t bytes, uint64_t src_off,
                  uint64_t dst_off) {
    ARG_CHECK(src && dst && src->bytes >= src_off + bytes && dst->bytes >= dst_off + bytes);
    HIP_CHECK(hipMemcpy((uint8_t*)dst->ptr + dst_off, (const uint8_t*)src->ptr + src_off,
                        bytes, hipMemcpyDeviceToDevice));
    return GPUE_OK;
}

__global__ void k_add_u64(unsigned long long* p, unsigned long long v) { *p += v; }

// SUM(a[i]*b[i]) + COUNT over a gathered match chunk — the agg sink's update
// for the chunked (unfused) config-2 plan: Aggregator::update_batch with a
// sum<int64> state (reference be/src/exprs/agg/sum.h:45-181). Block-reduce in
// LDS, one atomic per block.
__global__ void k_sum_prod_u32(const uint32_t* __restrict__ a,
                               const uint32_t* __restrict__ b, uint64_t n,
                               int64_t* __restrict__ acc) {
    __shared__ long long red[BLOCK / 64];
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    long long local = 0;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
        local += (long long)a[i] * b[i];
    for (int off = 32; off > 0; off >>= 1)
        local += __shfl_down(local, off, 64);
    if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = local;
    __syncthreads();
    if (threadIdx.x == 0) {
        long long s = 0;
        for (int w = 0; w < BLOCK / 64; w++) s += red[w];
        atomicAdd((unsigned long long*)acc, (unsigned long long)s);
    }
}

extern "C" int gpue_sum_prod_u32(gpue_session* s, gpue_dbuf* a, gpue_dbuf* b, uint64_t n,
                                 gpue_dbuf* acc /*i64[2]: sum, count*/);
int gpue_sum_prod_u32(gpue_session* s, gpue_dbuf* a, gpue_dbuf* b, uint64_t n,
                      gpue_dbuf* acc) {
    ARG_CHECK(s && a && b && acc && acc->bytes >= 16);
    ARG_CHECK(a->bytes >= n * 4 && b->bytes >= n * 4);
    if (n > 0)
        hipLaunchKernelGGL(k_sum_prod_u32, dim3(grid_stream(n)), dim3(BLOCK), 0, s->stream,
                           (const uint32_t*)a->ptr, (const uint32_t*)b->ptr, n,
                           (int64_t*)acc->ptr);
    hipLaunchKernelGGL(k_add_u64, dim3(1), dim3(1), 0, s->stream,
                       (unsigned long long*)acc->ptr + 1, (unsigned long long)n);
    HIP_CHECK(hipGetLastError());
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// hipGraph step replay: capture a step's launch sequence once, replay it per
// step with a single hipGraphLaunch — removes the per-launch host cost that
// dominates small (SF10-sized) steps. The CDNA-native analog of the
// reference's driver time-slice batching.
// ---------------------------------------------------------------------------
struct gpue_graph {
    hipGraph_t graph = nullptr;
    hipGraphExec_t exec = nullptr;
};

extern "C" {
int gpue_graph_begin(gpue_session* s);
int gpue_graph_end(gpue_session* s, gpue_graph** out);
int gpue_graph_launch(gpue_session* s, gpue_graph* g);
void gpue_graph_destroy(gpue_graph* g);
}

int gpue_graph_begin(gpue_session* s) {
    ARG_CHECK(s);
    HIP_CHECK(hipStreamBeginCapture(s->stream, hipStreamCaptureModeThreadLocal));
    return GPUE_OK;
}

int gpue_graph_end(gpue_session* s, gpue_graph** out) {
    ARG_CHECK(s && out);
    gpue_graph* g = new gpue_graph();
    hipError_t err = hipStreamEndCapture(s->stream, &g->graph);
    if (err != hipSuccess) {
        delete g;
        snprintf(g_err, sizeof(g_err), "hipStreamEndCapture failed: %s",
                 hipGetErrorString(err));
        return GPUE_ERR_HIP;
    }
    err = hipGraphInstantiate(&g->exec, g->graph, nullptr, nullptr, 0);
    if (err != hipSuccess) {
        (void)hipGraphDestroy(g->graph);
        delete g;
        snprintf(g_err, sizeof(g_err), "hipGraphInstantiate failed: %s",
                 hipGetErrorString(err));
        return GPUE_ERR_HIP;
    }
    *out = g;
    return GPUE_OK;
}

int gpue_graph_launch(gpue_session* s, gpue_graph* g) {
    ARG_CHECK(s && g && g->exec);
    HIP_CHECK(hipGraphLaunch(g->exec, s->stream));
    return GPUE_OK;
}

void gpue_graph_destroy(gpue_graph* g) {
    if (!g) return;
    if (g->exec) (void)hipGraphExecDestroy(g->exec);
    if (g->graph) (void)hipGraphDestroy(g->graph);
    delete g;
}

// ---------------------------------------------------------------------------
// deterministic synthetic generator — splitmix64 finalizer, counter-based.
// MUST stay identical to oracle/oracle.c sm64/orc_gen_u64 and the numpy
// restatement in starrocks_amd/gen.py.
// ---------------------------------------------------------------------------
__host__ __device__ static inline uint64_t sm64(uint64_t x) {
    x += 0x9E3779B97F4A7C15ull;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
    return x ^ (x >> 31);
}
__host__ __device__ static inline uint64_t gen_u64(uint64_t seed, uint64_t tag, uint64_t i) {
    return sm64(seed + tag * 0x9E3779B97F4A7C15ull + i);
}

// tags shared with oracle/oracle.c
enum { TAG_ORDERDATE = 1, TAG_EXTPRICE = 2, TAG_DISCOUNT = 3,
       TAG_PARTKEY = 4, TAG_SUPPKEY = 5, TAG_REVENUE = 6,
       TAG_PCAT = 7, TAG_PBRD = 8, TAG_SREG = 9,
       TAG_CUSTKEY = 10, TAG_SUPPCOST = 11, TAG_CREG = 12,
       TAG_SNAT = 13, TAG_SCITY = 14 };
static constexpr int N_DAYS = 2556;
static constexpr uint32_t N_PARTS_SF100 = 1400000u;
static constexpr uint32_t N_SUPPS_SF100 = 200000u;
static constexpr uint32_t N_CUSTS_SF100 = 3000000u;
enum { TAG_LOKEY = 15, TAG_LEXT = 16, TAG_LDISC = 17, TAG_LSHIP = 18,
       TAG_OCUST = 19, TAG_ODATE = 20, TAG_CMKT = 21 };
// 16-byte space-padded dictionary (identical to oracle MKT_SEGMENTS)
__constant__ char MKT_SEGMENTS_DEV[5][16] = {
    {'A','U','T','O','M','O','B','I','L','E',' ',' ',' ',' ',' ',' '},
    {'B','U','I','L','D','I','N','G',' ',' ',' ',' ',' ',' ',' ',' '},
    {'F','U','R','N','I','T','U','R','E',' ',' ',' ',' ',' ',' ',' '},
    {'M','A','C','H','I','N','E','R','Y',' ',' ',' ',' ',' ',' ',' '},
    {'H','O','U','S','E','H','O','L','D',' ',' ',' ',' ',' ',' ',' '}};

__global__ void k_gen_u32_mod(uint32_t* out, uint64_t seed, uint64_t tag,
                              uint64_t row_start, uint64_t n, uint32_t mod, uint32_t add) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        uint64_t v = gen_u64(seed, tag, row_start + i);
        out[i] = (mod ? (uint32_t)(v % mod) : (uint32_t)v) + add;
    }
}

__global__ void k_gen_i64(int64_t* out, uint64_t seed, uint64_t tag,
                          uint64_t row_start, uint64_t n) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
        out[i] = (int64_t)gen_u64(seed, tag, row_start + i);
}

__global__ void k_gen_lineorder_q1(int32_t* od, int32_t* ep, int32_t* dc,
                                   const int32_t* __restrict__ datekey,
                                   uint64_t seed, uint64_t row_start, uint64_t n) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        uint64_t r = row_start + i;
        od[i] = datekey[gen_u64(seed, TAG_ORDERDATE, r) % N_DAYS];
        ep[i] = (int32_t)(gen_u64(seed, TAG_EXTPRICE, r) % 100000u) + 1;
        dc[i] = (int32_t)(gen_u64(seed, TAG_DISCOUNT, r) % 11u);
    }
}

__global__ void k_gen_lineorder_q21(int32_t* pk, int32_t* sk, int32_t* od, int32_t* rv,
                                    const int32_t* __restrict__ datekey,
                                    uint64_t seed, uint64_t row_start, uint64_t n) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        uint64_t r = row_start + i;
        pk[i] = (int32_t)(gen_u64(seed, TAG_PARTKEY, r) % N_PARTS_SF100) + 1;
        sk[i] = (int32_t)(gen_u64(seed, TAG_SUPPKEY, r) % N_SUPPS_SF100) + 1;
        od[i] = datekey[gen_u64(seed, TAG_ORDERDATE, r) % N_DAYS];
        rv[i] = (int32_t)(gen_u64(seed, TAG_REVENUE, r) % 10000000u);
    }
}

__global__ void k_gen_lineorder_q43(int32_t* ck, int32_t* sk, int32_t* pk, int32_t* od,
                                    int32_t* rv, int32_t* sc,
                                    const int32_t* __restrict__ datekey,
                                    uint64_t seed, uint64_t row_start, uint64_t n) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        uint64_t r = row_start + i;
        ck[i] = (int32_t)(gen_u64(seed, TAG_CUSTKEY, r) % N_CUSTS_SF100) + 1;
        sk[i] = (int32_t)(gen_u64(seed, TAG_SUPPKEY, r) % N_SUPPS_SF100) + 1;
        pk[i] = (int32_t)(gen_u64(seed, TAG_PARTKEY, r) % N_PARTS_SF100) + 1;
        od[i] = datekey[gen_u64(seed, TAG_ORDERDATE, r) % N_DAYS];
        rv[i] = (int32_t)(gen_u64(seed, TAG_REVENUE, r) % 10000000u);
        sc[i] = (int32_t)(gen_u64(seed, TAG_SUPPCOST, r) % 100000u) + 1;
    }
}

// host-side Gregorian calendar (identical to oracle orc_gen_dates)
static void host_gen_dates(int32_t n_days, int32_t* datekey, int32_t* dyear) {
    static const int MDAYS[12] = {31, 28, 31, 30, 31, 30, 31, 31, 30, 31, 30, 31};
    int y = 1992, m = 1, d = 1;
    for (int32_t i = 0; i < n_days; i++) {
        datekey[i] = y * 10000 + m * 100 + d;
        if (dyear) dyear[i] = y;
        int leap = (y % 4 == 0 && y % 100 != 0) || y % 400 == 0;
        int md = MDAYS[m - 1] + (m == 2 && leap);
        if (++d > md) { d = 1; if (++m > 12) { m = 1; y++; } }
    }
}

static int ensure_datekey(gpue_session* s) {
    if (s->d_datekey) return GPUE_OK;
    int32_t host[N_DAYS];
    host_gen_dates(N_DAYS, host, nullptr);
    HIP_CHECK(hipMalloc(&s->d_datekey, N_DAYS * sizeof(int32_t)));
    HIP_CHECK(hipMemcpy(s->d_datekey, host, N_DAYS * sizeof(int32_t), hipMemcpyHostToDevice));
    return GPUE_OK;
}

int gpue_gen_u32_mod(gpue_session* s, gpue_dbuf* out, uint64_t seed, uint64_t tag,
                     uint64_t row_start, uint64_t n, uint32_t mod, uint32_t add) {
    ARG_CHECK(s && out && out->bytes >= n * 4);
    hipLaunchKernelGGL(k_gen_u32_mod, dim3(grid_stream(n)), dim3(BLOCK), 0, s->stream,
                       (uint32_t*)out->ptr, seed, tag, row_start, n, mod, add);
    HIP_CHECK(hipGetLastError());
    return GPUE_OK;
}

int gpue_gen_i64(gpue_session* s, gpue_dbuf* out, uint64_t seed, uint64_t tag,
                 uint64_t row_start, uint64_t n) {
    ARG_CHECK(s && out && out->bytes >= n * 8);
    hipLaunchKernelGGL(k_gen_i64, dim3(grid_stream(n)), dim3(BLOCK), 0, s->stream,
                       (int64_t*)out->ptr, seed, tag, row_start, n);
    HIP_CHECK(hipGetLastError());
    return GPUE_OK;
}

int gpue_gen_lineorder_q1(gpue_session* s, uint64_t seed, uint64_t row_start, uint64_t n,
                          gpue_dbuf* od, gpue_dbuf* ep, gpue_dbuf* dc) {
    ARG_CHECK(s && od && ep && dc && od->bytes >= n * 4 && ep->bytes >= n * 4 && dc->bytes >= n * 4);
    int rc = ensure_datekey(s);
    if (rc != GPUE_OK) return rc;
    hipLaunchKernelGGL(k_gen_lineorder_q1, dim3(grid_stream(n)), dim3(BLOCK), 0, s->stream,
                       (int32_t*)od->ptr, (int32_t*)ep->ptr, (int32_t*)dc->ptr,
                       s->d_datekey, seed, row_start, n);
    HIP_CHECK(hipGetLastError());
    return GPUE_OK;
}

extern "C" int gpue_gen_lineorder_q43(gpue_session* s, uint64_t seed, uint64_t row_start,
                                      uint64_t n, gpue_dbuf* ck, gpue_dbuf* sk, gpue_dbuf* pk,
                                      gpue_dbuf* od, gpue_dbuf* rv, gpue_dbuf* sc);
int gpue_gen_lineorder_q43(gpue_session* s, uint64_t seed, uint64_t row_start, uint64_t n,
                           gpue_dbuf* ck, gpue_dbuf* sk, gpue_dbuf* pk, gpue_dbuf* od,
                           gpue_dbuf* rv, gpue_dbuf* sc) {
    ARG_CHECK(s && ck && sk && pk && od && rv && sc);
    ARG_CHECK(ck->bytes >= n * 4 && sk->bytes >= n * 4 && pk->bytes >= n * 4 &&
              od->bytes >= n * 4 && rv->bytes >= n * 4 && sc->bytes >= n * 4);
    int rc = ensure_datekey(s);
    if (rc != GPUE_OK) return rc;
    hipLaunchKernelGGL(k_gen_lineorder_q43, dim3(grid_stream(n)), dim3(BLOCK), 0, s->stream,
                       (int32_t*)ck->ptr, (int32_t*)sk->ptr, (int32_t*)pk->ptr,
                       (int32_t*)od->ptr, (int32_t*)rv->ptr, (int32_t*)sc->ptr,
                       s->d_datekey, seed, row_start, n);
    HIP_CHECK(hipGetLastError());
    return GPUE_OK;
}

int gpue_gen_lineorder_q21(gpue_session* s, uint64_t seed, uint64_t row_start, uint64_t n,
                           gpue_dbuf* pk, gpue_dbuf* sk, gpue_dbuf* od, gpue_dbuf* rv) {
    ARG_CHECK(s && pk && sk && od && rv && pk->bytes >= n * 4 && sk->bytes >= n * 4 &&
              od->bytes >= n * 4 && rv->bytes >= n * 4);
    int rc = ensure_datekey(s);
    if (rc != GPUE_OK) return rc;
    hipLaunchKernelGGL(k_gen_lineorder_q21, dim3(grid_stream(n)), dim3(BLOCK), 0, s->stream,
                       (int32_t*)pk->ptr, (int32_t*)sk->ptr, (int32_t*)od->ptr,
                       (int32_t*)rv->ptr, s->d_datekey, seed, row_start, n);
    HIP_CHECK(hipGetLastError());
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// scan + predicate filter: ordered (stable) stream compaction.
// Replaces SIMD::Filter::filter_range (reference base/simd/filter.h:26-38):
// the AVX-512 vpcompress idiom maps to wave64 ballot + popcount-prefix; the
// sequential output order is preserved by tiling: block b owns the contiguous
// row range [b*tile, (b+1)*tile) and writes at exclusive-scanned offsets.
// ---------------------------------------------------------------------------
__global__ void k_filter_count(const int64_t* __restrict__ in, uint64_t n, int64_t theta,
                               uint64_t tile, uint64_t* __restrict__ block_counts) {
    uint64_t lo = (uint64_t)blockIdx.x * tile;
    uint64_t hi = min(lo + tile, n);
    uint64_t c = 0;
    for (uint64_t i = lo + threadIdx.x; i < hi; i += blockDim.x)
        c += (in[i] < theta);
    // wave reduce then block reduce
    for (int off = WAVE / 2; off > 0; off >>= 1)
        c += __shfl_down((unsigned long long)c, off, WAVE);
    __shared__ uint64_t wsum[BLOCK / WAVE];
    int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;
    if (lane == 0) wsum[wid] = c;
    __syncthreads();
    if (threadIdx.x == 0) {
        uint64_t t = 0;
        for (int w = 0; w < BLOCK / WAVE; w++) t += wsum[w];
        block_counts[blockIdx.x] = t;
    }
}

// single-block exclusive scan of nb (<= MAX_GRID+1) entries; thread 0
// sequential — nb is tiny and this launch is noise next to the data passes
__global__ void k_scan_small(uint64_t* data, uint32_t n) {
    if (threadIdx.x == 0 && blockIdx.x == 0) {
        uint64_t acc = 0;
        for (uint32_t i = 0; i < n; i++) {
            uint64_t v = data[i];
            data[i] = acc;
            acc += v;
        }
        data[n] = acc; // total
    }
}

__global__ void k_filter_emit(const int64_t* __restrict__ in, uint64_t n, int64_t theta,
                              uint64_t tile, const uint64_t* __restrict__ block_offsets,
                              int64_t* __restrict__ out) {
    uint64_t lo = (uint64_t)blockIdx.x * tile;
    uint64_t hi = min(lo + tile, n);
    __shared__ uint64_t wbase[BLOCK / WAVE + 1];
    uint64_t offset = block_offsets[blockIdx.x];
    int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;
    for (uint64_t base = lo; base < hi; base += blockDim.x) {
        uint64_t i = base + threadIdx.x;
        bool pred = (i < hi) && (in[i] < theta);
        uint64_t mask = __ballot(pred);
        uint32_t wcount = __popcll(mask);
        if (lane == 0) wbase[wid] = wcount;
        __syncthreads();
        if (threadIdx.x == 0) {
            uint64_t acc = offset;
            for (int w = 0; w < BLOCK / WAVE; w++) {
                uint64_t v = wbase[w];
                wbase[w] = acc;
                acc += v;
            }
            wbase[BLOCK / WAVE] = acc;
        }
        __syncthreads();
        if (pred) {
            uint32_t rank = __popcll(mask & ((1ull << lane) - 1));
            out[wbase[wid] + rank] = in[i];
        }
        offset = wbase[BLOCK / WAVE];
        __syncthreads();
    }
}

// ---------------------------------------------------------------------------
// Single-pass ordered compaction — decoupled lookback. Each block claims
// consecutive tiles via an atomic ticket (a claimed tile's predecessor was
// claimed earlier by a resident block => forward progress); the tile's
// {flag,count} pack into ONE 64-bit atomic word, so publish/consume need no
// separate fence (all consumed data is in-word; guide §6 G16 concerns
// multi-word hand-offs). Passing values stage compactly in LDS, one
// coalesced read + one coalesced write per element — the reference's
// algorithmic byte count exactly. Spins are bounded: on timeout the kernel
// sets an error flag and exits (host returns GPUE_ERR_HIP).
// ---------------------------------------------------------------------------
// rows-per-thread and threads-per-block are template parameters of the
// lookback kernels below (r02 sweep: 32 x 1024 default)
static constexpr unsigned long long FILT_AGG = 1ull << 62;    // aggregate available
static constexpr unsigned long long FILT_PREFIX = 2ull << 62; // inclusive prefix available
static constexpr unsigned long long FILT_CNT_MASK = (1ull << 62) - 1;

template <int ITEMS, int TPB = BLOCK> // rows/thread and threads/block:
                     // larger tiles -> fewer lookback pipeline stages (the
                     // per-tile publish/observe latency chain dominates at
                     // 488 K tiles)
__global__ __launch_bounds__(TPB) void
k_filter_lookback(const int64_t* __restrict__ in, uint64_t n, int64_t theta,
                                  int64_t* __restrict__ out,
                                  unsigned long long* __restrict__ tile_desc,
                                  unsigned long long* __restrict__ ticket,
                                  unsigned long long* __restrict__ total_out,
                                  unsigned long long* __restrict__ error_out) {
    // thread t owns the ITEMS consecutive rows [lo + 8t, lo + 8t + 8): output
    // order == thread order == row order; ONE block scan per tile.
    __shared__ uint64_t sh_tile;
    __shared__ unsigned long long sh_excl;
    __shared__ uint32_t wsum[TPB / WAVE];
    const uint64_t TILE = (uint64_t)TPB * ITEMS;
    const uint64_t n_tiles = (n + TILE - 1) / TILE;
    int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;
    (void)ticket;
    (void)sh_tile;
    // static ascending tiles: the host sizes the grid to be FULLY RESIDENT
    // (occupancy-queried), so tile t's predecessor is always executing on
    // another resident block — no ticket atomic (a single ticket address
    // serializes ~12 ns × n_tiles: measured 6 ms of a 7 ms launch at 488 K
    // tiles). The bounded spin turns any residency miscount into an error
    // return instead of a hang.
    for (uint64_t t = blockIdx.x; t < n_tiles; t += gridDim.x) {
        uint64_t lo = t * TILE;
        uint64_t hi = min(lo + TILE, n);
        uint64_t my = lo + (uint64_t)threadIdx.x * ITEMS;
        int64_t v[ITEMS];
        uint32_t pm = 0; // predicate bitmask over my rows (ITEMS <= 32)
        if (my + ITEMS <= hi) {
            const longlong2* p2 = (const longlong2*)(in + my);
            #pragma unroll
            for (int j = 0; j < ITEMS / 2; j++) {
                longlong2 w = p2[j];
                v[2 * j] = w.x;
                v[2 * j + 1] = w.y;
            }
            #pragma unroll
            for (int j = 0; j < ITEMS; j++) pm |= (v[j] < theta) << j;
        } else {
            for (int j = 0; j < ITEMS; j++) {
                uint64_t i = my + j;
                v[j] = (i < hi) ? in[i] : 0;
                pm |= ((i < hi) && (v[j] < theta)) << j;
            }
        }
        uint32_t c = __popc(pm);
        // block exclusive scan of per-thread counts: wave shfl-scan + wave
        // totals combined by thread 0
        uint32_t pre = c;
        for (int off = 1; off < WAVE; off <<= 1) {
            uint32_t up = __shfl_up(pre, off, WAVE);
            if (lane >= off) pre += up;
        }
        uint32_t wave_total = __shfl(pre, WAVE - 1, WAVE);
        uint32_t my_excl = pre - c;
        if (lane == WAVE - 1) wsum[wid] = pre; // inclusive wave total
        __syncthreads();
        uint32_t wave_base = 0;
        for (int w = 0; w < wid; w++) wave_base += wsum[w];
        uint32_t tile_count = 0;
        for (int w = 0; w < TPB / WAVE; w++) tile_count += wsum[w];
        (void)wave_total;
        // WAVE-PARALLEL lookback (wave 0): 64 predecessor descriptors per
        // step — a serial walk scans O(grid) aggregates per tile and
        // measured 7.5 ms at 488 K tiles
        if (wid == 0) {
            if (t == 0) {
                if (lane == 0) {
                    __hip_atomic_store(&tile_desc[0], FILT_PREFIX | tile_count,
                                       __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
                    sh_excl = 0;
                }
            } else {
                if (lane == 0)
                    __hip_atomic_store(&tile_desc[t], FILT_AGG | tile_count,
                                       __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
                unsigned long long running = 0;
                int64_t base = (int64_t)t - 1; // lane l inspects base - l
                uint64_t spins = 0;
                for (;;) {
                    int64_t idx = base - lane;
                    // before tile 0: a virtual PREFIX of 0
                    unsigned long long d =
                        idx >= 0 ? __hip_atomic_load(&tile_desc[idx], __ATOMIC_RELAXED,
                                                     __HIP_MEMORY_SCOPE_AGENT)
                                 : FILT_PREFIX;
                    unsigned long long flag = d & ~FILT_CNT_MASK;
                    uint64_t prefix_mask = __ballot(flag == FILT_PREFIX);
                    uint64_t invalid_mask = __ballot(flag == 0);
                    int first_prefix = prefix_mask ? (__ffsll((unsigned long long)prefix_mask) - 1) : WAVE;
                    int first_invalid = invalid_mask ? (__ffsll((unsigned long long)invalid_mask) - 1) : WAVE;
                    if (first_prefix < first_invalid) {
                        unsigned long long contrib =
                            (lane <= first_prefix) ? (d & FILT_CNT_MASK) : 0;
                        for (int off = WAVE / 2; off > 0; off >>= 1)
                            contrib += __shfl_down(contrib, off, WAVE);
                        running += __shfl(contrib, 0, WAVE);
                        break;
                    }
                    if (first_invalid == WAVE) { // full window of aggregates
                        unsigned long long contrib = d & FILT_CNT_MASK;
                        for (int off = WAVE / 2; off > 0; off >>= 1)
                            contrib += __shfl_down(contrib, off, WAVE);
                        running += __shfl(contrib, 0, WAVE);
                        base -= WAVE;
                        continue;
                    }
                    if (++spins > (1ull << 28)) { // bounded: flag error, bail out
                        if (lane == 0) atomicOr(error_out, 1ull);
                        running = 0;
                        break;
                    }
                }
                if (lane == 0) {
                    __hip_atomic_store(&tile_desc[t], FILT_PREFIX | (running + tile_count),
                                       __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
                    sh_excl = running;
                }
            }
            if (lane == 0 && t == n_tiles - 1) *total_out = sh_excl + tile_count;
        }
        __syncthreads();
        uint64_t w = sh_excl + wave_base + my_excl;
        #pragma unroll
        for (int j = 0; j < ITEMS; j++)
            if (pm & (1u << j)) out[w++] = v[j];
        __syncthreads();
    }
}

// Wave-coalesced layout variant (GPUE_FILT_WL): the baseline kernel gives
// each THREAD 32 consecutive rows, so every 16 B load instruction touches 64
// DISTINCT 256 B-strided cache lines — 4x the TA line-request work per byte
// of a lane-coalesced stream (the coalesced 3-stream ubench runs 5658 GB/s
// where the filter runs 3784). Here each WAVE owns a contiguous 2048-row
// chunk and lanes advance through it in lockstep 1 KB steps (64 lanes x
// 16 B contiguous per instruction); emit order is reconstructed per step
// from two ballots (even/odd element ranks), with the running wave offset
// chained in a wave-uniform register — same registers (v[32]), same
// descriptor protocol and grid sizing as the baseline.
template <int TPB>
__global__ __launch_bounds__(TPB) void
k_filter_lookback_wl(const int64_t* __restrict__ in, uint64_t n, int64_t theta,
                     int64_t* __restrict__ out,
                     unsigned long long* __restrict__ tile_desc,
                     unsigned long long* __restrict__ ticket,
                     unsigned long long* __restrict__ total_out,
                     unsigned long long* __restrict__ error_out) {
    __shared__ unsigned long long sh_excl;
    __shared__ uint32_t wsum[TPB / WAVE];
    constexpr int STEPS = 16;             // 16 steps x 2 values = 32 rows/thread
    constexpr uint64_t CHUNK = (uint64_t)STEPS * 2 * WAVE; // 2048 rows per wave
    const uint64_t TILE = (uint64_t)(TPB / WAVE) * CHUNK;
    const uint64_t n_tiles = (n + TILE - 1) / TILE;
    int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;
    (void)ticket;
    for (uint64_t t = blockIdx.x; t < n_tiles; t += gridDim.x) {
        uint64_t lo = t * TILE;
        uint64_t hi = min(lo + TILE, n);
        uint64_t lo_w = lo + (uint64_t)wid * CHUNK;
        int64_t v[2 * STEPS];
        uint32_t wave_total = 0;
        if (lo_w + CHUNK <= hi) { // full chunk: unguarded 1 KB coalesced steps
            const longlong2* p2 = (const longlong2*)(in + lo_w);
            #pragma unroll
            for (int j = 0; j < STEPS; j++) {
                longlong2 w = p2[j * WAVE + lane];
                v[2 * j] = w.x;
                v[2 * j + 1] = w.y;
                wave_total += __popcll(__ballot(w.x < theta));
                wave_total += __popcll(__ballot(w.y < theta));
            }
        } else {
            #pragma unroll
            for (int j = 0; j < STEPS; j++) {
                uint64_t r0 = lo_w + (uint64_t)j * 2 * WAVE + 2 * lane;
                // OOB elements get v == theta: the LT predicate is false
                v[2 * j] = r0 < hi ? in[r0] : theta;
                v[2 * j + 1] = r0 + 1 < hi ? in[r0 + 1] : theta;
                wave_total += __popcll(__ballot(v[2 * j] < theta));
                wave_total += __popcll(__ballot(v[2 * j + 1] < theta));
            }
        }
        if (lane == 0) wsum[wid] = wave_total;
        __syncthreads();
        uint32_t wave_base = 0;
        for (int w = 0; w < wid; w++) wave_base += wsum[w];
        uint32_t tile_count = 0;
        for (int w = 0; w < TPB / WAVE; w++) tile_count += wsum[w];
        // lookback — identical protocol to the baseline kernel
        if (wid == 0) {
            if (t == 0) {
                if (lane == 0) {
                    __hip_atomic_store(&tile_desc[0], FILT_PREFIX | tile_count,
                                       __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
                    sh_excl = 0;
                }
            } else {
                if (lane == 0)
                    __hip_atomic_store(&tile_desc[t], FILT_AGG | tile_count,
                                       __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
                unsigned long long running = 0;
                int64_t base = (int64_t)t - 1;
                uint64_t spins = 0;
                for (;;) {
                    int64_t idx = base - lane;
                    unsigned long long d =
                        idx >= 0 ? __hip_atomic_load(&tile_desc[idx], __ATOMIC_RELAXED,
                                                     __HIP_MEMORY_SCOPE_AGENT)
                                 : FILT_PREFIX;
                    unsigned long long flag = d & ~FILT_CNT_MASK;
                    uint64_t prefix_mask = __ballot(flag == FILT_PREFIX);
                    uint64_t invalid_mask = __ballot(flag == 0);
                    int first_prefix = prefix_mask ? (__ffsll((unsigned long long)prefix_mask) - 1) : WAVE;
                    int first_invalid = invalid_mask ? (__ffsll((unsigned long long)invalid_mask) - 1) : WAVE;
                    if (first_prefix < first_invalid) {
                        unsigned long long contrib =
                            (lane <= first_prefix) ? (d & FILT_CNT_MASK) : 0;
                        for (int off = WAVE / 2; off > 0; off >>= 1)
                            contrib += __shfl_down(contrib, off, WAVE);
                        running += __shfl(contrib, 0, WAVE);
                        break;
                    }
                    if (first_invalid == WAVE) {
                        unsigned long long contrib = d & FILT_CNT_MASK;
                        for (int off = WAVE / 2; off > 0; off >>= 1)
                            contrib += __shfl_down(contrib, off, WAVE);
                        running += __shfl(contrib, 0, WAVE);
                        base -= WAVE;
                        continue;
                    }
                    if (++spins > (1ull << 28)) {
                        if (lane == 0) atomicOr(error_out, 1ull);
                        running = 0;
                        break;
                    }
                }
                if (lane == 0) {
                    __hip_atomic_store(&tile_desc[t], FILT_PREFIX | (running + tile_count),
                                       __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
                    sh_excl = running;
                }
            }
            if (lane == 0 && t == n_tiles - 1) *total_out = sh_excl + tile_count;
        }
        __syncthreads();
        // in-order emit: per step, two ballots rank the 128 contiguous rows
        // (order = (lane, element)); the wave's running offset is uniform
        uint64_t running = sh_excl + wave_base;
        uint64_t lml = ((uint64_t)1 << lane) - 1;
        #pragma unroll
        for (int j = 0; j < STEPS; j++) {
            bool p0 = v[2 * j] < theta;
            bool p1 = v[2 * j + 1] < theta;
            uint64_t m0 = __ballot(p0);
            uint64_t m1 = __ballot(p1);
            uint32_t below = __popcll(m0 & lml) + __popcll(m1 & lml);
            if (p0) out[running + below] = v[2 * j];
            if (p1) out[running + below + (p0 ? 1u : 0u)] = v[2 * j + 1];
            running += __popcll(m0) + __popcll(m1);
        }
        __syncthreads();
    }
}

// Re-read emit variant (GPUE_FILT_RR) — MEASURED WORSE at every selectivity
// (s=0.01: 2.83 vs 2.14 ms; s=0.5: 12.6 vs 4.45 — the emit re-reads miss L2
// at an 8 GB footprint; profiles/r02_filter_items_sweep.log). Keeping the
// 32 values in registers is the right design; this stays env-gated as the
// recorded negative.
template <int ITEMS, int TPB>
__global__ __launch_bounds__(TPB) void
k_filter_lookback_rr(const int64_t* __restrict__ in, uint64_t n, int64_t theta,
                     int64_t* __restrict__ out,
                     unsigned long long* __restrict__ tile_desc,
                     unsigned long long* __restrict__ ticket,
                     unsigned long long* __restrict__ total_out,
                     unsigned long long* __restrict__ error_out) {
    __shared__ unsigned long long sh_excl;
    __shared__ uint32_t wsum[TPB / WAVE];
    const uint64_t TILE = (uint64_t)TPB * ITEMS;
    const uint64_t n_tiles = (n + TILE - 1) / TILE;
    int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;
    (void)ticket;
    for (uint64_t t = blockIdx.x; t < n_tiles; t += gridDim.x) {
        uint64_t lo = t * TILE;
        uint64_t hi = min(lo + TILE, n);
        uint64_t my = lo + (uint64_t)threadIdx.x * ITEMS;
        uint32_t pm = 0;
        if (my + ITEMS <= hi) {
            const longlong2* p2 = (const longlong2*)(in + my);
            #pragma unroll
            for (int j = 0; j < ITEMS / 2; j++) {
                longlong2 w = p2[j];
                pm |= (w.x < theta) << (2 * j);
                pm |= (w.y < theta) << (2 * j + 1);
            }
        } else {
            for (int j = 0; j < ITEMS; j++) {
                uint64_t i = my + j;
                pm |= ((i < hi) && (in[i] < theta)) << j;
            }
        }
        uint32_t c = __popc(pm);
        uint32_t pre = c;
        for (int off = 1; off < WAVE; off <<= 1) {
            uint32_t up = __shfl_up(pre, off, WAVE);
            if (lane >= off) pre += up;
        }
        uint32_t my_excl = pre - c;
        if (lane == WAVE - 1) wsum[wid] = pre;
        __syncthreads();
        uint32_t wave_base = 0;
        for (int w = 0; w < wid; w++) wave_base += wsum[w];
        uint32_t tile_count = 0;
        for (int w = 0; w < TPB / WAVE; w++) tile_count += wsum[w];
        if (wid == 0) {
            if (t == 0) {
                if (lane == 0) {
                    __hip_atomic_store(&tile_desc[0], FILT_PREFIX | tile_count,
                                       __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
                    sh_excl = 0;
                }
            } else {
                if (lane == 0)
                    __hip_atomic_store(&tile_desc[t], FILT_AGG | tile_count,
                                       __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
                unsigned long long running = 0;
                int64_t base = (int64_t)t - 1;
                uint64_t spins = 0;
                for (;;) {
                    int64_t idx = base - lane;
                    unsigned long long d =
                        idx >= 0 ? __hip_atomic_load(&tile_desc[idx], __ATOMIC_RELAXED,
                                                     __HIP_MEMORY_SCOPE_AGENT)
                                 : FILT_PREFIX;
                    unsigned long long flag = d & ~FILT_CNT_MASK;
                    uint64_t prefix_mask = __ballot(flag == FILT_PREFIX);
                    uint64_t invalid_mask = __ballot(flag == 0);
                    int first_prefix = prefix_mask ? (__ffsll((unsigned long long)prefix_mask) - 1) : WAVE;
                    int first_invalid = invalid_mask ? (__ffsll((unsigned long long)invalid_mask) - 1) : WAVE;
                    if (first_prefix < first_invalid) {
                        unsigned long long contrib =
                            (lane <= first_prefix) ? (d & FILT_CNT_MASK) : 0;
                        for (int off = WAVE / 2; off > 0; off >>= 1)
                            contrib += __shfl_down(contrib, off, WAVE);
                        running += __shfl(contrib, 0, WAVE);
                        break;
                    }
                    if (first_invalid == WAVE) {
                        unsigned long long contrib = d & FILT_CNT_MASK;
                        for (int off = WAVE / 2; off > 0; off >>= 1)
                            contrib += __shfl_down(contrib, off, WAVE);
                        running += __shfl(contrib, 0, WAVE);
                        base -= WAVE;
                        continue;
                    }
                    if (++spins > (1ull << 28)) {
                        if (lane == 0) atomicOr(error_out, 1ull);
                        running = 0;
                        break;
                    }
                }
                if (lane == 0) {
                    __hip_atomic_store(&tile_desc[t], FILT_PREFIX | (running + tile_count),
                                       __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
                    sh_excl = running;
                }
            }
            if (lane == 0 && t == n_tiles - 1) *total_out = sh_excl + tile_count;
        }
        __syncthreads();
        uint64_t w = sh_excl + wave_base + my_excl;
        #pragma unroll
        for (int j = 0; j < ITEMS; j++)
            if (pm & (1u << j)) out[w++] = in[my + j]; // re-read (mostly L2)
        __syncthreads();
    }
}

// Software-pipelined lookback (GPUE_FILT_PIPE) — MEASURED NEGATIVE
// (5.37 vs 2.14 ms at s=0.01, every TPB/ITEMS combo): the double tile
// buffer (v[2][32] = 128 VGPRs + machinery) spills to scratch and the
// scratch traffic swamps the ~7.5 us/tile walk stall it was meant to hide.
// Kept env-gated as the recorded negative; the register-resident
// single-buffer form (32 rows x 1024 threads) remains the winner.
template <int ITEMS, int TPB>
__global__ __launch_bounds__(TPB) void
k_filter_lookback_pipe(const int64_t* __restrict__ in, uint64_t n, int64_t theta,
                       int64_t* __restrict__ out,
                       unsigned long long* __restrict__ tile_desc,
                       unsigned long long* __restrict__ ticket,
                       unsigned long long* __restrict__ total_out,
                       unsigned long long* __restrict__ error_out) {
    __shared__ unsigned long long sh_excl;
    __shared__ uint32_t wsum[TPB / WAVE];
    const uint64_t TILE = (uint64_t)TPB * ITEMS;
    const uint64_t n_tiles = (n + TILE - 1) / TILE;
    int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;
    (void)ticket;
    int64_t v[2][ITEMS];
    auto load_tile = [&](uint64_t t, int64_t* dst) {
        uint64_t lo = t * TILE;
        uint64_t hi = min(lo + TILE, n);
        uint64_t my = lo + (uint64_t)threadIdx.x * ITEMS;
        if (my + ITEMS <= hi) {
            const longlong2* p2 = (const longlong2*)(in + my);
            #pragma unroll
            for (int j = 0; j < ITEMS / 2; j++) {
                longlong2 w = p2[j];
                dst[2 * j] = w.x;
                dst[2 * j + 1] = w.y;
            }
        } else {
            for (int j = 0; j < ITEMS; j++) {
                uint64_t i = my + j;
                dst[j] = (i < hi) ? in[i] : theta; // >= theta: never matches
            }
        }
    };
    uint64_t t = blockIdx.x;
    if (t < n_tiles) load_tile(t, v[0]);
    int cur = 0;
    for (; t < n_tiles; t += gridDim.x, cur ^= 1) {
        uint64_t lo = t * TILE;
        uint64_t hi = min(lo + TILE, n);
        uint64_t my = lo + (uint64_t)threadIdx.x * ITEMS;
        uint32_t pm = 0;
        #pragma unroll
        for (int j = 0; j < ITEMS; j++)
            pm |= ((my + j < hi) && (v[cur][j] < theta)) << j;
        uint32_t c = __popc(pm);
        uint32_t pre = c;
        for (int off = 1; off < WAVE; off <<= 1) {
            uint32_t up = __shfl_up(pre, off, WAVE);
            if (lane >= off) pre += up;
        }
        uint32_t my_excl = pre - c;
        if (lane == WAVE - 1) wsum[wid] = pre;
        __syncthreads();
        uint32_t wave_base = 0;
        for (int w = 0; w < wid; w++) wave_base += wsum[w];
        uint32_t tile_count = 0;
        for (int w = 0; w < TPB / WAVE; w++) tile_count += wsum[w];
        // publish AGG immediately, then ISSUE next tile's loads so their
        // latency overlaps the lookback walk below
        if (wid == 0 && lane == 0 && t > 0)
            __hip_atomic_store(&tile_desc[t], FILT_AGG | tile_count, __ATOMIC_RELAXED,
                               __HIP_MEMORY_SCOPE_AGENT);
        if (t + gridDim.x < n_tiles) load_tile(t + gridDim.x, v[cur ^ 1]);
        if (wid == 0) {
            if (t == 0) {
                if (lane == 0) {
                    __hip_atomic_store(&tile_desc[0], FILT_PREFIX | tile_count,
                                       __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
                    sh_excl = 0;
                }
            } else {
                unsigned long long running = 0;
                int64_t base = (int64_t)t - 1;
                uint64_t spins = 0;
                for (;;) {
                    int64_t idx = base - lane;
                    unsigned long long d =
                        idx >= 0 ? __hip_atomic_load(&tile_desc[idx], __ATOMIC_RELAXED,
                                                     __HIP_MEMORY_SCOPE_AGENT)
                                 : FILT_PREFIX;
                    unsigned long long flag = d & ~FILT_CNT_MASK;
                    uint64_t prefix_mask = __ballot(flag == FILT_PREFIX);
                    uint64_t invalid_mask = __ballot(flag == 0);
                    int first_prefix = prefix_mask ? (__ffsll((unsigned long long)prefix_mask) - 1) : WAVE;
                    int first_invalid = invalid_mask ? (__ffsll((unsigned long long)invalid_mask) - 1) : WAVE;
                    if (first_prefix < first_invalid) {
                        unsigned long long contrib =
                            (lane <= first_prefix) ? (d & FILT_CNT_MASK) : 0;
                        for (int off = WAVE / 2; off > 0; off >>= 1)
                            contrib += __shfl_down(contrib, off, WAVE);
                        running += __shfl(contrib, 0, WAVE);
                        break;
                    }
                    if (first_invalid == WAVE) {
                        unsigned long long contrib = d & FILT_CNT_MASK;
                        for (int off = WAVE / 2; off > 0; off >>= 1)
                            contrib += __shfl_down(contrib, off, WAVE);
                        running += __shfl(contrib, 0, WAVE);
                        base -= WAVE;
                        continue;
                    }
                    if (++spins > (1ull << 28)) {
                        if (lane == 0) atomicOr(error_out, 1ull);
                        running = 0;
                        break;
                    }
                }
                if (lane == 0) {
                    __hip_atomic_store(&tile_desc[t], FILT_PREFIX | (running + tile_count),
                                       __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
                    sh_excl = running;
                }
            }
            if (lane == 0 && t == n_tiles - 1) *total_out = sh_excl + tile_count;
        }
        __syncthreads();
        uint64_t w = sh_excl + wave_base + my_excl;
        #pragma unroll
        for (int j = 0; j < ITEMS; j++)
            if (pm & (1u << j)) out[w++] = v[cur][j];
        __syncthreads();
    }
}

extern "C" int gpue_scan_filter_i64_lt_sp(gpue_session* s, gpue_dbuf* in, uint64_t n,
                                          int64_t theta, gpue_dbuf* out, uint64_t* out_count);
int gpue_scan_filter_i64_lt_sp(gpue_session* s, gpue_dbuf* in, uint64_t n, int64_t theta,
                               gpue_dbuf* out, uint64_t* out_count) {
    ARG_CHECK(s && in && out && out_count && in->bytes >= n * 8 && n > 0);
    // r02 sweep (profiles/r02_filter_items_sweep.log): 32 rows/thread wins at
    // every selectivity (s=0.01: 3.46 -> 2.40 ms, +44%) — larger tiles mean
    // 4x fewer lookback pipeline stages
    int items = env_cap("GPUE_FILT_ITEMS", 32);
    if (items != 8 && items != 16 && items != 32) items = 16;
    // wave-coalesced layout is the measured default (r02: s=0.01 2.13 ->
    // 1.73 ms, s=0.5 4.45 -> 2.64 ms — profiles/r02_filter_wl.log); an
    // explicit GPUE_FILT_WL=0 or a non-default ITEMS sweep selects the
    // per-thread-contiguous baseline
    const char* wl_env = getenv("GPUE_FILT_WL");
    bool use_wl = wl_env ? atoi(wl_env) != 0 : items == 32;
    if (use_wl) items = 32; // the wl kernel's tile is TPB*32
    // TPB sweep (same log): 1024-thread blocks (32 K-row tiles) win at every
    // selectivity — s=0.01 2.14 ms (3784 GB/s), and single-pass now beats
    // the two-pass form even at s=0.5 (4.45 vs 4.81 ms)
    int tpb = env_cap("GPUE_FILT_TPB", 1024);
    if (tpb != 256 && tpb != 512 && tpb != 1024) tpb = BLOCK;
    uint64_t tile = (uint64_t)tpb * items;
    uint64_t n_tiles = (n + tile - 1) / tile;
    unsigned long long* d_desc = nullptr;
    unsigned long long* d_misc = nullptr; // {ticket(unused), total, error}
    HIP_CHECK(hipMalloc(&d_desc, n_tiles * 8));
    HIP_CHECK(hipMalloc(&d_misc, 3 * 8));
    HIP_CHECK(hipMemsetAsync(d_desc, 0, n_tiles * 8, s->stream));
    HIP_CHECK(hipMemsetAsync(d_misc, 0, 3 * 8, s->stream));
    auto kern = items == 8 ? k_filter_lookback<8, 256>
               : items == 32 ? k_filter_lookback<32, 256> : k_filter_lookback<16, 256>;
    if (tpb == 512) kern = items == 32 ? k_filter_lookback<32, 512>
                                       : k_filter_lookback<16, 512>;
    else if (tpb == 1024) kern = items == 32 ? k_filter_lookback<32, 1024>
                                             : k_filter_lookback<16, 1024>;
    const char* pipe = getenv("GPUE_FILT_PIPE");
    if (pipe && atoi(pipe)) {
        if (tpb == 1024) kern = items == 32 ? k_filter_lookback_pipe<32, 1024>
                                            : k_filter_lookback_pipe<16, 1024>;
        else if (tpb == 512) kern = items == 32 ? k_filter_lookback_pipe<32, 512>
                                                : k_filter_lookback_pipe<16, 512>;
        else kern = items == 32 ? k_filter_lookback_pipe<32, 256>
                                : k_filter_lookback_pipe<16, 256>;
    }
    if (use_wl) {
        kern = tpb == 1024 ? k_filter_lookback_wl<1024>
               : tpb == 512 ? k_filter_lookback_wl<512> : k_filter_lookback_wl<256>;
    }
    const char* rr = getenv("GPUE_FILT_RR");
    if (rr && atoi(rr)) {
        kern = items == 32 ? k_filter_lookback_rr<32, 256> : k_filter_lookback_rr<16, 256>;
        if (tpb == 512) kern = items == 32 ? k_filter_lookback_rr<32, 512>
                                           : k_filter_lookback_rr<16, 512>;
        else if (tpb == 1024) kern = items == 32 ? k_filter_lookback_rr<32, 1024>
                                                 : k_filter_lookback_rr<16, 1024>;
    }
    // fully-resident grid for the static-assignment lookback (see kernel
    // comment); the occupancy API can over-report by one block per CU on
    // SGPR-heavy 256-thread kernels (MI355X_MICROARCH.md) — subtract one
    int blocks_per_cu = 0;
    HIP_CHECK(hipOccupancyMaxActiveBlocksPerMultiprocessor(&blocks_per_cu, kern, tpb, 0));
    if (blocks_per_cu > 1) blocks_per_cu -= 1;
    if (blocks_per_cu < 1) blocks_per_cu = 1;
    hipDeviceProp_t props;
    HIP_CHECK(hipGetDeviceProperties(&props, s->device));
    uint64_t resident = (uint64_t)props.multiProcessorCount * blocks_per_cu;
    uint32_t nb = (uint32_t)(n_tiles < resident ? n_tiles : resident);
    hipLaunchKernelGGL(kern, dim3(nb), dim3(tpb), 0, s->stream,
                       (const int64_t*)in->ptr, n, theta, (int64_t*)out->ptr, d_desc,
                       d_misc, d_misc + 1, d_misc + 2);
    unsigned long long h_misc[3];
    HIP_CHECK(hipMemcpyAsync(h_misc, d_misc, 3 * 8, hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    (void)hipFree(d_desc);
    (void)hipFree(d_misc);
    if (h_misc[2]) {
        snprintf(g_err, sizeof(g_err), "filter lookback spin timeout (predecessor stuck)");
        return GPUE_ERR_HIP;
    }
    *out_count = h_misc[1];
    return GPUE_OK;
}

int gpue_scan_filter_i64_lt(gpue_session* s, gpue_dbuf* in, uint64_t n, int64_t theta,
                            gpue_dbuf* out, uint64_t* out_count) {
    ARG_CHECK(s && in && out && out_count && in->bytes >= n * 8);
    uint32_t nb = grid_for(n);
    uint64_t tile = (n + nb - 1) / nb;
    uint64_t* d_counts = nullptr;
    HIP_CHECK(hipMalloc(&d_counts, (nb + 1) * sizeof(uint64_t)));
    hipLaunchKernelGGL(k_filter_count, dim3(nb), dim3(BLOCK), 0, s->stream,
                       (const int64_t*)in->ptr, n, theta, tile, d_counts);
    hipLaunchKernelGGL(k_scan_small, dim3(1), dim3(1), 0, s->stream, d_counts, nb);
    hipLaunchKernelGGL(k_filter_emit, dim3(nb), dim3(BLOCK), 0, s->stream,
                       (const int64_t*)in->ptr, n, theta, tile, d_counts,
                       (int64_t*)out->ptr);
    uint64_t total = 0;
    HIP_CHECK(hipMemcpyAsync(&total, d_counts + nb, sizeof(uint64_t),
                             hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    (void)hipFree(d_counts);
    *out_count = total;
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// hash-join build — RANGE_DIRECT_MAPPING (reference
// join_hash_map_method.hpp:625-707). min/max by device reduction, then either
// a payload scatter (fused fast path) or first/next chain scatter via
// atomicExch (the CPU's sequential next[i]=first[b]; first[b]=i becomes one
// atomic swap per row; chain ORDER is scatter-order — the match multiset is
// unchanged, SURVEY.md §7 hard part (b)).
// ---------------------------------------------------------------------------
__global__ void k_minmax_i32(const int32_t* __restrict__ keys, uint64_t n,
                             int32_t* mn_out, int32_t* mx_out) {
    int32_t mn = INT32_MAX, mx = INT32_MIN;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        int32_t v = keys[i];
        mn = min(mn, v);
        mx = max(mx, v);
    }
    for (int off = WAVE / 2; off > 0; off >>= 1) {
        mn = min(mn, __shfl_down(mn, off, WAVE));
        mx = max(mx, __shfl_down(mx, off, WAVE));
    }
    if ((threadIdx.x & (WAVE - 1)) == 0) {
        atomicMin(mn_out, mn);
        atomicMax(mx_out, mx);
    }
}

__global__ void k_build_payload(const int32_t* __restrict__ keys,
                                const uint32_t* __restrict__ payloads, uint64_t n,
                                int64_t min_key, uint32_t* __restrict__ first) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
        first[keys[i] - min_key] = payloads[i];
}

// Derive the semi-join bitset (1 bit/key) and the 16-bit payload copy from
// the u32 payload array — built once at build time, probed every launch.
// bounding range [min idx, max idx] of nonzero payload entries
__global__ void k_set_bounds(const uint32_t* __restrict__ first, uint64_t interval,
                             int32_t* __restrict__ lo_out, int32_t* __restrict__ hi_out) {
    int32_t lo = INT32_MAX, hi = INT32_MIN;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < interval; i += stride) {
        if (first[i] != 0) {
            lo = min(lo, (int32_t)i);
            hi = max(hi, (int32_t)i);
        }
    }
    for (int off = WAVE / 2; off > 0; off >>= 1) {
        lo = min(lo, __shfl_down(lo, off, WAVE));
        hi = max(hi, __shfl_down(hi, off, WAVE));
    }
    if ((threadIdx.x & (WAVE - 1)) == 0) {
        atomicMin(lo_out, lo);
        atomicMax(hi_out, hi);
    }
}

// Fold the passing-key bitset into the fixed 2^19-bit LDS prefilter
// (conservative: a fold-set bit means "some key with this masked index
// passes"). Built once per table; read-only at probe time.
// split two-probe fold: two independent 2^18-bit arrays in one 64 KB
// allocation — bloom math at 56 K passing keys: per-array density 0.214,
// double-probe false-positive 0.193^2 = 3.7% vs the single 2^19 fold's 6.1%
// wide split fold: two 2^19-bit arrays (128 KB total) for kernels whose LDS
// budget allows full-size folds — per-array density 0.107, double-probe
// false-positive 0.101^2 = 1.0%
__global__ void k_build_prefilter_split_w(const uint32_t* __restrict__ bitset,
                                          uint64_t set_interval,
                                          uint32_t* __restrict__ pf2) {
    const uint32_t M19 = (1u << 19) - 1;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < set_interval;
         i += stride) {
        if ((bitset[i >> 5] >> (i & 31)) & 1u) {
            uint32_t a = (uint32_t)i & M19;
            uint32_t b = ((uint32_t)i * 2654435761u) >> 13; // top 19 bits
            atomicOr(&pf2[a >> 5], 1u << (a & 31));
            atomicOr(&pf2[(1u << 14) + (b >> 5)], 1u << (b & 31));
        }
    }
}

__global__ void k_build_prefilter_split(const uint32_t* __restrict__ bitset,
                                        uint64_t set_interval,
                                        uint32_t* __restrict__ pf2) {
    const uint32_t M18 = (1u << 18) - 1;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < set_interval;
         i += stride) {
        if ((bitset[i >> 5] >> (i & 31)) & 1u) {
            uint32_t a = (uint32_t)i & M18;
            uint32_t b = ((uint32_t)i * 2654435761u) >> 14; // top 18 bits
            atomicOr(&pf2[a >> 5], 1u << (a & 31));
            atomicOr(&pf2[(1u << 13) + (b >> 5)], 1u << (b & 31));
        }
    }
}

__global__ void k_build_prefilter(const uint32_t* __restrict__ bitset, uint64_t set_interval,
                                  uint32_t* __restrict__ prefilter) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < set_interval;
         i += stride) {
        if ((bitset[i >> 5] >> (i & 31)) & 1u) {
            uint32_t f = (uint32_t)i & ((1u << 19) - 1);
            atomicOr(&prefilter[f >> 5], 1u << (f & 31));
        }
    }
}

__global__ void k_derive_bitset(const uint32_t* __restrict__ first, uint64_t offset,
                                uint64_t interval, uint32_t* __restrict__ bitset) {
    // bit j of word w == (first[offset + w*32 + j] != 0)
    uint64_t nwords = (interval + 31) / 32;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t w = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; w < nwords; w += stride) {
        uint32_t bits = 0;
        uint64_t base = w * 32;
        #pragma unroll 4
        for (int j = 0; j < 32; j++) {
            uint64_t idx = base + j;
            if (idx < interval && first[offset + idx] != 0) bits |= (1u << j);
        }
        bitset[w] = bits;
    }
}

__global__ void k_derive_u16(const uint32_t* __restrict__ first, uint64_t interval,
                             uint16_t* __restrict__ first16, uint32_t* __restrict__ overflow) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < interval; i += stride) {
        uint32_t v = first[i];
        if (v >= 65536u) atomicOr(overflow, 1u);
        first16[i] = (uint16_t)v;
    }
}

__global__ void k_build_range_direct(const int32_t* __restrict__ keys, uint64_t row_count,
                                     int64_t min_key, uint32_t* __restrict__ first,
                                     uint32_t* __restrict__ next) {
    // keys is 1-based (row 0 = sentinel); rows 1..row_count
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = 1 + (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i <= row_count;
         i += stride)
        next[i] = atomicExch(&first[keys[i] - min_key], (uint32_t)i);
}

static int join_table_minmax(gpue_session* s, const int32_t* d_keys, uint64_t n,
                             int64_t* mn_out, int64_t* mx_out) {
    int32_t h_init[2] = {INT32_MAX, INT32_MIN};
    int32_t* d_mm = nullptr;
    HIP_CHECK(hipMalloc(&d_mm, 2 * sizeof(int32_t)));
    HIP_CHECK(hipMemcpyAsync(d_mm, h_init, 2 * sizeof(int32_t), hipMemcpyHostToDevice, s->stream));
    hipLaunchKernelGGL(k_minmax_i32, dim3(grid_for(n)), dim3(BLOCK), 0, s->stream,
                       d_keys, n, d_mm, d_mm + 1);
    int32_t h_mm[2];
    HIP_CHECK(hipMemcpyAsync(h_mm, d_mm, 2 * sizeof(int32_t), hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    (void)hipFree(d_mm);
    *mn_out = h_mm[0];
    *mx_out = h_mm[1];
    return GPUE_OK;
}

int gpue_join_build_payload_i32(gpue_session* s, gpue_dbuf* keys, gpue_dbuf* payloads,
                                uint64_t n_rows, gpue_join_table** out) {
    ARG_CHECK(s && keys && payloads && out && n_rows > 0);
    ARG_CHECK(keys->bytes >= n_rows * 4 && payloads->bytes >= n_rows * 4);
    int64_t mn, mx;
    int rc = join_table_minmax(s, (const int32_t*)keys->ptr, n_rows, &mn, &mx);
    if (rc != GPUE_OK) return rc;
    uint64_t interval = (uint64_t)(mx - mn + 1);
    ARG_CHECK(interval < (1ull << 32)); // selector's RANGE_DIRECT gate (join_hash_table.cpp:287)
    gpue_join_table* t = new gpue_join_table();
    t->s = s; t->min_key = mn; t->max_key = mx;
    t->bucket_size = interval; t->row_count = n_rows;
    HIP_CHECK(hipMalloc(&t->first, interval * sizeof(uint32_t)));
    HIP_CHECK(hipMemsetAsync(t->first, 0, interval * sizeof(uint32_t), s->stream));
    hipLaunchKernelGGL(k_build_payload, dim3(grid_for(n_rows)), dim3(BLOCK), 0, s->stream,
                       (const int32_t*)keys->ptr, (const uint32_t*)payloads->ptr, n_rows,
                       mn, t->first);
    HIP_CHECK(hipGetLastError());
    // derive the probe-side structures: set bounds first,
    // then the bitset over just the passing keys' bounding range
    int32_t h_bounds[2] = {INT32_MAX, INT32_MIN};
    int32_t* d_bounds = nullptr;
    HIP_CHECK(hipMalloc(&d_bounds, 2 * sizeof(int32_t)));
    HIP_CHECK(hipMemcpyAsync(d_bounds, h_bounds, sizeof(h_bounds), hipMemcpyHostToDevice,
                             s->stream));
    hipLaunchKernelGGL(k_set_bounds, dim3(grid_for(interval)), dim3(BLOCK), 0, s->stream,
                       t->first, interval, d_bounds, d_bounds + 1);
    HIP_CHECK(hipMemcpyAsync(h_bounds, d_bounds, sizeof(h_bounds), hipMemcpyDeviceToHost,
                             s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    (void)hipFree(d_bounds);
    uint64_t set_off, set_interval;
    if (h_bounds[0] > h_bounds[1]) { // empty pass set
        t->set_min = mn;
        t->set_max = mn;
        set_off = 0;
        set_interval = 1;
    } else {
        t->set_min = mn + h_bounds[0];
        t->set_max = mn + h_bounds[1];
        set_off = (uint64_t)h_bounds[0];
        set_interval = (uint64_t)(h_bounds[1] - h_bounds[0] + 1);
    }
    uint64_t nwords = (set_interval + 31) / 32;
    HIP_CHECK(hipMalloc(&t->bitset, nwords * sizeof(uint32_t)));
    hipLaunchKernelGGL(k_derive_bitset, dim3(grid_for(nwords)), dim3(BLOCK), 0, s->stream,
                       t->first, set_off, set_interval, t->bitset);
    HIP_CHECK(hipMalloc(&t->prefilter, (1u << 19) / 8));
    HIP_CHECK(hipMemsetAsync(t->prefilter, 0, (1u << 19) / 8, s->stream));
    hipLaunchKernelGGL(k_build_prefilter, dim3(grid_for(set_interval)), dim3(BLOCK), 0,
                       s->stream, t->bitset, set_interval, t->prefilter);
    HIP_CHECK(hipMalloc(&t->prefilter2, (1u << 19) / 8));
    HIP_CHECK(hipMemsetAsync(t->prefilter2, 0, (1u << 19) / 8, s->stream));
    hipLaunchKernelGGL(k_build_prefilter_split, dim3(grid_for(set_interval)), dim3(BLOCK),
                       0, s->stream, t->bitset, set_interval, t->prefilter2);
    HIP_CHECK(hipMalloc(&t->prefilter2w, (1u << 20) / 8));
    HIP_CHECK(hipMemsetAsync(t->prefilter2w, 0, (1u << 20) / 8, s->stream));
    hipLaunchKernelGGL(k_build_prefilter_split_w, dim3(grid_for(set_interval)), dim3(BLOCK),
                       0, s->stream, t->bitset, set_interval, t->prefilter2w);
    uint32_t* d_ovf = nullptr;
    HIP_CHECK(hipMalloc(&d_ovf, sizeof(uint32_t)));
    HIP_CHECK(hipMemsetAsync(d_ovf, 0, sizeof(uint32_t), s->stream));
    HIP_CHECK(hipMalloc(&t->first16, interval * sizeof(uint16_t)));
    hipLaunchKernelGGL(k_derive_u16, dim3(grid_for(interval)), dim3(BLOCK), 0, s->stream,
                       t->first, interval, t->first16, d_ovf);
    uint32_t ovf = 0;
    HIP_CHECK(hipMemcpyAsync(&ovf, d_ovf, sizeof(uint32_t), hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    (void)hipFree(d_ovf);
    if (ovf) { (void)hipFree(t->first16); t->first16 = nullptr; }
    *out = t;
    return GPUE_OK;
}

int gpue_join_build_range_direct_i32(gpue_session* s, gpue_dbuf* keys, uint64_t row_count,
                                     gpue_join_table** out) {
    ARG_CHECK(s && keys && out && row_count > 0);
    ARG_CHECK(keys->bytes >= (row_count + 1) * 4);
    // min/max over rows 1..row_count (row 0 is the sentinel default row)
    int64_t mn, mx;
    int rc = join_table_minmax(s, (const int32_t*)keys->ptr + 1, row_count, &mn, &mx);
    if (rc != GPUE_OK) return rc;
    uint64_t interval = (uint64_t)(mx - mn + 1);
    ARG_CHECK(interval < (1ull << 32));
    gpue_join_table* t = new gpue_join_table();
    t->s = s; t->min_key = mn; t->max_key = mx;
    t->bucket_size = interval; t->row_count = row_count;
    t->kind = gpue_join_table::RANGE_DIRECT;
    HIP_CHECK(hipMalloc(&t->first, interval * sizeof(uint32_t)));
    HIP_CHECK(hipMalloc(&t->next, (row_count + 1) * sizeof(uint32_t)));
    HIP_CHECK(hipMemsetAsync(t->first, 0, interval * sizeof(uint32_t), s->stream));
    HIP_CHECK(hipMemsetAsync(t->next, 0, (row_count + 1) * sizeof(uint32_t), s->stream));
    hipLaunchKernelGGL(k_build_range_direct, dim3(grid_for(row_count)), dim3(BLOCK), 0,
                       s->stream, (const int32_t*)keys->ptr, row_count, mn, t->first, t->next);
    HIP_CHECK(hipGetLastError());
    *out = t;
    return GPUE_OK;
}

void gpue_join_table_destroy(gpue_join_table* t) {
    if (!t) return;
    if (t->first) (void)hipFree(t->first);
    if (t->next) (void)hipFree(t->next);
    if (t->bitset) (void)hipFree(t->bitset);
    if (t->first16) (void)hipFree(t->first16);
    if (t->prefilter) (void)hipFree(t->prefilter);
    if (t->prefilter2) (void)hipFree(t->prefilter2);
    if (t->prefilter2w) (void)hipFree(t->prefilter2w);
    if (t->dense_groups) (void)hipFree(t->dense_groups);
    if (t->build_keys) (void)hipFree(t->build_keys);
    if (t->key_bytes) (void)hipFree(t->key_bytes);
    if (t->key_offsets) (void)hipFree(t->key_offsets);
    if (t->key_nulls) (void)hipFree(t->key_nulls);
    if (t->build_keys64) (void)hipFree(t->build_keys64);
    if (t->build_keys128) (void)hipFree(t->build_keys128);
    delete t;
}

int gpue_join_table_minmax(gpue_join_table* t, int64_t* min_out, int64_t* max_out) {
    ARG_CHECK(t && min_out && max_out);
    *min_out = t->min_key;
    *max_out = t->max_key;
    return GPUE_OK;
}

int gpue_join_table_first_d2h(gpue_join_table* t, uint32_t* dst, uint64_t n_entries) {
    ARG_CHECK(t && dst && n_entries <= t->bucket_size);
    HIP_CHECK(hipMemcpy(dst, t->first, n_entries * sizeof(uint32_t), hipMemcpyDeviceToHost));
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// BUCKET_CHAINED — the selector's generic fallback (reference
// join_hash_map_method.hpp:37-120): bucket = JoinKeyHash (multiplicative,
// join_hash_map_helper.h:36-44, bit-identical to the oracle), scatter
// next[i] = atomicExch(&first[b], i). Chain ORDER under shared buckets is
// scatter-order (the CPU's is reverse insertion order) — the emitted match
// multiset is identical and that is the result (SURVEY.md §7 hard part (a)).
// ---------------------------------------------------------------------------
__device__ static inline uint32_t join_hash_u32(uint32_t v, uint32_t num_log_buckets) {
    v ^= v >> (32 - num_log_buckets);
    return (v * 2654435761u) >> (32 - num_log_buckets);
}

__global__ void k_build_bucket_chained(const uint32_t* __restrict__ keys, uint64_t row_count,
                                       uint32_t log_bucket_size, uint32_t* __restrict__ first,
                                       uint32_t* __restrict__ next) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = 1 + (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i <= row_count;
         i += stride)
        next[i] = atomicExch(&first[join_hash_u32(keys[i], log_bucket_size)], (uint32_t)i);
}

// 8-byte (BIGINT) key variants — JoinKeyHash<8> (join_hash_map_helper.h:
// 46-55, multiplier 11400714819323198485; pinned by the oracle's stats64
// KATs), same chained structure with u64 build-key compares.
__device__ static inline uint32_t join_hash_u64_dev(uint64_t v, uint32_t num_log_buckets) {
    v ^= v >> (64 - num_log_buckets);
    return (uint32_t)((v * 11400714819323198485ull) >> (64 - num_log_buckets));
}

__global__ void k_build_bucket_chained64(const uint64_t* __restrict__ keys,
                                         uint64_t row_count, uint32_t log_bucket_size,
                                         uint32_t* __restrict__ first,
                                         uint32_t* __restrict__ next) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = 1 + (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i <= row_count;
         i += stride)
        next[i] = atomicExch(&first[join_hash_u64_dev(keys[i], log_bucket_size)],
                             (uint32_t)i);
}

__global__ void k_probe_count_bc64(const uint64_t* __restrict__ probe_keys, uint64_t n,
                                   uint32_t log_bucket_size,
                                   const uint32_t* __restrict__ first,
                                   const uint32_t* __restrict__ next,
                                   const uint64_t* __restrict__ build_keys, int mode,
                                   uint32_t* __restrict__ row_counts) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        uint64_t k = probe_keys[i];
        uint32_t b = first[join_hash_u64_dev(k, log_bucket_size)];
        uint32_t c = 0;
        while (b != 0) {
            c += (build_keys[b] == k);
            b = next[b];
        }
        row_counts[i] = join_mode_count(c, mode);
    }
}

__global__ void k_probe_emit_bc64(const uint64_t* __restrict__ probe_keys, uint64_t n,
                                  uint32_t log_bucket_size,
                                  const uint32_t* __restrict__ first,
                                  const uint32_t* __restrict__ next,
                                  const uint64_t* __restrict__ build_keys, int mode,
                                  const uint32_t* __restrict__ row_counts,
                                  const uint64_t* __restrict__ row_offsets,
                                  uint32_t* __restrict__ out_probe,
                                  uint32_t* __restrict__ out_build) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        if (row_counts[i] == 0) continue;
        uint64_t pos = row_offsets[i];
        uint64_t k = probe_keys[i];
        uint32_t b = first[join_hash_u64_dev(k, log_bucket_size)];
        uint32_t c = 0;
        while (b != 0) {
            if (build_keys[b] == k) {
                if (mode == 0 || mode == 3 || (mode == 1 && c == 0)) {
                    out_probe[pos] = (uint32_t)i;
                    out_build[pos] = b;
                    pos++;
                }
                c++;
                if (mode == 1 || mode == 2) break;
            }
            b = next[b];
        }
        if (c == 0 && (mode == 2 || mode == 3)) {
            out_probe[pos] = (uint32_t)i;
            out_build[pos] = 0;
        }
    }
}

__global__ void k_build_bucket_chained_nulls(const uint32_t* __restrict__ keys,
                                             const uint8_t* __restrict__ is_nulls,
                                             uint64_t row_count, uint32_t log_bucket_size,
                                             uint32_t* __restrict__ first,
                                             uint32_t* __restrict__ next) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = 1 + (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i <= row_count;
         i += stride) {
        if (is_nulls[i]) continue; // null build rows never enter a chain
        next[i] = atomicExch(&first[join_hash_u32(keys[i], log_bucket_size)], (uint32_t)i);
    }
}

// pack two int32 key columns into one 8-byte key (the selector's
// SERIALIZED_FIXED_SIZE_BIGINT path, join_hash_map_helper.h:112-136)
__global__ void k_pack_keys_2xi32(const int32_t* __restrict__ a,
                                  const int32_t* __restrict__ b, uint64_t n,
                                  uint64_t* __restrict__ out) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
        out[i] = (uint64_t)(uint32_t)a[i] | ((uint64_t)(uint32_t)b[i] << 32);
}

extern "C" int gpue_pack_keys_2xi32(gpue_session* s, gpue_dbuf* a, gpue_dbuf* b, uint64_t n,
                                    gpue_dbuf* out);
int gpue_pack_keys_2xi32(gpue_session* s, gpue_dbuf* a, gpue_dbuf* b, uint64_t n,
                         gpue_dbuf* out) {
    ARG_CHECK(s && a && b && out && out->bytes >= n * 8);
    hipLaunchKernelGGL(k_pack_keys_2xi32, dim3(grid_stream(n)), dim3(BLOCK), 0, s->stream,
                       (const int32_t*)a->ptr, (const int32_t*)b->ptr, n,
                       (uint64_t*)out->ptr);
    HIP_CHECK(hipGetLastError());
    return GPUE_OK;
}

// calc_bucket_size (join_hash_map_helper.h:68-78): NormalizeCapacity(n+(n-1)/4)+1
static uint32_t calc_bucket_size(uint32_t size) {
    uint64_t expect = (uint64_t)size + (size - 1) / 4;
    if (expect >= (1ull << 31)) return 1u << 31;
    uint64_t norm = expect ? (~0ull >> __builtin_clzll(expect)) : 1;
    return (uint32_t)(norm + 1);
}

// ---------------------------------------------------------------------------
// JoinHashMapSelector restatement (reference join_hash_table.cpp:164-344) —
// the automatic key-constructor + map-method decision the reference makes at
// JoinHashTable::build time. Pure host logic; enum values in gpue.h. The
// decision is the REFERENCE's (CPU cache-size thresholds as parameters, the
// session flags defaulting true per SessionVariable.java:2060-2067);
// gpue_join_build_auto_i32 below maps the decision onto the GPU table kinds.
// ---------------------------------------------------------------------------
extern "C" {
int gpue_join_select_key_constructor(int num_keys, const int32_t* fixed_sizes,
                                     const uint8_t* null_safe,
                                     int enable_fixed_size_string,
                                     int32_t* packed_bytes_out);
int gpue_join_select_varchar_constructor(int32_t max_size, int enable_fixed_size_string);
int gpue_join_select_method(int key_constructor, int lt_class, uint64_t row_count,
                            int64_t min_value, int64_t max_value, int mode,
                            int with_other_conjunct, int enable_range_direct,
                            int enable_linear_chained, uint64_t l2_size,
                            uint64_t l3_size);
}

// _determine_key_constructor (join_hash_table.cpp:164-229)
int gpue_join_select_key_constructor(int num_keys, const int32_t* fixed_sizes,
                                     const uint8_t* null_safe,
                                     int enable_fixed_size_string,
                                     int32_t* packed_bytes_out) {
    (void)enable_fixed_size_string;
    *packed_bytes_out = 0;
    if (num_keys == 1 && !null_safe[0]) { // :175 single non-null-safe key
        int32_t sz = fixed_sizes[0];
        if (sz > 0) {
            *packed_bytes_out = sz;
            return GPUE_KEYCON_ONE_KEY;
        }
        return GPUE_KEYCON_ONE_KEY_VARCHAR;
    }
    // :199-229 multi-key / null-safe: pack fixed widths (+1 B per null-safe)
    int64_t total = 0;
    for (int i = 0; i < num_keys; i++) {
        int32_t cur = fixed_sizes[i];
        if (cur <= 0) return GPUE_KEYCON_SERIALIZED_VARCHAR;
        cur += null_safe[i] ? 1 : 0;
        total += cur;
    }
    if (total > 16) { // > 16 B never fixed-packs; packed width unused
        return GPUE_KEYCON_SERIALIZED_VARCHAR;
    }
    *packed_bytes_out = (int32_t)total;
    if (total <= 4) return GPUE_KEYCON_FIXED_INT;
    if (total <= 8) return GPUE_KEYCON_FIXED_BIGINT;
    if (total <= 16) return GPUE_KEYCON_FIXED_LARGEINT;
    return GPUE_KEYCON_SERIALIZED_VARCHAR;
}

// single-varchar fixed-size-string refinement (:178-194 via
// _get_binary_column_max_size :121-162)
int gpue_join_select_varchar_constructor(int32_t max_size, int enable_fixed_size_string) {
    if (!enable_fixed_size_string || max_size <= 0) return GPUE_KEYCON_ONE_KEY_VARCHAR;
    if (max_size <= 4) return GPUE_KEYCON_FIXED_INT;
    if (max_size <= 8) return GPUE_KEYCON_FIXED_BIGINT;
    if (max_size <= 16) return GPUE_KEYCON_FIXED_LARGEINT;
    return GPUE_KEYCON_ONE_KEY_VARCHAR;
}

// _determine_hash_map_method (:231-256) + _try_use_range_direct_mapping
// (:270-321) + _try_use_linear_chained (:323-344)
int gpue_join_select_method(int key_constructor, int lt_class, uint64_t row_count,
                            int64_t min_value, int64_t max_value, int mode,
                            int with_other_conjunct, int enable_range_direct,
                            int enable_linear_chained, uint64_t l2_size,
                            uint64_t l3_size) {
    if (lt_class == GPUE_LT_TINY) return GPUE_JM_DIRECT; // :239
    const bool semi_or_anti_no_conj =
        (mode == GPUE_JOIN_LEFT_SEMI || mode == GPUE_JOIN_LEFT_ANTI) && !with_other_conjunct;
    uint64_t rc1 = row_count + 1;
    const uint64_t bucket_size =
        calc_bucket_size(rc1 > 0xFFFFFFFFull ? 0xFFFFFFFFu : (uint32_t)rc1);
    if (key_constructor == GPUE_KEYCON_ONE_KEY &&
        (lt_class == GPUE_LT_INT || lt_class == GPUE_LT_BIGINT)) { // :242
        if (enable_range_direct && row_count > 0 &&
            !(min_value == INT64_MIN && max_value == INT64_MAX)) { // :283 overflow guard
            uint64_t interval = (uint64_t)max_value - (uint64_t)min_value + 1;
            if (interval < 0xFFFFFFFFull) { // :288
                if (semi_or_anti_no_conj) { // :301 one bit vs 8 B first+next
                    uint64_t memory = (interval + 7) / 8;
                    if (memory <= bucket_size * 64 || memory <= l3_size / 2)
                        return GPUE_JM_RANGE_DIRECT_SET;
                } else {
                    if (interval <= bucket_size || interval <= l2_size) // :307
                        return GPUE_JM_RANGE_DIRECT;
                    if (interval / 4 + row_count * 4 <=
                        (bucket_size + bucket_size / 10) * 4) // :310-317
                        return GPUE_JM_DENSE_RANGE_DIRECT;
                }
            }
        }
    }
    // :323-344 — 24-bit fp-packed bucket index cap (join_hash_map_method.h:135)
    if (enable_linear_chained && bucket_size <= 0xFFFFFFu)
        return semi_or_anti_no_conj ? GPUE_JM_LINEAR_CHAINED_SET : GPUE_JM_LINEAR_CHAINED;
    return GPUE_JM_BUCKET_CHAINED; // fallback :258-263 (asof out of scope)
}

extern "C" int gpue_join_build_bucket_chained_nulls_u32(gpue_session* s, gpue_dbuf* keys,
                                                        gpue_dbuf* is_nulls,
                                                        uint64_t row_count,
                                                        gpue_join_table** out);
int gpue_join_build_bucket_chained_nulls_u32(gpue_session* s, gpue_dbuf* keys,
                                             gpue_dbuf* is_nulls, uint64_t row_count,
                                             gpue_join_table** out) {
    ARG_CHECK(s && keys && is_nulls && out && row_count > 0 && row_count + 1 < (1ull << 31));
    ARG_CHECK(keys->bytes >= (row_count + 1) * 4 && is_nulls->bytes >= row_count + 1);
    gpue_join_table* t = new gpue_join_table();
    t->s = s;
    t->kind = gpue_join_table::BUCKET_CHAINED;
    t->row_count = row_count;
    t->bucket_size = calc_bucket_size((uint32_t)(row_count + 1));
    t->log_bucket_size = (uint32_t)__builtin_ctzll(t->bucket_size);
    HIP_CHECK(hipMalloc(&t->first, t->bucket_size * sizeof(uint32_t)));
    HIP_CHECK(hipMalloc(&t->next, (row_count + 1) * sizeof(uint32_t)));
    HIP_CHECK(hipMalloc(&t->build_keys, (row_count + 1) * sizeof(uint32_t)));
    HIP_CHECK(hipMemsetAsync(t->first, 0, t->bucket_size * sizeof(uint32_t), s->stream));
    HIP_CHECK(hipMemsetAsync(t->next, 0, (row_count + 1) * sizeof(uint32_t), s->stream));
    HIP_CHECK(hipMemcpyAsync(t->build_keys, keys->ptr, (row_count + 1) * sizeof(uint32_t),
                             hipMemcpyDeviceToDevice, s->stream));
    hipLaunchKernelGGL(k_build_bucket_chained_nulls, dim3(grid_for(row_count)), dim3(BLOCK),
                       0, s->stream, t->build_keys, (const uint8_t*)is_nulls->ptr, row_count,
                       t->log_bucket_size, t->first, t->next);
    HIP_CHECK(hipGetLastError());
    *out = t;
    return GPUE_OK;
}

extern "C" int gpue_join_build_bucket_chained_u32(gpue_session* s, gpue_dbuf* keys,
                                                  uint64_t row_count, gpue_join_table** out);
int gpue_join_build_bucket_chained_u32(gpue_session* s, gpue_dbuf* keys, uint64_t row_count,
                                       gpue_join_table** out) {
    ARG_CHECK(s && keys && out && row_count > 0 && row_count + 1 < (1ull << 31));
    ARG_CHECK(keys->bytes >= (row_count + 1) * 4);
    gpue_join_table* t = new gpue_join_table();
    t->s = s;
    t->kind = gpue_join_table::BUCKET_CHAINED;
    t->row_count = row_count;
    t->bucket_size = calc_bucket_size((uint32_t)(row_count + 1));
    t->log_bucket_size = (uint32_t)__builtin_ctzll(t->bucket_size);
    HIP_CHECK(hipMalloc(&t->first, t->bucket_size * sizeof(uint32_t)));
    HIP_CHECK(hipMalloc(&t->next, (row_count + 1) * sizeof(uint32_t)));
    HIP_CHECK(hipMalloc(&t->build_keys, (row_count + 1) * sizeof(uint32_t)));
    HIP_CHECK(hipMemsetAsync(t->first, 0, t->bucket_size * sizeof(uint32_t), s->stream));
    HIP_CHECK(hipMemsetAsync(t->next, 0, (row_count + 1) * sizeof(uint32_t), s->stream));
    HIP_CHECK(hipMemcpyAsync(t->build_keys, keys->ptr, (row_count + 1) * sizeof(uint32_t),
                             hipMemcpyDeviceToDevice, s->stream));
    hipLaunchKernelGGL(k_build_bucket_chained, dim3(grid_for(row_count)), dim3(BLOCK), 0,
                       s->stream, t->build_keys, row_count, t->log_bucket_size, t->first,
                       t->next);
    HIP_CHECK(hipGetLastError());
    *out = t;
    return GPUE_OK;
}

// forward decls: the count->scan->emit helpers are defined with the probe
// section further down this file
__global__ void k_block_sums_u32(const uint32_t* __restrict__ counts, uint64_t n,
                                 uint64_t tile, uint64_t* __restrict__ block_sums);
__global__ void k_scan_small(uint64_t* __restrict__ sums, uint64_t n_blocks);
__global__ void k_scan_offsets(const uint32_t* __restrict__ counts, uint64_t n, uint64_t tile,
                               const uint64_t* __restrict__ block_offsets,
                               uint64_t* __restrict__ offsets);

extern "C" {
int gpue_join_build_bucket_chained_u64(gpue_session* s, gpue_dbuf* keys /*u64, 1-based*/,
                                       uint64_t row_count, gpue_join_table** out);
int gpue_join_probe_emit_mode_u64(gpue_session* s, gpue_join_table* t, gpue_dbuf* probe_keys,
                                  uint64_t n_rows, int mode, gpue_dbuf* out_probe_idx,
                                  gpue_dbuf* out_build_idx, uint64_t* match_count);
}

int gpue_join_build_bucket_chained_u64(gpue_session* s, gpue_dbuf* keys, uint64_t row_count,
                                       gpue_join_table** out) {
    ARG_CHECK(s && keys && out && row_count > 0 && row_count + 1 < (1ull << 31));
    ARG_CHECK(keys->bytes >= (row_count + 1) * 8);
    gpue_join_table* t = new gpue_join_table();
    t->s = s;
    t->kind = gpue_join_table::BUCKET_CHAINED64;
    t->row_count = row_count;
    t->bucket_size = calc_bucket_size((uint32_t)(row_count + 1));
    t->log_bucket_size = (uint32_t)__builtin_ctzll(t->bucket_size);
    HIP_CHECK(hipMalloc(&t->first, t->bucket_size * sizeof(uint32_t)));
    HIP_CHECK(hipMalloc(&t->next, (row_count + 1) * sizeof(uint32_t)));
    HIP_CHECK(hipMalloc(&t->build_keys64, (row_count + 1) * sizeof(uint64_t)));
    HIP_CHECK(hipMemsetAsync(t->first, 0, t->bucket_size * sizeof(uint32_t), s->stream));
    HIP_CHECK(hipMemsetAsync(t->next, 0, (row_count + 1) * sizeof(uint32_t), s->stream));
    HIP_CHECK(hipMemcpyAsync(t->build_keys64, keys->ptr, (row_count + 1) * sizeof(uint64_t),
                             hipMemcpyDeviceToDevice, s->stream));
    hipLaunchKernelGGL(k_build_bucket_chained64, dim3(grid_for(row_count)), dim3(BLOCK), 0,
                       s->stream, t->build_keys64, row_count, t->log_bucket_size, t->first,
                       t->next);
    HIP_CHECK(hipGetLastError());
    *out = t;
    return GPUE_OK;
}

int gpue_join_probe_emit_mode_u64(gpue_session* s, gpue_join_table* t, gpue_dbuf* probe_keys,
                                  uint64_t n_rows, int mode, gpue_dbuf* out_probe_idx,
                                  gpue_dbuf* out_build_idx, uint64_t* match_count) {
    ARG_CHECK(s && t && probe_keys && match_count);
    ARG_CHECK(t->kind == gpue_join_table::BUCKET_CHAINED64);
    ARG_CHECK(mode >= 0 && mode <= 3);
    ARG_CHECK(probe_keys->bytes >= n_rows * 8);
    uint32_t nb = grid_for(n_rows);
    uint64_t tile = (n_rows + nb - 1) / nb;
    uint32_t* d_counts = nullptr;
    uint64_t* d_bsums = nullptr;
    uint64_t* d_offsets = nullptr;
    HIP_CHECK(hipMalloc(&d_counts, n_rows * 4));
    HIP_CHECK(hipMalloc(&d_bsums, (nb + 1) * 8));
    hipLaunchKernelGGL(k_probe_count_bc64, dim3(nb), dim3(BLOCK), 0, s->stream,
                       (const uint64_t*)probe_keys->ptr, n_rows, t->log_bucket_size,
                       t->first, t->next, t->build_keys64, mode, d_counts);
    hipLaunchKernelGGL(k_block_sums_u32, dim3(nb), dim3(BLOCK), 0, s->stream, d_counts,
                       n_rows, tile, d_bsums);
    hipLaunchKernelGGL(k_scan_small, dim3(1), dim3(1), 0, s->stream, d_bsums, nb);
    uint64_t total = 0;
    HIP_CHECK(hipMemcpyAsync(&total, d_bsums + nb, 8, hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    *match_count = total;
    if (out_probe_idx && out_build_idx && total > 0) {
        ARG_CHECK(out_probe_idx->bytes >= total * 4 && out_build_idx->bytes >= total * 4);
        HIP_CHECK(hipMalloc(&d_offsets, n_rows * 8));
        hipLaunchKernelGGL(k_scan_offsets, dim3(nb), dim3(BLOCK), 0, s->stream, d_counts,
                           n_rows, tile, d_bsums, d_offsets);
        hipLaunchKernelGGL(k_probe_emit_bc64, dim3(nb), dim3(BLOCK), 0, s->stream,
                           (const uint64_t*)probe_keys->ptr, n_rows, t->log_bucket_size,
                           t->first, t->next, t->build_keys64, mode, d_counts, d_offsets,
                           (uint32_t*)out_probe_idx->ptr, (uint32_t*)out_build_idx->ptr);
        HIP_CHECK(hipStreamSynchronize(s->stream));
        (void)hipFree(d_offsets);
    }
    (void)hipFree(d_counts);
    (void)hipFree(d_bsums);
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// LINEAR_CHAINED — the selector's preferred method under the 16 M-bucket cap
// (reference join_hash_map_method.h:118-150, .hpp:125-368): hash in space
// bucket<<8, 8-bit fingerprint packed in the top byte of first[], linear
// probing with triangular increment, same-key rows chained via next[].
// Parallel build claims slots with CAS; concurrent colliding inserts may
// land keys at different probe offsets than the CPU's sequential order, and
// chain order is scatter-order — probes follow the same (fp,key) search so
// the match multiset is identical (SURVEY.md §7 hard part (b)).
// ---------------------------------------------------------------------------
static constexpr uint32_t LC_FP_BITS = 8u;
static constexpr uint32_t LC_DATA_MASK = 0x00FFFFFFu;

__global__ void k_build_linear_chained(const uint32_t* __restrict__ keys, uint64_t row_count,
                                       uint32_t log_bucket_size, uint32_t bucket_mask,
                                       uint32_t* __restrict__ first,
                                       uint32_t* __restrict__ next) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = 1 + (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i <= row_count;
         i += stride) {
        uint32_t k = keys[i];
        uint32_t hash = join_hash_u32(k, log_bucket_size + LC_FP_BITS);
        uint32_t fp = hash << (32 - LC_FP_BITS);
        uint32_t b = hash >> LC_FP_BITS;
        uint32_t probe_times = 1;
        for (;;) {
            uint32_t cur = first[b];
            if (cur == 0) {
                next[i] = 0;
                uint32_t old = atomicCAS(&first[b], 0u, fp | (uint32_t)i);
                if (old == 0) break;
                cur = old; // someone claimed it — fall through to compare
            }
            if ((cur & ~LC_DATA_MASK) == fp && keys[cur & LC_DATA_MASK] == k) {
                // same key: push-front onto its chain
                uint32_t old = cur;
                for (;;) {
                    next[i] = old & LC_DATA_MASK;
                    uint32_t seen = atomicCAS(&first[b], old, fp | (uint32_t)i);
                    if (seen == old) break;
                    old = seen; // head moved (same key/fp by construction)
                }
                break;
            }
            b = (b + probe_times) & bucket_mask;
            probe_times++;
        }
    }
}

extern "C" int gpue_join_build_linear_chained_u32(gpue_session* s, gpue_dbuf* keys,
                                                  uint64_t row_count, gpue_join_table** out);
int gpue_join_build_linear_chained_u32(gpue_session* s, gpue_dbuf* keys, uint64_t row_count,
                                       gpue_join_table** out) {
    ARG_CHECK(s && keys && out && row_count > 0 && row_count < LC_DATA_MASK);
    ARG_CHECK(keys->bytes >= (row_count + 1) * 4);
    gpue_join_table* t = new gpue_join_table();
    t->s = s;
    t->kind = gpue_join_table::LINEAR_CHAINED;
    t->row_count = row_count;
    t->bucket_size = calc_bucket_size((uint32_t)(row_count + 1));
    ARG_CHECK(t->bucket_size <= LC_DATA_MASK); // the 16M cap (join_hash_map_method.h:136)
    t->log_bucket_size = (uint32_t)__builtin_ctzll(t->bucket_size);
    HIP_CHECK(hipMalloc(&t->first, t->bucket_size * sizeof(uint32_t)));
    HIP_CHECK(hipMalloc(&t->next, (row_count + 1) * sizeof(uint32_t)));
    HIP_CHECK(hipMalloc(&t->build_keys, (row_count + 1) * sizeof(uint32_t)));
    HIP_CHECK(hipMemsetAsync(t->first, 0, t->bucket_size * sizeof(uint32_t), s->stream));
    HIP_CHECK(hipMemsetAsync(t->next, 0, (row_count + 1) * sizeof(uint32_t), s->stream));
    HIP_CHECK(hipMemcpyAsync(t->build_keys, keys->ptr, (row_count + 1) * sizeof(uint32_t),
                             hipMemcpyDeviceToDevice, s->stream));
    hipLaunchKernelGGL(k_build_linear_chained, dim3(grid_for(row_count)), dim3(BLOCK), 0,
                       s->stream, t->build_keys, row_count, t->log_bucket_size,
                       (uint32_t)(t->bucket_size - 1), t->first, t->next);
    HIP_CHECK(hipGetLastError());
    *out = t;
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// DENSE_RANGE_DIRECT_MAPPING (join_hash_map_method.h:378,
// .hpp:781-904): the interval is compressed 16:1 by rank/select — per
// 32-key group a {start_index, bitset} pair (8 B per 32 interval
// positions), first[] sized by the number of PRESENT keys. Probe: bit test
// then first[start + popcount(bits below)]; chains hold identical keys so
// the walk needs no compare. The selector picks this when the interval
// exceeds bucket/L2 but 2 bits/position + 4 B/row still beats
// bucket-chained (+10%) — on GPU it also shrinks the random-probe footprint
// toward L2 residency.
// ---------------------------------------------------------------------------
__global__ void k_dense_bitsets(const int32_t* __restrict__ keys, uint64_t row_count,
                                int64_t mn, uint2* __restrict__ groups) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = 1 + (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i <= row_count;
         i += stride) {
        uint32_t b = (uint32_t)(keys[i] - mn);
        atomicOr(&groups[b >> 5].y, 1u << (b & 31));
    }
}

__global__ void k_dense_popcounts(const uint2* __restrict__ groups, uint64_t ngroups,
                                  uint32_t* __restrict__ counts) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t g = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; g < ngroups;
         g += stride)
        counts[g] = __popc(groups[g].y);
}

__global__ void k_dense_starts(uint2* __restrict__ groups, uint64_t ngroups,
                               const uint64_t* __restrict__ offsets) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t g = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; g < ngroups;
         g += stride)
        groups[g].x = (uint32_t)offsets[g];
}

__device__ static inline uint32_t dense_slot(const uint2* __restrict__ groups,
                                             uint32_t bucket_num) {
    uint2 g = groups[bucket_num >> 5];
    uint32_t below = g.y & ((1u << (bucket_num & 31)) - 1);
    return g.x + __popc(below);
}

__global__ void k_dense_chains(const int32_t* __restrict__ keys, uint64_t row_count,
                               int64_t mn, const uint2* __restrict__ groups,
                               uint32_t* __restrict__ first, uint32_t* __restrict__ next) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = 1 + (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i <= row_count;
         i += stride) {
        uint32_t slot = dense_slot(groups, (uint32_t)(keys[i] - mn));
        next[i] = atomicExch(&first[slot], (uint32_t)i);
    }
}

// dense chain head per probe key (lookup_init, .hpp:906-940): 0 when out of
// range or bit clear
__device__ static inline uint32_t dense_head(const int32_t* __restrict__ pk, uint64_t i,
                                             int64_t mn, int64_t mx,
                                             const uint2* __restrict__ groups,
                                             const uint32_t* __restrict__ first) {
    int64_t k = pk[i];
    if (k < mn || k > mx) return 0;
    uint32_t b = (uint32_t)(k - mn);
    uint2 g = groups[b >> 5];
    if (!((g.y >> (b & 31)) & 1u)) return 0;
    uint32_t below = g.y & ((1u << (b & 31)) - 1);
    return first[g.x + __popc(below)];
}

__global__ void k_probe_count_dense(const int32_t* __restrict__ probe_keys, uint64_t n,
                                    int64_t mn, int64_t mx,
                                    const uint2* __restrict__ groups,
                                    const uint32_t* __restrict__ first,
                                    const uint32_t* __restrict__ next, int mode,
                                    uint32_t* __restrict__ row_counts) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        uint32_t b = dense_head(probe_keys, i, mn, mx, groups, first);
        uint32_t c = 0;
        while (b != 0) { // chains hold identical keys: no compare
            c++;
            b = next[b];
        }
        row_counts[i] = join_mode_count(c, mode);
    }
}

__global__ void k_probe_emit_dense(const int32_t* __restrict__ probe_keys, uint64_t n,
                                   int64_t mn, int64_t mx,
                                   const uint2* __restrict__ groups,
                                   const uint32_t* __restrict__ first,
                                   const uint32_t* __restrict__ next, int mode,
                                   const uint32_t* __restrict__ row_counts,
                                   const uint64_t* __restrict__ row_offsets,
                                   uint32_t* __restrict__ out_probe,
                                   uint32_t* __restrict__ out_build) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        if (row_counts[i] == 0) continue;
        uint64_t pos = row_offsets[i];
        uint32_t b = dense_head(probe_keys, i, mn, mx, groups, first);
        if (b == 0) { // unmatched emit (ANTI/OUTER)
            out_probe[pos] = (uint32_t)i;
            out_build[pos] = 0;
            continue;
        }
        if (mode == 2) continue;
        while (b != 0) {
            out_probe[pos] = (uint32_t)i;
            out_build[pos] = b;
            pos++;
            if (mode == 1) break;
            b = next[b];
        }
    }
}

extern "C" int gpue_join_build_dense_range_direct_i32(gpue_session* s, gpue_dbuf* keys,
                                                      uint64_t row_count,
                                                      gpue_join_table** out);
int gpue_join_build_dense_range_direct_i32(gpue_session* s, gpue_dbuf* keys,
                                           uint64_t row_count, gpue_join_table** out) {
    ARG_CHECK(s && keys && out && row_count > 0);
    ARG_CHECK(keys->bytes >= (row_count + 1) * 4);
    int64_t mn, mx;
    int rc = join_table_minmax(s, (const int32_t*)keys->ptr + 1, row_count, &mn, &mx);
    if (rc != GPUE_OK) return rc;
    uint64_t interval = (uint64_t)(mx - mn + 1);
    ARG_CHECK(interval < (1ull << 32));
    uint64_t ngroups = (interval + 31) / 32;
    gpue_join_table* t = new gpue_join_table();
    t->s = s;
    t->kind = gpue_join_table::DENSE_RANGE_DIRECT;
    t->min_key = mn;
    t->max_key = mx;
    t->row_count = row_count;
    HIP_CHECK(hipMalloc(&t->dense_groups, ngroups * sizeof(uint2)));
    HIP_CHECK(hipMemsetAsync(t->dense_groups, 0, ngroups * sizeof(uint2), s->stream));
    hipLaunchKernelGGL(k_dense_bitsets, dim3(grid_for(row_count)), dim3(BLOCK), 0,
                       s->stream, (const int32_t*)keys->ptr, row_count, mn,
                       t->dense_groups);
    // start_index = exclusive scan of per-group popcounts (the loop at
    // .hpp:878-881), via the block-sums scan used by the partition kernels
    uint32_t* d_counts = nullptr;
    uint64_t* d_bsums = nullptr;
    uint64_t* d_offsets = nullptr;
    uint32_t nb = grid_for(ngroups);
    uint64_t tile = (ngroups + nb - 1) / nb;
    HIP_CHECK(hipMalloc(&d_counts, ngroups * 4));
    HIP_CHECK(hipMalloc(&d_bsums, (nb + 1) * 8));
    HIP_CHECK(hipMalloc(&d_offsets, ngroups * 8));
    hipLaunchKernelGGL(k_dense_popcounts, dim3(nb), dim3(BLOCK), 0, s->stream,
                       t->dense_groups, ngroups, d_counts);
    hipLaunchKernelGGL(k_block_sums_u32, dim3(nb), dim3(BLOCK), 0, s->stream, d_counts,
                       ngroups, tile, d_bsums);
    hipLaunchKernelGGL(k_scan_small, dim3(1), dim3(1), 0, s->stream, d_bsums, nb);
    hipLaunchKernelGGL(k_scan_offsets, dim3(nb), dim3(BLOCK), 0, s->stream, d_counts,
                       ngroups, tile, d_bsums, d_offsets);
    hipLaunchKernelGGL(k_dense_starts, dim3(nb), dim3(BLOCK), 0, s->stream,
                       t->dense_groups, ngroups, d_offsets);
    uint64_t used = 0;
    HIP_CHECK(hipMemcpyAsync(&used, d_bsums + nb, 8, hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    t->bucket_size = used; // number of present keys = first[] size
    HIP_CHECK(hipMalloc(&t->first, (used ? used : 1) * 4));
    HIP_CHECK(hipMalloc(&t->next, (row_count + 1) * 4));
    HIP_CHECK(hipMemsetAsync(t->first, 0, (used ? used : 1) * 4, s->stream));
    HIP_CHECK(hipMemsetAsync(t->next, 0, (row_count + 1) * 4, s->stream));
    hipLaunchKernelGGL(k_dense_chains, dim3(grid_for(row_count)), dim3(BLOCK), 0,
                       s->stream, (const int32_t*)keys->ptr, row_count, mn,
                       t->dense_groups, t->first, t->next);
    HIP_CHECK(hipGetLastError());
    (void)hipFree(d_counts);
    (void)hipFree(d_bsums);
    (void)hipFree(d_offsets);
    *out = t;
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// Auto build (JoinHashTable::build's selector step, join_hash_table.cpp:
// 350-380 -> JoinHashMapSelector): min/max the build keys on device, decide
// with the reference's rules, build the matching GPU table. Decision->layout
// mapping documented in gpue.h.
// ---------------------------------------------------------------------------
extern "C" int gpue_join_build_auto_i32(gpue_session* s, gpue_dbuf* keys,
                                        uint64_t row_count, int mode,
                                        int with_other_conjunct, uint64_t l2_size,
                                        uint64_t l3_size, gpue_join_table** out,
                                        int* method_out);
int gpue_join_build_auto_i32(gpue_session* s, gpue_dbuf* keys, uint64_t row_count,
                             int mode, int with_other_conjunct, uint64_t l2_size,
                             uint64_t l3_size, gpue_join_table** out, int* method_out) {
    ARG_CHECK(s && keys && out && row_count > 0);
    ARG_CHECK(keys->bytes >= (row_count + 1) * 4);
    if (l2_size == 0) l2_size = 4ull << 20;   // MI355X XCD L2
    if (l3_size == 0) l3_size = 256ull << 20; // Infinity Cache
    int64_t mn = 0, mx = 0;
    int rc = join_table_minmax(s, (const int32_t*)keys->ptr + 1, row_count, &mn, &mx);
    if (rc != GPUE_OK) return rc;
    int method = gpue_join_select_method(GPUE_KEYCON_ONE_KEY, GPUE_LT_INT, row_count,
                                         mn, mx, mode, with_other_conjunct,
                                         /*enable_range_direct=*/1,
                                         /*enable_linear_chained=*/1, l2_size, l3_size);
    if (method_out) *method_out = method;
    switch (method) {
        case GPUE_JM_DIRECT:
        case GPUE_JM_RANGE_DIRECT:
        case GPUE_JM_RANGE_DIRECT_SET:
            return gpue_join_build_range_direct_i32(s, keys, row_count, out);
        case GPUE_JM_DENSE_RANGE_DIRECT:
            return gpue_join_build_dense_range_direct_i32(s, keys, row_count, out);
        case GPUE_JM_LINEAR_CHAINED:
        case GPUE_JM_LINEAR_CHAINED_SET:
            return gpue_join_build_linear_chained_u32(s, keys, row_count, out);
        default:
            return gpue_join_build_bucket_chained_u32(s, keys, row_count, out);
    }
}

// 8-byte-key auto build: run the selector with LT_BIGINT and report its
// decision; every physical tier maps onto the u64 bucket-chained table (the
// u64 key domain has no range-direct/linear specialization in this engine —
// dense int64 dims in our configs are i32-valued and take the i32 paths).
__global__ void k_minmax_i64(const int64_t* __restrict__ keys, uint64_t n,
                             long long* __restrict__ mn, long long* __restrict__ mx) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    long long lmn = 0x7FFFFFFFFFFFFFFFll, lmx = 0x8000000000000000ll;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        long long v = keys[i];
        lmn = v < lmn ? v : lmn;
        lmx = v > lmx ? v : lmx;
    }
    for (int off = WAVE / 2; off > 0; off >>= 1) {
        long long a = __shfl_down(lmn, off, WAVE);
        long long b = __shfl_down(lmx, off, WAVE);
        lmn = a < lmn ? a : lmn;
        lmx = b > lmx ? b : lmx;
    }
    if ((threadIdx.x & (WAVE - 1)) == 0) {
        atomicMin(mn, lmn);
        atomicMax(mx, lmx);
    }
}

extern "C" int gpue_join_build_auto_u64(gpue_session* s, gpue_dbuf* keys,
                                        uint64_t row_count, int mode,
                                        int with_other_conjunct, uint64_t l2_size,
                                        uint64_t l3_size, gpue_join_table** out,
                                        int* method_out);
int gpue_join_build_auto_u64(gpue_session* s, gpue_dbuf* keys, uint64_t row_count,
                             int mode, int with_other_conjunct, uint64_t l2_size,
                             uint64_t l3_size, gpue_join_table** out, int* method_out) {
    ARG_CHECK(s && keys && out && row_count > 0);
    ARG_CHECK(keys->bytes >= (row_count + 1) * 8);
    if (l2_size == 0) l2_size = 4ull << 20;
    if (l3_size == 0) l3_size = 256ull << 20;
    long long h_init[2] = {0x7FFFFFFFFFFFFFFFll, (long long)0x8000000000000000ll};
    long long* d_mm = nullptr;
    HIP_CHECK(hipMalloc(&d_mm, 16));
    HIP_CHECK(hipMemcpyAsync(d_mm, h_init, 16, hipMemcpyHostToDevice, s->stream));
    hipLaunchKernelGGL(k_minmax_i64, dim3(grid_for(row_count)), dim3(BLOCK), 0, s->stream,
                       (const int64_t*)keys->ptr + 1, row_count, d_mm, d_mm + 1);
    long long h_mm[2];
    HIP_CHECK(hipMemcpyAsync(h_mm, d_mm, 16, hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    (void)hipFree(d_mm);
    int method = gpue_join_select_method(GPUE_KEYCON_ONE_KEY, GPUE_LT_BIGINT, row_count,
                                         h_mm[0], h_mm[1], mode, with_other_conjunct, 1, 1,
                                         l2_size, l3_size);
    if (method_out) *method_out = method;
    return gpue_join_build_bucket_chained_u64(s, keys, row_count, out);
}

// linear-probe lookup to the chain head, then chain walk (lookup_init,
// join_hash_map_method.hpp:286-368): chains hold identical keys, so the walk
// needs no further compare
__device__ static inline uint32_t lc_lookup_head(uint32_t k, uint32_t log_bucket_size,
                                                 uint32_t bucket_mask,
                                                 const uint32_t* __restrict__ first,
                                                 const uint32_t* __restrict__ build_keys) {
    uint32_t hash = join_hash_u32(k, log_bucket_size + LC_FP_BITS);
    uint32_t fp = hash << (32 - LC_FP_BITS);
    uint32_t b = hash >> LC_FP_BITS;
    uint32_t probe_times = 1;
    for (;;) {
        uint32_t cur = first[b];
        if (cur == 0) return 0;
        if ((cur & ~LC_DATA_MASK) == fp && build_keys[cur & LC_DATA_MASK] == k)
            return cur & LC_DATA_MASK;
        b = (b + probe_times) & bucket_mask;
        probe_times++;
    }
}

__global__ void k_probe_count_lc(const uint32_t* __restrict__ probe_keys, uint64_t n,
                                 uint32_t log_bucket_size, uint32_t bucket_mask,
                                 const uint32_t* __restrict__ first,
                                 const uint32_t* __restrict__ next,
                                 const uint32_t* __restrict__ build_keys, int mode,
                                 uint32_t* __restrict__ row_counts) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        uint32_t b = lc_lookup_head(probe_keys[i], log_bucket_size, bucket_mask, first,
                                    build_keys);
        uint32_t c = 0;
        while (b != 0) {
            c++;
            b = next[b];
        }
        row_counts[i] = join_mode_count(c, mode);
    }
}

__global__ void k_probe_emit_lc(const uint32_t* __restrict__ probe_keys, uint64_t n,
                                uint32_t log_bucket_size, uint32_t bucket_mask,
                                const uint32_t* __restrict__ first,
                                const uint32_t* __restrict__ next,
                                const uint32_t* __restrict__ build_keys, int mode,
                                const uint32_t* __restrict__ row_counts,
                                const uint64_t* __restrict__ row_offsets,
                                uint32_t* __restrict__ out_probe,
                                uint32_t* __restrict__ out_build) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        if (row_counts[i] == 0) continue;
        uint64_t pos = row_offsets[i];
        uint32_t b = lc_lookup_head(probe_keys[i], log_bucket_size, bucket_mask, first,
                                    build_keys);
        if (b == 0) { // unmatched emit (ANTI/OUTER)
            out_probe[pos] = (uint32_t)i;
            out_build[pos] = 0;
            continue;
        }
        if (mode == 2) continue;
        while (b != 0) {
            out_probe[pos] = (uint32_t)i;
            out_build[pos] = b;
            pos++;
            if (mode == 1) break;
            b = next[b];
        }
    }
}

// probe with key-equality compare along the chain (lookup_init +
// _probe_from_ht with AreKeysInChainIdentical == false)
__global__ void k_probe_count_bc(const uint32_t* __restrict__ probe_keys, uint64_t n,
                                 uint32_t log_bucket_size, const uint32_t* __restrict__ first,
                                 const uint32_t* __restrict__ next,
                                 const uint32_t* __restrict__ build_keys, int mode,
                                 uint32_t* __restrict__ row_counts) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        uint32_t k = probe_keys[i];
        uint32_t b = first[join_hash_u32(k, log_bucket_size)];
        uint32_t c = 0;
        while (b != 0) {
            c += (build_keys[b] == k);
            b = next[b];
        }
        row_counts[i] = join_mode_count(c, mode);
    }
}

__global__ void k_probe_emit_bc(const uint32_t* __restrict__ probe_keys, uint64_t n,
                                uint32_t log_bucket_size, const uint32_t* __restrict__ first,
                                const uint32_t* __restrict__ next,
                                const uint32_t* __restrict__ build_keys, int mode,
                                const uint32_t* __restrict__ row_counts,
                                const uint64_t* __restrict__ row_offsets,
                                uint32_t* __restrict__ out_probe,
                                uint32_t* __restrict__ out_build) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        if (row_counts[i] == 0) continue;
        uint32_t k = probe_keys[i];
        uint64_t pos = row_offsets[i];
        uint32_t b = first[join_hash_u32(k, log_bucket_size)];
        bool any = false;
        while (b != 0) {
            if (build_keys[b] == k) {
                any = true;
                if (mode != 2) {
                    out_probe[pos] = (uint32_t)i;
                    out_build[pos] = b;
                    pos++;
                }
                if (mode == 1 || mode == 2) break;
            }
            b = next[b];
        }
        if (!any && (mode == 2 || mode == 3)) {
            out_probe[pos] = (uint32_t)i;
            out_build[pos] = 0;
        }
    }
}

// ---------------------------------------------------------------------------
// hash-join probe: lookup + chain-walk emit (reference
// join_hash_map_method.hpp:688-707 lookup_init + join_hash_map.hpp:717-795
// _probe_from_ht). Two-phase count/emit replaces the CPU's resumable cursor
// (SURVEY.md §7 hard part (c)); output ordered by probe row.
// ---------------------------------------------------------------------------
// block-wise scan for row offsets: pass A block sums, host scan, pass B add
__global__ void k_block_sums_u32(const uint32_t* __restrict__ v, uint64_t n, uint64_t tile,
                                 uint64_t* __restrict__ sums) {
    uint64_t lo = (uint64_t)blockIdx.x * tile, hi = min(lo + tile, n);
    uint64_t c = 0;
    for (uint64_t i = lo + threadIdx.x; i < hi; i += blockDim.x) c += v[i];
    for (int off = WAVE / 2; off > 0; off >>= 1)
        c += __shfl_down((unsigned long long)c, off, WAVE);
    __shared__ uint64_t wsum[BLOCK / WAVE];
    int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;
    if (lane == 0) wsum[wid] = c;
    __syncthreads();
    if (threadIdx.x == 0) {
        uint64_t t = 0;
        for (int w = 0; w < BLOCK / WAVE; w++) t += wsum[w];
        sums[blockIdx.x] = t;
    }
}

// Range-direct chains hold IDENTICAL keys by construction
// (AreKeysInChainIdentical, join_hash_map_method.h:264), so the probe needs
// no build-key compare: count = chain length. The public entry re-walks with
// that simplification.
__global__ void k_probe_count_rd(const int32_t* __restrict__ probe_keys, uint64_t n,
                                 int64_t mn, int64_t mx, const uint32_t* __restrict__ first,
                                 const uint32_t* __restrict__ next, int mode,
                                 uint32_t* __restrict__ row_counts) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        int32_t k = probe_keys[i];
        uint32_t c = 0;
        if (k >= mn && k <= mx) {
            uint32_t b = first[k - mn];
            while (b != 0) {
                c++;
                b = next[b];
            }
        }
        row_counts[i] = join_mode_count(c, mode);
    }
}

__global__ void k_probe_emit_rd(const int32_t* __restrict__ probe_keys, uint64_t n,
                                int64_t mn, int64_t mx, const uint32_t* __restrict__ first,
                                const uint32_t* __restrict__ next, int mode,
                                const uint32_t* __restrict__ row_counts,
                                const uint64_t* __restrict__ row_offsets,
                                uint32_t* __restrict__ out_probe,
                                uint32_t* __restrict__ out_build) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        if (row_counts[i] == 0) continue;
        uint64_t pos = row_offsets[i];
        int32_t k = probe_keys[i];
        uint32_t b = (k >= mn && k <= mx) ? first[k - mn] : 0;
        if (b == 0) { // only reachable for ANTI/OUTER: the unmatched emit
            out_probe[pos] = (uint32_t)i;
            out_build[pos] = 0;
            continue;
        }
        if (mode == 2) continue; // matched row contributes nothing to ANTI
        while (b != 0) {
            out_probe[pos] = (uint32_t)i;
            out_build[pos] = b;
            pos++;
            if (mode == 1) break; // SEMI: first match only
            b = next[b];
        }
    }
}

// exclusive scan of u32 counts into u64 offsets — block sums + thread0 scan +
// per-block add; test-path sizes (≤ tens of M rows), not the bench hot loop
__global__ void k_scan_offsets(const uint32_t* __restrict__ counts, uint64_t n, uint64_t tile,
                               const uint64_t* __restrict__ block_offsets,
                               uint64_t* __restrict__ offsets) {
    uint64_t lo = (uint64_t)blockIdx.x * tile, hi = min(lo + tile, n);
    if (threadIdx.x == 0) {
        uint64_t acc = block_offsets[blockIdx.x];
        for (uint64_t i = lo; i < hi; i++) {
            offsets[i] = acc;
            acc += counts[i];
        }
    }
}

extern "C" int gpue_join_probe_emit_mode_i32(gpue_session* s, gpue_join_table* t,
                                             gpue_dbuf* probe_keys, uint64_t n_rows, int mode,
                                             gpue_dbuf* out_probe_idx,
                                             gpue_dbuf* out_build_idx, uint64_t* match_count);

int gpue_join_probe_emit_i32(gpue_session* s, gpue_join_table* t, gpue_dbuf* probe_keys,
                             uint64_t n_rows, gpue_dbuf* out_probe_idx,
                             gpue_dbuf* out_build_idx, uint64_t* match_count) {
    return gpue_join_probe_emit_mode_i32(s, t, probe_keys, n_rows, 0, out_probe_idx,
                                         out_build_idx, match_count);
}

int gpue_join_probe_emit_mode_i32(gpue_session* s, gpue_join_table* t, gpue_dbuf* probe_keys,
                                  uint64_t n_rows, int mode, gpue_dbuf* out_probe_idx,
                                  gpue_dbuf* out_build_idx, uint64_t* match_count) {
    ARG_CHECK(s && t && probe_keys && match_count && t->next != nullptr);
    ARG_CHECK(mode >= 0 && mode <= 3);
    ARG_CHECK(probe_keys->bytes >= n_rows * 4);
    uint32_t nb = grid_for(n_rows);
    uint64_t tile = (n_rows + nb - 1) / nb;
    uint32_t* d_counts = nullptr;
    uint64_t* d_bsums = nullptr;
    uint64_t* d_offsets = nullptr;
    HIP_CHECK(hipMalloc(&d_counts, n_rows * sizeof(uint32_t)));
    HIP_CHECK(hipMalloc(&d_bsums, (nb + 1) * sizeof(uint64_t)));
    if (t->kind == gpue_join_table::LINEAR_CHAINED) {
        hipLaunchKernelGGL(k_probe_count_lc, dim3(nb), dim3(BLOCK), 0, s->stream,
                           (const uint32_t*)probe_keys->ptr, n_rows, t->log_bucket_size,
                           (uint32_t)(t->bucket_size - 1), t->first, t->next, t->build_keys,
                           mode, d_counts);
    } else if (t->kind == gpue_join_table::BUCKET_CHAINED) {
        hipLaunchKernelGGL(k_probe_count_bc, dim3(nb), dim3(BLOCK), 0, s->stream,
                           (const uint32_t*)probe_keys->ptr, n_rows, t->log_bucket_size,
                           t->first, t->next, t->build_keys, mode, d_counts);
    } else if (t->kind == gpue_join_table::DENSE_RANGE_DIRECT) {
        hipLaunchKernelGGL(k_probe_count_dense, dim3(nb), dim3(BLOCK), 0, s->stream,
                           (const int32_t*)probe_keys->ptr, n_rows, t->min_key, t->max_key,
                           t->dense_groups, t->first, t->next, mode, d_counts);
    } else
    hipLaunchKernelGGL(k_probe_count_rd, dim3(nb), dim3(BLOCK), 0, s->stream,
                       (const int32_t*)probe_keys->ptr, n_rows, t->min_key, t->max_key,
                       t->first, t->next, mode, d_counts);
    hipLaunchKernelGGL(k_block_sums_u32, dim3(nb), dim3(BLOCK), 0, s->stream,
                       d_counts, n_rows, tile, d_bsums);
    hipLaunchKernelGGL(k_scan_small, dim3(1), dim3(1), 0, s->stream, d_bsums, nb);
    uint64_t total = 0;
    HIP_CHECK(hipMemcpyAsync(&total, d_bsums + nb, sizeof(uint64_t), hipMemcpyDeviceToHost,
                             s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    *match_count = total;
    if (out_probe_idx && out_build_idx && total > 0) {
        ARG_CHECK(out_probe_idx->bytes >= total * 4 && out_build_idx->bytes >= total * 4);
        HIP_CHECK(hipMalloc(&d_offsets, n_rows * sizeof(uint64_t)));
        hipLaunchKernelGGL(k_scan_offsets, dim3(nb), dim3(BLOCK), 0, s->stream,
                           d_counts, n_rows, tile, d_bsums, d_offsets);
        if (t->kind == gpue_join_table::LINEAR_CHAINED) {
            hipLaunchKernelGGL(k_probe_emit_lc, dim3(nb), dim3(BLOCK), 0, s->stream,
                               (const uint32_t*)probe_keys->ptr, n_rows, t->log_bucket_size,
                               (uint32_t)(t->bucket_size - 1), t->first, t->next,
                               t->build_keys, mode, d_counts, d_offsets,
                               (uint32_t*)out_probe_idx->ptr, (uint32_t*)out_build_idx->ptr);
        } else if (t->kind == gpue_join_table::BUCKET_CHAINED) {
            hipLaunchKernelGGL(k_probe_emit_bc, dim3(nb), dim3(BLOCK), 0, s->stream,
                               (const uint32_t*)probe_keys->ptr, n_rows, t->log_bucket_size,
                               t->first, t->next, t->build_keys, mode, d_counts, d_offsets,
                               (uint32_t*)out_probe_idx->ptr, (uint32_t*)out_build_idx->ptr);
        } else if (t->kind == gpue_join_table::DENSE_RANGE_DIRECT) {
            hipLaunchKernelGGL(k_probe_emit_dense, dim3(nb), dim3(BLOCK), 0, s->stream,
                               (const int32_t*)probe_keys->ptr, n_rows, t->min_key,
                               t->max_key, t->dense_groups, t->first, t->next, mode,
                               d_counts, d_offsets, (uint32_t*)out_probe_idx->ptr,
                               (uint32_t*)out_build_idx->ptr);
        } else
        hipLaunchKernelGGL(k_probe_emit_rd, dim3(nb), dim3(BLOCK), 0, s->stream,
                           (const int32_t*)probe_keys->ptr, n_rows, t->min_key, t->max_key,
                           t->first, t->next, mode, d_counts, d_offsets,
                           (uint32_t*)out_probe_idx->ptr, (uint32_t*)out_build_idx->ptr);
        HIP_CHECK(hipStreamSynchronize(s->stream));
        (void)hipFree(d_offsets);
    }
    (void)hipFree(d_counts);
    (void)hipFree(d_bsums);
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// Dictionary-encoded binary page decode (binary_dict_page.cpp:229-280): the
// data page's int32 codewords (bitshuffle layer handled by
// gpue_page_decode_bshuf_lz4_i32) index the dict page's distinct strings;
// output is a BinaryColumn (bytes + uint32 offsets). Two-phase: per-row
// lengths -> scan -> parallel byte copy.
// ---------------------------------------------------------------------------
__global__ void k_dict_code_lengths(const int32_t* __restrict__ codes, uint64_t n,
                                    const uint32_t* __restrict__ doff,
                                    uint32_t* __restrict__ lens) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        uint32_t c = (uint32_t)codes[i];
        lens[i] = doff[c + 1] - doff[c];
    }
}

__global__ void k_dict_emit(const int32_t* __restrict__ codes, uint64_t n,
                            const uint8_t* __restrict__ dbytes,
                            const uint32_t* __restrict__ doff,
                            const uint64_t* __restrict__ row_offsets, uint64_t total,
                            uint8_t* __restrict__ out_bytes,
                            uint32_t* __restrict__ out_offsets) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    uint64_t tid = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (tid == 0) out_offsets[n] = (uint32_t)total;
    for (uint64_t i = tid; i < n; i += stride) {
        uint32_t c = (uint32_t)codes[i];
        uint32_t len = doff[c + 1] - doff[c];
        uint64_t pos = row_offsets[i];
        out_offsets[i] = (uint32_t)pos;
        const uint8_t* src = dbytes + doff[c];
        for (uint32_t k = 0; k < len; k++) out_bytes[pos + k] = src[k];
    }
}

extern "C" int gpue_dict_decode_binary(gpue_session* s, gpue_dbuf* dict_bytes,
                                       gpue_dbuf* dict_offsets, gpue_dbuf* codes,
                                       uint64_t n, gpue_dbuf* out_bytes,
                                       gpue_dbuf* out_offsets, uint64_t* total_bytes);
int gpue_dict_decode_binary(gpue_session* s, gpue_dbuf* dict_bytes, gpue_dbuf* dict_offsets,
                            gpue_dbuf* codes, uint64_t n, gpue_dbuf* out_bytes,
                            gpue_dbuf* out_offsets, uint64_t* total_bytes) {
    ARG_CHECK(s && dict_bytes && dict_offsets && codes && total_bytes);
    ARG_CHECK(codes->bytes >= n * 4);
    uint32_t nb = grid_for(n);
    uint64_t tile = (n + nb - 1) / nb;
    uint32_t* d_lens = nullptr;
    uint64_t* d_bsums = nullptr;
    uint64_t* d_off = nullptr;
    HIP_CHECK(hipMalloc(&d_lens, n * 4));
    HIP_CHECK(hipMalloc(&d_bsums, (nb + 1) * 8));
    hipLaunchKernelGGL(k_dict_code_lengths, dim3(nb), dim3(BLOCK), 0, s->stream,
                       (const int32_t*)codes->ptr, n, (const uint32_t*)dict_offsets->ptr,
                       d_lens);
    hipLaunchKernelGGL(k_block_sums_u32, dim3(nb), dim3(BLOCK), 0, s->stream, d_lens, n,
                       tile, d_bsums);
    hipLaunchKernelGGL(k_scan_small, dim3(1), dim3(1), 0, s->stream, d_bsums, nb);
    uint64_t total = 0;
    HIP_CHECK(hipMemcpyAsync(&total, d_bsums + nb, 8, hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    *total_bytes = total;
    if (out_bytes && out_offsets) {
        ARG_CHECK(out_bytes->bytes >= total && out_offsets->bytes >= (n + 1) * 4);
        HIP_CHECK(hipMalloc(&d_off, n * 8));
        hipLaunchKernelGGL(k_scan_offsets, dim3(nb), dim3(BLOCK), 0, s->stream, d_lens, n,
                           tile, d_bsums, d_off);
        hipLaunchKernelGGL(k_dict_emit, dim3(nb), dim3(BLOCK), 0, s->stream,
                           (const int32_t*)codes->ptr, n, (const uint8_t*)dict_bytes->ptr,
                           (const uint32_t*)dict_offsets->ptr, d_off, total,
                           (uint8_t*)out_bytes->ptr, (uint32_t*)out_offsets->ptr);
        HIP_CHECK(hipStreamSynchronize(s->stream));
        (void)hipFree(d_off);
    }
    (void)hipFree(d_lens);
    (void)hipFree(d_bsums);
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// Multi-conjunct evaluation with the reference's EAGER-PRUNE strategy
// (chunk_predicate_evaluator.cpp:31-80): per conjunct, AND-merge a boolean
// filter and count survivors; all-true skips, all-false short-circuits;
// when zeros exceed max(0.8*rows, 1024) all columns compact NOW so later
// conjuncts touch fewer rows; final compaction applies a pending filter.
// ops: 0 EQ(lo), 1 LT(hi), 2 BETWEEN[lo,hi].
// ---------------------------------------------------------------------------
__global__ void k_gather_u32(const uint32_t* __restrict__ in,
                             const uint32_t* __restrict__ idx, uint64_t n,
                             uint32_t* __restrict__ out);

__global__ void k_eval_pred_i32(const int32_t* __restrict__ col, uint64_t n, int op,
                                int32_t lo, int32_t hi, uint8_t* __restrict__ filter,
                                unsigned long long* __restrict__ true_count) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    unsigned long long local = 0;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        int32_t v = col[i];
        uint8_t pass = op == 0 ? (v == lo) : op == 1 ? (v < hi) : (v >= lo && v <= hi);
        uint8_t merged = filter[i] & pass;
        filter[i] = merged;
        local += merged;
    }
    for (int off = 32; off > 0; off >>= 1)
        local += __shfl_down(local, off, WAVE);
    if ((threadIdx.x & (WAVE - 1)) == 0 && local)
        atomicAdd(true_count, local);
}

__global__ void k_mask_block_counts(const uint8_t* __restrict__ filter, uint64_t n,
                                    uint64_t tile, uint64_t* __restrict__ block_sums) {
    __shared__ unsigned long long red[BLOCK / WAVE];
    uint64_t lo_ = (uint64_t)blockIdx.x * tile, hi_ = min(lo_ + tile, n);
    unsigned long long c = 0;
    for (uint64_t i = lo_ + threadIdx.x; i < hi_; i += blockDim.x) c += filter[i];
    for (int off = 32; off > 0; off >>= 1) c += __shfl_down(c, off, WAVE);
    if ((threadIdx.x & (WAVE - 1)) == 0) red[threadIdx.x / WAVE] = c;
    __syncthreads();
    if (threadIdx.x == 0) {
        unsigned long long t = 0;
        for (int w = 0; w < BLOCK / WAVE; w++) t += red[w];
        block_sums[blockIdx.x] = t;
    }
}

// stable compaction of row indexes from the u8 filter (serial within tile —
// prune points are rare; correctness-first like k_scan_offsets)
__global__ void k_mask_emit_idx(const uint8_t* __restrict__ filter, uint64_t n,
                                uint64_t tile, const uint64_t* __restrict__ block_offsets,
                                uint32_t* __restrict__ row_idx) {
    uint64_t lo_ = (uint64_t)blockIdx.x * tile, hi_ = min(lo_ + tile, n);
    if (threadIdx.x == 0) {
        uint64_t pos = block_offsets[blockIdx.x];
        for (uint64_t i = lo_; i < hi_; i++)
            if (filter[i]) row_idx[pos++] = (uint32_t)i;
    }
}

__global__ void k_eval_pred_i64(const int64_t* __restrict__ col, uint64_t n, int op,
                                int64_t lo, int64_t hi, uint8_t* __restrict__ filter,
                                unsigned long long* __restrict__ true_count) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    unsigned long long local = 0;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        int64_t v = col[i];
        uint8_t pass = op == 0 ? (v == lo) : op == 1 ? (v < hi) : (v >= lo && v <= hi);
        uint8_t merged = filter[i] & pass;
        filter[i] = merged;
        local += merged;
    }
    for (int off = 32; off > 0; off >>= 1)
        local += __shfl_down(local, off, WAVE);
    if ((threadIdx.x & (WAVE - 1)) == 0 && local)
        atomicAdd(true_count, local);
}

__global__ void k_gather_u64(const uint64_t* __restrict__ in,
                             const uint32_t* __restrict__ idx, uint64_t n,
                             uint64_t* __restrict__ out);

extern "C" int gpue_eval_conjuncts_i32(gpue_session* s, gpue_dbuf** cols, int n_cols,
                                       uint64_t n_rows, const int32_t* pred_col,
                                       const int32_t* pred_op, const int32_t* pred_lo,
                                       const int32_t* pred_hi, int n_preds,
                                       uint64_t* out_rows);
int gpue_eval_conjuncts_i32(gpue_session* s, gpue_dbuf** cols, int n_cols, uint64_t n_rows,
                            const int32_t* pred_col, const int32_t* pred_op,
                            const int32_t* pred_lo, const int32_t* pred_hi, int n_preds,
                            uint64_t* out_rows) {
    ARG_CHECK(s && cols && n_cols > 0 && pred_col && pred_op && pred_lo && pred_hi &&
              out_rows);
    uint64_t n = n_rows;
    uint8_t* d_filter = nullptr;
    unsigned long long* d_tc = nullptr;
    HIP_CHECK(hipMalloc(&d_filter, n_rows ? n_rows : 1));
    HIP_CHECK(hipMalloc(&d_tc, 8));
    HIP_CHECK(hipMemsetAsync(d_filter, 1, n_rows ? n_rows : 1, s->stream));
    uint64_t zero_count = 0;

    auto compact_all = [&](uint64_t keep) -> int {
        uint32_t nb = grid_for(n);
        uint64_t tile = (n + nb - 1) / nb;
        uint64_t* d_bs = nullptr;
        uint32_t* d_idx = nullptr;
        int32_t* d_tmp = nullptr;
        HIP_CHECK(hipMalloc(&d_bs, (nb + 1) * 8));
        HIP_CHECK(hipMalloc(&d_idx, keep * 4));
        HIP_CHECK(hipMalloc(&d_tmp, keep * 4));
        hipLaunchKernelGGL(k_mask_block_counts, dim3(nb), dim3(BLOCK), 0, s->stream,
                           d_filter, n, tile, d_bs);
        hipLaunchKernelGGL(k_scan_small, dim3(1), dim3(1), 0, s->stream, d_bs, nb);
        hipLaunchKernelGGL(k_mask_emit_idx, dim3(nb), dim3(BLOCK), 0, s->stream, d_filter,
                           n, tile, d_bs, d_idx);
        for (int k = 0; k < n_cols; k++) {
            hipLaunchKernelGGL(k_gather_u32, dim3(grid_stream(keep)), dim3(BLOCK), 0,
                               s->stream, (const uint32_t*)cols[k]->ptr, d_idx, keep,
                               (uint32_t*)d_tmp);
            HIP_CHECK(hipMemcpyAsync(cols[k]->ptr, d_tmp, keep * 4,
                                     hipMemcpyDeviceToDevice, s->stream));
        }
        HIP_CHECK(hipStreamSynchronize(s->stream));
        (void)hipFree(d_bs);
        (void)hipFree(d_idx);
        (void)hipFree(d_tmp);
        n = keep;
        HIP_CHECK(hipMemsetAsync(d_filter, 1, n ? n : 1, s->stream));
        zero_count = 0;
        return GPUE_OK;
    };

    for (int p = 0; p < n_preds && n > 0; p++) {
        ARG_CHECK(pred_col[p] >= 0 && pred_col[p] < n_cols);
        HIP_CHECK(hipMemsetAsync(d_tc, 0, 8, s->stream));
        hipLaunchKernelGGL(k_eval_pred_i32, dim3(grid_stream(n)), dim3(BLOCK), 0, s->stream,
                           (const int32_t*)cols[pred_col[p]]->ptr, n, pred_op[p],
                           pred_lo[p], pred_hi[p], d_filter, d_tc);
        unsigned long long tc = 0;
        HIP_CHECK(hipMemcpyAsync(&tc, d_tc, 8, hipMemcpyDeviceToHost, s->stream));
        HIP_CHECK(hipStreamSynchronize(s->stream));
        // NOTE: tc counts MERGED survivors (filter & pass) — a conjunct whose
        // own column is all-true over already-filtered rows leaves tc ==
        // previous survivor count, matching the reference's merge+count_zero
        if (tc == 0) { n = 0; break; }
        zero_count = n - tc;
        uint64_t prune_threshold = n * 8 / 10 > 1024 ? n * 8 / 10 : 1024;
        if (zero_count > prune_threshold) {
            int rc = compact_all(tc);
            if (rc != GPUE_OK) return rc;
        }
    }
    if (n > 0 && zero_count > 0) {
        int rc = compact_all(n - zero_count);
        if (rc != GPUE_OK) return rc;
    }
    (void)hipFree(d_filter);
    (void)hipFree(d_tc);
    *out_rows = n;
    return GPUE_OK;
}

extern "C" int gpue_eval_conjuncts_i64(gpue_session* s, gpue_dbuf** cols, int n_cols,
                                       uint64_t n_rows, const int32_t* pred_col,
                                       const int32_t* pred_op, const int64_t* pred_lo,
                                       const int64_t* pred_hi, int n_preds,
                                       uint64_t* out_rows);
int gpue_eval_conjuncts_i64(gpue_session* s, gpue_dbuf** cols, int n_cols, uint64_t n_rows,
                            const int32_t* pred_col, const int32_t* pred_op,
                            const int64_t* pred_lo, const int64_t* pred_hi, int n_preds,
                            uint64_t* out_rows) {
    ARG_CHECK(s && cols && n_cols > 0 && pred_col && pred_op && pred_lo && pred_hi &&
              out_rows);
    uint64_t n = n_rows;
    uint8_t* d_filter = nullptr;
    unsigned long long* d_tc = nullptr;
    HIP_CHECK(hipMalloc(&d_filter, n_rows ? n_rows : 1));
    HIP_CHECK(hipMalloc(&d_tc, 8));
    HIP_CHECK(hipMemsetAsync(d_filter, 1, n_rows ? n_rows : 1, s->stream));
    uint64_t zero_count = 0;

    auto compact_all = [&](uint64_t keep) -> int {
        uint32_t nb = grid_for(n);
        uint64_t tile = (n + nb - 1) / nb;
        uint64_t* d_bs = nullptr;
        uint32_t* d_idx = nullptr;
        int64_t* d_tmp = nullptr;
        HIP_CHECK(hipMalloc(&d_bs, (nb + 1) * 8));
        HIP_CHECK(hipMalloc(&d_idx, keep * 4));
        HIP_CHECK(hipMalloc(&d_tmp, keep * 8));
        hipLaunchKernelGGL(k_mask_block_counts, dim3(nb), dim3(BLOCK), 0, s->stream,
                           d_filter, n, tile, d_bs);
        hipLaunchKernelGGL(k_scan_small, dim3(1), dim3(1), 0, s->stream, d_bs, nb);
        hipLaunchKernelGGL(k_mask_emit_idx, dim3(nb), dim3(BLOCK), 0, s->stream, d_filter,
                           n, tile, d_bs, d_idx);
        for (int k = 0; k < n_cols; k++) {
            hipLaunchKernelGGL(k_gather_u64, dim3(grid_stream(keep)), dim3(BLOCK), 0,
                               s->stream, (const uint64_t*)cols[k]->ptr, d_idx, keep,
                               (uint64_t*)d_tmp);
            HIP_CHECK(hipMemcpyAsync(cols[k]->ptr, d_tmp, keep * 8,
                                     hipMemcpyDeviceToDevice, s->stream));
        }
        HIP_CHECK(hipStreamSynchronize(s->stream));
        (void)hipFree(d_bs);
        (void)hipFree(d_idx);
        (void)hipFree(d_tmp);
        n = keep;
        HIP_CHECK(hipMemsetAsync(d_filter, 1, n ? n : 1, s->stream));
        zero_count = 0;
        return GPUE_OK;
    };

    for (int p = 0; p < n_preds && n > 0; p++) {
        ARG_CHECK(pred_col[p] >= 0 && pred_col[p] < n_cols);
        HIP_CHECK(hipMemsetAsync(d_tc, 0, 8, s->stream));
        hipLaunchKernelGGL(k_eval_pred_i64, dim3(grid_stream(n)), dim3(BLOCK), 0, s->stream,
                           (const int64_t*)cols[pred_col[p]]->ptr, n, pred_op[p],
                           pred_lo[p], pred_hi[p], d_filter, d_tc);
        unsigned long long tc = 0;
        HIP_CHECK(hipMemcpyAsync(&tc, d_tc, 8, hipMemcpyDeviceToHost, s->stream));
        HIP_CHECK(hipStreamSynchronize(s->stream));
        if (tc == 0) { n = 0; break; }
        zero_count = n - tc;
        uint64_t prune_threshold = n * 8 / 10 > 1024 ? n * 8 / 10 : 1024;
        if (zero_count > prune_threshold) {
            int rc = compact_all(tc);
            if (rc != GPUE_OK) return rc;
        }
    }
    if (n > 0 && zero_count > 0) {
        int rc = compact_all(n - zero_count);
        if (rc != GPUE_OK) return rc;
    }
    (void)hipFree(d_filter);
    (void)hipFree(d_tc);
    *out_rows = n;
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// SERIALIZED_VARCHAR / Slice keys (the selector's last constructor branch):
// JoinKeyHash<Slice> = crc_hash_32(bytes, len, 0x811C9DC5) & (bucket_size-1)
// (join_hash_map_helper.h:57-64; CRC32-C + phmap_mix<4>, bitwise device
// form identical to the oracle's table form), chains walked with a byte
// compare (hash + verify). Columns are BinaryColumn-shaped: bytes + uint32
// offsets (binary_column.h:458-459), rows 1-based with row 0 = empty.
// ---------------------------------------------------------------------------
__device__ static inline uint32_t crc_hash_32_dev(const uint8_t* p, uint32_t len,
                                                  uint32_t seed) {
    uint32_t h = seed;
    for (uint32_t i = 0; i < len; i++) {
        h ^= p[i];
        #pragma unroll
        for (int k = 0; k < 8; k++)
            h = (h >> 1) ^ (0x82F63B78u & (uint32_t) - (int)(h & 1));
    }
    uint64_t l = (uint64_t)h * 0xcc9e2d51ull;
    return (uint32_t)(l ^ (l >> 32));
}

__global__ void k_build_varchar(const uint8_t* __restrict__ bytes,
                                const uint32_t* __restrict__ offsets,
                                const uint8_t* __restrict__ is_nulls /*may be null*/,
                                uint64_t row_count, uint32_t bucket_mask,
                                uint32_t* __restrict__ first, uint32_t* __restrict__ next) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = 1 + (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i <= row_count;
         i += stride) {
        if (is_nulls && is_nulls[i]) { next[i] = 0; continue; }
        uint32_t len = offsets[i + 1] - offsets[i];
        uint32_t b = crc_hash_32_dev(bytes + offsets[i], len, 0x811C9DC5u) & bucket_mask;
        next[i] = atomicExch(&first[b], (uint32_t)i);
    }
}

__device__ static inline bool slice_eq(const uint8_t* a, uint32_t alen, const uint8_t* b,
                                       uint32_t blen) {
    if (alen != blen) return false;
    for (uint32_t k = 0; k < alen; k++)
        if (a[k] != b[k]) return false;
    return true;
}

__global__ void k_probe_count_vc(const uint8_t* __restrict__ pbytes,
                                 const uint32_t* __restrict__ poffsets, uint64_t n,
                                 uint32_t bucket_mask, const uint32_t* __restrict__ first,
                                 const uint32_t* __restrict__ next,
                                 const uint8_t* __restrict__ bbytes,
                                 const uint32_t* __restrict__ boffsets,
                                 const uint8_t* __restrict__ pnulls /*may be null*/,
                                 int mode, uint32_t* __restrict__ row_counts) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        if (mode == 6 && pnulls && pnulls[i]) {
            // NULL_AWARE_LEFT_ANTI (NOT IN): null probe rows excluded
            // (join_hash_map.hpp:1225-1240)
            row_counts[i] = 0;
            continue;
        }
        uint32_t len = poffsets[i + 1] - poffsets[i];
        uint32_t b = crc_hash_32_dev(pbytes + poffsets[i], len, 0x811C9DC5u) & bucket_mask;
        uint32_t j = (pnulls && pnulls[i]) ? 0u : first[b];
        uint32_t c = 0;
        while (j != 0) {
            c += slice_eq(bbytes + boffsets[j], boffsets[j + 1] - boffsets[j],
                          pbytes + poffsets[i], len);
            j = next[j];
        }
        row_counts[i] = join_mode_count(c, mode == 6 ? 2 : mode);
    }
}

__global__ void k_probe_emit_vc(const uint8_t* __restrict__ pbytes,
                                const uint32_t* __restrict__ poffsets, uint64_t n,
                                uint32_t bucket_mask, const uint32_t* __restrict__ first,
                                const uint32_t* __restrict__ next,
                                const uint8_t* __restrict__ bbytes,
                                const uint32_t* __restrict__ boffsets,
                                const uint8_t* __restrict__ pnulls /*may be null*/,
                                int mode, const uint32_t* __restrict__ row_counts,
                                const uint64_t* __restrict__ row_offsets,
                                uint32_t* __restrict__ out_probe,
                                uint32_t* __restrict__ out_build) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        if (row_counts[i] == 0) continue;
        uint64_t pos = row_offsets[i];
        uint32_t len = poffsets[i + 1] - poffsets[i];
        uint32_t b = crc_hash_32_dev(pbytes + poffsets[i], len, 0x811C9DC5u) & bucket_mask;
        uint32_t j = (pnulls && pnulls[i]) ? 0u : first[b];
        uint32_t c = 0;
        while (j != 0) {
            if (slice_eq(bbytes + boffsets[j], boffsets[j + 1] - boffsets[j],
                         pbytes + poffsets[i], len)) {
                if (mode == 0 || mode == 3 || (mode == 1 && c == 0)) {
                    out_probe[pos] = (uint32_t)i;
                    out_build[pos] = j;
                    pos++;
                }
                c++;
                if (mode == 1 || mode == 2) break;
            }
            j = next[j];
        }
        if (c == 0 && (mode == 2 || mode == 3)) { // unmatched emit, build 0 = NULL
            out_probe[pos] = (uint32_t)i;
            out_build[pos] = 0;
        }
    }
}

extern "C" {
int gpue_join_build_varchar(gpue_session* s, gpue_dbuf* bytes, gpue_dbuf* offsets,
                            uint64_t row_count, gpue_join_table** out);
int gpue_join_probe_emit_varchar(gpue_session* s, gpue_join_table* t, gpue_dbuf* pbytes,
                                 gpue_dbuf* poffsets, uint64_t n_rows,
                                 gpue_dbuf* out_probe_idx, gpue_dbuf* out_build_idx,
                                 uint64_t* match_count);
int gpue_join_probe_emit_varchar_mode(gpue_session* s, gpue_join_table* t, gpue_dbuf* pbytes,
                                      gpue_dbuf* poffsets, uint64_t n_rows, int mode,
                                      gpue_dbuf* out_probe_idx, gpue_dbuf* out_build_idx,
                                      uint64_t* match_count);
int gpue_join_probe_emit_varchar_nulls(gpue_session* s, gpue_join_table* t, gpue_dbuf* pbytes,
                                       gpue_dbuf* poffsets, gpue_dbuf* pnulls,
                                       uint64_t n_rows, int mode, gpue_dbuf* out_probe_idx,
                                       gpue_dbuf* out_build_idx, uint64_t* match_count);
int gpue_join_build_varchar_nulls(gpue_session* s, gpue_dbuf* bytes, gpue_dbuf* offsets,
                                  gpue_dbuf* is_nulls, uint64_t row_count,
                                  gpue_join_table** out);
}

int gpue_join_build_varchar(gpue_session* s, gpue_dbuf* bytes, gpue_dbuf* offsets,
                            uint64_t row_count, gpue_join_table** out) {
    return gpue_join_build_varchar_nulls(s, bytes, offsets, nullptr, row_count, out);
}

int gpue_join_build_varchar_nulls(gpue_session* s, gpue_dbuf* bytes, gpue_dbuf* offsets,
                                  gpue_dbuf* is_nulls, uint64_t row_count,
                                  gpue_join_table** out) {
    ARG_CHECK(s && bytes && offsets && out && row_count > 0);
    ARG_CHECK(offsets->bytes >= (row_count + 2) * 4);
    gpue_join_table* t = new gpue_join_table();
    t->s = s;
    t->kind = gpue_join_table::VARCHAR;
    t->row_count = row_count;
    t->bucket_size = calc_bucket_size((uint32_t)(row_count + 1));
    t->log_bucket_size = (uint32_t)__builtin_ctzll(t->bucket_size);
    HIP_CHECK(hipMalloc(&t->first, t->bucket_size * sizeof(uint32_t)));
    HIP_CHECK(hipMalloc(&t->next, (row_count + 1) * sizeof(uint32_t)));
    HIP_CHECK(hipMalloc(&t->key_bytes, bytes->bytes));
    HIP_CHECK(hipMalloc(&t->key_offsets, (row_count + 2) * sizeof(uint32_t)));
    HIP_CHECK(hipMemsetAsync(t->first, 0, t->bucket_size * sizeof(uint32_t), s->stream));
    HIP_CHECK(hipMemsetAsync(t->next, 0, (row_count + 1) * sizeof(uint32_t), s->stream));
    HIP_CHECK(hipMemcpyAsync(t->key_bytes, bytes->ptr, bytes->bytes,
                             hipMemcpyDeviceToDevice, s->stream));
    HIP_CHECK(hipMemcpyAsync(t->key_offsets, offsets->ptr, (row_count + 2) * 4,
                             hipMemcpyDeviceToDevice, s->stream));
    if (is_nulls) {
        ARG_CHECK(is_nulls->bytes >= row_count + 1);
        HIP_CHECK(hipMalloc(&t->key_nulls, row_count + 1));
        HIP_CHECK(hipMemcpyAsync(t->key_nulls, is_nulls->ptr, row_count + 1,
                                 hipMemcpyDeviceToDevice, s->stream));
    }
    hipLaunchKernelGGL(k_build_varchar, dim3(grid_for(row_count)), dim3(BLOCK), 0, s->stream,
                       t->key_bytes, t->key_offsets, t->key_nulls, row_count,
                       (uint32_t)(t->bucket_size - 1), t->first, t->next);
    HIP_CHECK(hipGetLastError());
    *out = t;
    return GPUE_OK;
}

int gpue_join_probe_emit_varchar(gpue_session* s, gpue_join_table* t, gpue_dbuf* pbytes,
                                 gpue_dbuf* poffsets, uint64_t n_rows,
                                 gpue_dbuf* out_probe_idx, gpue_dbuf* out_build_idx,
                                 uint64_t* match_count) {
    return gpue_join_probe_emit_varchar_mode(s, t, pbytes, poffsets, n_rows, 0,
                                             out_probe_idx, out_build_idx, match_count);
}

int gpue_join_probe_emit_varchar_mode(gpue_session* s, gpue_join_table* t, gpue_dbuf* pbytes,
                                      gpue_dbuf* poffsets, uint64_t n_rows, int mode,
                                      gpue_dbuf* out_probe_idx, gpue_dbuf* out_build_idx,
                                      uint64_t* match_count) {
    return gpue_join_probe_emit_varchar_nulls(s, t, pbytes, poffsets, nullptr, n_rows, mode,
                                              out_probe_idx, out_build_idx, match_count);
}

int gpue_join_probe_emit_varchar_nulls(gpue_session* s, gpue_join_table* t, gpue_dbuf* pbytes,
                                       gpue_dbuf* poffsets, gpue_dbuf* pnulls,
                                       uint64_t n_rows, int mode, gpue_dbuf* out_probe_idx,
                                       gpue_dbuf* out_build_idx, uint64_t* match_count) {
    ARG_CHECK(s && t && pbytes && poffsets && match_count);
    ARG_CHECK((mode >= 0 && mode <= 3) || mode == 6);
    ARG_CHECK(t->kind == gpue_join_table::VARCHAR);
    ARG_CHECK(poffsets->bytes >= (n_rows + 1) * 4);
    uint32_t nb = grid_for(n_rows);
    uint64_t tile = (n_rows + nb - 1) / nb;
    uint32_t* d_counts = nullptr;
    uint64_t* d_bsums = nullptr;
    uint64_t* d_offsets = nullptr;
    HIP_CHECK(hipMalloc(&d_counts, n_rows * 4));
    HIP_CHECK(hipMalloc(&d_bsums, (nb + 1) * 8));
    hipLaunchKernelGGL(k_probe_count_vc, dim3(nb), dim3(BLOCK), 0, s->stream,
                       (const uint8_t*)pbytes->ptr, (const uint32_t*)poffsets->ptr, n_rows,
                       (uint32_t)(t->bucket_size - 1), t->first, t->next, t->key_bytes,
                       t->key_offsets, pnulls ? (const uint8_t*)pnulls->ptr : nullptr,
                       mode, d_counts);
    hipLaunchKernelGGL(k_block_sums_u32, dim3(nb), dim3(BLOCK), 0, s->stream, d_counts,
                       n_rows, tile, d_bsums);
    hipLaunchKernelGGL(k_scan_small, dim3(1), dim3(1), 0, s->stream, d_bsums, nb);
    uint64_t total = 0;
    HIP_CHECK(hipMemcpyAsync(&total, d_bsums + nb, 8, hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    *match_count = total;
    if (out_probe_idx && out_build_idx && total > 0) {
        ARG_CHECK(out_probe_idx->bytes >= total * 4 && out_build_idx->bytes >= total * 4);
        HIP_CHECK(hipMalloc(&d_offsets, n_rows * 8));
        hipLaunchKernelGGL(k_scan_offsets, dim3(nb), dim3(BLOCK), 0, s->stream, d_counts,
                           n_rows, tile, d_bsums, d_offsets);
        hipLaunchKernelGGL(k_probe_emit_vc, dim3(nb), dim3(BLOCK), 0, s->stream,
                           (const uint8_t*)pbytes->ptr, (const uint32_t*)poffsets->ptr,
                           n_rows, (uint32_t)(t->bucket_size - 1), t->first, t->next,
                           t->key_bytes, t->key_offsets,
                           pnulls ? (const uint8_t*)pnulls->ptr : nullptr,
                           mode == 6 ? 2 : mode, d_counts,
                           d_offsets, (uint32_t*)out_probe_idx->ptr,
                           (uint32_t*)out_build_idx->ptr);
        HIP_CHECK(hipStreamSynchronize(s->stream));
        (void)hipFree(d_offsets);
    }
    (void)hipFree(d_counts);
    (void)hipFree(d_bsums);
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// 16-byte (LARGEINT / SERIALIZED_FIXED_SIZE_LARGEINT) bucket-chained join —
// the generic JoinKeyHash<T,16> path: crc_hash_32 over the 16 key bytes,
// seed CRC_SEED, masked by (bucket_size-1) (join_hash_map_helper.h:23-30).
// Functionally equal to the varchar path with fixed 16-byte slices (the
// oracle pin reuses exactly that), minus the per-row offsets indirection.
// ---------------------------------------------------------------------------
__global__ void k_build_bucket_chained128(const ulonglong2* __restrict__ keys,
                                          uint64_t row_count, uint32_t bucket_mask,
                                          uint32_t* __restrict__ first,
                                          uint32_t* __restrict__ next) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = 1 + (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i <= row_count;
         i += stride) {
        ulonglong2 k = keys[i];
        uint32_t b = crc_hash_32_dev((const uint8_t*)&k, 16, 0x811C9DC5u) & bucket_mask;
        next[i] = atomicExch(&first[b], (uint32_t)i);
    }
}

__global__ void k_probe_count_u128(const ulonglong2* __restrict__ probe_keys, uint64_t n,
                                   uint32_t bucket_mask, const uint32_t* __restrict__ first,
                                   const uint32_t* __restrict__ next,
                                   const ulonglong2* __restrict__ build_keys, int mode,
                                   uint32_t* __restrict__ row_counts) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        ulonglong2 k = probe_keys[i];
        uint32_t b = first[crc_hash_32_dev((const uint8_t*)&k, 16, 0x811C9DC5u) & bucket_mask];
        uint32_t c = 0;
        while (b != 0) {
            ulonglong2 bk = build_keys[b];
            c += (bk.x == k.x) & (bk.y == k.y);
            b = next[b];
        }
        row_counts[i] = join_mode_count(c, mode);
    }
}

__global__ void k_probe_emit_u128(const ulonglong2* __restrict__ probe_keys, uint64_t n,
                                  uint32_t bucket_mask, const uint32_t* __restrict__ first,
                                  const uint32_t* __restrict__ next,
                                  const ulonglong2* __restrict__ build_keys, int mode,
                                  const uint32_t* __restrict__ row_counts,
                                  const uint64_t* __restrict__ row_offsets,
                                  uint32_t* __restrict__ out_probe,
                                  uint32_t* __restrict__ out_build) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        if (row_counts[i] == 0) continue;
        uint64_t pos = row_offsets[i];
        ulonglong2 k = probe_keys[i];
        uint32_t b = first[crc_hash_32_dev((const uint8_t*)&k, 16, 0x811C9DC5u) & bucket_mask];
        bool any = false;
        while (b != 0) {
            ulonglong2 bk = build_keys[b];
            if ((bk.x == k.x) & (bk.y == k.y)) {
                any = true;
                if (mode != 2) {
                    out_probe[pos] = (uint32_t)i;
                    out_build[pos] = b;
                    pos++;
                }
                if (mode == 1 || mode == 2) break;
            }
            b = next[b];
        }
        if (!any && (mode == 2 || mode == 3)) {
            out_probe[pos] = (uint32_t)i;
            out_build[pos] = 0;
        }
    }
}

extern "C" {
int gpue_join_build_bucket_chained_u128(gpue_session* s, gpue_dbuf* keys /*16 B, 1-based*/,
                                        uint64_t row_count, gpue_join_table** out);
int gpue_join_probe_emit_mode_u128(gpue_session* s, gpue_join_table* t,
                                   gpue_dbuf* probe_keys, uint64_t n_rows, int mode,
                                   gpue_dbuf* out_probe_idx, gpue_dbuf* out_build_idx,
                                   uint64_t* match_count);
}

int gpue_join_build_bucket_chained_u128(gpue_session* s, gpue_dbuf* keys,
                                        uint64_t row_count, gpue_join_table** out) {
    ARG_CHECK(s && keys && out && row_count > 0 && row_count + 1 < (1ull << 31));
    ARG_CHECK(keys->bytes >= (row_count + 1) * 16);
    gpue_join_table* t = new gpue_join_table();
    t->s = s;
    t->kind = gpue_join_table::BUCKET_CHAINED128;
    t->row_count = row_count;
    t->bucket_size = calc_bucket_size((uint32_t)(row_count + 1));
    t->log_bucket_size = (uint32_t)__builtin_ctzll(t->bucket_size);
    HIP_CHECK(hipMalloc(&t->first, t->bucket_size * sizeof(uint32_t)));
    HIP_CHECK(hipMalloc(&t->next, (row_count + 1) * sizeof(uint32_t)));
    HIP_CHECK(hipMalloc(&t->build_keys128, (row_count + 1) * sizeof(ulonglong2)));
    HIP_CHECK(hipMemsetAsync(t->first, 0, t->bucket_size * sizeof(uint32_t), s->stream));
    HIP_CHECK(hipMemsetAsync(t->next, 0, (row_count + 1) * sizeof(uint32_t), s->stream));
    HIP_CHECK(hipMemcpyAsync(t->build_keys128, keys->ptr,
                             (row_count + 1) * sizeof(ulonglong2),
                             hipMemcpyDeviceToDevice, s->stream));
    hipLaunchKernelGGL(k_build_bucket_chained128, dim3(grid_for(row_count)), dim3(BLOCK), 0,
                       s->stream, t->build_keys128, row_count,
                       (uint32_t)(t->bucket_size - 1), t->first, t->next);
    HIP_CHECK(hipGetLastError());
    *out = t;
    return GPUE_OK;
}

int gpue_join_probe_emit_mode_u128(gpue_session* s, gpue_join_table* t, gpue_dbuf* probe_keys,
                                   uint64_t n_rows, int mode, gpue_dbuf* out_probe_idx,
                                   gpue_dbuf* out_build_idx, uint64_t* match_count) {
    ARG_CHECK(s && t && probe_keys && match_count);
    ARG_CHECK(t->kind == gpue_join_table::BUCKET_CHAINED128);
    ARG_CHECK(mode >= 0 && mode <= 3);
    ARG_CHECK(probe_keys->bytes >= n_rows * 16);
    uint32_t nb = grid_for(n_rows);
    uint64_t tile = (n_rows + nb - 1) / nb;
    uint32_t* d_counts = nullptr;
    uint64_t* d_bsums = nullptr;
    uint64_t* d_offsets = nullptr;
    HIP_CHECK(hipMalloc(&d_counts, n_rows * 4));
    HIP_CHECK(hipMalloc(&d_bsums, (nb + 1) * 8));
    hipLaunchKernelGGL(k_probe_count_u128, dim3(nb), dim3(BLOCK), 0, s->stream,
                       (const ulonglong2*)probe_keys->ptr, n_rows,
                       (uint32_t)(t->bucket_size - 1), t->first, t->next, t->build_keys128,
                       mode, d_counts);
    hipLaunchKernelGGL(k_block_sums_u32, dim3(nb), dim3(BLOCK), 0, s->stream, d_counts,
                       n_rows, tile, d_bsums);
    hipLaunchKernelGGL(k_scan_small, dim3(1), dim3(1), 0, s->stream, d_bsums, nb);
    uint64_t total = 0;
    HIP_CHECK(hipMemcpyAsync(&total, d_bsums + nb, 8, hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    *match_count = total;
    if (out_probe_idx && out_build_idx && total > 0) {
        ARG_CHECK(out_probe_idx->bytes >= total * 4 && out_build_idx->bytes >= total * 4);
        HIP_CHECK(hipMalloc(&d_offsets, n_rows * 8));
        hipLaunchKernelGGL(k_scan_offsets, dim3(nb), dim3(BLOCK), 0, s->stream, d_counts,
                           n_rows, tile, d_bsums, d_offsets);
        hipLaunchKernelGGL(k_probe_emit_u128, dim3(nb), dim3(BLOCK), 0, s->stream,
                           (const ulonglong2*)probe_keys->ptr, n_rows,
                           (uint32_t)(t->bucket_size - 1), t->first, t->next,
                           t->build_keys128, mode, d_counts, d_offsets,
                           (uint32_t*)out_probe_idx->ptr, (uint32_t*)out_build_idx->ptr);
        HIP_CHECK(hipStreamSynchronize(s->stream));
        (void)hipFree(d_offsets);
    }
    (void)hipFree(d_counts);
    (void)hipFree(d_bsums);
    return GPUE_OK;
}

// SERIALIZED_FIXED_SIZE_LARGEINT packing for two 8-byte key columns
// (join_key_constructor.h:40-153 / serialize_batch_at_interval: keys
// concatenated little-endian into one 16-byte value)
__global__ void k_pack_keys_2xi64(const uint64_t* __restrict__ a,
                                  const uint64_t* __restrict__ b, uint64_t n,
                                  ulonglong2* __restrict__ out) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
        out[i] = make_ulonglong2(a[i], b[i]);
}

extern "C" int gpue_pack_keys_2xi64(gpue_session* s, gpue_dbuf* a, gpue_dbuf* b,
                                    uint64_t n, gpue_dbuf* out);
int gpue_pack_keys_2xi64(gpue_session* s, gpue_dbuf* a, gpue_dbuf* b, uint64_t n,
                         gpue_dbuf* out) {
    ARG_CHECK(s && a && b && out);
    ARG_CHECK(a->bytes >= n * 8 && b->bytes >= n * 8 && out->bytes >= n * 16);
    hipLaunchKernelGGL(k_pack_keys_2xi64, dim3(grid_for(n)), dim3(BLOCK), 0, s->stream,
                       (const uint64_t*)a->ptr, (const uint64_t*)b->ptr, n,
                       (ulonglong2*)out->ptr);
    HIP_CHECK(hipGetLastError());
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// SimdBlockFilter — the reference's split-block bloom runtime filter
// (runtime_filter.h:79-232, runtime_filter.cpp:26-36,114-124; upstream
// fastfilter_cpp). 32 B buckets of 8 lanes; one bit per lane from
// (key*SALT[i])>>27; inserted hash = phmap_mix<8>(value) for integer keys
// (runtime_filter.h:1271-1276). Build uses atomicOr (order-independent, so
// the directory is BIT-IDENTICAL to the oracle's serial build); probe is the
// scan-side early-prune the reference pushes to scan operators
// (operator.h:188-199). Oracle restatement: oracle.c orc_sbf_*.
// ---------------------------------------------------------------------------
__constant__ static const uint32_t SBF_SALT[8] = {0x47b6137bu, 0x44974d91u, 0x8824ad5bu,
                                                  0xa2b7289du, 0x705495c7u, 0x2df1424bu,
                                                  0x9efc4947u, 0x5c6bfb31u};

__device__ static inline uint64_t phmap_mix8_dev(uint64_t a) {
    const uint64_t k = 0xde5fb9d2630458e9ull;
    return a * k + __umul64hi(a, k);
}

__global__ void k_sbf_build_i32(const int32_t* __restrict__ keys, uint64_t n,
                                int32_t log_num_buckets, uint32_t* __restrict__ dir) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        uint64_t h = phmap_mix8_dev((uint64_t)(int64_t)keys[i]);
        uint32_t bucket = (uint32_t)(h & ((1ull << log_num_buckets) - 1));
        uint32_t key = (uint32_t)(h >> log_num_buckets);
        #pragma unroll
        for (int j = 0; j < 8; j++)
            atomicOr(&dir[bucket * 8 + j], 1u << ((key * SBF_SALT[j]) >> 27));
    }
}

__global__ void k_sbf_test_i32(const int32_t* __restrict__ keys, uint64_t n,
                               const uint32_t* __restrict__ dir, int32_t log_num_buckets,
                               uint8_t* __restrict__ out) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        uint64_t h = phmap_mix8_dev((uint64_t)(int64_t)keys[i]);
        uint32_t bucket = (uint32_t)(h & ((1ull << log_num_buckets) - 1));
        uint32_t key = (uint32_t)(h >> log_num_buckets);
        const uint32_t* b = &dir[bucket * 8];
        uint4 lo = *(const uint4*)b;
        uint4 hi = *(const uint4*)(b + 4);
        uint32_t pass = 1;
        const uint32_t* lanes = &lo.x;
        #pragma unroll
        for (int j = 0; j < 4; j++)
            pass &= (lanes[j] >> ((key * SBF_SALT[j]) >> 27)) & 1u;
        const uint32_t* lanes2 = &hi.x;
        #pragma unroll
        for (int j = 0; j < 4; j++)
            pass &= (lanes2[j] >> ((key * SBF_SALT[j + 4]) >> 27)) & 1u;
        out[i] = (uint8_t)pass;
    }
}

extern "C" {
int gpue_sbf_build_i32(gpue_session* s, gpue_dbuf* keys, uint64_t n,
                       int32_t log_num_buckets, gpue_dbuf* directory);
int gpue_sbf_test_i32(gpue_session* s, gpue_dbuf* keys, uint64_t n, gpue_dbuf* directory,
                      int32_t log_num_buckets, gpue_dbuf* out);
}

int gpue_sbf_build_i32(gpue_session* s, gpue_dbuf* keys, uint64_t n, int32_t log_num_buckets,
                       gpue_dbuf* directory) {
    ARG_CHECK(s && keys && directory && log_num_buckets >= 1 && log_num_buckets < 32);
    ARG_CHECK(directory->bytes >= (32ull << log_num_buckets));
    ARG_CHECK(keys->bytes >= n * 4);
    HIP_CHECK(hipMemsetAsync(directory->ptr, 0, 32ull << log_num_buckets, s->stream));
    hipLaunchKernelGGL(k_sbf_build_i32, dim3(grid_for(n)), dim3(BLOCK), 0, s->stream,
                       (const int32_t*)keys->ptr, n, log_num_buckets,
                       (uint32_t*)directory->ptr);
    HIP_CHECK(hipGetLastError());
    return GPUE_OK;
}

int gpue_sbf_test_i32(gpue_session* s, gpue_dbuf* keys, uint64_t n, gpue_dbuf* directory,
                      int32_t log_num_buckets, gpue_dbuf* out) {
    ARG_CHECK(s && keys && directory && out && log_num_buckets >= 1 && log_num_buckets < 32);
    ARG_CHECK(keys->bytes >= n * 4 && out->bytes >= n);
    hipLaunchKernelGGL(k_sbf_test_i32, dim3(grid_stream(n)), dim3(BLOCK), 0, s->stream,
                       (const int32_t*)keys->ptr, n, (const uint32_t*)directory->ptr,
                       log_num_buckets, (uint8_t*)out->ptr);
    HIP_CHECK(hipGetLastError());
    return GPUE_OK;
}

// RIGHT SEMI/ANTI over Slice keys: probe pass marks matched BUILD rows
// (atomicOr into a bitset), then compact matched/unmatched build rows.
__global__ void k_probe_mark_vc(const uint8_t* __restrict__ pbytes,
                                const uint32_t* __restrict__ poffsets, uint64_t n,
                                uint32_t bucket_mask, const uint32_t* __restrict__ first,
                                const uint32_t* __restrict__ next,
                                const uint8_t* __restrict__ bbytes,
                                const uint32_t* __restrict__ boffsets,
                                uint32_t* __restrict__ matched_bits) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        uint32_t len = poffsets[i + 1] - poffsets[i];
        uint32_t b = crc_hash_32_dev(pbytes + poffsets[i], len, 0x811C9DC5u) & bucket_mask;
        for (uint32_t j = first[b]; j != 0; j = next[j])
            if (slice_eq(bbytes + boffsets[j], boffsets[j + 1] - boffsets[j],
                         pbytes + poffsets[i], len))
                atomicOr(&matched_bits[j >> 5], 1u << (j & 31));
    }
}

extern "C" int gpue_join_probe_right_varchar(gpue_session* s, gpue_join_table* t,
                                             gpue_dbuf* pbytes, gpue_dbuf* poffsets,
                                             uint64_t n_rows, int anti,
                                             gpue_dbuf* out_build_idx, uint64_t* count);
__global__ void k_right_emit_bits(const uint32_t* __restrict__ matched_bits,
                                  uint64_t build_rows, int anti,
                                  unsigned long long* __restrict__ cursor,
                                  uint32_t* __restrict__ out) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t j = 1 + (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; j <= build_rows;
         j += stride) {
        uint32_t hit = (matched_bits[j >> 5] >> (j & 31)) & 1u;
        if ((anti && !hit) || (!anti && hit)) {
            unsigned long long pos = atomicAdd(cursor, 1ull);
            if (out) out[pos] = (uint32_t)j;
        }
    }
}

int gpue_join_probe_right_varchar(gpue_session* s, gpue_join_table* t, gpue_dbuf* pbytes,
                                  gpue_dbuf* poffsets, uint64_t n_rows, int anti,
                                  gpue_dbuf* out_build_idx, uint64_t* count) {
    ARG_CHECK(s && t && pbytes && poffsets && count);
    ARG_CHECK(t->kind == gpue_join_table::VARCHAR);
    uint64_t nwords = (t->row_count + 1 + 31) / 32;
    uint32_t* d_bits = nullptr;
    unsigned long long* d_cursor = nullptr;
    HIP_CHECK(hipMalloc(&d_bits, nwords * 4));
    HIP_CHECK(hipMalloc(&d_cursor, 8));
    HIP_CHECK(hipMemsetAsync(d_bits, 0, nwords * 4, s->stream));
    HIP_CHECK(hipMemsetAsync(d_cursor, 0, 8, s->stream));
    hipLaunchKernelGGL(k_probe_mark_vc, dim3(grid_for(n_rows)), dim3(BLOCK), 0, s->stream,
                       (const uint8_t*)pbytes->ptr, (const uint32_t*)poffsets->ptr, n_rows,
                       (uint32_t)(t->bucket_size - 1), t->first, t->next, t->key_bytes,
                       t->key_offsets, d_bits);
    if (out_build_idx) {
        ARG_CHECK(out_build_idx->bytes >= t->row_count * 4);
        hipLaunchKernelGGL(k_right_emit_bits, dim3(grid_for(t->row_count)), dim3(BLOCK), 0,
                           s->stream, d_bits, t->row_count, anti, d_cursor,
                           (uint32_t*)out_build_idx->ptr);
    } else {
        hipLaunchKernelGGL(k_right_emit_bits, dim3(grid_for(t->row_count)), dim3(BLOCK), 0,
                           s->stream, d_bits, t->row_count, anti, d_cursor, nullptr);
    }
    unsigned long long c = 0;
    HIP_CHECK(hipMemcpyAsync(&c, d_cursor, 8, hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    (void)hipFree(d_bits);
    (void)hipFree(d_cursor);
    *count = c;
    return GPUE_OK;
}

// Nullable probe (lookup_init is_nulls path + per-type semantics): a null
// probe key matches nothing — INNER/SEMI emit nothing for it, ANTI/OUTER
// emit the unmatched (i, 0) row.
__global__ void k_probe_count_bc_nulls(const uint32_t* __restrict__ probe_keys,
                                       const uint8_t* __restrict__ is_nulls, uint64_t n,
                                       uint32_t log_bucket_size,
                                       const uint32_t* __restrict__ first,
                                       const uint32_t* __restrict__ next,
                                       const uint32_t* __restrict__ build_keys, int mode,
                                       uint32_t* __restrict__ row_counts) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        uint32_t c = 0;
        if (!is_nulls[i]) {
            uint32_t k = probe_keys[i];
            uint32_t b = first[join_hash_u32(k, log_bucket_size)];
            while (b != 0) {
                c += (build_keys[b] == k);
                b = next[b];
            }
        } else if (mode == 6) {
            // NULL_AWARE_LEFT_ANTI (the NOT IN lowering,
            // join_hash_map.hpp:1225-1240): null probe rows are EXCLUDED —
            // NULL NOT IN (...) is never true — where plain LEFT_ANTI emits
            // them as unmatched
            row_counts[i] = 0;
            continue;
        }
        row_counts[i] = join_mode_count(c, mode == 6 ? 2 : mode);
    }
}

__global__ void k_probe_emit_bc_nulls(const uint32_t* __restrict__ probe_keys,
                                      const uint8_t* __restrict__ is_nulls, uint64_t n,
                                      uint32_t log_bucket_size,
                                      const uint32_t* __restrict__ first,
                                      const uint32_t* __restrict__ next,
                                      const uint32_t* __restrict__ build_keys, int mode,
                                      const uint32_t* __restrict__ row_counts,
                                      const uint64_t* __restrict__ row_offsets,
                                      uint32_t* __restrict__ out_probe,
                                      uint32_t* __restrict__ out_build) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    if (mode == 6) mode = 2; // null-aware anti: null rows already have count 0
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        if (row_counts[i] == 0) continue;
        uint64_t pos = row_offsets[i];
        bool any = false;
        if (!is_nulls[i]) {
            uint32_t k = probe_keys[i];
            uint32_t b = first[join_hash_u32(k, log_bucket_size)];
            while (b != 0) {
                if (build_keys[b] == k) {
                    any = true;
                    if (mode != 2) {
                        out_probe[pos] = (uint32_t)i;
                        out_build[pos] = b;
                        pos++;
                    }
                    if (mode == 1 || mode == 2) break;
                }
                b = next[b];
            }
        }
        if (!any && (mode == 2 || mode == 3)) {
            out_probe[pos] = (uint32_t)i;
            out_build[pos] = 0;
        }
    }
}

extern "C" int gpue_join_probe_emit_nulls_i32(gpue_session* s, gpue_join_table* t,
                                              gpue_dbuf* probe_keys, gpue_dbuf* probe_nulls,
                                              uint64_t n_rows, int mode,
                                              gpue_dbuf* out_probe_idx,
                                              gpue_dbuf* out_build_idx, uint64_t* match_count);
int gpue_join_probe_emit_nulls_i32(gpue_session* s, gpue_join_table* t, gpue_dbuf* probe_keys,
                                   gpue_dbuf* probe_nulls, uint64_t n_rows, int mode,
                                   gpue_dbuf* out_probe_idx, gpue_dbuf* out_build_idx,
                                   uint64_t* match_count) {
    ARG_CHECK(s && t && probe_keys && probe_nulls && match_count);
    ARG_CHECK(t->kind == gpue_join_table::BUCKET_CHAINED && t->next != nullptr);
    ARG_CHECK(probe_nulls->bytes >= n_rows);
    uint32_t nb = grid_for(n_rows);
    uint64_t tile = (n_rows + nb - 1) / nb;
    uint32_t* d_counts = nullptr;
    uint64_t* d_bsums = nullptr;
    uint64_t* d_offsets = nullptr;
    HIP_CHECK(hipMalloc(&d_counts, n_rows * sizeof(uint32_t)));
    HIP_CHECK(hipMalloc(&d_bsums, (nb + 1) * sizeof(uint64_t)));
    hipLaunchKernelGGL(k_probe_count_bc_nulls, dim3(nb), dim3(BLOCK), 0, s->stream,
                       (const uint32_t*)probe_keys->ptr, (const uint8_t*)probe_nulls->ptr,
                       n_rows, t->log_bucket_size, t->first, t->next, t->build_keys, mode,
                       d_counts);
    hipLaunchKernelGGL(k_block_sums_u32, dim3(nb), dim3(BLOCK), 0, s->stream, d_counts,
                       n_rows, tile, d_bsums);
    hipLaunchKernelGGL(k_scan_small, dim3(1), dim3(1), 0, s->stream, d_bsums, nb);
    uint64_t total = 0;
    HIP_CHECK(hipMemcpyAsync(&total, d_bsums + nb, 8, hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    *match_count = total;
    if (out_probe_idx && out_build_idx && total > 0) {
        ARG_CHECK(out_probe_idx->bytes >= total * 4 && out_build_idx->bytes >= total * 4);
        HIP_CHECK(hipMalloc(&d_offsets, n_rows * sizeof(uint64_t)));
        hipLaunchKernelGGL(k_scan_offsets, dim3(nb), dim3(BLOCK), 0, s->stream, d_counts,
                           n_rows, tile, d_bsums, d_offsets);
        hipLaunchKernelGGL(k_probe_emit_bc_nulls, dim3(nb), dim3(BLOCK), 0, s->stream,
                           (const uint32_t*)probe_keys->ptr,
                           (const uint8_t*)probe_nulls->ptr, n_rows, t->log_bucket_size,
                           t->first, t->next, t->build_keys, mode, d_counts, d_offsets,
                           (uint32_t*)out_probe_idx->ptr, (uint32_t*)out_build_idx->ptr);
        HIP_CHECK(hipStreamSynchronize(s->stream));
        (void)hipFree(d_offsets);
    }
    (void)hipFree(d_counts);
    (void)hipFree(d_bsums);
    return GPUE_OK;
}

// RIGHT SEMI/ANTI (join_hash_map.hpp _probe_from_ht_for_right_*): the probe
// pass marks matched BUILD rows; output = matched (semi) / unmatched (anti)
// build rows.
__global__ void k_right_mark_rd(const int32_t* __restrict__ probe_keys, uint64_t n,
                                int64_t mn, int64_t mx, const uint32_t* __restrict__ first,
                                const uint32_t* __restrict__ next,
                                uint32_t* __restrict__ matched_bits) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        int32_t k = probe_keys[i];
        if (k < mn || k > mx) continue;
        uint32_t b = first[k - mn];
        while (b != 0) {
            atomicOr(&matched_bits[b >> 5], 1u << (b & 31));
            b = next[b];
        }
    }
}

// DENSE_RANGE_DIRECT right-mark: rank/select head + chain walk, no compare
__global__ void k_right_mark_dense(const int32_t* __restrict__ probe_keys, uint64_t n,
                                   int64_t mn, int64_t mx,
                                   const uint2* __restrict__ groups,
                                   const uint32_t* __restrict__ first,
                                   const uint32_t* __restrict__ next,
                                   uint32_t* __restrict__ matched_bits) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        uint32_t b = dense_head(probe_keys, i, mn, mx, groups, first);
        while (b != 0) {
            atomicOr(&matched_bits[b >> 5], 1u << (b & 31));
            b = next[b];
        }
    }
}

__global__ void k_right_mark_keys(const uint32_t* __restrict__ probe_keys, uint64_t n,
                                  int is_linear, uint32_t log_bucket_size,
                                  uint32_t bucket_mask, const uint32_t* __restrict__ first,
                                  const uint32_t* __restrict__ next,
                                  const uint32_t* __restrict__ build_keys,
                                  uint32_t* __restrict__ matched_bits) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        uint32_t k = probe_keys[i];
        uint32_t b = is_linear
                         ? lc_lookup_head(k, log_bucket_size, bucket_mask, first, build_keys)
                         : first[join_hash_u32(k, log_bucket_size)];
        while (b != 0) {
            if (is_linear || build_keys[b] == k)
                atomicOr(&matched_bits[b >> 5], 1u << (b & 31));
            b = next[b];
        }
    }
}

__global__ void k_right_row_counts(const uint32_t* __restrict__ matched_bits,
                                   uint64_t build_rows, int anti,
                                   uint32_t* __restrict__ row_counts) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t j = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; j < build_rows;
         j += stride) {
        uint32_t b = (uint32_t)j + 1; // build rows are 1-based
        uint32_t m = (matched_bits[b >> 5] >> (b & 31)) & 1u;
        row_counts[j] = anti ? (1u - m) : m;
    }
}

__global__ void k_right_emit(const uint32_t* __restrict__ row_counts,
                             const uint64_t* __restrict__ row_offsets, uint64_t build_rows,
                             uint32_t* __restrict__ out_build) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t j = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; j < build_rows;
         j += stride)
        if (row_counts[j]) out_build[row_offsets[j]] = (uint32_t)j + 1;
}

extern "C" int gpue_join_probe_right_i32(gpue_session* s, gpue_join_table* t,
                                         gpue_dbuf* probe_keys, uint64_t n_rows, int anti,
                                         gpue_dbuf* out_build_idx, uint64_t* count);
int gpue_join_probe_right_i32(gpue_session* s, gpue_join_table* t, gpue_dbuf* probe_keys,
                              uint64_t n_rows, int anti, gpue_dbuf* out_build_idx,
                              uint64_t* count) {
    ARG_CHECK(s && t && probe_keys && count && t->next != nullptr);
    uint64_t nwords = (t->row_count + 1 + 31) / 32;
    uint32_t* d_bits = nullptr;
    uint32_t* d_counts = nullptr;
    uint64_t* d_bsums = nullptr;
    uint64_t* d_offsets = nullptr;
    HIP_CHECK(hipMalloc(&d_bits, nwords * 4));
    HIP_CHECK(hipMemsetAsync(d_bits, 0, nwords * 4, s->stream));
    if (t->kind == gpue_join_table::RANGE_DIRECT) {
        hipLaunchKernelGGL(k_right_mark_rd, dim3(grid_for(n_rows)), dim3(BLOCK), 0, s->stream,
                           (const int32_t*)probe_keys->ptr, n_rows, t->min_key, t->max_key,
                           t->first, t->next, d_bits);
    } else if (t->kind == gpue_join_table::DENSE_RANGE_DIRECT) {
        hipLaunchKernelGGL(k_right_mark_dense, dim3(grid_for(n_rows)), dim3(BLOCK), 0,
                           s->stream, (const int32_t*)probe_keys->ptr, n_rows, t->min_key,
                           t->max_key, t->dense_groups, t->first, t->next, d_bits);
    } else {
        hipLaunchKernelGGL(k_right_mark_keys, dim3(grid_for(n_rows)), dim3(BLOCK), 0,
                           s->stream, (const uint32_t*)probe_keys->ptr, n_rows,
                           t->kind == gpue_join_table::LINEAR_CHAINED ? 1 : 0,
                           t->log_bucket_size, (uint32_t)(t->bucket_size - 1), t->first,
                           t->next, t->build_keys, d_bits);
    }
    uint64_t br = t->row_count;
    uint32_t nb = grid_for(br);
    uint64_t tile = (br + nb - 1) / nb;
    HIP_CHECK(hipMalloc(&d_counts, br * 4));
    HIP_CHECK(hipMalloc(&d_bsums, (nb + 1) * 8));
    hipLaunchKernelGGL(k_right_row_counts, dim3(nb), dim3(BLOCK), 0, s->stream, d_bits, br,
                       anti, d_counts);
    hipLaunchKernelGGL(k_block_sums_u32, dim3(nb), dim3(BLOCK), 0, s->stream, d_counts, br,
                       tile, d_bsums);
    hipLaunchKernelGGL(k_scan_small, dim3(1), dim3(1), 0, s->stream, d_bsums, nb);
    uint64_t total = 0;
    HIP_CHECK(hipMemcpyAsync(&total, d_bsums + nb, 8, hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    *count = total;
    if (out_build_idx && total > 0) {
        ARG_CHECK(out_build_idx->bytes >= total * 4);
        HIP_CHECK(hipMalloc(&d_offsets, br * 8));
        hipLaunchKernelGGL(k_scan_offsets, dim3(nb), dim3(BLOCK), 0, s->stream, d_counts, br,
                           tile, d_bsums, d_offsets);
        hipLaunchKernelGGL(k_right_emit, dim3(nb), dim3(BLOCK), 0, s->stream, d_counts,
                           d_offsets, br, (uint32_t*)out_build_idx->ptr);
        HIP_CHECK(hipStreamSynchronize(s->stream));
        (void)hipFree(d_offsets);
    }
    (void)hipFree(d_bits);
    (void)hipFree(d_counts);
    (void)hipFree(d_bsums);
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// fused probe + aggregate pipelines
// ---------------------------------------------------------------------------
// Semi-join probe from an LDS-resident bitset: the u32-payload gather version
// below measured issue-bound at ~2.0 TB/s (PMC traffic == algorithmic bytes,
// profiles/pmc_q1*, so not bandwidth) — 64 random lanes per wave serialize in
// the vector-memory path. The date dim needs only membership (SSB Q1 is a
// semi join — the reference's RANGE_DIRECT_MAPPING_SET, 1 bit/key,
// join_hash_table.cpp:297-303), and its bitset (61130 bits = 7.6 KB) stages
// in LDS once per block: probes become ds_read, off the TA path entirely.
__global__ void k_q1_join_sum_bitset(const int32_t* __restrict__ od,
                                     const int32_t* __restrict__ ep,
                                     const int32_t* __restrict__ dc, uint64_t n,
                                     int64_t mn, int64_t mx,
                                     const uint32_t* __restrict__ gbits,
                                     unsigned long long* __restrict__ sum_out,
                                     unsigned long long* __restrict__ cnt_out) {
    extern __shared__ uint32_t lbits[];
    const uint32_t interval = (uint32_t)(mx - mn + 1);
    const uint32_t nwords = (interval + 31) / 32;
    for (uint32_t w = threadIdx.x; w < nwords; w += blockDim.x) lbits[w] = gbits[w];
    __syncthreads();
    int64_t sum = 0;
    uint64_t cnt = 0;
    const uint64_t n4 = n / 4;
    const int4* __restrict__ od4 = (const int4*)od;
    const int4* __restrict__ ep4 = (const int4*)ep;
    const int4* __restrict__ dc4 = (const int4*)dc;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    // Two independent quads in flight per iteration: without this the next
    // iteration's loads only issue after the current LDS-probe/accumulate
    // tail (loop-carried serialization) — the 3-stream ubench ceiling needs
    // ≥6 outstanding loads per wave. Probe phase batches all 8 ds_reads; a
    // ?:/if around a load would emit per-element branches + vmcnt(0) waits
    // (guide §5 item 4(c); verified in the .s of the branchy form).
    auto quad = [&](int4 k4, int4 e4, int4 d4) {
        uint32_t idx[4], w[4];
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            idx[j] = (uint32_t)((&k4.x)[j] - (int32_t)mn);
            uint32_t cidx = idx[j] < interval ? idx[j] : 0u;
            w[j] = lbits[cidx >> 5] >> (cidx & 31);
        }
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            int64_t pass = (int64_t)((idx[j] < interval) & (w[j] & 1u));
            sum += ((int64_t)(&e4.x)[j] * (&d4.x)[j]) & -pass;
            cnt += (uint64_t)pass;
        }
    };
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    for (; i + stride < n4; i += 2 * stride) {
        int4 ka = od4[i], ea = ep4[i], da = dc4[i];
        uint64_t i2 = i + stride;
        int4 kb = od4[i2], eb = ep4[i2], db = dc4[i2];
        quad(ka, ea, da);
        quad(kb, eb, db);
    }
    for (; i < n4; i += stride) quad(od4[i], ep4[i], dc4[i]);
    uint64_t tid = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    for (uint64_t i = n4 * 4 + tid; i < n; i += stride) {
        uint32_t idx = (uint32_t)(od[i] - (int32_t)mn);
        if (idx < interval && ((lbits[idx >> 5] >> (idx & 31)) & 1u)) {
            sum += (int64_t)ep[i] * dc[i];
            cnt++;
        }
    }
    for (int off = WAVE / 2; off > 0; off >>= 1) {
        sum += __shfl_down((long long)sum, off, WAVE);
        cnt += __shfl_down((unsigned long long)cnt, off, WAVE);
    }
    if ((threadIdx.x & (WAVE - 1)) == 0) {
        atomicAdd(sum_out, (unsigned long long)sum);
        atomicAdd(cnt_out, (unsigned long long)cnt);
    }
}

// 16 B/lane vectorized (Guideline 13: 4×int32 per column per iteration) with
// UNCONDITIONAL column loads — a load guarded by the probe branch serializes
// behind per-element vmcnt waits (guide §5 item 4(c)); the probe gather is
// made branchless by clamping the index into the (L2-resident) dim array.
__global__ void k_q1_join_sum(const int32_t* __restrict__ od, const int32_t* __restrict__ ep,
                              const int32_t* __restrict__ dc, uint64_t n,
                              int64_t mn, int64_t mx, const uint32_t* __restrict__ dfirst,
                              unsigned long long* __restrict__ sum_out,
                              unsigned long long* __restrict__ cnt_out) {
    int64_t sum = 0;
    uint64_t cnt = 0;
    const uint64_t n4 = n / 4;
    const uint32_t interval = (uint32_t)(mx - mn + 1);
    const int4* __restrict__ od4 = (const int4*)od;
    const int4* __restrict__ ep4 = (const int4*)ep;
    const int4* __restrict__ dc4 = (const int4*)dc;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
        int4 k4 = od4[i];
        int4 e4 = ep4[i];
        int4 d4 = dc4[i];
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            int32_t k = (&k4.x)[j];
            uint32_t idx = (uint32_t)(k - (int32_t)mn);
            uint32_t cidx = idx < interval ? idx : 0;
            bool pass = (idx < interval) & (dfirst[cidx] != 0);
            sum += pass ? (int64_t)(&e4.x)[j] * (&d4.x)[j] : 0;
            cnt += pass;
        }
    }
    // scalar tail (n % 4 rows), first few threads of the grid
    uint64_t tid = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    for (uint64_t i = n4 * 4 + tid; i < n; i += stride) {
        int32_t k = od[i];
        uint32_t idx = (uint32_t)(k - (int32_t)mn);
        if (idx < interval && dfirst[idx] != 0) {
            sum += (int64_t)ep[i] * dc[i];
            cnt++;
        }
    }
    for (int off = WAVE / 2; off > 0; off >>= 1) {
        sum += __shfl_down((long long)sum, off, WAVE);
        cnt += __shfl_down((unsigned long long)cnt, off, WAVE);
    }
    if ((threadIdx.x & (WAVE - 1)) == 0) {
        atomicAdd(sum_out, (unsigned long long)sum); // mod-2^64 add == signed add
        atomicAdd(cnt_out, (unsigned long long)cnt);
    }
}


// choose the LDS-bitset probe when the dim bitset fits comfortably in LDS
// (<= 32 KiB leaves >= 4 blocks/CU); fall back to the payload-gather kernel
static void launch_q1(gpue_session* s, gpue_join_table* dates, const int32_t* od,
                      const int32_t* ep, const int32_t* dc, uint64_t n,
                      unsigned long long* sum_out, unsigned long long* cnt_out) {
    uint64_t set_interval = (uint64_t)(dates->set_max - dates->set_min + 1);
    uint64_t nwords = (set_interval + 31) / 32;
    if (dates->bitset && dates->set_max >= dates->set_min && nwords * 4 <= 32768) {
        hipLaunchKernelGGL(k_q1_join_sum_bitset, dim3(grid_stream(n)), dim3(BLOCK),
                           nwords * 4, s->stream, od, ep, dc, n, dates->set_min,
                           dates->set_max, dates->bitset, sum_out, cnt_out);
    } else {
        hipLaunchKernelGGL(k_q1_join_sum, dim3(grid_stream(n)), dim3(BLOCK), 0, s->stream,
                           od, ep, dc, n, dates->min_key, dates->max_key, dates->first,
                           sum_out, cnt_out);
    }
}

// Accumulating form: NO per-step memset — the caller keeps a persistent acc
// and differences successive readbacks on the host. Removes one launch from
// the 2-op step (the SF10 line is launch-overhead-sensitive, §4b).
extern "C" int gpue_q1_join_sum_accum(gpue_session* s, gpue_join_table* dates, gpue_dbuf* od,
                                      gpue_dbuf* ep, gpue_dbuf* dc, uint64_t n,
                                      gpue_dbuf* acc);
int gpue_q1_join_sum_accum(gpue_session* s, gpue_join_table* dates, gpue_dbuf* od,
                           gpue_dbuf* ep, gpue_dbuf* dc, uint64_t n, gpue_dbuf* acc) {
    ARG_CHECK(s && dates && od && ep && dc && acc && acc->bytes >= 16);
    launch_q1(s, dates, (const int32_t*)od->ptr, (const int32_t*)ep->ptr,
              (const int32_t*)dc->ptr, n, (unsigned long long*)acc->ptr,
              (unsigned long long*)acc->ptr + 1);
    HIP_CHECK(hipGetLastError());
    return GPUE_OK;
}

int gpue_q1_join_sum(gpue_session* s, gpue_join_table* dates, gpue_dbuf* od, gpue_dbuf* ep,
                     gpue_dbuf* dc, uint64_t n, int64_t* sum_out, uint64_t* match_count_out) {
    ARG_CHECK(s && dates && od && ep && dc && sum_out && match_count_out);
    ARG_CHECK(od->bytes >= n * 4 && ep->bytes >= n * 4 && dc->bytes >= n * 4);
    unsigned long long h_acc[2] = {0, 0};
    unsigned long long* d_acc = nullptr;
    HIP_CHECK(hipMalloc(&d_acc, 2 * sizeof(unsigned long long)));
    HIP_CHECK(hipMemcpyAsync(d_acc, h_acc, sizeof(h_acc), hipMemcpyHostToDevice, s->stream));
    launch_q1(s, dates, (const int32_t*)od->ptr, (const int32_t*)ep->ptr,
              (const int32_t*)dc->ptr, n, d_acc, d_acc + 1);
    HIP_CHECK(hipMemcpyAsync(h_acc, d_acc, sizeof(h_acc), hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    (void)hipFree(d_acc);
    *sum_out = (int64_t)h_acc[0];
    *match_count_out = h_acc[1];
    return GPUE_OK;
}

// Async variant for the bench hot loop: zero + launch into a caller-owned
// 16-byte accumulator dbuf ({sum, count}), no sync, no transient allocation.
extern "C" int gpue_q1_join_sum_async(gpue_session* s, gpue_join_table* dates, gpue_dbuf* od,
                                      gpue_dbuf* ep, gpue_dbuf* dc, uint64_t n, gpue_dbuf* acc);
int gpue_q1_join_sum_async(gpue_session* s, gpue_join_table* dates, gpue_dbuf* od,
                           gpue_dbuf* ep, gpue_dbuf* dc, uint64_t n, gpue_dbuf* acc) {
    ARG_CHECK(s && dates && od && ep && dc && acc && acc->bytes >= 16);
    HIP_CHECK(hipMemsetAsync(acc->ptr, 0, 16, s->stream));
    launch_q1(s, dates, (const int32_t*)od->ptr, (const int32_t*)ep->ptr,
              (const int32_t*)dc->ptr, n, (unsigned long long*)acc->ptr,
              (unsigned long long*)acc->ptr + 1);
    HIP_CHECK(hipGetLastError());
    return GPUE_OK;
}

static constexpr int NG_Q21 = 7 * 1000; // (d_year-1992) in [0,7) × p_brand in [0,1000)

// 1024-thread blocks: the 56 KB LDS group array would cap a 256-thread block
// at 2 blocks/CU = 8 waves/CU — far too few to hide the random pfirst/sfirst
// gather latency. 1024 threads × 2 blocks/CU = 32 waves/CU (full) at 112 KB
// LDS/CU (< 160 KB). Columns are read 16 B/lane (int4, Guideline 13) with
// unconditional loads; only the dependent dim gathers stay branchy (they
// carry the 1/25 × 1/5 selectivity, so skipping them saves real traffic).
static constexpr int BLOCK_Q21 = 1024;

__global__ __launch_bounds__(BLOCK_Q21) void
k_q21_star_agg(const int32_t* __restrict__ pk, const int32_t* __restrict__ sk,
               const int32_t* __restrict__ od, const int32_t* __restrict__ rv,
               uint64_t n, const uint32_t* __restrict__ pbits, int64_t psmin,
               uint64_t psint, const uint16_t* __restrict__ pfirst,
               const uint32_t* __restrict__ sbits, int64_t ssmin, uint64_t ssint,
               const uint16_t* __restrict__ dfirst, int64_t dmin,
               unsigned long long* __restrict__ group_sums) {
    __shared__ unsigned long long g[NG_Q21]; // 56 KB LDS
    for (int j = threadIdx.x; j < NG_Q21; j += blockDim.x) g[j] = 0;
    __syncthreads();
    const uint64_t n4 = n / 4;
    const int4* __restrict__ pk4 = (const int4*)pk;
    const int4* __restrict__ sk4 = (const int4*)sk;
    const int4* __restrict__ od4 = (const int4*)od;
    const int4* __restrict__ rv4 = (const int4*)rv;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    // SQ counters (profiles/pmc_sq_q21): 64 % of wave cycles parked on memory
    // waits at full occupancy — latency-bound. Two independent quads per
    // iteration double the loads in flight per wave.
    auto quad = [&](int4 p4, int4 s4, int4 o4, int4 r4) {
        // phase 1: runtime-filter probe — the part table's 1-bit membership
        // set (175 KB, L2-resident; the reference pushes exactly this filter
        // to the scan, runtime_filter.h:79 / SURVEY.md §8f) replaces a 2.8 MB
        // payload gather for the 96 % of rows the category filter rejects.
        // All four word-gathers issue back-to-back (guide §5 item 4(c)).
        uint32_t pb[4], pin[4];
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            uint32_t idx = (uint32_t)((&p4.x)[j] - psmin);
            pin[j] = idx < psint;
            uint32_t cidx = pin[j] ? idx : 0u;
            pb[j] = pbits[cidx >> 5] >> (cidx & 31);
        }
        // phase 2: survivors (≈4 %) gather the brand payload and probe the
        // supplier bitset (25 KB) and date payload — branching here skips
        // real traffic
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            if (!(pin[j] & pb[j] & 1u)) continue;
            uint32_t brand1 = pfirst[(&p4.x)[j] - 1];
            uint32_t sidx = (uint32_t)((&s4.x)[j] - ssmin);
            if (sidx >= ssint || !((sbits[sidx >> 5] >> (sidx & 31)) & 1u)) continue;
            uint32_t year1 = dfirst[(&o4.x)[j] - dmin];
            atomicAdd(&g[(year1 - 1) * 1000 + (brand1 - 1)],
                      (unsigned long long)(int64_t)(&r4.x)[j]);
        }
    };
    // (a 2-deep pk prefetch pipeline measured 9 % SLOWER here — the extra
    // registers/branches cost more than the overlapped latency buys;
    // q43-style deferred stream loads behind the part mask measured 2565 vs
    // 2792 GB/s — q21 is LATENCY-bound, so shortening the issue window by
    // loading all four streams up front beats saving sk/od/rv lines)
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    for (; i + stride < n4; i += 2 * stride) {
        int4 pa = pk4[i], sa = sk4[i], oa = od4[i], ra = rv4[i];
        uint64_t i2 = i + stride;
        int4 pb_ = pk4[i2], sb = sk4[i2], ob = od4[i2], rb = rv4[i2];
        quad(pa, sa, oa, ra);
        quad(pb_, sb, ob, rb);
    }
    for (; i < n4; i += stride) quad(pk4[i], sk4[i], od4[i], rv4[i]);
    uint64_t tid = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    for (uint64_t i = n4 * 4 + tid; i < n; i += stride) {
        uint32_t pidx = (uint32_t)(pk[i] - psmin);
        if (pidx >= psint || !((pbits[pidx >> 5] >> (pidx & 31)) & 1u)) continue;
        uint32_t brand1 = pfirst[pk[i] - 1];
        uint32_t sidx = (uint32_t)(sk[i] - ssmin);
        if (sidx >= ssint || !((sbits[sidx >> 5] >> (sidx & 31)) & 1u)) continue;
        uint32_t year1 = dfirst[od[i] - dmin];
        atomicAdd(&g[(year1 - 1) * 1000 + (brand1 - 1)], (unsigned long long)(int64_t)rv[i]);
    }
    __syncthreads();
    for (int j = threadIdx.x; j < NG_Q21; j += blockDim.x)
        if (g[j] != 0) atomicAdd(&group_sums[j], g[j]);
}

// Experiment variant (GPUE_Q21_GLOB=1): no LDS group array — passing rows
// (~0.8 %) atomicAdd straight into the 56 KB global/L2-resident group
// buffer, letting the kernel run at streaming geometry (grid_stream, 256-
// thread blocks, no LDS occupancy cap).
__global__ void k_q21_star_agg_glob(const int32_t* __restrict__ pk,
                                    const int32_t* __restrict__ sk,
                                    const int32_t* __restrict__ od,
                                    const int32_t* __restrict__ rv, uint64_t n,
                                    const uint32_t* __restrict__ pbits, int64_t psmin,
                                    uint64_t psint, const uint16_t* __restrict__ pfirst,
                                    const uint32_t* __restrict__ sbits, int64_t ssmin,
                                    uint64_t ssint, const uint16_t* __restrict__ dfirst,
                                    int64_t dmin,
                                    unsigned long long* __restrict__ group_sums) {
    const uint64_t n4 = n / 4;
    const int4* __restrict__ pk4 = (const int4*)pk;
    const int4* __restrict__ sk4 = (const int4*)sk;
    const int4* __restrict__ od4 = (const int4*)od;
    const int4* __restrict__ rv4 = (const int4*)rv;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    auto quad = [&](int4 p4, int4 s4, int4 o4, int4 r4) {
        uint32_t pb[4], pin[4];
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            uint32_t idx = (uint32_t)((&p4.x)[j] - psmin);
            pin[j] = idx < psint;
            uint32_t cidx = pin[j] ? idx : 0u;
            pb[j] = pbits[cidx >> 5] >> (cidx & 31);
        }
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            if (!(pin[j] & pb[j] & 1u)) continue;
            uint32_t brand1 = pfirst[(&p4.x)[j] - 1];
            uint32_t sidx = (uint32_t)((&s4.x)[j] - ssmin);
            if (sidx >= ssint || !((sbits[sidx >> 5] >> (sidx & 31)) & 1u)) continue;
            uint32_t year1 = dfirst[(&o4.x)[j] - dmin];
            atomicAdd(&group_sums[(year1 - 1) * 1000 + (brand1 - 1)],
                      (unsigned long long)(int64_t)(&r4.x)[j]);
        }
    };
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    for (; i + stride < n4; i += 2 * stride) {
        int4 pa = pk4[i], sa = sk4[i], oa = od4[i], ra = rv4[i];
        uint64_t i2 = i + stride;
        int4 pb_ = pk4[i2], sb = sk4[i2], ob = od4[i2], rb = rv4[i2];
        quad(pa, sa, oa, ra);
        quad(pb_, sb, ob, rb);
    }
    for (; i < n4; i += stride) quad(pk4[i], sk4[i], od4[i], rv4[i]);
    uint64_t tid = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    for (uint64_t r = n4 * 4 + tid; r < n; r += stride) {
        uint32_t pidx = (uint32_t)(pk[r] - psmin);
        if (pidx >= psint || !((pbits[pidx >> 5] >> (pidx & 31)) & 1u)) continue;
        uint32_t brand1 = pfirst[pk[r] - 1];
        uint32_t sidx = (uint32_t)(sk[r] - ssmin);
        if (sidx >= ssint || !((sbits[sidx >> 5] >> (sidx & 31)) & 1u)) continue;
        uint32_t year1 = dfirst[od[r] - dmin];
        atomicAdd(&group_sums[(year1 - 1) * 1000 + (brand1 - 1)],
                  (unsigned long long)(int64_t)rv[r]);
    }
}

// ---------------------------------------------------------------------------
// LDS-prefilter q21 (round-2 item 3 — the stream/gather overlap lever).
// Measured state of the fused kernel: the 4-column stream leg
// (1.57 ms, HBM-bound) and the per-row part-bitset L2 gather leg (2.32 ms)
// are fully ADDITIVE, and TCC counters put the gathers at an 87% L2 hit rate
// — the leg is bound by the TA's DIVERGENT-ADDRESS request rate (~1
// lane/cycle), not by misses. The fix is to answer the per-row part filter
// from LDS, whose random-access rate is bank-parallel (~20x the TA's rate
// for divergent words):
//   - the build side folds the part bitset (175 KB over the passing-key
//     range) into a 2^19-bit = 64 KB PREFILTER by index masking; a clear bit
//     is definitely-not-in-category, a set bit is "maybe" (fold factor ~2.7
//     keys/bit at SF100 => ~10% maybes at a 4% true rate). This is the same
//     conservative-filter idea as the reference's SimdBlockFilter runtime
//     filter (runtime_filter.h:79), applied to the scan's dim filter.
//   - each block stages the prefilter in LDS (64 KB) next to the 56 KB group
//     array (120 KB/block, 1 block/CU, 16 waves — streams saturate HBM from
//     4 waves/CU, cf. the q1 kernel) and tests all four quad keys with LDS
//     reads. Only "maybe" rows (~10%) gather the u16 brand payload from L2,
//     whose zero value encodes the exact category filter (payload tables:
//     first[key-min] = brand+1 | 0) — so the exact 175 KB
//     bitset is never probed at all and the TA leg shrinks ~10x.
// Parity: identical emissions — prefilter false => bitset false => brand 0.
// ---------------------------------------------------------------------------
static constexpr int PF_LOG2 = 19;                      // 2^19 bits = 64 KB
static constexpr uint32_t PF_WORDS = (1u << PF_LOG2) / 32;
static constexpr uint32_t PF_MASK = (1u << PF_LOG2) - 1;

// LDSG: group sums in LDS (56 KB, 1 block/CU) vs global atomics (frees the
// LDS for a 2nd block -> 32 waves/CU). NT: non-temporal stream loads (leave
// L2 to the u16 payload the maybe-rows gather).
template <bool LDSG, bool NT>
__global__ __launch_bounds__(BLOCK_Q21) void
k_q21_star_agg_pf(const int32_t* __restrict__ pk, const int32_t* __restrict__ sk,
                  const int32_t* __restrict__ od, const int32_t* __restrict__ rv,
                  uint64_t n, const uint32_t* __restrict__ prefilter, int64_t psmin,
                  uint64_t psint, const uint16_t* __restrict__ pfirst,
                  const uint32_t* __restrict__ sbits, int64_t ssmin, uint64_t ssint,
                  const uint16_t* __restrict__ dfirst, int64_t dmin,
                  unsigned long long* __restrict__ group_sums) {
    __shared__ uint32_t pf[PF_WORDS];        // 64 KB folded part filter
    __shared__ unsigned long long g[LDSG ? NG_Q21 : 1]; // 56 KB group sums
    for (uint32_t w = threadIdx.x; w < PF_WORDS; w += blockDim.x) pf[w] = prefilter[w];
    if (LDSG)
        for (int j = threadIdx.x; j < NG_Q21; j += blockDim.x) g[j] = 0;
    __syncthreads();
    const uint64_t n4 = n / 4;
    const int4* __restrict__ pk4 = (const int4*)pk;
    const int4* __restrict__ sk4 = (const int4*)sk;
    const int4* __restrict__ od4 = (const int4*)od;
    const int4* __restrict__ rv4 = (const int4*)rv;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    auto quad = [&](int4 p4, int4 s4, int4 o4, int4 r4) {
        uint32_t maybe[4];
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            uint32_t idx = (uint32_t)((&p4.x)[j] - psmin);
            bool in = idx < psint;
            uint32_t fidx = (in ? idx : 0u) & PF_MASK;
            maybe[j] = in & (pf[fidx >> 5] >> (fidx & 31)) & 1u;
        }
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            if (!maybe[j]) continue;
            uint32_t brand1 = pfirst[(&p4.x)[j] - 1]; // 0 = fails the category filter
            if (!brand1) continue;
            uint32_t sidx = (uint32_t)((&s4.x)[j] - ssmin);
            if (sidx >= ssint || !((sbits[sidx >> 5] >> (sidx & 31)) & 1u)) continue;
            uint32_t year1 = dfirst[(&o4.x)[j] - dmin];
            unsigned long long* dst = LDSG ? &g[(year1 - 1) * 1000 + (brand1 - 1)]
                                           : &group_sums[(year1 - 1) * 1000 + (brand1 - 1)];
            atomicAdd(dst, (unsigned long long)(int64_t)(&r4.x)[j]);
        }
    };
    auto ld4 = [&](const int4* p, uint64_t i) {
        if (!NT) return p[i];
        // the builtin rejects HIP vector types; two u64 NT loads keep the
        // 16-B dwordx4-equivalent width
        const uint64_t* q = (const uint64_t*)(p + i);
        uint64_t lo = __builtin_nontemporal_load(q);
        uint64_t hi = __builtin_nontemporal_load(q + 1);
        int4 v;
        v.x = (int32_t)lo; v.y = (int32_t)(lo >> 32);
        v.z = (int32_t)hi; v.w = (int32_t)(hi >> 32);
        return v;
    };
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    for (; i + stride < n4; i += 2 * stride) {
        int4 pa = ld4(pk4, i), sa = ld4(sk4, i), oa = ld4(od4, i), ra = ld4(rv4, i);
        uint64_t i2 = i + stride;
        int4 pb_ = ld4(pk4, i2), sb = ld4(sk4, i2), ob = ld4(od4, i2), rb = ld4(rv4, i2);
        quad(pa, sa, oa, ra);
        quad(pb_, sb, ob, rb);
    }
    for (; i < n4; i += stride) quad(pk4[i], sk4[i], od4[i], rv4[i]);
    uint64_t tid = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    for (uint64_t r = n4 * 4 + tid; r < n; r += stride) {
        uint32_t idx = (uint32_t)(pk[r] - psmin);
        if (idx >= psint) continue;
        uint32_t fidx = idx & PF_MASK;
        if (!((pf[fidx >> 5] >> (fidx & 31)) & 1u)) continue;
        uint32_t brand1 = pfirst[pk[r] - 1];
        if (!brand1) continue;
        uint32_t sidx = (uint32_t)(sk[r] - ssmin);
        if (sidx >= ssint || !((sbits[sidx >> 5] >> (sidx & 31)) & 1u)) continue;
        uint32_t year1 = dfirst[od[r] - dmin];
        unsigned long long* dst = LDSG ? &g[(year1 - 1) * 1000 + (brand1 - 1)]
                                       : &group_sums[(year1 - 1) * 1000 + (brand1 - 1)];
        atomicAdd(dst, (unsigned long long)(int64_t)rv[r]);
    }
    if (LDSG) {
        __syncthreads();
        for (int j = threadIdx.x; j < NG_Q21; j += blockDim.x)
            if (g[j] != 0) atomicAdd(&group_sums[j], g[j]);
    }
}

// ---------------------------------------------------------------------------
// Wave-queue variant of the prefilter kernel (GPUE_Q21_PF=4). The
// decomposition (tools/q21_decomp.py r02) shows streams + LDS prefilter
// test run at 1.48 ms — phase 2 alone costs the remaining ~0.8 ms, because
// with a 10% maybe rate nearly EVERY wave executes every sparse branch body
// with ~6 active lanes (wave divergence, not memory). Fix: compact maybes
// into a per-wave LDS queue ({pk,sk,od,rv} int4 entries, ballot+rank push)
// and drain a FULL 64-lane wave of survivors at a time — phase-2 instruction
// episodes drop ~8x and every confirm gather issues with all lanes active.
// 64 KB pf + 56 KB groups + 32 KB queues = 152 KB LDS, 1 block/CU.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(BLOCK_Q21) void
k_q21_star_agg_pfq(const int32_t* __restrict__ pk, const int32_t* __restrict__ sk,
                   const int32_t* __restrict__ od, const int32_t* __restrict__ rv,
                   uint64_t n, const uint32_t* __restrict__ prefilter, int64_t psmin,
                   uint64_t psint, const uint16_t* __restrict__ pfirst,
                   const uint32_t* __restrict__ sbits, int64_t ssmin, uint64_t ssint,
                   const uint16_t* __restrict__ dfirst, int64_t dmin,
                   unsigned long long* __restrict__ group_sums) {
    __shared__ uint32_t pf[PF_WORDS];              // 64 KB folded part filter
    __shared__ unsigned long long g[NG_Q21];       // 56 KB group sums
    __shared__ int4 wq[BLOCK_Q21 / WAVE][128];     // 32 KB per-wave maybe queues
    for (uint32_t w = threadIdx.x; w < PF_WORDS; w += blockDim.x) pf[w] = prefilter[w];
    for (int j = threadIdx.x; j < NG_Q21; j += blockDim.x) g[j] = 0;
    __syncthreads();
    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x & (WAVE - 1);
    uint32_t wqn = 0; // wave-uniform queue fill (ballot counts are uniform)
    const uint64_t n4 = n / 4;
    const int4* __restrict__ pk4 = (const int4*)pk;
    const int4* __restrict__ sk4 = (const int4*)sk;
    const int4* __restrict__ od4 = (const int4*)od;
    const int4* __restrict__ rv4 = (const int4*)rv;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    auto ld4 = [&](const int4* p, uint64_t i) {
        const uint64_t* q = (const uint64_t*)(p + i);
        uint64_t lo = __builtin_nontemporal_load(q);
        uint64_t hi = __builtin_nontemporal_load(q + 1);
        int4 v;
        v.x = (int32_t)lo; v.y = (int32_t)(lo >> 32);
        v.z = (int32_t)hi; v.w = (int32_t)(hi >> 32);
        return v;
    };
    // drain one full wave of queued maybes: every filter executes with all
    // lanes carrying real candidates
    auto drain64 = [&]() {
        int4 e = wq[wid][wqn - 64 + lane];
        uint32_t brand1 = pfirst[e.x - 1]; // 0 = fails the exact category filter
        if (brand1) {
            uint32_t sidx = (uint32_t)(e.y - ssmin);
            if (sidx < ssint && ((sbits[sidx >> 5] >> (sidx & 31)) & 1u)) {
                uint32_t year1 = dfirst[e.z - dmin];
                atomicAdd(&g[(year1 - 1) * 1000 + (brand1 - 1)],
                          (unsigned long long)(int64_t)e.w);
            }
        }
        wqn -= 64;
    };
    // IMPORTANT: every loop condition below is WAVE-UNIFORM (based on the
    // wave's base index, not the lane's) — the queue bookkeeping (wqn) is a
    // wave-uniform register, so a lane-dependent trip count would desync it
    auto quad = [&](int4 p4, int4 s4, int4 o4, int4 r4, bool inb) {
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            uint32_t idx = (uint32_t)((&p4.x)[j] - psmin);
            bool in = inb & (idx < psint);
            uint32_t fidx = (in ? idx : 0u) & PF_MASK;
            bool maybe = in & (pf[fidx >> 5] >> (fidx & 31)) & 1u;
            uint64_t m = __ballot(maybe);
            if (m) {
                uint32_t rank = __popcll(m & ((1ull << lane) - 1));
                if (maybe)
                    wq[wid][wqn + rank] =
                        make_int4((&p4.x)[j], (&s4.x)[j], (&o4.x)[j], (&r4.x)[j]);
                wqn += __popcll(m);
                if (wqn >= 64) drain64();
            }
        }
    };
    uint64_t base = (uint64_t)blockIdx.x * blockDim.x + (uint64_t)wid * WAVE;
    uint64_t i = base + lane;
    // full-wave main loop: both quads of every lane in bounds
    for (; base + stride + WAVE <= n4; base += 2 * stride, i += 2 * stride) {
        int4 pa = ld4(pk4, i), sa = ld4(sk4, i), oa = ld4(od4, i), ra = ld4(rv4, i);
        uint64_t i2 = i + stride;
        int4 pb_ = ld4(pk4, i2), sb = ld4(sk4, i2), ob = ld4(od4, i2), rb = ld4(rv4, i2);
        quad(pa, sa, oa, ra, true);
        quad(pb_, sb, ob, rb, true);
    }
    // wave-uniform predicated remainder (single quads, OOB lanes masked)
    for (; base < n4; base += stride, i += stride) {
        bool inb = i < n4;
        int4 z = make_int4(0, 0, 0, 0);
        int4 p4 = inb ? pk4[i] : z, s4 = inb ? sk4[i] : z;
        int4 o4 = inb ? od4[i] : z, r4 = inb ? rv4[i] : z;
        quad(p4, s4, o4, r4, inb);
    }
    // drain the partial tail (lanes < wqn active)
    if (wqn > 0 && lane < (int)wqn) {
        int4 e = wq[wid][lane];
        uint32_t brand1 = pfirst[e.x - 1];
        if (brand1) {
            uint32_t sidx = (uint32_t)(e.y - ssmin);
            if (sidx < ssint && ((sbits[sidx >> 5] >> (sidx & 31)) & 1u)) {
                uint32_t year1 = dfirst[e.z - dmin];
                atomicAdd(&g[(year1 - 1) * 1000 + (brand1 - 1)],
                          (unsigned long long)(int64_t)e.w);
            }
        }
    }
    // scalar row tail (n % 4): exact path, no queue
    uint64_t tid = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    for (uint64_t r = n4 * 4 + tid; r < n; r += stride) {
        uint32_t idx = (uint32_t)(pk[r] - psmin);
        if (idx >= psint) continue;
        uint32_t fidx = idx & PF_MASK;
        if (!((pf[fidx >> 5] >> (fidx & 31)) & 1u)) continue;
        uint32_t brand1 = pfirst[pk[r] - 1];
        if (!brand1) continue;
        uint32_t sidx = (uint32_t)(sk[r] - ssmin);
        if (sidx >= ssint || !((sbits[sidx >> 5] >> (sidx & 31)) & 1u)) continue;
        uint32_t year1 = dfirst[od[r] - dmin];
        atomicAdd(&g[(year1 - 1) * 1000 + (brand1 - 1)], (unsigned long long)(int64_t)rv[r]);
    }
    __syncthreads();
    for (int j = threadIdx.x; j < NG_Q21; j += blockDim.x)
        if (g[j] != 0) atomicAdd(&group_sums[j], g[j]);
}

// ---------------------------------------------------------------------------
// Split two-probe prefilter variant (GPUE_Q21_PF=6, experiment): bloom-k=2
// over two 2^18-bit folds — predicted maybe rate 10.1% -> ~7.7% (25% fewer
// drains) at the cost of a second LDS read + hash per streamed value.
// Same wave-queue structure as k_q21_star_agg_pfq; `prefilter` here is the
// table's prefilter2 (split layout).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(BLOCK_Q21) void
k_q21_star_agg_pfq2(const int32_t* __restrict__ pk, const int32_t* __restrict__ sk,
                   const int32_t* __restrict__ od, const int32_t* __restrict__ rv,
                   uint64_t n, const uint32_t* __restrict__ prefilter, int64_t psmin,
                   uint64_t psint, const uint16_t* __restrict__ pfirst,
                   const uint32_t* __restrict__ sbits, int64_t ssmin, uint64_t ssint,
                   const uint16_t* __restrict__ dfirst, int64_t dmin,
                   unsigned long long* __restrict__ group_sums) {
    __shared__ uint32_t pfa[1 << 13];              // 32 KB fold A (i & M18)
    __shared__ uint32_t pfb[1 << 13];              // 32 KB fold B (hash18(i))
    __shared__ unsigned long long g[NG_Q21];       // 56 KB group sums
    __shared__ int4 wq[BLOCK_Q21 / WAVE][128];     // 32 KB per-wave maybe queues
    for (uint32_t w = threadIdx.x; w < (1u << 13); w += blockDim.x) {
        pfa[w] = prefilter[w];
        pfb[w] = prefilter[(1u << 13) + w];
    }
    for (int j = threadIdx.x; j < NG_Q21; j += blockDim.x) g[j] = 0;
    __syncthreads();
    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x & (WAVE - 1);
    uint32_t wqn = 0; // wave-uniform queue fill (ballot counts are uniform)
    const uint64_t n4 = n / 4;
    const int4* __restrict__ pk4 = (const int4*)pk;
    const int4* __restrict__ sk4 = (const int4*)sk;
    const int4* __restrict__ od4 = (const int4*)od;
    const int4* __restrict__ rv4 = (const int4*)rv;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    auto ld4 = [&](const int4* p, uint64_t i) {
        const uint64_t* q = (const uint64_t*)(p + i);
        uint64_t lo = __builtin_nontemporal_load(q);
        uint64_t hi = __builtin_nontemporal_load(q + 1);
        int4 v;
        v.x = (int32_t)lo; v.y = (int32_t)(lo >> 32);
        v.z = (int32_t)hi; v.w = (int32_t)(hi >> 32);
        return v;
    };
    // drain one full wave of queued maybes: every filter executes with all
    // lanes carrying real candidates
    auto drain64 = [&]() {
        int4 e = wq[wid][wqn - 64 + lane];
        uint32_t brand1 = pfirst[e.x - 1]; // 0 = fails the exact category filter
        if (brand1) {
            uint32_t sidx = (uint32_t)(e.y - ssmin);
            if (sidx < ssint && ((sbits[sidx >> 5] >> (sidx & 31)) & 1u)) {
                uint32_t year1 = dfirst[e.z - dmin];
                atomicAdd(&g[(year1 - 1) * 1000 + (brand1 - 1)],
                          (unsigned long long)(int64_t)e.w);
            }
        }
        wqn -= 64;
    };
    // IMPORTANT: every loop condition below is WAVE-UNIFORM (based on the
    // wave's base index, not the lane's) — the queue bookkeeping (wqn) is a
    // wave-uniform register, so a lane-dependent trip count would desync it
    auto quad = [&](int4 p4, int4 s4, int4 o4, int4 r4, bool inb) {
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            uint32_t idx = (uint32_t)((&p4.x)[j] - psmin);
            bool in = inb & (idx < psint);
            uint32_t ia = (in ? idx : 0u) & ((1u << 18) - 1);
            uint32_t ib = ((in ? idx : 0u) * 2654435761u) >> 14;
            bool maybe = in & (pfa[ia >> 5] >> (ia & 31)) & (pfb[ib >> 5] >> (ib & 31)) & 1u;
            uint64_t m = __ballot(maybe);
            if (m) {
                uint32_t rank = __popcll(m & ((1ull << lane) - 1));
                if (maybe)
                    wq[wid][wqn + rank] =
                        make_int4((&p4.x)[j], (&s4.x)[j], (&o4.x)[j], (&r4.x)[j]);
                wqn += __popcll(m);
                if (wqn >= 64) drain64();
            }
        }
    };
    uint64_t base = (uint64_t)blockIdx.x * blockDim.x + (uint64_t)wid * WAVE;
    uint64_t i = base + lane;
    // full-wave main loop: both quads of every lane in bounds
    for (; base + stride + WAVE <= n4; base += 2 * stride, i += 2 * stride) {
        int4 pa = ld4(pk4, i), sa = ld4(sk4, i), oa = ld4(od4, i), ra = ld4(rv4, i);
        uint64_t i2 = i + stride;
        int4 pb_ = ld4(pk4, i2), sb = ld4(sk4, i2), ob = ld4(od4, i2), rb = ld4(rv4, i2);
        quad(pa, sa, oa, ra, true);
        quad(pb_, sb, ob, rb, true);
    }
    // wave-uniform predicated remainder (single quads, OOB lanes masked)
    for (; base < n4; base += stride, i += stride) {
        bool inb = i < n4;
        int4 z = make_int4(0, 0, 0, 0);
        int4 p4 = inb ? pk4[i] : z, s4 = inb ? sk4[i] : z;
        int4 o4 = inb ? od4[i] : z, r4 = inb ? rv4[i] : z;
        quad(p4, s4, o4, r4, inb);
    }
    // drain the partial tail (lanes < wqn active)
    if (wqn > 0 && lane < (int)wqn) {
        int4 e = wq[wid][lane];
        uint32_t brand1 = pfirst[e.x - 1];
        if (brand1) {
            uint32_t sidx = (uint32_t)(e.y - ssmin);
            if (sidx < ssint && ((sbits[sidx >> 5] >> (sidx & 31)) & 1u)) {
                uint32_t year1 = dfirst[e.z - dmin];
                atomicAdd(&g[(year1 - 1) * 1000 + (brand1 - 1)],
                          (unsigned long long)(int64_t)e.w);
            }
        }
    }
    // scalar row tail (n % 4): exact path, no queue
    uint64_t tid = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    for (uint64_t r = n4 * 4 + tid; r < n; r += stride) {
        uint32_t idx = (uint32_t)(pk[r] - psmin);
        if (idx >= psint) continue;
        uint32_t ia = idx & ((1u << 18) - 1);
        uint32_t ib = (idx * 2654435761u) >> 14;
        if (!((pfa[ia >> 5] >> (ia & 31)) & (pfb[ib >> 5] >> (ib & 31)) & 1u)) continue;
        uint32_t brand1 = pfirst[pk[r] - 1];
        if (!brand1) continue;
        uint32_t sidx = (uint32_t)(sk[r] - ssmin);
        if (sidx >= ssint || !((sbits[sidx >> 5] >> (sidx & 31)) & 1u)) continue;
        uint32_t year1 = dfirst[od[r] - dmin];
        atomicAdd(&g[(year1 - 1) * 1000 + (brand1 - 1)], (unsigned long long)(int64_t)rv[r]);
    }
    __syncthreads();
    for (int j = threadIdx.x; j < NG_Q21; j += blockDim.x)
        if (g[j] != 0) atomicAdd(&group_sums[j], g[j]);
}


// ---------------------------------------------------------------------------
// Deferred-load wave-queue q21 (GPUE_Q21_PF=7, experiment): the q43 form
// applied to q21 — stream ONLY pk (2.4 GB instead of 9.6 GB), queue {pk,row}
// maybes through the split two-probe prefilter, and let drains load sk (4%
// of rows), then od/rv (~0.8%) on demand. Line-fetch estimate: 2.4 GB pk +
// ~1.15 GB sk lines + ~0.6 GB od/rv lines = ~4.1 GB — if drain issue-rate
// holds like q43's, this beats the full-stream 1.57 ms.
// LDS: 2x32 KB folds + 56 KB groups + 16 KB {pk,row} queues = 136 KB.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(BLOCK_Q21) void
k_q21_star_agg_pfq3(const int32_t* __restrict__ pk, const int32_t* __restrict__ sk,
                    const int32_t* __restrict__ od, const int32_t* __restrict__ rv,
                    uint64_t n, const uint32_t* __restrict__ prefilter, int64_t psmin,
                    uint64_t psint, const uint16_t* __restrict__ pfirst,
                    const uint32_t* __restrict__ sbits, int64_t ssmin, uint64_t ssint,
                    const uint16_t* __restrict__ dfirst, int64_t dmin,
                    unsigned long long* __restrict__ group_sums) {
    __shared__ uint32_t pfa[1 << 13];
    __shared__ uint32_t pfb[1 << 13];
    __shared__ unsigned long long g[NG_Q21];
    __shared__ int2 wq[BLOCK_Q21 / WAVE][128]; // {pk, row32}
    for (uint32_t w = threadIdx.x; w < (1u << 13); w += blockDim.x) {
        pfa[w] = prefilter[w];
        pfb[w] = prefilter[(1u << 13) + w];
    }
    for (int j = threadIdx.x; j < NG_Q21; j += blockDim.x) g[j] = 0;
    __syncthreads();
    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x & (WAVE - 1);
    uint32_t wqn = 0; // wave-uniform (ballot counts); loops are wave-uniform
    const uint64_t n4 = n / 4;
    const int4* __restrict__ pk4 = (const int4*)pk;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    auto process = [&](int2 e) { // confirm cascade with on-demand loads
        uint32_t brand1 = pfirst[e.x - 1]; // exact category filter (0 = fail)
        if (!brand1) return;
        uint32_t r = (uint32_t)e.y;
        uint32_t sidx = (uint32_t)(sk[r] - ssmin);
        if (sidx >= ssint || !((sbits[sidx >> 5] >> (sidx & 31)) & 1u)) return;
        uint32_t year1 = dfirst[od[r] - dmin];
        atomicAdd(&g[(year1 - 1) * 1000 + (brand1 - 1)],
                  (unsigned long long)(int64_t)rv[r]);
    };
    auto push = [&](int32_t key, uint32_t row, bool inb) {
        uint32_t idx = (uint32_t)(key - psmin);
        bool in = inb & (idx < psint);
        uint32_t ia = (in ? idx : 0u) & ((1u << 18) - 1);
        uint32_t ib = ((in ? idx : 0u) * 2654435761u) >> 14;
        bool maybe = in & (pfa[ia >> 5] >> (ia & 31)) & (pfb[ib >> 5] >> (ib & 31)) & 1u;
        uint64_t m = __ballot(maybe);
        if (m) {
            uint32_t rank = __popcll(m & ((1ull << lane) - 1));
            if (maybe) wq[wid][wqn + rank] = make_int2(key, (int32_t)row);
            wqn += __popcll(m);
            if (wqn >= 64) {
                process(wq[wid][wqn - 64 + lane]);
                wqn -= 64;
            }
        }
    };
    auto ld4nt = [&](const int4* p, uint64_t i) {
        const uint64_t* q = (const uint64_t*)(p + i);
        uint64_t lo = __builtin_nontemporal_load(q);
        uint64_t hi = __builtin_nontemporal_load(q + 1);
        int4 v;
        v.x = (int32_t)lo; v.y = (int32_t)(lo >> 32);
        v.z = (int32_t)hi; v.w = (int32_t)(hi >> 32);
        return v;
    };
    uint64_t base = (uint64_t)blockIdx.x * blockDim.x + (uint64_t)wid * WAVE;
    uint64_t i = base + lane;
    for (; base + stride + WAVE <= n4; base += 2 * stride, i += 2 * stride) {
        int4 pa = ld4nt(pk4, i);
        uint64_t i2 = i + stride;
        int4 pb_ = ld4nt(pk4, i2);
        #pragma unroll
        for (int j = 0; j < 4; j++) push((&pa.x)[j], (uint32_t)(i * 4 + j), true);
        #pragma unroll
        for (int j = 0; j < 4; j++) push((&pb_.x)[j], (uint32_t)(i2 * 4 + j), true);
    }
    for (; base < n4; base += stride, i += stride) {
        bool inb = i < n4;
        int4 p4 = inb ? pk4[i] : make_int4(0, 0, 0, 0);
        #pragma unroll
        for (int j = 0; j < 4; j++) push((&p4.x)[j], (uint32_t)(i * 4 + j), inb);
    }
    if (wqn > 0 && lane < (int)wqn) process(wq[wid][lane]);
    // scalar row tail (n % 4): exact path
    uint64_t tid = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    for (uint64_t r = n4 * 4 + tid; r < n; r += stride) {
        uint32_t idx = (uint32_t)(pk[r] - psmin);
        if (idx >= psint) continue;
        uint32_t brand1 = pfirst[pk[r] - 1];
        if (!brand1) continue;
        uint32_t sidx = (uint32_t)(sk[r] - ssmin);
        if (sidx >= ssint || !((sbits[sidx >> 5] >> (sidx & 31)) & 1u)) continue;
        uint32_t year1 = dfirst[od[r] - dmin];
        atomicAdd(&g[(year1 - 1) * 1000 + (brand1 - 1)], (unsigned long long)(int64_t)rv[r]);
    }
    __syncthreads();
    for (int j = threadIdx.x; j < NG_Q21; j += blockDim.x)
        if (g[j] != 0) atomicAdd(&group_sums[j], g[j]);
}

// ---------------------------------------------------------------------------
// Two-stream pipelined q21 (GPUE_Q21_PIPE=1): the fused kernel's streaming
// leg (1.57 ms) and its part-probe gather leg (2.32 ms) measured fully
// ADDITIVE (profiles/q21_decomp.log) — so split them into an operator pair
// and overlap across chunks: K_A (pk stream + part runtime-filter probe +
// brand payload gather -> u16 per row) runs on stream2 for chunk i while
// K_B (sk/od/rv streams + brand u16 + supplier/date probes + LDS group agg)
// consumes chunk i-1 on the session stream, linked by events.
// ---------------------------------------------------------------------------
__global__ void k_q21_pass_payload(const int32_t* __restrict__ pk, uint64_t n,
                                   const uint32_t* __restrict__ pbits, int64_t psmin,
                                   uint64_t psint, const uint16_t* __restrict__ pfirst,
                                   uint16_t* __restrict__ brand_out) {
    const uint64_t n4 = n / 4;
    const int4* __restrict__ pk4 = (const int4*)pk;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    for (; i + stride < n4; i += 2 * stride) {
        int4 pa = pk4[i];
        int4 pb_ = pk4[i + stride];
        ushort4 oa, ob;
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            uint32_t idx = (uint32_t)((&pa.x)[j] - psmin);
            bool in = idx < psint;
            uint32_t cidx = in ? idx : 0u;
            bool pass = in & ((pbits[cidx >> 5] >> (cidx & 31)) & 1u);
            (&oa.x)[j] = pass ? pfirst[(&pa.x)[j] - 1] : (uint16_t)0;
        }
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            uint32_t idx = (uint32_t)((&pb_.x)[j] - psmin);
            bool in = idx < psint;
            uint32_t cidx = in ? idx : 0u;
            bool pass = in & ((pbits[cidx >> 5] >> (cidx & 31)) & 1u);
            (&ob.x)[j] = pass ? pfirst[(&pb_.x)[j] - 1] : (uint16_t)0;
        }
        ((ushort4*)brand_out)[i] = oa;
        ((ushort4*)brand_out)[i + stride] = ob;
    }
    for (; i < n4; i += stride) {
        int4 p4 = pk4[i];
        ushort4 o;
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            uint32_t idx = (uint32_t)((&p4.x)[j] - psmin);
            bool in = idx < psint;
            uint32_t cidx = in ? idx : 0u;
            bool pass = in & ((pbits[cidx >> 5] >> (cidx & 31)) & 1u);
            (&o.x)[j] = pass ? pfirst[(&p4.x)[j] - 1] : (uint16_t)0;
        }
        ((ushort4*)brand_out)[i] = o;
    }
    // tail rows (n % 4)
    uint64_t tid = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    for (uint64_t r = n4 * 4 + tid; r < n; r += stride) {
        uint32_t idx = (uint32_t)(pk[r] - psmin);
        bool in = idx < psint;
        uint32_t cidx = in ? idx : 0u;
        bool pass = in & ((pbits[cidx >> 5] >> (cidx & 31)) & 1u);
        brand_out[r] = pass ? pfirst[pk[r] - 1] : (uint16_t)0;
    }
}

__global__ __launch_bounds__(BLOCK_Q21) void
k_q21_phase2(const uint16_t* __restrict__ brand, const int32_t* __restrict__ sk,
             const int32_t* __restrict__ od, const int32_t* __restrict__ rv, uint64_t n,
             const uint32_t* __restrict__ sbits, int64_t ssmin, uint64_t ssint,
             const uint16_t* __restrict__ dfirst, int64_t dmin,
             unsigned long long* __restrict__ group_sums) {
    __shared__ unsigned long long g[NG_Q21];
    for (int j = threadIdx.x; j < NG_Q21; j += blockDim.x) g[j] = 0;
    __syncthreads();
    const uint64_t n4 = n / 4;
    const ushort4* __restrict__ br4 = (const ushort4*)brand;
    const int4* __restrict__ sk4 = (const int4*)sk;
    const int4* __restrict__ od4 = (const int4*)od;
    const int4* __restrict__ rv4 = (const int4*)rv;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
        ushort4 b4 = br4[i];
        int4 s4 = sk4[i];
        int4 o4 = od4[i];
        int4 r4 = rv4[i];
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            uint16_t brand1 = (&b4.x)[j];
            if (brand1 == 0) continue;
            uint32_t sidx = (uint32_t)((&s4.x)[j] - ssmin);
            if (sidx >= ssint || !((sbits[sidx >> 5] >> (sidx & 31)) & 1u)) continue;
            uint32_t year1 = dfirst[(&o4.x)[j] - dmin];
            atomicAdd(&g[(year1 - 1) * 1000 + (brand1 - 1)],
                      (unsigned long long)(int64_t)(&r4.x)[j]);
        }
    }
    uint64_t tid = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    for (uint64_t r = n4 * 4 + tid; r < n; r += stride) {
        uint16_t brand1 = brand[r];
        if (brand1 == 0) continue;
        uint32_t sidx = (uint32_t)(sk[r] - ssmin);
        if (sidx >= ssint || !((sbits[sidx >> 5] >> (sidx & 31)) & 1u)) continue;
        uint32_t year1 = dfirst[od[r] - dmin];
        atomicAdd(&g[(year1 - 1) * 1000 + (brand1 - 1)], (unsigned long long)(int64_t)rv[r]);
    }
    __syncthreads();
    for (int j = threadIdx.x; j < NG_Q21; j += blockDim.x)
        if (g[j] != 0) atomicAdd(&group_sums[j], g[j]);
}

extern "C" int gpue_q21_star_agg_pipe(gpue_session* s, gpue_join_table* parts,
                                      gpue_join_table* supps, gpue_join_table* dates,
                                      gpue_dbuf* pk, gpue_dbuf* sk, gpue_dbuf* od,
                                      gpue_dbuf* rv, uint64_t n, gpue_dbuf* brand_scratch,
                                      gpue_dbuf* group_sums, int n_chunks);
int gpue_q21_star_agg_pipe(gpue_session* s, gpue_join_table* parts, gpue_join_table* supps,
                           gpue_join_table* dates, gpue_dbuf* pk, gpue_dbuf* sk,
                           gpue_dbuf* od, gpue_dbuf* rv, uint64_t n,
                           gpue_dbuf* brand_scratch, gpue_dbuf* group_sums, int n_chunks) {
    ARG_CHECK(s && parts && supps && dates && pk && sk && od && rv && brand_scratch &&
              group_sums);
    ARG_CHECK(parts->first16 && parts->bitset && supps->bitset && dates->first16);
    ARG_CHECK(brand_scratch->bytes >= n * 2);
    ARG_CHECK(n_chunks >= 1 && n_chunks <= gpue_session::N_CHUNK_EVENTS);
    HIP_CHECK(hipMemsetAsync(group_sums->ptr, 0, NG_Q21 * sizeof(int64_t), s->stream));
    uint64_t chunk = ((n / n_chunks) + 3) & ~3ull; // 4-row aligned chunks
    for (int c = 0; c < n_chunks; c++) {
        uint64_t lo = (uint64_t)c * chunk;
        if (lo >= n) break;
        uint64_t len = min(chunk, n - lo);
        hipLaunchKernelGGL(k_q21_pass_payload, dim3(grid_for(len)), dim3(BLOCK), 0,
                           s->stream2, (const int32_t*)pk->ptr + lo, len, parts->bitset,
                           parts->set_min, (uint64_t)(parts->set_max - parts->set_min + 1),
                           parts->first16, (uint16_t*)brand_scratch->ptr + lo);
        HIP_CHECK(hipEventRecord(s->chunk_ev[c], s->stream2));
        HIP_CHECK(hipStreamWaitEvent(s->stream, s->chunk_ev[c], 0));
        hipLaunchKernelGGL(k_q21_phase2, dim3(env_cap("GPUE_GRID_WIDE", 512)),
                           dim3(BLOCK_Q21), 0, s->stream,
                           (const uint16_t*)brand_scratch->ptr + lo,
                           (const int32_t*)sk->ptr + lo, (const int32_t*)od->ptr + lo,
                           (const int32_t*)rv->ptr + lo, len, supps->bitset,
                           supps->set_min, (uint64_t)(supps->set_max - supps->set_min + 1),
                           dates->first16, dates->min_key,
                           (unsigned long long*)group_sums->ptr);
    }
    HIP_CHECK(hipGetLastError());
    return GPUE_OK;
}

static bool q21_glob() {
    static int v = -1;
    if (v < 0) {
        const char* e = getenv("GPUE_Q21_GLOB");
        v = (e && atoi(e)) ? 1 : 0;
    }
    return v == 1;
}

// GPUE_Q21_PF: 1 (default) = LDS-prefilter kernel, 0 = the round-1 fused
// L2-bitset kernel
static bool q21_pf() {
    static int v = -1;
    if (v < 0) {
        const char* e = getenv("GPUE_Q21_PF");
        v = (e && !atoi(e)) ? 0 : 1;
    }
    return v == 1;
}

extern "C" int gpue_q21_star_agg_async(gpue_session* s, gpue_join_table* parts,
                                       gpue_join_table* supps, gpue_join_table* dates,
                                       gpue_dbuf* pk, gpue_dbuf* sk, gpue_dbuf* od,
                                       gpue_dbuf* rv, uint64_t n, gpue_dbuf* group_sums);
int gpue_q21_star_agg_async(gpue_session* s, gpue_join_table* parts, gpue_join_table* supps,
                            gpue_join_table* dates, gpue_dbuf* pk, gpue_dbuf* sk,
                            gpue_dbuf* od, gpue_dbuf* rv, uint64_t n, gpue_dbuf* group_sums) {
    ARG_CHECK(s && parts && supps && dates && pk && sk && od && rv && group_sums);
    ARG_CHECK(group_sums->bytes >= NG_Q21 * sizeof(int64_t));
    ARG_CHECK(parts->min_key == 1 && supps->min_key == 1);
    ARG_CHECK(parts->first16 && supps->first16 && dates->first16);
    HIP_CHECK(hipMemsetAsync(group_sums->ptr, 0, NG_Q21 * sizeof(int64_t), s->stream));
    if (q21_pf() && parts->prefilter) {
        const char* e = getenv("GPUE_Q21_PF");
        int mode = e ? atoi(e) : 7; // deferred-load default (the q43 form:
                                    // stream pk only, drain sk/od/rv on
                                    // demand): 1.106-1.133 vs mode 6's
                                    // 1.559-1.560 ms interleaved (r02 late).
                                    // 6 = full-stream + split fold; 4 =
                                    // single-fold wave-queue; 1 = prefilter
        // NT streams measured 2.267 vs 2.575 ms (frac 0.54 vs 0.47,
        // gpurun_out/q21_var_sweep.log r02) — default on; the global-group
        // 2-block variant (mode 2) measured 3.2 ms and stays for evidence
        const char* nt = getenv("GPUE_Q21_PF_NT");
        bool use_nt = !nt || atoi(nt);
        int def_grid = mode == 2 ? 512 : 256; // global-group variant fits 2 blocks/CU
        // GPUE_Q21_PF_TPB: threads per block (the kernel is blockDim-agnostic;
        // 120 KB LDS pins it at 1 block/CU regardless, so TPB sets waves/CU:
        // 1024 -> 16, 512 -> 8, 256 -> 4 — the q1 streaming geometry)
        int tpb = env_cap("GPUE_Q21_PF_TPB", BLOCK_Q21);
        auto launch = [&](auto kern, int grid) {
            hipLaunchKernelGGL(kern, dim3(env_cap("GPUE_GRID_PF", grid)),
                               dim3(tpb), 0, s->stream, (const int32_t*)pk->ptr,
                               (const int32_t*)sk->ptr, (const int32_t*)od->ptr,
                               (const int32_t*)rv->ptr, n, parts->prefilter, parts->set_min,
                               (uint64_t)(parts->set_max - parts->set_min + 1),
                               parts->first16, supps->bitset, supps->set_min,
                               (uint64_t)(supps->set_max - supps->set_min + 1),
                               dates->first16, dates->min_key,
                               (unsigned long long*)group_sums->ptr);
        };
        if (mode == 7) { // deferred-load (q43-form) experiment
            hipLaunchKernelGGL(k_q21_star_agg_pfq3, dim3(env_cap("GPUE_GRID_PF", def_grid)),
                               dim3(tpb), 0, s->stream, (const int32_t*)pk->ptr,
                               (const int32_t*)sk->ptr, (const int32_t*)od->ptr,
                               (const int32_t*)rv->ptr, n, parts->prefilter2,
                               parts->set_min,
                               (uint64_t)(parts->set_max - parts->set_min + 1),
                               parts->first16, supps->bitset, supps->set_min,
                               (uint64_t)(supps->set_max - supps->set_min + 1),
                               dates->first16, dates->min_key,
                               (unsigned long long*)group_sums->ptr);
        }
        else if (mode == 6) { // split two-probe prefilter experiment
            hipLaunchKernelGGL(k_q21_star_agg_pfq2, dim3(env_cap("GPUE_GRID_PF", def_grid)),
                               dim3(tpb), 0, s->stream, (const int32_t*)pk->ptr,
                               (const int32_t*)sk->ptr, (const int32_t*)od->ptr,
                               (const int32_t*)rv->ptr, n, parts->prefilter2,
                               parts->set_min,
                               (uint64_t)(parts->set_max - parts->set_min + 1),
                               parts->first16, supps->bitset, supps->set_min,
                               (uint64_t)(supps->set_max - supps->set_min + 1),
                               dates->first16, dates->min_key,
                               (unsigned long long*)group_sums->ptr);
        }
        else if (mode == 4) launch(k_q21_star_agg_pfq, def_grid); // wave-queue variant
        else if (mode == 2 && use_nt) launch(k_q21_star_agg_pf<false, true>, def_grid);
        else if (mode == 2) launch(k_q21_star_agg_pf<false, false>, def_grid);
        else if (use_nt) launch(k_q21_star_agg_pf<true, true>, def_grid);
        else launch(k_q21_star_agg_pf<true, false>, def_grid);
        HIP_CHECK(hipGetLastError());
        return GPUE_OK;
    }
    if (q21_glob()) {
        hipLaunchKernelGGL(k_q21_star_agg_glob, dim3(grid_stream(n)), dim3(BLOCK), 0,
                           s->stream, (const int32_t*)pk->ptr, (const int32_t*)sk->ptr,
                           (const int32_t*)od->ptr, (const int32_t*)rv->ptr, n,
                           parts->bitset, parts->set_min,
                           (uint64_t)(parts->set_max - parts->set_min + 1), parts->first16,
                           supps->bitset, supps->set_min,
                           (uint64_t)(supps->set_max - supps->set_min + 1), dates->first16,
                           dates->min_key, (unsigned long long*)group_sums->ptr);
    } else
    hipLaunchKernelGGL(k_q21_star_agg, dim3(env_cap("GPUE_GRID_WIDE", 512)), dim3(BLOCK_Q21), 0, s->stream,
                       (const int32_t*)pk->ptr, (const int32_t*)sk->ptr,
                       (const int32_t*)od->ptr, (const int32_t*)rv->ptr, n,
                       parts->bitset, parts->set_min,
                       (uint64_t)(parts->set_max - parts->set_min + 1), parts->first16,
                       supps->bitset, supps->set_min,
                       (uint64_t)(supps->set_max - supps->set_min + 1), dates->first16,
                       dates->min_key, (unsigned long long*)group_sums->ptr);
    HIP_CHECK(hipGetLastError());
    return GPUE_OK;
}

int gpue_q21_star_agg(gpue_session* s, gpue_join_table* parts, gpue_join_table* supps,
                      gpue_join_table* dates, gpue_dbuf* pk, gpue_dbuf* sk, gpue_dbuf* od,
                      gpue_dbuf* rv, uint64_t n, int64_t* group_sums_out) {
    ARG_CHECK(s && parts && supps && dates && pk && sk && od && rv && group_sums_out);
    ARG_CHECK(pk->bytes >= n * 4 && sk->bytes >= n * 4 && od->bytes >= n * 4 && rv->bytes >= n * 4);
    ARG_CHECK(parts->min_key == 1 && supps->min_key == 1); // keys are 1..N
    ARG_CHECK(parts->first16 && supps->first16 && dates->first16); // payloads < 65536
    unsigned long long* d_g = nullptr;
    HIP_CHECK(hipMalloc(&d_g, NG_Q21 * sizeof(unsigned long long)));
    // route through the same kernel dispatch as the bench path (LDS
    // prefilter default, GPUE_Q21_PF/GPUE_Q21_GLOB variants) so parity
    // tests exercise whatever kernel the bench actually runs
    gpue_dbuf tmp{s, d_g, NG_Q21 * sizeof(unsigned long long), false};
    int rc = gpue_q21_star_agg_async(s, parts, supps, dates, pk, sk, od, rv, n, &tmp);
    if (rc != GPUE_OK) {
        (void)hipFree(d_g);
        return rc;
    }
    HIP_CHECK(hipMemcpyAsync(group_sums_out, d_g, NG_Q21 * sizeof(int64_t),
                             hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    (void)hipFree(d_g);
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// Config 4 — SSB Q4.3: 4-way star join (customer ⋈ supplier ⋈ part ⋈ date)
// + GROUP BY (d_year, s_city, p_brand) as compact filtered indexes:
// gid = (dpay-1)*400 + (spay-1)*40 + (ppay-1), 2×10×40 = 800 groups
// (SURVEY.md §8a cfg 4 ≈ 800 groups). SUM(lo_revenue − lo_supplycost).
// Probe order by rejection rate × table cost: part bitset (1/25, 175 KB L2)
// batched for every row; survivors probe supplier bitset (1/25, 25 KB),
// customer bitset (1/5, 375 KB), date payload (2/7).
// ---------------------------------------------------------------------------
static constexpr int NG_Q43 = 800;

// PF: stage the 64 KB folded part prefilter in LDS (same structure as
// k_q21_star_agg_pf) and answer the per-row part test there; only ~10%
// "maybe" rows confirm against the exact L2 bitset, so the quad mask stays
// EXACT and the deferred-stream-load savings (P(line skipped) = 0.52) are
// preserved while the TA divergent-request leg shrinks ~10x. 64+6.4 KB LDS
// -> two 1024-thread blocks/CU (32 waves).
template <bool PF>
__global__ __launch_bounds__(BLOCK_Q21) void
k_q43_star_agg(const int32_t* __restrict__ ck, const int32_t* __restrict__ sk,
               const int32_t* __restrict__ pk, const int32_t* __restrict__ od,
               const int32_t* __restrict__ rv, const int32_t* __restrict__ sc,
               uint64_t n, const uint32_t* __restrict__ prefilter,
               const uint32_t* __restrict__ cbits, int64_t csmin, uint64_t csint,
               const uint32_t* __restrict__ sbits, int64_t ssmin, uint64_t ssint,
               const uint16_t* __restrict__ sfirst,
               const uint32_t* __restrict__ pbits, int64_t psmin, uint64_t psint,
               const uint16_t* __restrict__ pfirst,
               const uint16_t* __restrict__ dfirst, int64_t dmin,
               unsigned long long* __restrict__ group_sums) {
    __shared__ unsigned long long g[NG_Q43];
    __shared__ uint32_t pf[PF ? PF_WORDS : 1];
    if (PF)
        for (uint32_t w = threadIdx.x; w < PF_WORDS; w += blockDim.x) pf[w] = prefilter[w];
    for (int j = threadIdx.x; j < NG_Q43; j += blockDim.x) g[j] = 0;
    __syncthreads();
    const uint64_t n4 = n / 4;
    const int4* __restrict__ ck4 = (const int4*)ck;
    const int4* __restrict__ sk4 = (const int4*)sk;
    const int4* __restrict__ pk4 = (const int4*)pk;
    const int4* __restrict__ od4 = (const int4*)od;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    // The part bitset (1/25 selective) gates everything: test it from pk
    // alone, and only LOAD the other five streams for quads with a survivor.
    // A 64 B line holds 16 rows; P(no survivor in a line) = (24/25)^16 = 0.52,
    // so about half the ck/sk/od/rv/sc lines are never fetched (GPUE env
    // A/B-measured).
    auto quad_mask = [&](int4 p4) -> uint32_t {
        uint32_t m = 0;
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            uint32_t idx = (uint32_t)((&p4.x)[j] - psmin);
            uint32_t cidx = idx < psint ? idx : 0u;
            if (PF) {
                uint32_t fidx = cidx & PF_MASK;
                bool maybe = (idx < psint) & (pf[fidx >> 5] >> (fidx & 31)) & 1u;
                if (maybe) m |= ((pbits[cidx >> 5] >> (cidx & 31)) & 1u) << j;
            } else {
                m |= ((idx < psint) & (pbits[cidx >> 5] >> (cidx & 31)) & 1u) << j;
            }
        }
        return m;
    };
    // rv/sc feed only the final accumulate: after all four filters the pass
    // rate is ~1e-4, so load them per surviving ROW as scalars — those two
    // streams are then almost never fetched at all.
    auto quad_rest = [&](uint32_t m, uint64_t q, int4 c4, int4 s4, int4 p4, int4 o4) {
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            if (!((m >> j) & 1u)) continue;
            uint32_t sidx = (uint32_t)((&s4.x)[j] - ssmin);
            if (sidx >= ssint || !((sbits[sidx >> 5] >> (sidx & 31)) & 1u)) continue;
            uint32_t cidx = (uint32_t)((&c4.x)[j] - csmin);
            if (cidx >= csint || !((cbits[cidx >> 5] >> (cidx & 31)) & 1u)) continue;
            uint32_t dpay = dfirst[(&o4.x)[j] - dmin];
            if (dpay == 0) continue;
            uint32_t ppay = pfirst[(&p4.x)[j] - 1];
            uint32_t spay = sfirst[(&s4.x)[j] - 1];
            uint64_t r = q * 4 + j;
            atomicAdd(&g[(dpay - 1) * 400 + (spay - 1) * 40 + (ppay - 1)],
                      (unsigned long long)((int64_t)rv[r] - sc[r]));
        }
    };
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    for (; i + stride < n4; i += 2 * stride) {
        int4 pa = pk4[i];
        uint64_t i2 = i + stride;
        int4 pb_ = pk4[i2];
        uint32_t ma = quad_mask(pa), mb = quad_mask(pb_);
        if (ma) quad_rest(ma, i, ck4[i], sk4[i], pa, od4[i]);
        if (mb) quad_rest(mb, i2, ck4[i2], sk4[i2], pb_, od4[i2]);
    }
    for (; i < n4; i += stride) {
        int4 p4 = pk4[i];
        uint32_t m = quad_mask(p4);
        if (m) quad_rest(m, i, ck4[i], sk4[i], p4, od4[i]);
    }
    // scalar tail
    uint64_t tid = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    for (uint64_t r = n4 * 4 + tid; r < n; r += stride) {
        uint32_t pidx = (uint32_t)(pk[r] - psmin);
        if (pidx >= psint || !((pbits[pidx >> 5] >> (pidx & 31)) & 1u)) continue;
        uint32_t sidx = (uint32_t)(sk[r] - ssmin);
        if (sidx >= ssint || !((sbits[sidx >> 5] >> (sidx & 31)) & 1u)) continue;
        uint32_t cidx = (uint32_t)(ck[r] - csmin);
        if (cidx >= csint || !((cbits[cidx >> 5] >> (cidx & 31)) & 1u)) continue;
        uint32_t dpay = dfirst[od[r] - dmin];
        if (dpay == 0) continue;
        uint32_t ppay = pfirst[pk[r] - 1];
        uint32_t spay = sfirst[sk[r] - 1];
        atomicAdd(&g[(dpay - 1) * 400 + (spay - 1) * 40 + (ppay - 1)],
                  (unsigned long long)((int64_t)rv[r] - sc[r]));
    }
    __syncthreads();
    for (int j = threadIdx.x; j < NG_Q43; j += blockDim.x)
        if (g[j] != 0) atomicAdd(&group_sums[j], g[j]);
}

// ---------------------------------------------------------------------------
// Wave-queue q43 (GPUE_Q43_PF=4): only the pk stream is read unconditionally
// (2.4 GB); LDS-prefilter maybes (~10%) queue {pk, row} per wave and a FULL
// 64-lane drain confirms against the exact bitset, then loads sk/ck/od (and
// rv/sc for final survivors) as per-row scalars — the deferred-load idea
// taken to its limit, with phase 2 free of sparse-lane divergence. LDS:
// 64 KB pf + 6.4 KB groups + 16 KB queues = 87 KB, 1024-thr, 1 block/CU.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(BLOCK_Q21) void
k_q43_star_agg_pfq(const int32_t* __restrict__ ck, const int32_t* __restrict__ sk,
                   const int32_t* __restrict__ pk, const int32_t* __restrict__ od,
                   const int32_t* __restrict__ rv, const int32_t* __restrict__ sc,
                   uint64_t n, const uint32_t* __restrict__ prefilter,
                   const uint32_t* __restrict__ cbits, int64_t csmin, uint64_t csint,
                   const uint32_t* __restrict__ sbits, int64_t ssmin, uint64_t ssint,
                   const uint16_t* __restrict__ sfirst,
                   const uint32_t* __restrict__ pbits, int64_t psmin, uint64_t psint,
                   const uint16_t* __restrict__ pfirst,
                   const uint16_t* __restrict__ dfirst, int64_t dmin,
                   unsigned long long* __restrict__ group_sums) {
    __shared__ uint32_t pf[PF_WORDS];
    __shared__ unsigned long long g[NG_Q43];
    __shared__ int2 wq[BLOCK_Q21 / WAVE][128]; // {pk, row32}
    for (uint32_t w = threadIdx.x; w < PF_WORDS; w += blockDim.x) pf[w] = prefilter[w];
    for (int j = threadIdx.x; j < NG_Q43; j += blockDim.x) g[j] = 0;
    __syncthreads();
    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x & (WAVE - 1);
    uint32_t wqn = 0; // wave-uniform (ballot counts); loops below are wave-uniform
    const uint64_t n4 = n / 4;
    const int4* __restrict__ pk4 = (const int4*)pk;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    auto process = [&](int2 e) { // full filter chain for one queued candidate
        uint32_t idx = (uint32_t)(e.x - psmin);
        if (!((pbits[idx >> 5] >> (idx & 31)) & 1u)) return; // exact part filter
        uint32_t r = (uint32_t)e.y;
        int32_t skv = sk[r];
        uint32_t sidx = (uint32_t)(skv - ssmin);
        if (sidx >= ssint || !((sbits[sidx >> 5] >> (sidx & 31)) & 1u)) return;
        uint32_t cidx = (uint32_t)(ck[r] - csmin);
        if (cidx >= csint || !((cbits[cidx >> 5] >> (cidx & 31)) & 1u)) return;
        uint32_t dpay = dfirst[od[r] - dmin];
        if (dpay == 0) return;
        uint32_t ppay = pfirst[e.x - 1];
        uint32_t spay = sfirst[skv - 1];
        atomicAdd(&g[(dpay - 1) * 400 + (spay - 1) * 40 + (ppay - 1)],
                  (unsigned long long)((int64_t)rv[r] - sc[r]));
    };
    auto push = [&](int32_t key, uint32_t row, bool inb) {
        uint32_t idx = (uint32_t)(key - psmin);
        bool in = inb & (idx < psint);
        uint32_t fidx = (in ? idx : 0u) & PF_MASK;
        bool maybe = in & (pf[fidx >> 5] >> (fidx & 31)) & 1u;
        uint64_t m = __ballot(maybe);
        if (m) {
            uint32_t rank = __popcll(m & ((1ull << lane) - 1));
            if (maybe) wq[wid][wqn + rank] = make_int2(key, (int32_t)row);
            wqn += __popcll(m);
            if (wqn >= 64) {
                process(wq[wid][wqn - 64 + lane]);
                wqn -= 64;
            }
        }
    };
    uint64_t base = (uint64_t)blockIdx.x * blockDim.x + (uint64_t)wid * WAVE;
    uint64_t i = base + lane;
    for (; base + stride + WAVE <= n4; base += 2 * stride, i += 2 * stride) {
        int4 pa = pk4[i];
        uint64_t i2 = i + stride;
        int4 pb_ = pk4[i2];
        #pragma unroll
        for (int j = 0; j < 4; j++) push((&pa.x)[j], (uint32_t)(i * 4 + j), true);
        #pragma unroll
        for (int j = 0; j < 4; j++) push((&pb_.x)[j], (uint32_t)(i2 * 4 + j), true);
    }
    for (; base < n4; base += stride, i += stride) {
        bool inb = i < n4;
        int4 p4 = inb ? pk4[i] : make_int4(0, 0, 0, 0);
        #pragma unroll
        for (int j = 0; j < 4; j++) push((&p4.x)[j], (uint32_t)(i * 4 + j), inb);
    }
    if (wqn > 0 && lane < (int)wqn) process(wq[wid][lane]);
    uint64_t tid = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    for (uint64_t r = n4 * 4 + tid; r < n; r += stride) {
        uint32_t pidx = (uint32_t)(pk[r] - psmin);
        if (pidx >= psint || !((pbits[pidx >> 5] >> (pidx & 31)) & 1u)) continue;
        uint32_t sidx = (uint32_t)(sk[r] - ssmin);
        if (sidx >= ssint || !((sbits[sidx >> 5] >> (sidx & 31)) & 1u)) continue;
        uint32_t cidx = (uint32_t)(ck[r] - csmin);
        if (cidx >= csint || !((cbits[cidx >> 5] >> (cidx & 31)) & 1u)) continue;
        uint32_t dpay = dfirst[od[r] - dmin];
        if (dpay == 0) continue;
        uint32_t ppay = pfirst[pk[r] - 1];
        uint32_t spay = sfirst[sk[r] - 1];
        atomicAdd(&g[(dpay - 1) * 400 + (spay - 1) * 40 + (ppay - 1)],
                  (unsigned long long)((int64_t)rv[r] - sc[r]));
    }
    __syncthreads();
    for (int j = threadIdx.x; j < NG_Q43; j += blockDim.x)
        if (g[j] != 0) atomicAdd(&group_sums[j], g[j]);
}

// ---------------------------------------------------------------------------
// Wide split two-probe q43 (GPUE_Q43_PF=6): two full 2^19-bit folds (128 KB
// LDS — affordable because q43's group array is only 6.4 KB) cut the
// false-maybe rate 6.1% -> ~1.0%, nearly halving drain episodes. Same
// wave-queue structure; `prefilter` here is the table's prefilter2w.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(BLOCK_Q21) void
k_q43_star_agg_pfq2(const int32_t* __restrict__ ck, const int32_t* __restrict__ sk,
                   const int32_t* __restrict__ pk, const int32_t* __restrict__ od,
                   const int32_t* __restrict__ rv, const int32_t* __restrict__ sc,
                   uint64_t n, const uint32_t* __restrict__ prefilter,
                   const uint32_t* __restrict__ cbits, int64_t csmin, uint64_t csint,
                   const uint32_t* __restrict__ sbits, int64_t ssmin, uint64_t ssint,
                   const uint16_t* __restrict__ sfirst,
                   const uint32_t* __restrict__ pbits, int64_t psmin, uint64_t psint,
                   const uint16_t* __restrict__ pfirst,
                   const uint16_t* __restrict__ dfirst, int64_t dmin,
                   unsigned long long* __restrict__ group_sums) {
    __shared__ uint32_t pfa[PF_WORDS]; // 64 KB fold A (i & M19)
    __shared__ uint32_t pfb[PF_WORDS]; // 64 KB fold B (hash19(i))
    __shared__ unsigned long long g[NG_Q43];
    __shared__ int2 wq[BLOCK_Q21 / WAVE][128]; // {pk, row32}
    for (uint32_t w = threadIdx.x; w < PF_WORDS; w += blockDim.x) {
        pfa[w] = prefilter[w];
        pfb[w] = prefilter[PF_WORDS + w];
    }
    for (int j = threadIdx.x; j < NG_Q43; j += blockDim.x) g[j] = 0;
    __syncthreads();
    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x & (WAVE - 1);
    uint32_t wqn = 0; // wave-uniform (ballot counts); loops below are wave-uniform
    const uint64_t n4 = n / 4;
    const int4* __restrict__ pk4 = (const int4*)pk;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    auto process = [&](int2 e) { // full filter chain for one queued candidate
        uint32_t idx = (uint32_t)(e.x - psmin);
        if (!((pbits[idx >> 5] >> (idx & 31)) & 1u)) return; // exact part filter
        uint32_t r = (uint32_t)e.y;
        int32_t skv = sk[r];
        uint32_t sidx = (uint32_t)(skv - ssmin);
        if (sidx >= ssint || !((sbits[sidx >> 5] >> (sidx & 31)) & 1u)) return;
        uint32_t cidx = (uint32_t)(ck[r] - csmin);
        if (cidx >= csint || !((cbits[cidx >> 5] >> (cidx & 31)) & 1u)) return;
        uint32_t dpay = dfirst[od[r] - dmin];
        if (dpay == 0) return;
        uint32_t ppay = pfirst[e.x - 1];
        uint32_t spay = sfirst[skv - 1];
        atomicAdd(&g[(dpay - 1) * 400 + (spay - 1) * 40 + (ppay - 1)],
                  (unsigned long long)((int64_t)rv[r] - sc[r]));
    };
    auto push = [&](int32_t key, uint32_t row, bool inb) {
        uint32_t idx = (uint32_t)(key - psmin);
        bool in = inb & (idx < psint);
        uint32_t ia = (in ? idx : 0u) & PF_MASK;
        uint32_t ib = ((in ? idx : 0u) * 2654435761u) >> 13;
        bool maybe = in & (pfa[ia >> 5] >> (ia & 31)) & (pfb[ib >> 5] >> (ib & 31)) & 1u;
        uint64_t m = __ballot(maybe);
        if (m) {
            uint32_t rank = __popcll(m & ((1ull << lane) - 1));
            if (maybe) wq[wid][wqn + rank] = make_int2(key, (int32_t)row);
            wqn += __popcll(m);
            if (wqn >= 64) {
                process(wq[wid][wqn - 64 + lane]);
                wqn -= 64;
            }
        }
    };
    uint64_t base = (uint64_t)blockIdx.x * blockDim.x + (uint64_t)wid * WAVE;
    uint64_t i = base + lane;
    for (; base + stride + WAVE <= n4; base += 2 * stride, i += 2 * stride) {
        int4 pa = pk4[i];
        uint64_t i2 = i + stride;
        int4 pb_ = pk4[i2];
        #pragma unroll
        for (int j = 0; j < 4; j++) push((&pa.x)[j], (uint32_t)(i * 4 + j), true);
        #pragma unroll
        for (int j = 0; j < 4; j++) push((&pb_.x)[j], (uint32_t)(i2 * 4 + j), true);
    }
    for (; base < n4; base += stride, i += stride) {
        bool inb = i < n4;
        int4 p4 = inb ? pk4[i] : make_int4(0, 0, 0, 0);
        #pragma unroll
        for (int j = 0; j < 4; j++) push((&p4.x)[j], (uint32_t)(i * 4 + j), inb);
    }
    if (wqn > 0 && lane < (int)wqn) process(wq[wid][lane]);
    uint64_t tid = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    for (uint64_t r = n4 * 4 + tid; r < n; r += stride) {
        uint32_t pidx = (uint32_t)(pk[r] - psmin);
        if (pidx >= psint || !((pbits[pidx >> 5] >> (pidx & 31)) & 1u)) continue;
        uint32_t sidx = (uint32_t)(sk[r] - ssmin);
        if (sidx >= ssint || !((sbits[sidx >> 5] >> (sidx & 31)) & 1u)) continue;
        uint32_t cidx = (uint32_t)(ck[r] - csmin);
        if (cidx >= csint || !((cbits[cidx >> 5] >> (cidx & 31)) & 1u)) continue;
        uint32_t dpay = dfirst[od[r] - dmin];
        if (dpay == 0) continue;
        uint32_t ppay = pfirst[pk[r] - 1];
        uint32_t spay = sfirst[sk[r] - 1];
        atomicAdd(&g[(dpay - 1) * 400 + (spay - 1) * 40 + (ppay - 1)],
                  (unsigned long long)((int64_t)rv[r] - sc[r]));
    }
    __syncthreads();
    for (int j = threadIdx.x; j < NG_Q43; j += blockDim.x)
        if (g[j] != 0) atomicAdd(&group_sums[j], g[j]);
}

extern "C" int gpue_q43_star_agg_async(gpue_session* s, gpue_join_table* custs,
                                       gpue_join_table* supps, gpue_join_table* parts,
                                       gpue_join_table* dates, gpue_dbuf* ck, gpue_dbuf* sk,
                                       gpue_dbuf* pk, gpue_dbuf* od, gpue_dbuf* rv,
                                       gpue_dbuf* sc, uint64_t n, gpue_dbuf* group_sums);
extern "C" int gpue_q43_star_agg_accum_async(gpue_session* s, gpue_join_table* custs,
                                             gpue_join_table* supps, gpue_join_table* parts,
                                             gpue_join_table* dates, gpue_dbuf* ck,
                                             gpue_dbuf* sk, gpue_dbuf* pk, gpue_dbuf* od,
                                             gpue_dbuf* rv, gpue_dbuf* sc, uint64_t n,
                                             gpue_dbuf* group_sums);
int gpue_q43_star_agg_async(gpue_session* s, gpue_join_table* custs, gpue_join_table* supps,
                            gpue_join_table* parts, gpue_join_table* dates, gpue_dbuf* ck,
                            gpue_dbuf* sk, gpue_dbuf* pk, gpue_dbuf* od, gpue_dbuf* rv,
                            gpue_dbuf* sc, uint64_t n, gpue_dbuf* group_sums) {
    ARG_CHECK(s && custs && supps && parts && dates && ck && sk && pk && od && rv && sc);
    ARG_CHECK(group_sums && group_sums->bytes >= NG_Q43 * sizeof(int64_t));
    ARG_CHECK(supps->first16 && parts->first16 && dates->first16);
    ARG_CHECK(custs->bitset && supps->bitset && parts->bitset);
    ARG_CHECK(custs->min_key == 1 && supps->min_key == 1 && parts->min_key == 1);
    HIP_CHECK(hipMemsetAsync(group_sums->ptr, 0, NG_Q43 * sizeof(int64_t), s->stream));
    return gpue_q43_star_agg_accum_async(s, custs, supps, parts, dates, ck, sk, pk, od,
                                         rv, sc, n, group_sums);
}

// accumulate-only form (no zeroing): the chunked-exchange pipeline probes
// each received row-block as it lands, summing into the same group buffer
int gpue_q43_star_agg_accum_async(gpue_session* s, gpue_join_table* custs,
                                  gpue_join_table* supps, gpue_join_table* parts,
                                  gpue_join_table* dates, gpue_dbuf* ck, gpue_dbuf* sk,
                                  gpue_dbuf* pk, gpue_dbuf* od, gpue_dbuf* rv,
                                  gpue_dbuf* sc, uint64_t n, gpue_dbuf* group_sums) {
    ARG_CHECK(s && custs && supps && parts && dates && ck && sk && pk && od && rv && sc);
    ARG_CHECK(group_sums && group_sums->bytes >= NG_Q43 * sizeof(int64_t));
    ARG_CHECK(supps->first16 && parts->first16 && dates->first16);
    ARG_CHECK(custs->bitset && supps->bitset && parts->bitset);
    ARG_CHECK(custs->min_key == 1 && supps->min_key == 1 && parts->min_key == 1);
    const char* pfe = getenv("GPUE_Q43_PF");
    int pfm = pfe ? atoi(pfe) : 6; // wide split-fold default: 0.922 vs
                                   // 1.150 ms interleaved A/B (r02 late);
                                   // mode 4 = single-fold wave-queue
    bool use_pf = pfm && parts->prefilter;
    bool use_q = (pfm == 4 || pfm == 6) && parts->prefilter; // wave-queue forms
    int def_grid = use_q ? 256 : (use_pf ? 512 : 256); // queue: 1 block/CU
    auto kern = pfm == 6 && parts->prefilter2w
                    ? k_q43_star_agg_pfq2
                    : (use_q ? k_q43_star_agg_pfq
                             : (use_pf ? k_q43_star_agg<true> : k_q43_star_agg<false>));
    hipLaunchKernelGGL(kern, dim3(env_cap("GPUE_GRID_Q43", def_grid)), dim3(BLOCK_Q21), 0, s->stream,
                       (const int32_t*)ck->ptr, (const int32_t*)sk->ptr,
                       (const int32_t*)pk->ptr, (const int32_t*)od->ptr,
                       (const int32_t*)rv->ptr, (const int32_t*)sc->ptr, n,
                       pfm == 6 && parts->prefilter2w ? parts->prefilter2w
                                                      : parts->prefilter,
                       custs->bitset, custs->set_min,
                       (uint64_t)(custs->set_max - custs->set_min + 1),
                       supps->bitset, supps->set_min,
                       (uint64_t)(supps->set_max - supps->set_min + 1), supps->first16,
                       parts->bitset, parts->set_min,
                       (uint64_t)(parts->set_max - parts->set_min + 1), parts->first16,
                       dates->first16, dates->min_key,
                       (unsigned long long*)group_sums->ptr);
    HIP_CHECK(hipGetLastError());
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// exchange partition (reference exchange_sink_operator.cpp:611-660 semantics:
// FNV row hash -> ReduceOp channel -> counting-sort layout). Channel
// ASSIGNMENT is bit-identical to the reference; within a channel, rows are
// grouped by source block (ascending), order inside a block's group is
// emission order — the per-channel row SET is what routing correctness needs
// (SURVEY.md §8c).
// ---------------------------------------------------------------------------
// XXH3-64 4-to-8-byte path (exchange hash version 1,
// exchange_sink_operator.cpp:604-610; HashUtil::xx_hash3_64 =
// XXH3_64bits_withSeed). Restated from the published XXH3 spec (xxHash
// v0.8.x): XXH3_len_4to8_64b with the default kSecret words 8..23. Pinned
// to python-xxhash vectors via the oracle (tests/golden/xxh3_kats.json).
#define XXH3_SECRET8 0x1cad21f72c81017cull
#define XXH3_SECRET16 0xdb979083e96dd4deull
#define XXH3_M2 0x9FB21C651E98DF25ull
#define XXH3_SEED_32 0x9E3779B1u // HashUtil::XXH3_SEED_32 (hash_util.hpp:126)

__device__ static inline uint64_t xxh3_rrmxmx(uint64_t h, uint64_t len) {
    h ^= ((h << 49) | (h >> 15)) ^ ((h << 24) | (h >> 40));
    h *= XXH3_M2;
    h ^= (h >> 35) + len;
    h *= XXH3_M2;
    return h ^ (h >> 28);
}

// 4-byte input (one i32 key), seed = running per-column hash
__device__ static inline uint64_t xxh3_64_u32(uint32_t v, uint64_t seed) {
    uint32_t s32 = (uint32_t)seed;
    uint32_t sw = __builtin_bswap32(s32);
    seed ^= (uint64_t)sw << 32;
    uint64_t input64 = (uint64_t)v + ((uint64_t)v << 32); // in1 == in2 at len 4
    uint64_t bitflip = (XXH3_SECRET8 ^ XXH3_SECRET16) - seed;
    return xxh3_rrmxmx(input64 ^ bitflip, 4);
}

// 8-byte input (one i64 key) — the i64 exchange-hash hop; the oracle-side
// twin (orc_xxh3_hash_i64) is vector-pinned, this stays for the i64
// partition path when a workload needs it
__device__ [[maybe_unused]] static inline uint64_t xxh3_64_u64(uint64_t v, uint64_t seed) {
    uint32_t s32 = (uint32_t)seed;
    uint32_t sw = __builtin_bswap32(s32);
    seed ^= (uint64_t)sw << 32;
    uint32_t in1 = (uint32_t)v, in2 = (uint32_t)(v >> 32);
    uint64_t input64 = (uint64_t)in2 + ((uint64_t)in1 << 32);
    uint64_t bitflip = (XXH3_SECRET8 ^ XXH3_SECRET16) - seed;
    return xxh3_rrmxmx(input64 ^ bitflip, 8);
}

// zlib CRC32 (poly 0xEDB88320) — the bucket-shuffle hash path
// (exchange_sink_operator.cpp:617-622 via HashUtil::zlib_crc_hash, seed 0).
// Pinned against python zlib through the oracle restatement.
__device__ static inline uint32_t zlib_crc32_u32(uint32_t key, uint32_t seed) {
    uint32_t crc = ~seed;
    #pragma unroll
    for (int b = 0; b < 4; b++) {
        crc ^= (key >> (8 * b)) & 0xFFu;
        #pragma unroll
        for (int k = 0; k < 8; k++)
            crc = (crc >> 1) ^ (0xEDB88320u & (0u - (crc & 1u)));
    }
    return ~crc;
}

__device__ static inline uint32_t fnv_u32(uint32_t key, uint32_t seed) {
    uint32_t h = seed;
    #pragma unroll
    for (int b = 0; b < 4; b++) {
        h = (((key >> (8 * b)) & 0xFF) ^ h) * 16777619u;
    }
    return h;
}

__device__ static inline uint32_t fnv_u64(uint64_t key, uint32_t seed) {
    uint32_t h = seed;
    #pragma unroll
    for (int b = 0; b < 8; b++) h = (((uint32_t)(key >> (8 * b)) & 0xFFu) ^ h) * 16777619u;
    return h;
}

static constexpr uint32_t MAX_CH = 64;
__global__ void k_partition_scan(const uint32_t* __restrict__ hist, uint32_t nb,
                                 uint32_t nch, uint64_t* __restrict__ offsets);

template <int HV> // 0 = FNV (default/back-compat), 1 = xxh3 (version 1)
__global__ void k_partition_hist(const uint32_t* __restrict__ keys, uint64_t n, uint64_t tile,
                                 uint32_t nch, uint32_t* __restrict__ block_hist) {
    __shared__ uint32_t h[MAX_CH];
    for (uint32_t c = threadIdx.x; c < nch; c += blockDim.x) h[c] = 0;
    __syncthreads();
    uint64_t lo = (uint64_t)blockIdx.x * tile, hi = min(lo + tile, n);
    for (uint64_t i = lo + threadIdx.x; i < hi; i += blockDim.x) {
        uint32_t hash = HV == 2   ? zlib_crc32_u32(keys[i], 0)
                        : HV == 1 ? (uint32_t)xxh3_64_u32(keys[i], XXH3_SEED_32)
                                  : fnv_u32(keys[i], 0x811C9DC5u);
        uint32_t ch = (uint32_t)(((uint64_t)hash * nch) >> 32);
        atomicAdd(&h[ch], 1u);
    }
    __syncthreads();
    for (uint32_t c = threadIdx.x; c < nch; c += blockDim.x)
        block_hist[(uint64_t)blockIdx.x * nch + c] = h[c];
}

template <int HV>
__global__ void k_partition_emit(const uint32_t* __restrict__ keys, uint64_t n, uint64_t tile,
                                 uint32_t nch, const uint64_t* __restrict__ block_offsets,
                                 uint32_t* __restrict__ row_indexes) {
    __shared__ uint64_t cursor[MAX_CH];
    for (uint32_t c = threadIdx.x; c < nch; c += blockDim.x)
        cursor[c] = block_offsets[(uint64_t)blockIdx.x * nch + c];
    __syncthreads();
    uint64_t lo = (uint64_t)blockIdx.x * tile, hi = min(lo + tile, n);
    for (uint64_t i = lo + threadIdx.x; i < hi; i += blockDim.x) {
        uint32_t hash = HV == 2   ? zlib_crc32_u32(keys[i], 0)
                        : HV == 1 ? (uint32_t)xxh3_64_u32(keys[i], XXH3_SEED_32)
                                  : fnv_u32(keys[i], 0x811C9DC5u);
        uint32_t ch = (uint32_t)(((uint64_t)hash * nch) >> 32);
        uint64_t pos = atomicAdd((unsigned long long*)&cursor[ch], 1ull);
        row_indexes[pos] = (uint32_t)i;
    }
}

// Multi-column partition key: the exchange sink seeds FNV_SEED then CHAINS
// fnv_hash per partition column, each column using the running hash as its
// seed (exchange_sink_operator.cpp:611-617). Two-int32 variant.
__global__ void k_partition_hist_2xi32(const int32_t* __restrict__ a,
                                       const int32_t* __restrict__ b, uint64_t n,
                                       uint64_t tile, uint32_t nch,
                                       uint32_t* __restrict__ block_hist) {
    __shared__ uint32_t h[MAX_CH];
    for (uint32_t c = threadIdx.x; c < nch; c += blockDim.x) h[c] = 0;
    __syncthreads();
    uint64_t lo = (uint64_t)blockIdx.x * tile, hi = min(lo + tile, n);
    for (uint64_t i = lo + threadIdx.x; i < hi; i += blockDim.x) {
        uint32_t hash = fnv_u32((uint32_t)b[i], fnv_u32((uint32_t)a[i], 0x811C9DC5u));
        uint32_t ch = (uint32_t)(((uint64_t)hash * nch) >> 32);
        atomicAdd(&h[ch], 1u);
    }
    __syncthreads();
    for (uint32_t c = threadIdx.x; c < nch; c += blockDim.x)
        block_hist[(uint64_t)blockIdx.x * nch + c] = h[c];
}

__global__ void k_partition_emit_2xi32(const int32_t* __restrict__ a,
                                       const int32_t* __restrict__ b, uint64_t n,
                                       uint64_t tile, uint32_t nch,
                                       const uint64_t* __restrict__ block_offsets,
                                       uint32_t* __restrict__ row_indexes) {
    __shared__ uint64_t cursor[MAX_CH];
    for (uint32_t c = threadIdx.x; c < nch; c += blockDim.x)
        cursor[c] = block_offsets[(uint64_t)blockIdx.x * nch + c];
    __syncthreads();
    uint64_t lo = (uint64_t)blockIdx.x * tile, hi = min(lo + tile, n);
    for (uint64_t i = lo + threadIdx.x; i < hi; i += blockDim.x) {
        uint32_t hash = fnv_u32((uint32_t)b[i], fnv_u32((uint32_t)a[i], 0x811C9DC5u));
        uint32_t ch = (uint32_t)(((uint64_t)hash * nch) >> 32);
        uint64_t pos = atomicAdd((unsigned long long*)&cursor[ch], 1ull);
        row_indexes[pos] = (uint32_t)i;
    }
}

__global__ void k_partition_hist64(const uint64_t* __restrict__ keys, uint64_t n,
                                   uint64_t tile, uint32_t nch,
                                   uint32_t* __restrict__ block_hist) {
    __shared__ uint32_t h[MAX_CH];
    for (uint32_t c = threadIdx.x; c < nch; c += blockDim.x) h[c] = 0;
    __syncthreads();
    uint64_t lo = (uint64_t)blockIdx.x * tile, hi = min(lo + tile, n);
    for (uint64_t i = lo + threadIdx.x; i < hi; i += blockDim.x) {
        uint32_t ch = (uint32_t)(((uint64_t)fnv_u64(keys[i], 0x811C9DC5u) * nch) >> 32);
        atomicAdd(&h[ch], 1u);
    }
    __syncthreads();
    for (uint32_t c = threadIdx.x; c < nch; c += blockDim.x)
        block_hist[(uint64_t)blockIdx.x * nch + c] = h[c];
}

__global__ void k_partition_emit64(const uint64_t* __restrict__ keys, uint64_t n,
                                   uint64_t tile, uint32_t nch,
                                   const uint64_t* __restrict__ block_offsets,
                                   uint32_t* __restrict__ row_indexes) {
    __shared__ uint64_t cursor[MAX_CH];
    for (uint32_t c = threadIdx.x; c < nch; c += blockDim.x)
        cursor[c] = block_offsets[(uint64_t)blockIdx.x * nch + c];
    __syncthreads();
    uint64_t lo = (uint64_t)blockIdx.x * tile, hi = min(lo + tile, n);
    for (uint64_t i = lo + threadIdx.x; i < hi; i += blockDim.x) {
        uint32_t ch = (uint32_t)(((uint64_t)fnv_u64(keys[i], 0x811C9DC5u) * nch) >> 32);
        uint64_t pos = atomicAdd((unsigned long long*)&cursor[ch], 1ull);
        row_indexes[pos] = (uint32_t)i;
    }
}

// BIGINT-key variant (FNV over the 8 LE bytes, reference fnv_hash on an
// int64 key column)
extern "C" int gpue_partition_i64(gpue_session* s, gpue_dbuf* keys, uint64_t n, uint32_t nch,
                                  uint64_t* start_points_out, gpue_dbuf* row_indexes_out);
int gpue_partition_i64(gpue_session* s, gpue_dbuf* keys, uint64_t n, uint32_t nch,
                       uint64_t* start_points_out, gpue_dbuf* row_indexes_out) {
    ARG_CHECK(s && keys && start_points_out && row_indexes_out);
    ARG_CHECK(nch >= 1 && nch <= MAX_CH);
    ARG_CHECK(keys->bytes >= n * 8 && row_indexes_out->bytes >= n * 4);
    uint32_t nb = grid_for(n);
    uint64_t tile = (n + nb - 1) / nb;
    uint32_t* d_hist = nullptr;
    uint64_t* d_off = nullptr;
    HIP_CHECK(hipMalloc(&d_hist, (uint64_t)nb * nch * sizeof(uint32_t)));
    hipLaunchKernelGGL(k_partition_hist64, dim3(nb), dim3(BLOCK), 0, s->stream,
                       (const uint64_t*)keys->ptr, n, tile, nch, d_hist);
    uint32_t* h_hist = (uint32_t*)malloc((uint64_t)nb * nch * sizeof(uint32_t));
    uint64_t* h_off = (uint64_t*)malloc((uint64_t)nb * nch * sizeof(uint64_t));
    HIP_CHECK(hipMemcpyAsync(h_hist, d_hist, (uint64_t)nb * nch * sizeof(uint32_t),
                             hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    uint64_t acc = 0;
    for (uint32_t c = 0; c < nch; c++) {
        start_points_out[c] = acc;
        for (uint32_t b = 0; b < nb; b++) {
            h_off[(uint64_t)b * nch + c] = acc;
            acc += h_hist[(uint64_t)b * nch + c];
        }
    }
    start_points_out[nch] = acc;
    HIP_CHECK(hipMalloc(&d_off, (uint64_t)nb * nch * sizeof(uint64_t)));
    HIP_CHECK(hipMemcpyAsync(d_off, h_off, (uint64_t)nb * nch * sizeof(uint64_t),
                             hipMemcpyHostToDevice, s->stream));
    hipLaunchKernelGGL(k_partition_emit64, dim3(nb), dim3(BLOCK), 0, s->stream,
                       (const uint64_t*)keys->ptr, n, tile, nch, d_off,
                       (uint32_t*)row_indexes_out->ptr);
    HIP_CHECK(hipStreamSynchronize(s->stream));
    (void)hipFree(d_hist);
    (void)hipFree(d_off);
    free(h_hist);
    free(h_off);
    return GPUE_OK;
}

// async i64 form (device-side scan, no host round-trip) — see
// gpue_partition_i32_async
extern "C" int gpue_partition_i64_async(gpue_session* s, gpue_dbuf* keys, uint64_t n,
                                        uint32_t nch, gpue_dbuf* row_indexes_out,
                                        gpue_dbuf* scratch);
int gpue_partition_i64_async(gpue_session* s, gpue_dbuf* keys, uint64_t n, uint32_t nch,
                             gpue_dbuf* row_indexes_out, gpue_dbuf* scratch) {
    ARG_CHECK(s && keys && row_indexes_out && scratch);
    ARG_CHECK(nch >= 1 && nch <= MAX_CH);
    ARG_CHECK(keys->bytes >= n * 8 && row_indexes_out->bytes >= n * 4);
    uint32_t nb = grid_for(n);
    uint64_t tile = (n + nb - 1) / nb;
    ARG_CHECK(scratch->bytes >= (uint64_t)nb * nch * 12);
    uint32_t* d_hist = (uint32_t*)scratch->ptr;
    uint64_t* d_off = (uint64_t*)((uint8_t*)scratch->ptr + (uint64_t)nb * nch * 4);
    hipLaunchKernelGGL(k_partition_hist64, dim3(nb), dim3(BLOCK), 0, s->stream,
                       (const uint64_t*)keys->ptr, n, tile, nch, d_hist);
    hipLaunchKernelGGL(k_partition_scan, dim3(1), dim3(64), 0, s->stream, d_hist, nb, nch,
                       d_off);
    hipLaunchKernelGGL(k_partition_emit64, dim3(nb), dim3(BLOCK), 0, s->stream,
                       (const uint64_t*)keys->ptr, n, tile, nch, d_off,
                       (uint32_t*)row_indexes_out->ptr);
    HIP_CHECK(hipGetLastError());
    return GPUE_OK;
}

static int partition_i32_impl(gpue_session* s, gpue_dbuf* keys, uint64_t n, uint32_t nch,
                              uint64_t* start_points_out, gpue_dbuf* row_indexes_out,
                              int hash_version) {
    ARG_CHECK(s && keys && start_points_out && row_indexes_out);
    ARG_CHECK(nch >= 1 && nch <= MAX_CH);
    ARG_CHECK(keys->bytes >= n * 4 && row_indexes_out->bytes >= n * 4);
    uint32_t nb = grid_for(n);
    uint64_t tile = (n + nb - 1) / nb;
    uint32_t* d_hist = nullptr;
    uint64_t* d_off = nullptr;
    HIP_CHECK(hipMalloc(&d_hist, (uint64_t)nb * nch * sizeof(uint32_t)));
    if (hash_version == 2)
        hipLaunchKernelGGL(k_partition_hist<2>, dim3(nb), dim3(BLOCK), 0, s->stream,
                           (const uint32_t*)keys->ptr, n, tile, nch, d_hist);
    else if (hash_version == 1)
        hipLaunchKernelGGL(k_partition_hist<1>, dim3(nb), dim3(BLOCK), 0, s->stream,
                           (const uint32_t*)keys->ptr, n, tile, nch, d_hist);
    else
        hipLaunchKernelGGL(k_partition_hist<0>, dim3(nb), dim3(BLOCK), 0, s->stream,
                           (const uint32_t*)keys->ptr, n, tile, nch, d_hist);
    // host scan in channel-major order -> per-(block,channel) start offsets
    uint32_t* h_hist = (uint32_t*)malloc((uint64_t)nb * nch * sizeof(uint32_t));
    uint64_t* h_off = (uint64_t*)malloc((uint64_t)nb * nch * sizeof(uint64_t));
    HIP_CHECK(hipMemcpyAsync(h_hist, d_hist, (uint64_t)nb * nch * sizeof(uint32_t),
                             hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    uint64_t acc = 0;
    for (uint32_t c = 0; c < nch; c++) {
        start_points_out[c] = acc;
        for (uint32_t b = 0; b < nb; b++) {
            h_off[(uint64_t)b * nch + c] = acc;
            acc += h_hist[(uint64_t)b * nch + c];
        }
    }
    start_points_out[nch] = acc;
    HIP_CHECK(hipMalloc(&d_off, (uint64_t)nb * nch * sizeof(uint64_t)));
    HIP_CHECK(hipMemcpyAsync(d_off, h_off, (uint64_t)nb * nch * sizeof(uint64_t),
                             hipMemcpyHostToDevice, s->stream));
    if (hash_version == 2)
        hipLaunchKernelGGL(k_partition_emit<2>, dim3(nb), dim3(BLOCK), 0, s->stream,
                           (const uint32_t*)keys->ptr, n, tile, nch, d_off,
                           (uint32_t*)row_indexes_out->ptr);
    else if (hash_version == 1)
        hipLaunchKernelGGL(k_partition_emit<1>, dim3(nb), dim3(BLOCK), 0, s->stream,
                           (const uint32_t*)keys->ptr, n, tile, nch, d_off,
                           (uint32_t*)row_indexes_out->ptr);
    else
        hipLaunchKernelGGL(k_partition_emit<0>, dim3(nb), dim3(BLOCK), 0, s->stream,
                           (const uint32_t*)keys->ptr, n, tile, nch, d_off,
                           (uint32_t*)row_indexes_out->ptr);
    HIP_CHECK(hipStreamSynchronize(s->stream));
    (void)hipFree(d_hist);
    (void)hipFree(d_off);
    free(h_hist);
    free(h_off);
    return GPUE_OK;
}

int gpue_partition_i32(gpue_session* s, gpue_dbuf* keys, uint64_t n, uint32_t nch,
                       uint64_t* start_points_out, gpue_dbuf* row_indexes_out) {
    return partition_i32_impl(s, keys, n, nch, start_points_out, row_indexes_out, 0);
}

// Device-side channel-major exclusive scan of the per-(block,channel)
// histogram — replaces partition's host scan+sync for steady-state steps
// where the caller already knows the split sizes (they are static per
// shard): the async form enqueues hist -> scan -> emit with NO host
// round-trip, so the chunked-exchange pipeline never stalls the host.
__global__ void k_partition_scan(const uint32_t* __restrict__ hist, uint32_t nb,
                                 uint32_t nch, uint64_t* __restrict__ offsets) {
    if (blockIdx.x != 0 || threadIdx.x != 0) return;
    uint64_t acc = 0;
    for (uint32_t c = 0; c < nch; c++)
        for (uint32_t b = 0; b < nb; b++) {
            uint64_t idx = (uint64_t)b * nch + c;
            offsets[idx] = acc;
            acc += hist[idx];
        }
}

template <int HV>
static int partition_i32_async_impl(gpue_session* s, gpue_dbuf* keys, uint64_t n,
                                    uint32_t nch, gpue_dbuf* row_indexes_out,
                                    gpue_dbuf* scratch) {
    ARG_CHECK(s && keys && row_indexes_out && scratch);
    ARG_CHECK(nch >= 1 && nch <= MAX_CH);
    ARG_CHECK(keys->bytes >= n * 4 && row_indexes_out->bytes >= n * 4);
    uint32_t nb = grid_for(n);
    uint64_t tile = (n + nb - 1) / nb;
    // scratch holds hist (u32) + offsets (u64) per (block, channel)
    ARG_CHECK(scratch->bytes >= (uint64_t)nb * nch * 12);
    uint32_t* d_hist = (uint32_t*)scratch->ptr;
    uint64_t* d_off = (uint64_t*)((uint8_t*)scratch->ptr + (uint64_t)nb * nch * 4);
    hipLaunchKernelGGL(k_partition_hist<HV>, dim3(nb), dim3(BLOCK), 0, s->stream,
                       (const uint32_t*)keys->ptr, n, tile, nch, d_hist);
    hipLaunchKernelGGL(k_partition_scan, dim3(1), dim3(64), 0, s->stream, d_hist, nb, nch,
                       d_off);
    hipLaunchKernelGGL(k_partition_emit<HV>, dim3(nb), dim3(BLOCK), 0, s->stream,
                       (const uint32_t*)keys->ptr, n, tile, nch, d_off,
                       (uint32_t*)row_indexes_out->ptr);
    HIP_CHECK(hipGetLastError());
    return GPUE_OK;
}

extern "C" int gpue_partition_i32_async(gpue_session* s, gpue_dbuf* keys, uint64_t n,
                                        uint32_t nch, gpue_dbuf* row_indexes_out,
                                        gpue_dbuf* scratch);
int gpue_partition_i32_async(gpue_session* s, gpue_dbuf* keys, uint64_t n, uint32_t nch,
                             gpue_dbuf* row_indexes_out, gpue_dbuf* scratch) {
    return partition_i32_async_impl<0>(s, keys, n, nch, row_indexes_out, scratch);
}

// version-1 exchange hash (xxh3) partition — exchange_sink_operator.cpp:
// 604-610's `_exchange_hash_function_version == 1` branch
extern "C" int gpue_partition_xxh3_i32(gpue_session* s, gpue_dbuf* keys, uint64_t n,
                                       uint32_t nch, uint64_t* start_points_out,
                                       gpue_dbuf* row_indexes_out);
int gpue_partition_xxh3_i32(gpue_session* s, gpue_dbuf* keys, uint64_t n, uint32_t nch,
                            uint64_t* start_points_out, gpue_dbuf* row_indexes_out) {
    return partition_i32_impl(s, keys, n, nch, start_points_out, row_indexes_out, 1);
}

// bucket-shuffle hash path (zlib crc32, seed 0) — the third exchange hash
// (exchange_sink_operator.cpp:617-622)
extern "C" int gpue_partition_crc_i32(gpue_session* s, gpue_dbuf* keys, uint64_t n,
                                      uint32_t nch, uint64_t* start_points_out,
                                      gpue_dbuf* row_indexes_out);
int gpue_partition_crc_i32(gpue_session* s, gpue_dbuf* keys, uint64_t n, uint32_t nch,
                           uint64_t* start_points_out, gpue_dbuf* row_indexes_out) {
    return partition_i32_impl(s, keys, n, nch, start_points_out, row_indexes_out, 2);
}

// varchar (BinaryColumn) partition key: the exchange hashes the slice bytes
// (fnv_hash over BinaryColumn, FNV_SEED) — same counting-sort layout
__global__ void k_partition_hist_vc(const uint8_t* __restrict__ bytes,
                                    const uint32_t* __restrict__ offsets, uint64_t n,
                                    uint64_t tile, uint32_t nch,
                                    uint32_t* __restrict__ block_hist) {
    __shared__ uint32_t h[MAX_CH];
    for (uint32_t c = threadIdx.x; c < nch; c += blockDim.x) h[c] = 0;
    __syncthreads();
    uint64_t lo = (uint64_t)blockIdx.x * tile, hi = min(lo + tile, n);
    for (uint64_t i = lo + threadIdx.x; i < hi; i += blockDim.x) {
        uint32_t hash = 0x811C9DC5u;
        for (uint32_t b = offsets[i]; b < offsets[i + 1]; b++)
            hash = (bytes[b] ^ hash) * 16777619u;
        uint32_t ch = (uint32_t)(((uint64_t)hash * nch) >> 32);
        atomicAdd(&h[ch], 1u);
    }
    __syncthreads();
    for (uint32_t c = threadIdx.x; c < nch; c += blockDim.x)
        block_hist[(uint64_t)blockIdx.x * nch + c] = h[c];
}

__global__ void k_partition_emit_vc(const uint8_t* __restrict__ bytes,
                                    const uint32_t* __restrict__ offsets, uint64_t n,
                                    uint64_t tile, uint32_t nch,
                                    const uint64_t* __restrict__ block_offsets,
                                    uint32_t* __restrict__ row_indexes) {
    __shared__ uint64_t cursor[MAX_CH];
    for (uint32_t c = threadIdx.x; c < nch; c += blockDim.x)
        cursor[c] = block_offsets[(uint64_t)blockIdx.x * nch + c];
    __syncthreads();
    uint64_t lo = (uint64_t)blockIdx.x * tile, hi = min(lo + tile, n);
    for (uint64_t i = lo + threadIdx.x; i < hi; i += blockDim.x) {
        uint32_t hash = 0x811C9DC5u;
        for (uint32_t b = offsets[i]; b < offsets[i + 1]; b++)
            hash = (bytes[b] ^ hash) * 16777619u;
        uint32_t ch = (uint32_t)(((uint64_t)hash * nch) >> 32);
        uint64_t pos = atomicAdd((unsigned long long*)&cursor[ch], 1ull);
        row_indexes[pos] = (uint32_t)i;
    }
}

extern "C" int gpue_partition_varchar(gpue_session* s, gpue_dbuf* bytes, gpue_dbuf* offsets,
                                      uint64_t n, uint32_t nch, uint64_t* start_points_out,
                                      gpue_dbuf* row_indexes_out);
int gpue_partition_varchar(gpue_session* s, gpue_dbuf* bytes, gpue_dbuf* offsets,
                           uint64_t n, uint32_t nch, uint64_t* start_points_out,
                           gpue_dbuf* row_indexes_out) {
    ARG_CHECK(s && bytes && offsets && start_points_out && row_indexes_out);
    ARG_CHECK(nch >= 1 && nch <= MAX_CH);
    ARG_CHECK(offsets->bytes >= (n + 1) * 4 && row_indexes_out->bytes >= n * 4);
    uint32_t nb = grid_for(n);
    uint64_t tile = (n + nb - 1) / nb;
    uint32_t* d_hist = nullptr;
    uint64_t* d_off = nullptr;
    HIP_CHECK(hipMalloc(&d_hist, (uint64_t)nb * nch * sizeof(uint32_t)));
    hipLaunchKernelGGL(k_partition_hist_vc, dim3(nb), dim3(BLOCK), 0, s->stream,
                       (const uint8_t*)bytes->ptr, (const uint32_t*)offsets->ptr, n, tile,
                       nch, d_hist);
    uint32_t* h_hist = (uint32_t*)malloc((uint64_t)nb * nch * sizeof(uint32_t));
    uint64_t* h_off = (uint64_t*)malloc((uint64_t)nb * nch * sizeof(uint64_t));
    HIP_CHECK(hipMemcpyAsync(h_hist, d_hist, (uint64_t)nb * nch * sizeof(uint32_t),
                             hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    uint64_t acc = 0;
    for (uint32_t c = 0; c < nch; c++) {
        start_points_out[c] = acc;
        for (uint32_t b = 0; b < nb; b++) {
            h_off[(uint64_t)b * nch + c] = acc;
            acc += h_hist[(uint64_t)b * nch + c];
        }
    }
    start_points_out[nch] = acc;
    HIP_CHECK(hipMalloc(&d_off, (uint64_t)nb * nch * sizeof(uint64_t)));
    HIP_CHECK(hipMemcpyAsync(d_off, h_off, (uint64_t)nb * nch * sizeof(uint64_t),
                             hipMemcpyHostToDevice, s->stream));
    hipLaunchKernelGGL(k_partition_emit_vc, dim3(nb), dim3(BLOCK), 0, s->stream,
                       (const uint8_t*)bytes->ptr, (const uint32_t*)offsets->ptr, n, tile,
                       nch, d_off, (uint32_t*)row_indexes_out->ptr);
    HIP_CHECK(hipStreamSynchronize(s->stream));
    (void)hipFree(d_hist);
    (void)hipFree(d_off);
    free(h_hist);
    free(h_off);
    return GPUE_OK;
}


extern "C" int gpue_partition_2xi32(gpue_session* s, gpue_dbuf* a, gpue_dbuf* b,
                                    uint64_t n, uint32_t nch, uint64_t* start_points_out,
                                    gpue_dbuf* row_indexes_out);
int gpue_partition_2xi32(gpue_session* s, gpue_dbuf* a, gpue_dbuf* b, uint64_t n,
                         uint32_t nch, uint64_t* start_points_out,
                         gpue_dbuf* row_indexes_out) {
    ARG_CHECK(s && a && b && start_points_out && row_indexes_out);
    ARG_CHECK(nch >= 1 && nch <= MAX_CH);
    ARG_CHECK(a->bytes >= n * 4 && b->bytes >= n * 4 && row_indexes_out->bytes >= n * 4);
    uint32_t nb = grid_for(n);
    uint64_t tile = (n + nb - 1) / nb;
    uint32_t* d_hist = nullptr;
    uint64_t* d_off = nullptr;
    HIP_CHECK(hipMalloc(&d_hist, (uint64_t)nb * nch * sizeof(uint32_t)));
    hipLaunchKernelGGL(k_partition_hist_2xi32, dim3(nb), dim3(BLOCK), 0, s->stream,
                       (const int32_t*)a->ptr, (const int32_t*)b->ptr, n, tile, nch, d_hist);
    uint32_t* h_hist = (uint32_t*)malloc((uint64_t)nb * nch * sizeof(uint32_t));
    uint64_t* h_off = (uint64_t*)malloc((uint64_t)nb * nch * sizeof(uint64_t));
    HIP_CHECK(hipMemcpyAsync(h_hist, d_hist, (uint64_t)nb * nch * sizeof(uint32_t),
                             hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    uint64_t acc = 0;
    for (uint32_t c = 0; c < nch; c++) {
        start_points_out[c] = acc;
        for (uint32_t blk = 0; blk < nb; blk++) {
            h_off[(uint64_t)blk * nch + c] = acc;
            acc += h_hist[(uint64_t)blk * nch + c];
        }
    }
    start_points_out[nch] = acc;
    HIP_CHECK(hipMalloc(&d_off, (uint64_t)nb * nch * sizeof(uint64_t)));
    HIP_CHECK(hipMemcpyAsync(d_off, h_off, (uint64_t)nb * nch * sizeof(uint64_t),
                             hipMemcpyHostToDevice, s->stream));
    hipLaunchKernelGGL(k_partition_emit_2xi32, dim3(nb), dim3(BLOCK), 0, s->stream,
                       (const int32_t*)a->ptr, (const int32_t*)b->ptr, n, tile, nch, d_off,
                       (uint32_t*)row_indexes_out->ptr);
    HIP_CHECK(hipStreamSynchronize(s->stream));
    (void)hipFree(d_hist);
    (void)hipFree(d_off);
    free(h_hist);
    free(h_off);
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// streaming microbenchmarks — establish the achievable ceilings the fused
// kernels are judged against (read via gpue_ubench from tools/)
// ---------------------------------------------------------------------------
__global__ void k_ub_sum3(const int4* __restrict__ a, const int4* __restrict__ b,
                          const int4* __restrict__ c, uint64_t n4,
                          unsigned long long* __restrict__ out) {
    int64_t sum = 0;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
        int4 x = a[i], y = b[i], z = c[i];
        sum += (int64_t)x.x + x.y + x.z + x.w + y.x + y.y + y.z + y.w
             + (int64_t)z.x + z.y + z.z + z.w;
    }
    for (int off = WAVE / 2; off > 0; off >>= 1)
        sum += __shfl_down((long long)sum, off, WAVE);
    if ((threadIdx.x & (WAVE - 1)) == 0) atomicAdd(out, (unsigned long long)sum);
}

__global__ void k_ub_sum1(const int4* __restrict__ a, uint64_t n4,
                          unsigned long long* __restrict__ out) {
    int64_t sum = 0;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
        int4 x = a[i];
        sum += (int64_t)x.x + x.y + x.z + x.w;
    }
    for (int off = WAVE / 2; off > 0; off >>= 1)
        sum += __shfl_down((long long)sum, off, WAVE);
    if ((threadIdx.x & (WAVE - 1)) == 0) atomicAdd(out, (unsigned long long)sum);
}

// q21 decomposition probes: phase-1 only (pk stream + part-bitset gather)
// and streams-only (4 columns, no probe) — isolates which resource bounds
// the star-agg kernels
__global__ __launch_bounds__(BLOCK_Q21) void
k_ub_q21_phase1(const int4* __restrict__ pk4, uint64_t n4,
                const uint32_t* __restrict__ pbits, int64_t psmin, uint64_t psint,
                unsigned long long* __restrict__ out) {
    uint64_t acc = 0;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    for (; i + stride < n4; i += 2 * stride) {
        int4 pa = pk4[i];
        int4 pb_ = pk4[i + stride];
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            uint32_t idx = (uint32_t)((&pa.x)[j] - psmin);
            uint32_t cidx = idx < psint ? idx : 0u;
            acc += pbits[cidx >> 5] >> (cidx & 31) & 1u;
        }
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            uint32_t idx = (uint32_t)((&pb_.x)[j] - psmin);
            uint32_t cidx = idx < psint ? idx : 0u;
            acc += pbits[cidx >> 5] >> (cidx & 31) & 1u;
        }
    }
    for (; i < n4; i += stride) {
        int4 p4 = pk4[i];
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            uint32_t idx = (uint32_t)((&p4.x)[j] - psmin);
            uint32_t cidx = idx < psint ? idx : 0u;
            acc += pbits[cidx >> 5] >> (cidx & 31) & 1u;
        }
    }
    for (int off = WAVE / 2; off > 0; off >>= 1)
        acc += __shfl_down((unsigned long long)acc, off, WAVE);
    if ((threadIdx.x & (WAVE - 1)) == 0) atomicAdd(out, (unsigned long long)acc);
}

__global__ __launch_bounds__(BLOCK_Q21) void
k_ub_sum4(const int4* __restrict__ a, const int4* __restrict__ b,
          const int4* __restrict__ c, const int4* __restrict__ d, uint64_t n4,
          unsigned long long* __restrict__ out) {
    int64_t sum = 0;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
        int4 x = a[i], y = b[i], z = c[i], w = d[i];
        sum += (int64_t)x.x + x.y + x.z + x.w + y.x + y.y + y.z + y.w +
               (int64_t)z.x + z.y + z.z + z.w + w.x + w.y + w.z + w.w;
    }
    for (int off = WAVE / 2; off > 0; off >>= 1)
        sum += __shfl_down((long long)sum, off, WAVE);
    if ((threadIdx.x & (WAVE - 1)) == 0) atomicAdd(out, (unsigned long long)sum);
}

// PF-kernel decomposition legs (DESIGN.md §4d residual): same 120 KB LDS /
// 1024-thread / 1-block-per-CU geometry as k_q21_star_agg_pf, with phase 2
// removed. which=2: the 4 NT streams only; which=3: + the per-key LDS
// prefilter test (maybe-count accumulated to prevent DCE).
template <bool LDS_TEST>
__global__ __launch_bounds__(BLOCK_Q21) void
k_ub_q21_pf_legs(const int32_t* __restrict__ pk, const int32_t* __restrict__ sk,
                 const int32_t* __restrict__ od, const int32_t* __restrict__ rv,
                 uint64_t n, const uint32_t* __restrict__ prefilter, int64_t psmin,
                 uint64_t psint, unsigned long long* __restrict__ sink) {
    __shared__ uint32_t pf[PF_WORDS];
    __shared__ unsigned long long g[NG_Q21];
    for (uint32_t w = threadIdx.x; w < PF_WORDS; w += blockDim.x)
        pf[w] = prefilter ? prefilter[w] : 0u;
    for (int j = threadIdx.x; j < NG_Q21; j += blockDim.x) g[j] = 0;
    __syncthreads();
    const uint64_t n4 = n / 4;
    const int4* __restrict__ pk4 = (const int4*)pk;
    const int4* __restrict__ sk4 = (const int4*)sk;
    const int4* __restrict__ od4 = (const int4*)od;
    const int4* __restrict__ rv4 = (const int4*)rv;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    unsigned long long acc = 0;
    auto ld4 = [&](const int4* p, uint64_t i) {
        const uint64_t* q = (const uint64_t*)(p + i);
        uint64_t lo = __builtin_nontemporal_load(q);
        uint64_t hi = __builtin_nontemporal_load(q + 1);
        int4 v;
        v.x = (int32_t)lo; v.y = (int32_t)(lo >> 32);
        v.z = (int32_t)hi; v.w = (int32_t)(hi >> 32);
        return v;
    };
    auto quad = [&](int4 p4, int4 s4, int4 o4, int4 r4) {
        if (LDS_TEST) {
            #pragma unroll
            for (int j = 0; j < 4; j++) {
                uint32_t idx = (uint32_t)((&p4.x)[j] - psmin);
                bool in = idx < psint;
                uint32_t fidx = (in ? idx : 0u) & PF_MASK;
                acc += in & (pf[fidx >> 5] >> (fidx & 31)) & 1u;
            }
            acc += (unsigned long long)(uint32_t)(s4.x + o4.x + r4.x);
        } else {
            acc += (unsigned long long)(uint32_t)(p4.x + s4.x + o4.x + r4.x);
        }
    };
    uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    for (; i + stride < n4; i += 2 * stride) {
        int4 pa = ld4(pk4, i), sa = ld4(sk4, i), oa = ld4(od4, i), ra = ld4(rv4, i);
        uint64_t i2 = i + stride;
        int4 pb_ = ld4(pk4, i2), sb = ld4(sk4, i2), ob = ld4(od4, i2), rb = ld4(rv4, i2);
        quad(pa, sa, oa, ra);
        quad(pb_, sb, ob, rb);
    }
    for (; i < n4; i += stride) quad(pk4[i], sk4[i], od4[i], rv4[i]);
    for (int off = WAVE / 2; off > 0; off >>= 1)
        acc += __shfl_down(acc, off, WAVE);
    if ((threadIdx.x & (WAVE - 1)) == 0 && acc) atomicAdd(sink, acc);
    if (g[threadIdx.x & (NG_Q21 - 1)] > ~0ull - 1) sink[0] = 0; // keep g live
}

extern "C" int gpue_ubench_q21(gpue_session* s, int which, gpue_dbuf* pk, gpue_dbuf* sk,
                               gpue_dbuf* od, gpue_dbuf* rv, gpue_dbuf* pbits,
                               int64_t psmin, uint64_t psint, uint64_t n, int grid1024,
                               int reps, float* ms_out);
int gpue_ubench_q21(gpue_session* s, int which, gpue_dbuf* pk, gpue_dbuf* sk, gpue_dbuf* od,
                    gpue_dbuf* rv, gpue_dbuf* pbits, int64_t psmin, uint64_t psint,
                    uint64_t n, int grid1024, int reps, float* ms_out) {
    ARG_CHECK(s && pk && ms_out && reps > 0);
    uint64_t n4 = n / 4;
    unsigned long long* d_out = nullptr;
    HIP_CHECK(hipMalloc(&d_out, 8));
    HIP_CHECK(hipMemsetAsync(d_out, 0, 8, s->stream));
    for (int pass = 0; pass < 2; pass++) {
        if (pass == 1) HIP_CHECK(hipEventRecord(s->ev_start, s->stream));
        int iters = pass == 0 ? 1 : reps;
        for (int r = 0; r < iters; r++) {
            if (which == 0) {
                ARG_CHECK(pbits);
                hipLaunchKernelGGL(k_ub_q21_phase1, dim3(grid1024), dim3(BLOCK_Q21), 0,
                                   s->stream, (const int4*)pk->ptr, n4, (const uint32_t*)pbits->ptr,
                                   psmin, psint, d_out);
            } else if (which == 2 || which == 3) {
                ARG_CHECK(sk && od && rv);
                auto kern = which == 3 ? k_ub_q21_pf_legs<true> : k_ub_q21_pf_legs<false>;
                hipLaunchKernelGGL(kern, dim3(grid1024), dim3(BLOCK_Q21), 0, s->stream,
                                   (const int32_t*)pk->ptr, (const int32_t*)sk->ptr,
                                   (const int32_t*)od->ptr, (const int32_t*)rv->ptr, n,
                                   pbits ? (const uint32_t*)pbits->ptr : nullptr, psmin,
                                   psint, d_out);
            } else {
                ARG_CHECK(sk && od && rv);
                hipLaunchKernelGGL(k_ub_sum4, dim3(grid1024), dim3(BLOCK_Q21), 0, s->stream,
                                   (const int4*)pk->ptr, (const int4*)sk->ptr,
                                   (const int4*)od->ptr, (const int4*)rv->ptr, n4, d_out);
            }
        }
    }
    HIP_CHECK(hipEventRecord(s->ev_stop, s->stream));
    HIP_CHECK(hipEventSynchronize(s->ev_stop));
    float ms = 0;
    HIP_CHECK(hipEventElapsedTime(&ms, s->ev_start, s->ev_stop));
    (void)hipFree(d_out);
    *ms_out = ms / reps;
    return GPUE_OK;
}

// Random 1-bit gather microbench (VERDICT r01 weak #3 evidence request):
// n random word-probes into a bitset of the given footprint, int4-quad form
// (4 gathers in flight per lane — the exact access pattern of
// k_q3_probe_agg's order-bits leg). Reports the achieved probes/s so the
// q3 kernel's rate can be compared against the raw architectural rate at
// the same footprint.
__global__ void k_ub_bitgather(const uint32_t* __restrict__ idx, uint64_t n,
                               const uint32_t* __restrict__ bits, uint64_t nbits_mask,
                               unsigned long long* __restrict__ sink) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    unsigned long long acc = 0;
    const uint64_t n4 = n / 4;
    const uint4* __restrict__ idx4 = (const uint4*)idx;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
        uint4 q = idx4[i];
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            uint64_t b = (&q.x)[j] & nbits_mask;
            acc += (bits[b >> 5] >> (b & 31)) & 1u;
        }
    }
    for (int off = WAVE / 2; off > 0; off >>= 1)
        acc += __shfl_down(acc, off, WAVE);
    if ((threadIdx.x & (WAVE - 1)) == 0 && acc) atomicAdd(sink, acc);
}

extern "C" int gpue_ubench_bitgather(gpue_session* s, gpue_dbuf* idx, uint64_t n,
                                     gpue_dbuf* bits, uint64_t nbits_pow2, int reps,
                                     float* ms_out);
int gpue_ubench_bitgather(gpue_session* s, gpue_dbuf* idx, uint64_t n, gpue_dbuf* bits,
                          uint64_t nbits_pow2, int reps, float* ms_out) {
    ARG_CHECK(s && idx && bits && ms_out && reps > 0);
    ARG_CHECK(idx->bytes >= n * 4 && bits->bytes >= nbits_pow2 / 8);
    unsigned long long* d_sink = nullptr;
    HIP_CHECK(hipMalloc(&d_sink, 8));
    HIP_CHECK(hipMemsetAsync(d_sink, 0, 8, s->stream));
    hipLaunchKernelGGL(k_ub_bitgather, dim3(grid_capped(n / 4, MAX_GRID)), dim3(BLOCK), 0,
                       s->stream, (const uint32_t*)idx->ptr, n, (const uint32_t*)bits->ptr,
                       nbits_pow2 - 1, d_sink); // warm
    HIP_CHECK(hipEventRecord(s->ev_start, s->stream));
    for (int r = 0; r < reps; r++)
        hipLaunchKernelGGL(k_ub_bitgather, dim3(grid_capped(n / 4, MAX_GRID)), dim3(BLOCK),
                           0, s->stream, (const uint32_t*)idx->ptr, n,
                           (const uint32_t*)bits->ptr, nbits_pow2 - 1, d_sink);
    HIP_CHECK(hipEventRecord(s->ev_stop, s->stream));
    HIP_CHECK(hipEventSynchronize(s->ev_stop));
    float ms = 0;
    HIP_CHECK(hipEventElapsedTime(&ms, s->ev_start, s->ev_stop));
    (void)hipFree(d_sink);
    *ms_out = ms / reps;
    return GPUE_OK;
}

extern "C" int gpue_ubench(gpue_session* s, int which, gpue_dbuf* a, gpue_dbuf* b,
                           gpue_dbuf* c, uint64_t n_i32, int reps, float* ms_out);
int gpue_ubench(gpue_session* s, int which, gpue_dbuf* a, gpue_dbuf* b, gpue_dbuf* c,
                uint64_t n_i32, int reps, float* ms_out) {
    ARG_CHECK(s && a && ms_out && reps > 0);
    uint64_t n4 = n_i32 / 4;
    unsigned long long* d_out = nullptr;
    HIP_CHECK(hipMalloc(&d_out, sizeof(unsigned long long)));
    HIP_CHECK(hipMemsetAsync(d_out, 0, sizeof(unsigned long long), s->stream));
    // one untimed warm launch
    for (int pass = 0; pass < 2; pass++) {
        if (pass == 1) HIP_CHECK(hipEventRecord(s->ev_start, s->stream));
        int iters = pass == 0 ? 1 : reps;
        for (int r = 0; r < iters; r++) {
            if (which == 0) {
                hipLaunchKernelGGL(k_ub_sum1, dim3(grid_stream(n4)), dim3(BLOCK), 0, s->stream,
                                   (const int4*)a->ptr, n4, d_out);
            } else {
                ARG_CHECK(b && c);
                hipLaunchKernelGGL(k_ub_sum3, dim3(grid_stream(n4)), dim3(BLOCK), 0, s->stream,
                                   (const int4*)a->ptr, (const int4*)b->ptr,
                                   (const int4*)c->ptr, n4, d_out);
            }
        }
    }
    HIP_CHECK(hipEventRecord(s->ev_stop, s->stream));
    HIP_CHECK(hipEventSynchronize(s->ev_stop));
    float ms = 0;
    HIP_CHECK(hipEventElapsedTime(&ms, s->ev_start, s->ev_stop));
    (void)hipFree(d_out);
    *ms_out = ms / reps;
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// Generic hash aggregate — the AggHashMapWithKey::compute_agg_states +
// AggregateFunction::update_batch chain (reference be/src/exec/agg_hash_map.h
// :112-290, aggregator.cpp:937-959, exprs/agg/sum.h:45) for HIGH-cardinality
// GROUP BY, where the dense-group LDS shortcut of the SSB kernels does not
// apply. Open-addressing linear probing in HBM; group slots claimed with a
// 64-bit atomicCAS; SUM/COUNT accumulate with atomicAdd (order-independent
// => bit-exact int64). The CPU's lazy_emplace + prefetch pipeline maps to
// one CAS-claim + atomic update per row. Key sentinel: ~0ull.
// ---------------------------------------------------------------------------
static constexpr unsigned long long AGG_EMPTY = 0xFFFFFFFFFFFFFFFFull;

// ADVICE r01 guards: a real group key equal to the empty-slot sentinel is
// latched as an error (the row is skipped — the host fails the call, so the
// partial results never escape); every open-addressing walk is bounded at
// the table capacity and latches "table full" instead of spinning forever.
__device__ static inline bool agg_key_is_sentinel(unsigned long long k,
                                                  unsigned int* err) {
    if (k == AGG_EMPTY) {
        atomicOr(err, AGG_ERR_SENTINEL);
        return true;
    }
    return false;
}

struct gpue_agg_table {
    gpue_session* s;
    uint64_t cap;
    unsigned long long* slots;
    unsigned long long* sums;
    unsigned long long* counts;
    unsigned long long* cursor;
    unsigned long long* n_groups; // device count of claimed slots (grow trigger)
};

extern "C" {
int gpue_agg_table_create(gpue_session* s, uint64_t capacity, gpue_agg_table** out);
void gpue_agg_table_destroy(gpue_agg_table* t);
}

static int agg_table_reset(gpue_agg_table* t, bool with_counts = true);

int gpue_agg_table_create(gpue_session* s, uint64_t capacity, gpue_agg_table** out) {
    ARG_CHECK(s && out && capacity >= 16);
    uint64_t cap = 16;
    while (cap < capacity) cap <<= 1;
    gpue_agg_table* t = new gpue_agg_table{s, cap, nullptr, nullptr, nullptr, nullptr,
                                           nullptr};
    HIP_CHECK(hipMalloc(&t->slots, cap * 8));
    HIP_CHECK(hipMalloc(&t->sums, cap * 8));
    HIP_CHECK(hipMalloc(&t->counts, cap * 8));
    HIP_CHECK(hipMalloc(&t->cursor, 8));
    HIP_CHECK(hipMalloc(&t->n_groups, 8));
    // a fresh table must be READY: pushing into uninitialized slots would
    // walk garbage chains (a degenerate O(n*cap) crawl) and a garbage slot
    // aliasing a real key would silently corrupt that group's sum
    int rc = agg_table_reset(t, true);
    if (rc != GPUE_OK) return rc;
    *out = t;
    return GPUE_OK;
}

extern "C" int gpue_agg_table_reset(gpue_agg_table* t);

void gpue_agg_table_destroy(gpue_agg_table* t) {
    if (!t) return;
    (void)hipFree(t->slots);
    (void)hipFree(t->sums);
    (void)hipFree(t->counts);
    (void)hipFree(t->cursor);
    (void)hipFree(t->n_groups);
    delete t;
}

static int agg_table_reset(gpue_agg_table* t, bool with_counts) {
    HIP_CHECK(hipMemsetAsync(t->slots, 0xFF, t->cap * 8, t->s->stream));
    HIP_CHECK(hipMemsetAsync(t->sums, 0, t->cap * 8, t->s->stream));
    if (with_counts) HIP_CHECK(hipMemsetAsync(t->counts, 0, t->cap * 8, t->s->stream));
    HIP_CHECK(hipMemsetAsync(t->cursor, 0, 8, t->s->stream));
    HIP_CHECK(hipMemsetAsync(t->n_groups, 0, 8, t->s->stream));
    return GPUE_OK;
}

int gpue_agg_table_reset(gpue_agg_table* t) {
    ARG_CHECK(t);
    return agg_table_reset(t);
}

// ---------------------------------------------------------------------------
// Growable table — the MI355X analog of the reference's two-level
// conversion (Aggregator::try_convert_to_two_level_map, aggregator.cpp:
// 1237-1241; AggHashMapVariant::convert_to_two_level, agg_hash_variant.cpp:
// 318): when the map crosses a size threshold mid-stream, the reference
// swaps in a 16-way phmap parallel map to keep rehash/lock costs bounded.
// Our open-addressing CAS table has no locks, so the property to preserve
// is "a chunk push never overflows": callers check before each chunk
// (gpue_agg_table_ensure) and the table doubles + rehashes on device. Keys
// in the old table are unique, so the rehash claim never contends on
// payload writes and the per-slot stores need no atomics.
// ---------------------------------------------------------------------------
__global__ void k_agg_rehash(const unsigned long long* __restrict__ os,
                             const unsigned long long* __restrict__ osum,
                             const unsigned long long* __restrict__ ocnt, uint64_t ocap,
                             unsigned long long* __restrict__ ns,
                             unsigned long long* __restrict__ nsum,
                             unsigned long long* __restrict__ ncnt, uint64_t ncap_mask) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < ocap;
         i += stride) {
        unsigned long long k = os[i];
        if (k == AGG_EMPTY) continue;
        uint64_t s = ((k * 11400714819323198485ull) >> 32) & ncap_mask;
        while (atomicCAS(&ns[s], AGG_EMPTY, k) != AGG_EMPTY) s = (s + 1) & ncap_mask;
        nsum[s] = osum[i];
        ncnt[s] = ocnt[i];
    }
}

extern "C" {
int gpue_agg_table_size(gpue_session* s, gpue_agg_table* t, uint64_t* n_groups);
int gpue_agg_table_ensure(gpue_session* s, gpue_agg_table* t, uint64_t additional_rows);
}

int gpue_agg_table_size(gpue_session* s, gpue_agg_table* t, uint64_t* n_groups) {
    ARG_CHECK(s && t && n_groups);
    unsigned long long g = 0;
    HIP_CHECK(hipMemcpyAsync(&g, t->n_groups, 8, hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    *n_groups = g;
    return GPUE_OK;
}

int gpue_agg_table_ensure(gpue_session* s, gpue_agg_table* t, uint64_t additional_rows) {
    ARG_CHECK(s && t);
    uint64_t groups = 0;
    int rc = gpue_agg_table_size(s, t, &groups);
    if (rc != GPUE_OK) return rc;
    // worst case every incoming row is a new group; keep load factor <= 5/8
    uint64_t need = groups + additional_rows;
    if (need + need / 2 <= t->cap) return GPUE_OK;
    uint64_t ncap = t->cap;
    while (need + need / 2 > ncap) ncap <<= 1;
    unsigned long long *ns = nullptr, *nsum = nullptr, *ncnt = nullptr;
    HIP_CHECK(hipMalloc(&ns, ncap * 8));
    HIP_CHECK(hipMalloc(&nsum, ncap * 8));
    HIP_CHECK(hipMalloc(&ncnt, ncap * 8));
    HIP_CHECK(hipMemsetAsync(ns, 0xFF, ncap * 8, s->stream));
    HIP_CHECK(hipMemsetAsync(nsum, 0, ncap * 8, s->stream));
    HIP_CHECK(hipMemsetAsync(ncnt, 0, ncap * 8, s->stream));
    hipLaunchKernelGGL(k_agg_rehash, dim3(grid_for(t->cap)), dim3(BLOCK), 0, s->stream,
                       t->slots, t->sums, t->counts, t->cap, ns, nsum, ncnt, ncap - 1);
    HIP_CHECK(hipStreamSynchronize(s->stream));
    (void)hipFree(t->slots);
    (void)hipFree(t->sums);
    (void)hipFree(t->counts);
    t->slots = ns;
    t->sums = nsum;
    t->counts = ncnt;
    t->cap = ncap;
    return GPUE_OK;
}

__global__ void k_hash_agg_sum(const uint64_t* __restrict__ keys,
                               const int64_t* __restrict__ vals, uint64_t n,
                               unsigned long long* __restrict__ slots,
                               unsigned long long* __restrict__ sums,
                               unsigned long long* __restrict__ counts,
                               uint64_t cap_mask, unsigned int* __restrict__ err) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        unsigned long long k = keys[i];
        if (agg_key_is_sentinel(k, err)) continue;
        unsigned long long v = (unsigned long long)vals[i];
        uint64_t s = ((k * 11400714819323198485ull) >> 32) & cap_mask;
        uint64_t left = cap_mask + 1;
        for (;;) {
            unsigned long long cur = slots[s];
            if (cur == k) {
                atomicAdd(&sums[s], v);
                atomicAdd(&counts[s], 1ull);
                break;
            }
            if (cur == AGG_EMPTY) {
                unsigned long long old = atomicCAS(&slots[s], AGG_EMPTY, k);
                if (old == AGG_EMPTY || old == k) {
                    atomicAdd(&sums[s], v);
                    atomicAdd(&counts[s], 1ull);
                    break;
                }
            }
            s = (s + 1) & cap_mask;
            if (--left == 0) { atomicOr(err, AGG_ERR_FULL); break; }
        }
    }
}

// convert_hash_map_to_chunk analog (aggregator.cpp:1742-1816): iterate the
// table, emit keys + finalized states (emission order is table order —
// results are a set, compared key-sorted)
// Full aggregate-function state set (reference exprs/agg/: SUM, COUNT,
// MIN, MAX; AVG = SUM/COUNT at finalize, aggregate.h:136-269) and the
// decimal SUM widening to int128 (exprs/agg/sum.h:181: decimal64 inputs
// accumulate in int128 — here as a lo/hi pair with an explicit carry, two
// 64-bit atomics per update, exact mod 2^128 and order-independent).
__global__ void k_agg_init_minmax(long long* __restrict__ mins, long long* __restrict__ maxs,
                                  uint64_t cap) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t s = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; s < cap; s += stride) {
        mins[s] = 0x7FFFFFFFFFFFFFFFll;
        maxs[s] = 0x8000000000000000ll;
    }
}

__global__ void k_hash_agg_stats(const uint64_t* __restrict__ keys,
                                 const int64_t* __restrict__ vals, uint64_t n,
                                 unsigned long long* __restrict__ slots,
                                 unsigned long long* __restrict__ sums,
                                 unsigned long long* __restrict__ counts,
                                 long long* __restrict__ mins,
                                 long long* __restrict__ maxs, uint64_t cap_mask,
                                 unsigned int* __restrict__ err) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        unsigned long long k = keys[i];
        if (agg_key_is_sentinel(k, err)) continue;
        long long v = (long long)vals[i];
        uint64_t s = ((k * 11400714819323198485ull) >> 32) & cap_mask;
        uint64_t left = cap_mask + 1;
        bool ok = true;
        for (;;) {
            unsigned long long cur = slots[s];
            if (cur == k) break;
            if (cur == AGG_EMPTY) {
                unsigned long long old = atomicCAS(&slots[s], AGG_EMPTY, k);
                if (old == AGG_EMPTY || old == k) break;
            }
            s = (s + 1) & cap_mask;
            if (--left == 0) { atomicOr(err, AGG_ERR_FULL); ok = false; break; }
        }
        if (!ok) continue;
        atomicAdd(&sums[s], (unsigned long long)v);
        atomicAdd(&counts[s], 1ull);
        atomicMin(&mins[s], v);
        atomicMax(&maxs[s], v);
    }
}

__global__ void k_hash_agg_sum128(const uint64_t* __restrict__ keys,
                                  const int64_t* __restrict__ vals, uint64_t n,
                                  unsigned long long* __restrict__ slots,
                                  unsigned long long* __restrict__ lo,
                                  unsigned long long* __restrict__ hi, uint64_t cap_mask,
                                  unsigned int* __restrict__ err) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        unsigned long long k = keys[i];
        if (agg_key_is_sentinel(k, err)) continue;
        long long v = (long long)vals[i];
        uint64_t s = ((k * 11400714819323198485ull) >> 32) & cap_mask;
        uint64_t left = cap_mask + 1;
        bool ok = true;
        for (;;) {
            unsigned long long cur = slots[s];
            if (cur == k) break;
            if (cur == AGG_EMPTY) {
                unsigned long long old = atomicCAS(&slots[s], AGG_EMPTY, k);
                if (old == AGG_EMPTY || old == k) break;
            }
            s = (s + 1) & cap_mask;
            if (--left == 0) { atomicOr(err, AGG_ERR_FULL); ok = false; break; }
        }
        if (!ok) continue;
        unsigned long long vlo = (unsigned long long)v;
        long long vhi = v < 0 ? -1ll : 0ll; // sign extension of the int64 addend
        unsigned long long old_lo = atomicAdd(&lo[s], vlo);
        unsigned long long carry = (old_lo + vlo < old_lo) ? 1ull : 0ull;
        atomicAdd(&hi[s], (unsigned long long)vhi + carry);
    }
}

__global__ void k_hash_agg_emit_wide(const unsigned long long* __restrict__ slots,
                                     const unsigned long long* __restrict__ a,
                                     const unsigned long long* __restrict__ b,
                                     const long long* __restrict__ c,
                                     const long long* __restrict__ d, uint64_t cap,
                                     unsigned long long* __restrict__ cursor,
                                     uint64_t max_out, uint64_t* __restrict__ out_keys,
                                     int64_t* __restrict__ out_a, int64_t* __restrict__ out_b,
                                     int64_t* __restrict__ out_c, int64_t* __restrict__ out_d) {
    // same tiled two-phase structure as k_hash_agg_emit
    uint64_t nb = gridDim.x;
    uint64_t tile = (cap + nb - 1) / nb;
    uint64_t lo_ = (uint64_t)blockIdx.x * tile;
    uint64_t hi_ = min(lo_ + tile, cap);
    int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;
    __shared__ uint64_t wsum[BLOCK / WAVE + 1];
    uint64_t cnum = 0;
    for (uint64_t s = lo_ + threadIdx.x; s < hi_; s += blockDim.x)
        cnum += (slots[s] != AGG_EMPTY);
    for (int off = WAVE / 2; off > 0; off >>= 1)
        cnum += __shfl_down((unsigned long long)cnum, off, WAVE);
    if (lane == 0) wsum[wid] = cnum;
    __syncthreads();
    if (threadIdx.x == 0) {
        uint64_t t = 0;
        for (int w = 0; w < BLOCK / WAVE; w++) t += wsum[w];
        wsum[BLOCK / WAVE] = atomicAdd(cursor, (unsigned long long)t);
    }
    __syncthreads();
    uint64_t offset = wsum[BLOCK / WAVE];
    __shared__ uint64_t wbase[BLOCK / WAVE + 1];
    for (uint64_t base = lo_; base < hi_; base += blockDim.x) {
        uint64_t s = base + threadIdx.x;
        bool has = (s < hi_) && (slots[s] != AGG_EMPTY);
        uint64_t mask = __ballot(has);
        uint32_t wcount = __popcll(mask);
        if (lane == 0) wbase[wid] = wcount;
        __syncthreads();
        if (threadIdx.x == 0) {
            uint64_t acc = offset;
            for (int w = 0; w < BLOCK / WAVE; w++) {
                uint64_t v = wbase[w];
                wbase[w] = acc;
                acc += v;
            }
            wbase[BLOCK / WAVE] = acc;
        }
        __syncthreads();
        if (has) {
            uint64_t pos = wbase[wid] + __popcll(mask & ((1ull << lane) - 1));
            if (pos < max_out) {
                out_keys[pos] = slots[s];
                out_a[pos] = (int64_t)a[s];
                if (out_b) out_b[pos] = (int64_t)b[s];
                if (out_c) out_c[pos] = (int64_t)c[s];
                if (out_d) out_d[pos] = (int64_t)d[s];
            }
        }
        offset = wbase[BLOCK / WAVE];
        __syncthreads();
    }
}

extern "C" {
int gpue_hash_agg_stats_u64(gpue_session* s, gpue_dbuf* keys, gpue_dbuf* vals, uint64_t n,
                            uint64_t capacity_hint, gpue_dbuf* out_keys, gpue_dbuf* out_sums,
                            gpue_dbuf* out_counts, gpue_dbuf* out_mins, gpue_dbuf* out_maxs,
                            uint64_t max_out, uint64_t* n_groups);
int gpue_hash_agg_sum128_u64(gpue_session* s, gpue_dbuf* keys, gpue_dbuf* vals, uint64_t n,
                             uint64_t capacity_hint, gpue_dbuf* out_keys, gpue_dbuf* out_lo,
                             gpue_dbuf* out_hi, uint64_t max_out, uint64_t* n_groups);
}

int gpue_hash_agg_stats_u64(gpue_session* s, gpue_dbuf* keys, gpue_dbuf* vals, uint64_t n,
                            uint64_t capacity_hint, gpue_dbuf* out_keys, gpue_dbuf* out_sums,
                            gpue_dbuf* out_counts, gpue_dbuf* out_mins, gpue_dbuf* out_maxs,
                            uint64_t max_out, uint64_t* n_groups) {
    ARG_CHECK(s && keys && vals && out_keys && out_sums && out_counts && out_mins &&
              out_maxs && n_groups);
    uint64_t cap = 16;
    uint64_t want = capacity_hint ? capacity_hint : n * 2;
    while (cap < want) cap <<= 1;
    unsigned long long *d_slots = nullptr, *d_sums = nullptr, *d_counts = nullptr,
                       *d_cursor = nullptr;
    long long *d_mins = nullptr, *d_maxs = nullptr;
    HIP_CHECK(hipMalloc(&d_slots, cap * 8));
    HIP_CHECK(hipMalloc(&d_sums, cap * 8));
    HIP_CHECK(hipMalloc(&d_counts, cap * 8));
    HIP_CHECK(hipMalloc(&d_mins, cap * 8));
    HIP_CHECK(hipMalloc(&d_maxs, cap * 8));
    HIP_CHECK(hipMalloc(&d_cursor, 8));
    HIP_CHECK(hipMemsetAsync(d_slots, 0xFF, cap * 8, s->stream));
    HIP_CHECK(hipMemsetAsync(d_sums, 0, cap * 8, s->stream));
    HIP_CHECK(hipMemsetAsync(d_counts, 0, cap * 8, s->stream));
    HIP_CHECK(hipMemsetAsync(d_cursor, 0, 8, s->stream));
    hipLaunchKernelGGL(k_agg_init_minmax, dim3(grid_for(cap)), dim3(BLOCK), 0, s->stream,
                       d_mins, d_maxs, cap);
    hipLaunchKernelGGL(k_hash_agg_stats, dim3(grid_for(n)), dim3(BLOCK), 0, s->stream,
                       (const uint64_t*)keys->ptr, (const int64_t*)vals->ptr, n, d_slots,
                       d_sums, d_counts, d_mins, d_maxs, cap - 1, s->d_agg_err);
    hipLaunchKernelGGL(k_hash_agg_emit_wide, dim3(grid_for(cap)), dim3(BLOCK), 0, s->stream,
                       d_slots, d_sums, d_counts, d_mins, d_maxs, cap, d_cursor, max_out,
                       (uint64_t*)out_keys->ptr, (int64_t*)out_sums->ptr,
                       (int64_t*)out_counts->ptr, (int64_t*)out_mins->ptr,
                       (int64_t*)out_maxs->ptr);
    unsigned long long groups = 0;
    HIP_CHECK(hipMemcpyAsync(&groups, d_cursor, 8, hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    (void)hipFree(d_slots); (void)hipFree(d_sums); (void)hipFree(d_counts);
    (void)hipFree(d_mins); (void)hipFree(d_maxs); (void)hipFree(d_cursor);
    *n_groups = groups;
    int erc = agg_err_check(s, "agg_stats");
    if (erc != GPUE_OK) return erc;
    if (groups > max_out) {
        snprintf(g_err, sizeof(g_err), "agg_stats: %llu groups exceed max_out", groups);
        return GPUE_ERR_ARG;
    }
    return GPUE_OK;
}

int gpue_hash_agg_sum128_u64(gpue_session* s, gpue_dbuf* keys, gpue_dbuf* vals, uint64_t n,
                             uint64_t capacity_hint, gpue_dbuf* out_keys, gpue_dbuf* out_lo,
                             gpue_dbuf* out_hi, uint64_t max_out, uint64_t* n_groups) {
    ARG_CHECK(s && keys && vals && out_keys && out_lo && out_hi && n_groups);
    uint64_t cap = 16;
    uint64_t want = capacity_hint ? capacity_hint : n * 2;
    while (cap < want) cap <<= 1;
    unsigned long long *d_slots = nullptr, *d_lo = nullptr, *d_hi = nullptr,
                       *d_cursor = nullptr;
    HIP_CHECK(hipMalloc(&d_slots, cap * 8));
    HIP_CHECK(hipMalloc(&d_lo, cap * 8));
    HIP_CHECK(hipMalloc(&d_hi, cap * 8));
    HIP_CHECK(hipMalloc(&d_cursor, 8));
    HIP_CHECK(hipMemsetAsync(d_slots, 0xFF, cap * 8, s->stream));
    HIP_CHECK(hipMemsetAsync(d_lo, 0, cap * 8, s->stream));
    HIP_CHECK(hipMemsetAsync(d_hi, 0, cap * 8, s->stream));
    HIP_CHECK(hipMemsetAsync(d_cursor, 0, 8, s->stream));
    hipLaunchKernelGGL(k_hash_agg_sum128, dim3(grid_for(n)), dim3(BLOCK), 0, s->stream,
                       (const uint64_t*)keys->ptr, (const int64_t*)vals->ptr, n, d_slots,
                       d_lo, d_hi, cap - 1, s->d_agg_err);
    hipLaunchKernelGGL(k_hash_agg_emit_wide, dim3(grid_for(cap)), dim3(BLOCK), 0, s->stream,
                       d_slots, d_lo, d_hi, nullptr, nullptr, cap, d_cursor, max_out,
                       (uint64_t*)out_keys->ptr, (int64_t*)out_lo->ptr,
                       (int64_t*)out_hi->ptr, nullptr, nullptr);
    unsigned long long groups = 0;
    HIP_CHECK(hipMemcpyAsync(&groups, d_cursor, 8, hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    (void)hipFree(d_slots); (void)hipFree(d_lo); (void)hipFree(d_hi); (void)hipFree(d_cursor);
    *n_groups = groups;
    int erc = agg_err_check(s, "agg_sum128");
    if (erc != GPUE_OK) return erc;
    if (groups > max_out) {
        snprintf(g_err, sizeof(g_err), "agg_sum128: %llu groups exceed max_out", groups);
        return GPUE_ERR_ARG;
    }
    return GPUE_OK;
}

// convert_hash_map_to_chunk analog (aggregator.cpp:1742-1816). Tiled
// two-phase: each block counts its contiguous slot tile, takes ONE cursor
// atomic for its base, then emits with ballot/prefix placement. A per-wave
// (or per-slot — HW coalesces same-address wave atomics) returning cursor
// atomic costs a contended-line round trip per wave-iteration: measured
// 12.6 ms for a 67 M-slot table vs ~0.5 ms tiled (profiles q3 traces).
// Emission order is deterministic only per block; results are a set.
__global__ void k_hash_agg_emit(const unsigned long long* __restrict__ slots,
                                const unsigned long long* __restrict__ sums,
                                const unsigned long long* __restrict__ counts,
                                uint64_t cap, unsigned long long* __restrict__ cursor,
                                uint64_t max_out, uint64_t* __restrict__ out_keys,
                                int64_t* __restrict__ out_sums,
                                int64_t* __restrict__ out_counts) {
    uint64_t nb = gridDim.x;
    uint64_t tile = (cap + nb - 1) / nb;
    uint64_t lo = (uint64_t)blockIdx.x * tile;
    uint64_t hi = min(lo + tile, cap);
    int lane = threadIdx.x & (WAVE - 1), wid = threadIdx.x / WAVE;
    __shared__ uint64_t wsum[BLOCK / WAVE + 1];
    // phase 1: count this block's nonempty slots
    uint64_t c = 0;
    for (uint64_t s = lo + threadIdx.x; s < hi; s += blockDim.x)
        c += (slots[s] != AGG_EMPTY);
    for (int off = WAVE / 2; off > 0; off >>= 1)
        c += __shfl_down((unsigned long long)c, off, WAVE);
    if (lane == 0) wsum[wid] = c;
    __syncthreads();
    if (threadIdx.x == 0) {
        uint64_t t = 0;
        for (int w = 0; w < BLOCK / WAVE; w++) t += wsum[w];
        wsum[BLOCK / WAVE] = atomicAdd(cursor, (unsigned long long)t); // block base
    }
    __syncthreads();
    uint64_t offset = wsum[BLOCK / WAVE];
    // phase 2: emit at ballot/prefix positions (same structure as the
    // ordered filter emit)
    __shared__ uint64_t wbase[BLOCK / WAVE + 1];
    for (uint64_t base = lo; base < hi; base += blockDim.x) {
        uint64_t s = base + threadIdx.x;
        bool has = (s < hi) && (slots[s] != AGG_EMPTY);
        uint64_t mask = __ballot(has);
        uint32_t wcount = __popcll(mask);
        if (lane == 0) wbase[wid] = wcount;
        __syncthreads();
        if (threadIdx.x == 0) {
            uint64_t acc = offset;
            for (int w = 0; w < BLOCK / WAVE; w++) {
                uint64_t v = wbase[w];
                wbase[w] = acc;
                acc += v;
            }
            wbase[BLOCK / WAVE] = acc;
        }
        __syncthreads();
        if (has) {
            uint64_t pos = wbase[wid] + __popcll(mask & ((1ull << lane) - 1));
            if (pos < max_out) {
                out_keys[pos] = slots[s];
                out_sums[pos] = (int64_t)sums[s];
                if (out_counts != nullptr) out_counts[pos] = (int64_t)counts[s];
            }
        }
        offset = wbase[BLOCK / WAVE];
        __syncthreads();
    }
}

extern "C" int gpue_hash_agg_sum_u64(gpue_session* s, gpue_dbuf* keys, gpue_dbuf* vals,
                                     uint64_t n, uint64_t capacity_hint, gpue_dbuf* out_keys,
                                     gpue_dbuf* out_sums, gpue_dbuf* out_counts,
                                     uint64_t max_out, uint64_t* n_groups);
int gpue_hash_agg_sum_u64(gpue_session* s, gpue_dbuf* keys, gpue_dbuf* vals, uint64_t n,
                          uint64_t capacity_hint, gpue_dbuf* out_keys, gpue_dbuf* out_sums,
                          gpue_dbuf* out_counts, uint64_t max_out, uint64_t* n_groups) {
    ARG_CHECK(s && keys && vals && out_keys && out_sums && n_groups);
    ARG_CHECK(keys->bytes >= n * 8 && vals->bytes >= n * 8);
    uint64_t cap = 16;
    uint64_t want = capacity_hint ? capacity_hint : n * 2;
    while (cap < want) cap <<= 1;
    unsigned long long *d_slots = nullptr, *d_sums = nullptr, *d_counts = nullptr,
                       *d_cursor = nullptr;
    HIP_CHECK(hipMalloc(&d_slots, cap * 8));
    HIP_CHECK(hipMalloc(&d_sums, cap * 8));
    HIP_CHECK(hipMalloc(&d_counts, cap * 8));
    HIP_CHECK(hipMalloc(&d_cursor, 8));
    HIP_CHECK(hipMemsetAsync(d_slots, 0xFF, cap * 8, s->stream));
    HIP_CHECK(hipMemsetAsync(d_sums, 0, cap * 8, s->stream));
    HIP_CHECK(hipMemsetAsync(d_counts, 0, cap * 8, s->stream));
    HIP_CHECK(hipMemsetAsync(d_cursor, 0, 8, s->stream));
    hipLaunchKernelGGL(k_hash_agg_sum, dim3(grid_for(n)), dim3(BLOCK), 0, s->stream,
                       (const uint64_t*)keys->ptr, (const int64_t*)vals->ptr, n,
                       d_slots, d_sums, d_counts, cap - 1, s->d_agg_err);
    hipLaunchKernelGGL(k_hash_agg_emit, dim3(grid_for(cap)), dim3(BLOCK), 0, s->stream,
                       d_slots, d_sums, d_counts, cap, d_cursor, max_out,
                       (uint64_t*)out_keys->ptr, (int64_t*)out_sums->ptr,
                       out_counts ? (int64_t*)out_counts->ptr : nullptr);
    unsigned long long groups = 0;
    HIP_CHECK(hipMemcpyAsync(&groups, d_cursor, 8, hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    (void)hipFree(d_slots);
    (void)hipFree(d_sums);
    (void)hipFree(d_counts);
    (void)hipFree(d_cursor);
    *n_groups = groups;
    int erc = agg_err_check(s, "hash_agg");
    if (erc != GPUE_OK) return erc;
    if (groups > max_out) {
        snprintf(g_err, sizeof(g_err), "hash_agg: %llu groups exceed max_out %llu",
                 groups, (unsigned long long)max_out);
        return GPUE_ERR_ARG;
    }
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// Streaming pre-aggregation building blocks (aggregate_streaming_sink_
// operator.cpp:224-310 AUTO policy; the host-side state machine lives in
// starrocks_amd/pipeline.py StreamingAggOperator). The persistent
// gpue_agg_table is the operator's hash map:
//  - push: build_hash_map + compute_batch_agg_states (insert or update);
//    cnts == null means each row contributes count 1, otherwise the row is a
//    pre-aggregated partial (merge_batch semantics, exprs/agg/aggregate.h:
//    158-168) contributing its own count;
//  - probe: build_hash_map_with_selection's hit count (rows whose group
//    already exists — SIMD::count_zero(streaming_selection));
//  - update-only: _push_chunk_by_selective_preaggregation — aggregate rows
//    whose group exists, mark the rest for pass-through.
// ---------------------------------------------------------------------------
__global__ void k_hash_agg_push(const uint64_t* __restrict__ keys,
                                const int64_t* __restrict__ vals,
                                const int64_t* __restrict__ cnts, uint64_t n,
                                unsigned long long* __restrict__ slots,
                                unsigned long long* __restrict__ sums,
                                unsigned long long* __restrict__ counts, uint64_t cap_mask,
                                int update_only, uint8_t* __restrict__ miss_mask,
                                unsigned long long* __restrict__ hits_out,
                                unsigned int* __restrict__ err,
                                unsigned long long* __restrict__ ngroups) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    unsigned long long local_hits = 0;
    unsigned long long local_claims = 0;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        unsigned long long k = keys[i];
        if (agg_key_is_sentinel(k, err)) { if (miss_mask) miss_mask[i] = 1; continue; }
        unsigned long long v = vals ? (unsigned long long)vals[i] : 0ull;
        unsigned long long c = cnts ? (unsigned long long)cnts[i] : 1ull;
        uint64_t slot = ((k * 11400714819323198485ull) >> 32) & cap_mask;
        uint64_t left = cap_mask + 1;
        for (;;) {
            unsigned long long cur = slots[slot];
            if (cur == k) {
                if (vals) {
                    atomicAdd(&sums[slot], v);
                    if (!(update_only & 2)) atomicAdd(&counts[slot], c);
                }
                local_hits++;
                if (miss_mask) miss_mask[i] = 0;
                break;
            }
            if (cur == AGG_EMPTY) {
                if (update_only & 1) { // new group: leave for pass-through
                    if (miss_mask) miss_mask[i] = 1;
                    break;
                }
                unsigned long long old = atomicCAS(&slots[slot], AGG_EMPTY, k);
                if (old == AGG_EMPTY || old == k) {
                    if (vals) {
                        atomicAdd(&sums[slot], v);
                        if (!(update_only & 2)) atomicAdd(&counts[slot], c);
                    }
                    if (old == k) local_hits++;
                    else local_claims++;
                    if (miss_mask) miss_mask[i] = (old != k);
                    break;
                }
                continue; // lost the claim to another key: re-read this slot
            }
            slot = (slot + 1) & cap_mask;
            if (--left == 0) {
                atomicOr(err, AGG_ERR_FULL);
                if (miss_mask) miss_mask[i] = 1;
                break;
            }
        }
    }
    if (hits_out) {
        for (int off = 32; off > 0; off >>= 1)
            local_hits += __shfl_down(local_hits, off, WAVE);
        if ((threadIdx.x & (WAVE - 1)) == 0 && local_hits)
            atomicAdd(hits_out, local_hits);
    }
    for (int off = 32; off > 0; off >>= 1)
        local_claims += __shfl_down(local_claims, off, WAVE);
    if ((threadIdx.x & (WAVE - 1)) == 0 && local_claims && ngroups)
        atomicAdd(ngroups, local_claims);
}

extern "C" {
int gpue_hash_agg_push_u64(gpue_session* s, gpue_agg_table* at, gpue_dbuf* keys,
                           gpue_dbuf* vals, gpue_dbuf* cnts /*nullable*/, uint64_t n,
                           int update_only, gpue_dbuf* miss_mask /*nullable u8*/,
                           uint64_t* hits_out /*nullable*/);
int gpue_hash_agg_probe_hits_u64(gpue_session* s, gpue_agg_table* at, gpue_dbuf* keys,
                                 uint64_t n, uint64_t* hits_out);
int gpue_hash_agg_emit_u64(gpue_session* s, gpue_agg_table* at, gpue_dbuf* out_keys,
                           gpue_dbuf* out_sums, gpue_dbuf* out_counts /*nullable*/,
                           uint64_t max_out, uint64_t* n_groups);
}

static int hash_agg_push_impl(gpue_session* s, gpue_agg_table* at, gpue_dbuf* keys,
                              gpue_dbuf* vals, gpue_dbuf* cnts, uint64_t n, int update_only,
                              gpue_dbuf* miss_mask, uint64_t* hits_out) {
    ARG_CHECK(s && at && keys && keys->bytes >= n * 8);
    ARG_CHECK(update_only >= 0 && update_only <= 3);
    ARG_CHECK(!vals || vals->bytes >= n * 8);
    ARG_CHECK(!cnts || cnts->bytes >= n * 8);
    ARG_CHECK(!miss_mask || miss_mask->bytes >= n);
    unsigned long long* d_hits = nullptr;
    if (hits_out) {
        HIP_CHECK(hipMalloc(&d_hits, 8));
        HIP_CHECK(hipMemsetAsync(d_hits, 0, 8, s->stream));
    }
    hipLaunchKernelGGL(k_hash_agg_push, dim3(grid_for(n)), dim3(BLOCK), 0, s->stream,
                       (const uint64_t*)keys->ptr, vals ? (const int64_t*)vals->ptr : nullptr,
                       cnts ? (const int64_t*)cnts->ptr : nullptr, n, at->slots, at->sums,
                       at->counts, at->cap - 1, update_only,
                       miss_mask ? (uint8_t*)miss_mask->ptr : nullptr, d_hits,
                       s->d_agg_err, at->n_groups);
    HIP_CHECK(hipGetLastError());
    if (hits_out) {
        unsigned long long h = 0;
        HIP_CHECK(hipMemcpyAsync(&h, d_hits, 8, hipMemcpyDeviceToHost, s->stream));
        HIP_CHECK(hipStreamSynchronize(s->stream));
        (void)hipFree(d_hits);
        *hits_out = h;
    }
    return GPUE_OK;
}

int gpue_hash_agg_push_u64(gpue_session* s, gpue_agg_table* at, gpue_dbuf* keys,
                           gpue_dbuf* vals, gpue_dbuf* cnts, uint64_t n, int update_only,
                           gpue_dbuf* miss_mask, uint64_t* hits_out) {
    return hash_agg_push_impl(s, at, keys, vals, cnts, n, update_only, miss_mask, hits_out);
}

int gpue_hash_agg_probe_hits_u64(gpue_session* s, gpue_agg_table* at, gpue_dbuf* keys,
                                 uint64_t n, uint64_t* hits_out) {
    ARG_CHECK(hits_out);
    // probe-only: update_only with no vals — counts untouched, misses unmarked
    return hash_agg_push_impl(s, at, keys, nullptr, nullptr, n, 1, nullptr, hits_out);
}

int gpue_hash_agg_emit_u64(gpue_session* s, gpue_agg_table* at, gpue_dbuf* out_keys,
                           gpue_dbuf* out_sums, gpue_dbuf* out_counts, uint64_t max_out,
                           uint64_t* n_groups) {
    ARG_CHECK(s && at && out_keys && out_sums && n_groups);
    HIP_CHECK(hipMemsetAsync(at->cursor, 0, 8, s->stream));
    hipLaunchKernelGGL(k_hash_agg_emit, dim3(grid_for(at->cap)), dim3(BLOCK), 0, s->stream,
                       at->slots, at->sums, at->counts, at->cap, at->cursor, max_out,
                       (uint64_t*)out_keys->ptr, (int64_t*)out_sums->ptr,
                       out_counts ? (int64_t*)out_counts->ptr : nullptr);
    unsigned long long groups = 0;
    HIP_CHECK(hipMemcpyAsync(&groups, at->cursor, 8, hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    *n_groups = groups;
    int erc = agg_err_check(s, "hash_agg");
    if (erc != GPUE_OK) return erc;
    if (groups > max_out) {
        snprintf(g_err, sizeof(g_err), "hash_agg_emit: %llu groups exceed max_out %llu",
                 groups, (unsigned long long)max_out);
        return GPUE_ERR_ARG;
    }
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// Config 5 — TPC-H Q3-shaped: lineitem ⋈ orders ⋈ customer with a 16-byte
// dictionary-string filter (SERIALIZED_FIXED_SIZE_LARGEINT packing: two u64
// compares — reference join_hash_table.cpp:185-192), decimal revenue as
// scale-4 int64 (extendedprice cents × (100−discount)), and a
// HIGH-cardinality GROUP BY l_orderkey through the hash aggregate below.
// o_orderkey is dense 1..N (synthetic), so the orders "hash table" is the
// RANGE_DIRECT bitset over the passing orders.
// ---------------------------------------------------------------------------
__global__ void k_gen_lineitem_q3(int64_t* lk, int64_t* ext, int64_t* disc, int32_t* ship,
                                  const int32_t* __restrict__ datekey, uint64_t seed,
                                  uint64_t row_start, uint64_t n, uint64_t n_orders) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        uint64_t r = row_start + i;
        lk[i] = (int64_t)(gen_u64(seed, TAG_LOKEY, r) % n_orders) + 1;
        ext[i] = (int64_t)(gen_u64(seed, TAG_LEXT, r) % 10000000u) + 1;
        disc[i] = (int64_t)(gen_u64(seed, TAG_LDISC, r) % 11u);
        ship[i] = datekey[gen_u64(seed, TAG_LSHIP, r) % N_DAYS];
    }
}

__global__ void k_gen_orders_q3(int32_t* ocust, int32_t* odate,
                                const int32_t* __restrict__ datekey, uint64_t seed,
                                uint64_t n_orders, uint32_t n_custs) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t o = 1 + (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; o <= n_orders;
         o += stride) {
        ocust[o - 1] = (int32_t)(gen_u64(seed, TAG_OCUST, o) % n_custs) + 1;
        odate[o - 1] = datekey[gen_u64(seed, TAG_ODATE, o) % N_DAYS];
    }
}

__global__ void k_gen_cust_mkt16(uint8_t* out, uint64_t seed, uint32_t n_custs) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t c = 1 + (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; c <= n_custs;
         c += stride) {
        uint32_t seg = (uint32_t)(gen_u64(seed, TAG_CMKT, c) % 5u);
        ulonglong2 v = *reinterpret_cast<const ulonglong2*>(MKT_SEGMENTS_DEV[seg]);
        *reinterpret_cast<ulonglong2*>(out + (c - 1) * 16) = v;
    }
}

// 16-byte fixed-string equality -> bitset (two u64 compares per row)
__global__ void k_bits_str16_eq(const ulonglong2* __restrict__ col, uint64_t n,
                                unsigned long long la, unsigned long long lb,
                                uint32_t* __restrict__ bits) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        ulonglong2 v = col[i];
        if (v.x == la && v.y == lb) atomicOr(&bits[i >> 5], 1u << (i & 31));
    }
}

// Each lane evaluates 4 consecutive orders (int4 loads); 8-lane groups pack
// their 32 predicate bits into one word via shuffles — no atomics, each
// 32-bit bitset word written exactly once.
// Ballot variant (A/B via GPUE_Q3_OBITS_V2) — measured NEGATIVE: 8.3 ms vs
// 2.0 ms for the quad kernel at 450 M orders. One order per thread leaves a
// single outstanding cust-bitset gather per lane where the int4 quad form
// keeps four in flight — the pass is gather-latency-bound, so
// memory-level parallelism beats the simpler ballot assembly. Kept (off by
// default) as the documented experiment.
__global__ void k_q3_order_bits_ballot(const int32_t* __restrict__ ocust,
                                       const int32_t* __restrict__ odate, uint64_t n_orders,
                                       const uint32_t* __restrict__ cust_bits,
                                       int32_t cutoff, uint32_t* __restrict__ bits) {
    int lane = threadIdx.x & (WAVE - 1);
    uint64_t wave_id = ((uint64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
    uint64_t total_waves = (uint64_t)gridDim.x * blockDim.x / WAVE;
    for (uint64_t wb = wave_id * WAVE; wb < n_orders; wb += total_waves * WAVE) {
        uint64_t o = wb + lane;
        bool pass = false;
        if (o < n_orders && odate[o] < cutoff) {
            uint32_t c = (uint32_t)ocust[o] - 1;
            pass = (cust_bits[c >> 5] >> (c & 31)) & 1u;
        }
        unsigned long long mask = __ballot(pass);
        if (lane == 0) {
            bits[wb >> 5] = (uint32_t)mask;
            if (wb + 32 < n_orders) bits[(wb >> 5) + 1] = (uint32_t)(mask >> 32);
        }
    }
}

__global__ void k_q3_order_bits(const int32_t* __restrict__ ocust,
                                const int32_t* __restrict__ odate, uint64_t n_orders,
                                const uint32_t* __restrict__ cust_bits, int32_t cutoff,
                                uint32_t* __restrict__ bits) {
    uint64_t n4 = (n_orders + 3) / 4; // quads; caller sizes bits to whole words
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    int lane = threadIdx.x & (WAVE - 1);
    // wave-uniform loop bound (q - lane == this wave's first quad) so every
    // lane reaches the shuffles together
    for (uint64_t q = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; q - lane < n4;
         q += stride) {
        uint64_t base = q * 4;
        uint32_t b4 = 0;
        if (q < n4) {
            if (base + 4 <= n_orders) {
                // 16 B/lane vector loads for the two streamed columns; the
                // customer-bit gather is issued only for date-passing orders
                int4 d4 = ((const int4*)odate)[q];
                int4 c4 = ((const int4*)ocust)[q];
                #pragma unroll
                for (int j = 0; j < 4; j++) {
                    if ((&d4.x)[j] < cutoff) {
                        uint32_t c = (uint32_t)(&c4.x)[j] - 1;
                        b4 |= ((cust_bits[c >> 5] >> (c & 31)) & 1u) << j;
                    }
                }
            } else {
                for (int j = 0; j < 4; j++) {
                    uint64_t o = base + j;
                    if (o < n_orders && odate[o] < cutoff) {
                        uint32_t c = (uint32_t)ocust[o] - 1;
                        b4 |= ((cust_bits[c >> 5] >> (c & 31)) & 1u) << j;
                    }
                }
            }
        }
        uint32_t word = 0;
        #pragma unroll
        for (int k = 0; k < 8; k++)
            word |= (uint32_t)__shfl((int)b4, (lane & ~7) + k, WAVE) << (4 * k);
        // 8 quads = 32 orders = one bitset word, written exactly once by the
        // group leader (group base quad is 8-aligned: blockDim % 64 == 0)
        if (q < n4 && (lane & 7) == 0) bits[q / 8] = word;
    }
}

// Branchless double-quad variant (A/B via GPUE_Q3_OBITS_V3): the customer-
// bit gathers issue UNCONDITIONALLY with a pass-clamped index (the same
// guide §5 4(c) pattern as k_q1_join_sum_bitset — an if around a load emits
// per-element branches + vmcnt waits), and two quads run per iteration for
// 8 gathers in flight per lane.
__global__ void k_q3_order_bits_bl(const int32_t* __restrict__ ocust,
                                   const int32_t* __restrict__ odate, uint64_t n_orders,
                                   const uint32_t* __restrict__ cust_bits, int32_t cutoff,
                                   uint32_t* __restrict__ bits) {
    uint64_t n4 = (n_orders + 3) / 4;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    int lane = threadIdx.x & (WAVE - 1);
    auto quad_bits = [&](uint64_t q) -> uint32_t {
        uint32_t b4 = 0;
        uint64_t base = q * 4;
        if (base + 4 <= n_orders) {
            int4 d4 = ((const int4*)odate)[q];
            int4 c4 = ((const int4*)ocust)[q];
            uint32_t pass[4], w[4];
            #pragma unroll
            for (int j = 0; j < 4; j++) {
                pass[j] = (&d4.x)[j] < cutoff;
                uint32_t c = pass[j] ? (uint32_t)(&c4.x)[j] - 1 : 0u;
                w[j] = cust_bits[c >> 5] >> (c & 31);
            }
            #pragma unroll
            for (int j = 0; j < 4; j++) b4 |= (w[j] & pass[j] & 1u) << j;
        } else {
            for (int j = 0; j < 4; j++) {
                uint64_t o = base + j;
                if (o < n_orders && odate[o] < cutoff) {
                    uint32_t c = (uint32_t)ocust[o] - 1;
                    b4 |= ((cust_bits[c >> 5] >> (c & 31)) & 1u) << j;
                }
            }
        }
        return b4;
    };
    for (uint64_t q = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; q - lane < n4;
         q += 2 * stride) {
        uint64_t q2 = q + stride;
        uint32_t b4a = (q < n4) ? quad_bits(q) : 0u;
        uint32_t b4b = (q2 < n4) ? quad_bits(q2) : 0u;
        uint32_t wa = 0, wb = 0;
        #pragma unroll
        for (int k = 0; k < 8; k++) {
            wa |= (uint32_t)__shfl((int)b4a, (lane & ~7) + k, WAVE) << (4 * k);
            wb |= (uint32_t)__shfl((int)b4b, (lane & ~7) + k, WAVE) << (4 * k);
        }
        if (q < n4 && (lane & 7) == 0) bits[q / 8] = wa;
        if (q2 - lane < n4) {
            if (q2 < n4 && (lane & 7) == 0) bits[q2 / 8] = wb;
        }
    }
}

// fused lineitem filter + orders semi-probe + hash-agg insert.
// Scalar per-row form: a 16 B/lane two-row variant with batched bitset
// gathers measured 1188 vs 1759 GB/s — the ship filter (~50 % selective)
// makes half those gathers wasted traffic, and the conditional ext/disc
// loads serialize. COUNT is not in Q3's select list — one CAS-claim + one
// atomicAdd per passing row.
template <bool CT> // CT: CACHED ext/disc gathers — at ~9% survivor density
                   // adjacent survivors share 64 B lines, and the NT hint
                   // forces a re-fetch per survivor (the q21 mode-7 lesson)
__global__ void k_q3_probe_agg(const int64_t* __restrict__ lk,
                               const int64_t* __restrict__ ext,
                               const int64_t* __restrict__ disc,
                               const int32_t* __restrict__ ship, uint64_t n,
                               const uint32_t* __restrict__ order_bits, int32_t ship_cutoff,
                               unsigned long long* __restrict__ slots,
                               unsigned long long* __restrict__ sums,
                               unsigned long long* __restrict__ counts, uint64_t cap_mask,
                               unsigned int* __restrict__ err,
                               unsigned long long* __restrict__ ngroups) {
    (void)counts;
    unsigned long long local_claims = 0; // one atomic per wave at kernel end
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        // non-temporal loads on the four streamed columns: the decomp
        // instrument (tools/q3_decomp.py) shows the order-bits
        // random gather dominates (2.1 of 3.3 ms); streaming around L2 buys a
        // small (~1.4%) but consistent win by leaving L2 to the bitset.
        if (__builtin_nontemporal_load(ship + i) <= ship_cutoff) continue;
        unsigned long long k = (unsigned long long)__builtin_nontemporal_load(lk + i);
        uint64_t o = k - 1;
        if (!((order_bits[o >> 5] >> (o & 31)) & 1u)) continue;
        unsigned long long v = (unsigned long long)(
            (CT ? ext[i] : __builtin_nontemporal_load(ext + i)) *
            (100 - (CT ? disc[i] : __builtin_nontemporal_load(disc + i))));
        uint64_t s = ((k * 11400714819323198485ull) >> 32) & cap_mask;
        uint64_t left = cap_mask + 1;
        for (;;) {
            unsigned long long cur = slots[s];
            if (cur == k) { atomicAdd(&sums[s], v); break; }
            if (cur == AGG_EMPTY) {
                unsigned long long old = atomicCAS(&slots[s], AGG_EMPTY, k);
                if (old == AGG_EMPTY || old == k) {
                    if (old == AGG_EMPTY) local_claims++;
                    atomicAdd(&sums[s], v);
                    break;
                }
            }
            s = (s + 1) & cap_mask;
            if (--left == 0) { atomicOr(err, AGG_ERR_FULL); break; }
        }
    }
    // same-address claim-counter traffic wave-reduced (the round-1 cursor
    // lesson: per-item same-line atomics serialize at the TCC)
    for (int off = WAVE / 2; off > 0; off >>= 1)
        local_claims += __shfl_down(local_claims, off, WAVE);
    if ((threadIdx.x & (WAVE - 1)) == 0 && local_claims && ngroups)
        atomicAdd(ngroups, local_claims);
}

// ---------------------------------------------------------------------------
// Two-stage wave-queue q3 probe (GPUE_Q3_WQ): the fused kernel's legs run at
// low lane density (ship survivors 46%, order-bit survivors ~9%), so
// every wave issues every leg with mostly-idle lanes. Queue A compacts
// ship-passing rows; its full-wave drain loads lk and probes the order
// bitset with all 64 lanes carrying real candidates; survivors queue into B,
// whose full-wave drain loads ext/disc and does the hash-table insert.
// The 56 MB bitset random-gather itself stays architectural
// (profiles/r02_bitgather_ubench.json) — this removes the divergent-issue
// overhead around it. All loop/drain conditions are wave-uniform (the
// queue counters are wave-uniform registers).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(BLOCK) void
k_q3_probe_agg_wq(const int64_t* __restrict__ lk, const int64_t* __restrict__ ext,
                  const int64_t* __restrict__ disc, const int32_t* __restrict__ ship,
                  uint64_t n, const uint32_t* __restrict__ order_bits, int32_t ship_cutoff,
                  unsigned long long* __restrict__ slots,
                  unsigned long long* __restrict__ sums,
                  unsigned long long* __restrict__ counts, uint64_t cap_mask,
                  unsigned int* __restrict__ err,
                  unsigned long long* __restrict__ ngroups) {
    (void)counts;
    __shared__ uint32_t qa[BLOCK / WAVE][128]; // rows passing the ship filter
    __shared__ uint2 qb[BLOCK / WAVE][128];    // {lk32, row} passing the order bitset
    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x & (WAVE - 1);
    uint32_t wqa = 0, wqb = 0; // wave-uniform
    unsigned long long local_claims = 0;
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;

    auto insert = [&](uint32_t key32, uint32_t row) {
        unsigned long long k = key32;
        unsigned long long v = (unsigned long long)(__builtin_nontemporal_load(ext + row) *
                                                    (100 - __builtin_nontemporal_load(disc + row)));
        uint64_t s = ((k * 11400714819323198485ull) >> 32) & cap_mask;
        uint64_t left = cap_mask + 1;
        for (;;) {
            unsigned long long cur = slots[s];
            if (cur == k) { atomicAdd(&sums[s], v); break; }
            if (cur == AGG_EMPTY) {
                unsigned long long old = atomicCAS(&slots[s], AGG_EMPTY, k);
                if (old == AGG_EMPTY || old == k) {
                    if (old == AGG_EMPTY) local_claims++;
                    atomicAdd(&sums[s], v);
                    break;
                }
            }
            s = (s + 1) & cap_mask;
            if (--left == 0) { atomicOr(err, AGG_ERR_FULL); break; }
        }
    };
    // drain helpers take the count of VALID entries (wave-uniform); lanes
    // beyond it participate in the ballots with pass=false
    auto drainB = [&](uint32_t cnt) {
        bool valid = (uint32_t)lane < cnt;
        uint2 e = qb[wid][valid ? wqb - cnt + lane : 0];
        if (valid) insert(e.x, e.y);
        wqb -= cnt;
    };
    auto drainA = [&](uint32_t cnt) {
        bool valid = (uint32_t)lane < cnt;
        uint32_t r = qa[wid][valid ? wqa - cnt + lane : 0];
        uint64_t k = valid ? (uint64_t)__builtin_nontemporal_load(lk + r) : 1;
        uint64_t o = k - 1;
        bool pass = valid && ((order_bits[o >> 5] >> (o & 31)) & 1u);
        uint64_t m = __ballot(pass);
        wqa -= cnt;
        if (m) {
            uint32_t rank = __popcll(m & ((1ull << lane) - 1));
            if (pass) qb[wid][wqb + rank] = make_uint2((uint32_t)k, r);
            wqb += __popcll(m);
            if (wqb >= 64) drainB(64);
        }
    };
    uint64_t base = (uint64_t)blockIdx.x * blockDim.x + (uint64_t)wid * WAVE;
    uint64_t i = base + lane;
    for (; base < n; base += stride, i += stride) {
        bool inb = i < n;
        bool pass = inb && (__builtin_nontemporal_load(ship + (inb ? i : 0)) > ship_cutoff);
        uint64_t m = __ballot(pass);
        if (m) {
            uint32_t rank = __popcll(m & ((1ull << lane) - 1));
            if (pass) qa[wid][wqa + rank] = (uint32_t)i;
            wqa += __popcll(m);
            if (wqa >= 64) drainA(64);
        }
    }
    if (wqa > 0) drainA(wqa);
    if (wqb > 0) drainB(wqb);
    for (int off = WAVE / 2; off > 0; off >>= 1)
        local_claims += __shfl_down(local_claims, off, WAVE);
    if ((threadIdx.x & (WAVE - 1)) == 0 && local_claims && ngroups)
        atomicAdd(ngroups, local_claims);
}

// Decomposition instrument for q3's roofline attribution:
// legs bitmask accumulates per-leg cost — 1: ship stream, 2: +lk read &
// order-bits gather, 4: +ext/disc product, 8: +hash-table insert (the full
// kernel). Local accumulate + one atomic per block prevents DCE.
template <bool NT>  // NT: non-temporal stream loads — leave L2 to the bitset
__global__ void k_q3_legs(const int64_t* __restrict__ lk,
                          const int64_t* __restrict__ ext,
                          const int64_t* __restrict__ disc,
                          const int32_t* __restrict__ ship, uint64_t n,
                          const uint32_t* __restrict__ order_bits, int32_t ship_cutoff,
                          int legs, unsigned long long* __restrict__ slots,
                          unsigned long long* __restrict__ sums, uint64_t cap_mask,
                          unsigned long long* __restrict__ sink) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    unsigned long long acc = 0;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        int32_t sh = NT ? __builtin_nontemporal_load(ship + i) : ship[i];
        if (sh <= ship_cutoff) continue;
        acc++;
        if (!(legs & 2)) continue;
        unsigned long long k =
            (unsigned long long)(NT ? __builtin_nontemporal_load(lk + i) : lk[i]);
        uint64_t o = k - 1;
        if (!((order_bits[o >> 5] >> (o & 31)) & 1u)) continue;
        acc++;
        if (!(legs & 4)) continue;
        int64_t e_ = NT ? __builtin_nontemporal_load(ext + i) : ext[i];
        int64_t d_ = NT ? __builtin_nontemporal_load(disc + i) : disc[i];
        unsigned long long v = (unsigned long long)(e_ * (100 - d_));
        acc += v & 1;
        if (!(legs & 8)) continue;
        uint64_t s = ((k * 11400714819323198485ull) >> 32) & cap_mask;
        for (uint64_t tries = 0; tries <= cap_mask; tries++) {  // bounded: full table => stop, not hang
            unsigned long long cur = slots[s];
            if (cur == k) { atomicAdd(&sums[s], v); break; }
            if (cur == AGG_EMPTY) {
                unsigned long long old = atomicCAS(&slots[s], AGG_EMPTY, k);
                if (old == AGG_EMPTY || old == k) { atomicAdd(&sums[s], v); break; }
            }
            s = (s + 1) & cap_mask;
        }
    }
    for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off, 64);
    if (threadIdx.x == 0) atomicAdd(sink, acc);
}

// ---------------------------------------------------------------------------
// Partitioned q3 probe (round-2 lever 2 measured early): the 56 MB order-
// bits bitset misses L2 at 90% (profiles/r01f_tcc_hit_rates.txt), so the
// probe pays L3/HBM latency per gather. Two passes restore locality:
//   A) stream ship/lk/ext/disc; passing rows (ship filter) partition by
//      orderkey RANGE into nparts slices (the local-exchange repartition
//      idiom, local_exchange.h:71-260, keyed by range not hash so each
//      slice's bitset window is contiguous), materializing (key u32,
//      v = ext*(100-disc) i64);
//   B) one launch per partition probes its L2-resident bitset window and
//      inserts into the shared CAS agg table.
// Results are identical to the fused kernel (same inserts, order-free).
// ---------------------------------------------------------------------------
__global__ void k_q3_part_hist(const int64_t* __restrict__ lk,
                               const int32_t* __restrict__ ship, uint64_t n,
                               int32_t ship_cutoff, uint64_t slice, uint32_t nparts,
                               uint64_t tile, uint32_t* __restrict__ block_hist) {
    // wave-PRIVATE counters: same-address LDS atomic serialization measured
    // 4x the stream bound (profiles/q3_partitioned_probe_r01.txt); each wave
    // owns its own nparts counters, written out per (block, wave, partition)
    // so the emit can reserve per-wave ranges with no atomics at all.
    extern __shared__ uint32_t h[]; // [waves_per_block][nparts]
    const int wid = threadIdx.x / WAVE;
    const int nw = blockDim.x / WAVE;
    for (uint32_t c = threadIdx.x; c < (uint32_t)nw * nparts; c += blockDim.x) h[c] = 0;
    __syncthreads();
    uint64_t lo = (uint64_t)blockIdx.x * tile, hi = min(lo + tile, n);
    for (uint64_t i = lo + threadIdx.x; i < hi; i += blockDim.x) {
        if (__builtin_nontemporal_load(ship + i) <= ship_cutoff) continue;
        uint32_t p = (uint32_t)(((uint64_t)__builtin_nontemporal_load(lk + i) - 1) / slice);
        atomicAdd(&h[wid * nparts + p], 1u);
    }
    __syncthreads();
    for (uint32_t c = threadIdx.x; c < (uint32_t)nw * nparts; c += blockDim.x)
        block_hist[(uint64_t)blockIdx.x * nw * nparts + c] = h[c];
}

__global__ void k_q3_part_emit(const int64_t* __restrict__ lk,
                               const int64_t* __restrict__ ext,
                               const int64_t* __restrict__ disc,
                               const int32_t* __restrict__ ship, uint64_t n,
                               int32_t ship_cutoff, uint64_t slice, uint32_t nparts,
                               uint64_t tile, const uint64_t* __restrict__ wave_offsets,
                               uint32_t* __restrict__ out_keys,
                               long long* __restrict__ out_vals) {
    // per-(block,wave,partition) ranges reserved by the host scan: each
    // wave bumps its own LDS cursors — contention only among 64 lanes
    extern __shared__ uint64_t cur[]; // [waves_per_block][nparts]
    const int wid = threadIdx.x / WAVE;
    const int nw = blockDim.x / WAVE;
    for (uint32_t c = threadIdx.x; c < (uint32_t)nw * nparts; c += blockDim.x)
        cur[c] = wave_offsets[(uint64_t)blockIdx.x * nw * nparts + c];
    __syncthreads();
    uint64_t lo = (uint64_t)blockIdx.x * tile, hi = min(lo + tile, n);
    for (uint64_t i = lo + threadIdx.x; i < hi; i += blockDim.x) {
        if (__builtin_nontemporal_load(ship + i) <= ship_cutoff) continue;
        uint64_t k = (uint64_t)__builtin_nontemporal_load(lk + i);
        uint32_t p = (uint32_t)((k - 1) / slice);
        uint64_t pos = atomicAdd((unsigned long long*)&cur[wid * nparts + p], 1ull);
        out_keys[pos] = (uint32_t)k;
        out_vals[pos] = __builtin_nontemporal_load(ext + i) *
                        (100 - __builtin_nontemporal_load(disc + i));
    }
}

// v3 emit — LDS-STAGED tile sort (the standard GPU radix-partition form):
// each 1024-row tile is counted, prefix-summed, scattered into an LDS
// staging area CONTIGUOUS BY PARTITION, then flushed segment-by-segment to
// the block's per-partition global cursors — writes go out as ~384 B
// coalesced runs instead of isolated 12 B scatters (the cost that sank v1
// at 2.9 ms and v2 at worse; profiles/q3_partitioned_probe_r01.txt).
static constexpr int Q3P_TILE = 1024; // rows per tile = BLOCK * 4
__global__ void k_q3_part_emit_staged(const int64_t* __restrict__ lk,
                                      const int64_t* __restrict__ ext,
                                      const int64_t* __restrict__ disc,
                                      const int32_t* __restrict__ ship, uint64_t n,
                                      int32_t ship_cutoff, uint64_t slice, uint32_t nparts,
                                      uint64_t tile,
                                      const uint64_t* __restrict__ block_offsets,
                                      uint32_t* __restrict__ out_keys,
                                      long long* __restrict__ out_vals) {
    __shared__ uint32_t stage_k[Q3P_TILE];
    __shared__ long long stage_v[Q3P_TILE];
    __shared__ uint32_t cnt[256], cnt2[256], tile_off[257];
    __shared__ unsigned long long gcur[256];
    for (uint32_t c = threadIdx.x; c < nparts; c += blockDim.x)
        gcur[c] = block_offsets[(uint64_t)blockIdx.x * nparts + c];
    __syncthreads();
    uint64_t lo = (uint64_t)blockIdx.x * tile, hi = min(lo + tile, n);
    const int K = Q3P_TILE / BLOCK;
    for (uint64_t t0 = lo; t0 < hi; t0 += Q3P_TILE) {
        // 1) load this tile's rows into registers
        uint32_t rk[K];
        long long rv[K];
        uint32_t rp[K], pass[K];
        #pragma unroll
        for (int j = 0; j < K; j++) {
            uint64_t i = t0 + threadIdx.x + (uint64_t)j * blockDim.x;
            pass[j] = 0;
            if (i < hi && __builtin_nontemporal_load(ship + i) > ship_cutoff) {
                uint64_t k = (uint64_t)__builtin_nontemporal_load(lk + i);
                rk[j] = (uint32_t)k;
                rp[j] = (uint32_t)((k - 1) / slice);
                rv[j] = __builtin_nontemporal_load(ext + i) *
                        (100 - __builtin_nontemporal_load(disc + i));
                pass[j] = 1;
            }
        }
        // 2) tile histogram
        for (uint32_t c = threadIdx.x; c < nparts; c += blockDim.x) cnt[c] = cnt2[c] = 0;
        __syncthreads();
        #pragma unroll
        for (int j = 0; j < K; j++)
            if (pass[j]) atomicAdd(&cnt[rp[j]], 1u);
        __syncthreads();
        // 3) tile prefix (serial over <=256 partitions — trivial)
        if (threadIdx.x == 0) {
            uint32_t a = 0;
            for (uint32_t p = 0; p < nparts; p++) {
                tile_off[p] = a;
                a += cnt[p];
            }
            tile_off[nparts] = a;
        }
        __syncthreads();
        // 4) scatter into the partition-contiguous LDS staging area
        #pragma unroll
        for (int j = 0; j < K; j++)
            if (pass[j]) {
                uint32_t slot = tile_off[rp[j]] + atomicAdd(&cnt2[rp[j]], 1u);
                stage_k[slot] = rk[j];
                stage_v[slot] = rv[j];
            }
        __syncthreads();
        // 5) coalesced flush: stage index -> partition via 8-step search
        uint32_t total = tile_off[nparts];
        for (uint32_t j = threadIdx.x; j < total; j += blockDim.x) {
            uint32_t lo_p = 0, hi_p = nparts;
            while (hi_p - lo_p > 1) {
                uint32_t mid = (lo_p + hi_p) / 2;
                if (tile_off[mid] <= j) lo_p = mid; else hi_p = mid;
            }
            uint64_t dst = gcur[lo_p] + (j - tile_off[lo_p]);
            out_keys[dst] = stage_k[j];
            out_vals[dst] = stage_v[j];
        }
        __syncthreads();
        for (uint32_t c = threadIdx.x; c < nparts; c += blockDim.x) gcur[c] += cnt[c];
        __syncthreads();
    }
}

// fused pass B: every block is statically assigned ONE partition
// (block_part/block_base arrays) — one launch instead of nparts small
// launch-bound ones; consecutive blocks share a partition so each XCD's L2
// sees one bitset window at a time.
__global__ void k_q3_probe_slices_fused(const uint32_t* __restrict__ keys,
                                        const long long* __restrict__ vals,
                                        const uint32_t* __restrict__ block_part,
                                        const uint64_t* __restrict__ pstart,
                                        const uint32_t* __restrict__ blocks_per_part,
                                        const uint32_t* __restrict__ first_block_of_part,
                                        const uint32_t* __restrict__ order_bits,
                                        unsigned long long* __restrict__ slots,
                                        unsigned long long* __restrict__ sums,
                                        uint64_t cap_mask, unsigned int* __restrict__ err,
                                        unsigned long long* __restrict__ ngroups) {
    unsigned long long local_claims = 0;
    uint32_t p = block_part[blockIdx.x];
    uint64_t base = pstart[p];
    uint64_t cnt = pstart[p + 1] - base;
    uint32_t nb_p = blocks_per_part[p];
    uint32_t rel = blockIdx.x - first_block_of_part[p];
    uint64_t stride = (uint64_t)nb_p * blockDim.x;
    for (uint64_t i = (uint64_t)rel * blockDim.x + threadIdx.x; i < cnt; i += stride) {
        unsigned long long k = keys[base + i];
        uint64_t o = k - 1;
        if (!((order_bits[o >> 5] >> (o & 31)) & 1u)) continue;
        unsigned long long v = (unsigned long long)vals[base + i];
        uint64_t s = ((k * 11400714819323198485ull) >> 32) & cap_mask;
        uint64_t left = cap_mask + 1;
        for (;;) {
            unsigned long long cur_ = slots[s];
            if (cur_ == k) { atomicAdd(&sums[s], v); break; }
            if (cur_ == AGG_EMPTY) {
                unsigned long long old = atomicCAS(&slots[s], AGG_EMPTY, k);
                if (old == AGG_EMPTY || old == k) {
                    if (old == AGG_EMPTY) local_claims++;
                    atomicAdd(&sums[s], v);
                    break;
                }
            }
            s = (s + 1) & cap_mask;
            if (--left == 0) { atomicOr(err, AGG_ERR_FULL); break; }
        }
    }
    for (int off = WAVE / 2; off > 0; off >>= 1)
        local_claims += __shfl_down(local_claims, off, WAVE);
    if ((threadIdx.x & (WAVE - 1)) == 0 && local_claims && ngroups)
        atomicAdd(ngroups, local_claims);
}

extern "C" {
int gpue_gen_lineitem_q3(gpue_session* s, uint64_t seed, uint64_t row_start, uint64_t n,
                         uint64_t n_orders, gpue_dbuf* lk, gpue_dbuf* ext, gpue_dbuf* disc,
                         gpue_dbuf* ship);
int gpue_gen_orders_q3(gpue_session* s, uint64_t seed, uint64_t n_orders, uint32_t n_custs,
                       gpue_dbuf* ocust, gpue_dbuf* odate);
int gpue_gen_cust_mkt16(gpue_session* s, uint64_t seed, uint32_t n_custs, gpue_dbuf* out);
int gpue_bits_str16_eq(gpue_session* s, gpue_dbuf* col16, uint64_t n, const void* lit16,
                       gpue_dbuf* bits);
int gpue_q3_order_bits(gpue_session* s, gpue_dbuf* ocust, gpue_dbuf* odate, uint64_t n_orders,
                       gpue_dbuf* cust_bits, int32_t cutoff, gpue_dbuf* order_bits);
int gpue_q3_probe_agg(gpue_session* s, gpue_dbuf* lk, gpue_dbuf* ext, gpue_dbuf* disc,
                      gpue_dbuf* ship, uint64_t n, gpue_dbuf* order_bits, int32_t ship_cutoff,
                      uint64_t capacity_hint, gpue_dbuf* out_keys, gpue_dbuf* out_sums,
                      uint64_t max_out, uint64_t* n_groups);
}

extern "C" int gpue_q3_decomp(gpue_session* s, gpue_dbuf* lk, gpue_dbuf* ext,
                              gpue_dbuf* disc, gpue_dbuf* ship, uint64_t n,
                              gpue_dbuf* order_bits, int32_t ship_cutoff, int legs,
                              gpue_agg_table* at, gpue_dbuf* sink);
int gpue_q3_decomp(gpue_session* s, gpue_dbuf* lk, gpue_dbuf* ext, gpue_dbuf* disc,
                   gpue_dbuf* ship, uint64_t n, gpue_dbuf* order_bits, int32_t ship_cutoff,
                   int legs, gpue_agg_table* at, gpue_dbuf* sink) {
    ARG_CHECK(s && lk && ext && disc && ship && order_bits && sink && sink->bytes >= 8);
    ARG_CHECK(!(legs & 8) || at);
    hipLaunchKernelGGL((legs & 16) ? k_q3_legs<true> : k_q3_legs<false>,
                       dim3(grid_capped(n, env_cap("GPUE_GRID_Q3", MAX_GRID))),
                       dim3(BLOCK), 0, s->stream, (const int64_t*)lk->ptr,
                       (const int64_t*)ext->ptr, (const int64_t*)disc->ptr,
                       (const int32_t*)ship->ptr, n, (const uint32_t*)order_bits->ptr,
                       ship_cutoff, legs, at ? at->slots : nullptr,
                       at ? at->sums : nullptr, at ? at->cap - 1 : 0,
                       (unsigned long long*)sink->ptr);
    HIP_CHECK(hipGetLastError());
    return GPUE_OK;
}

int gpue_gen_lineitem_q3(gpue_session* s, uint64_t seed, uint64_t row_start, uint64_t n,
                         uint64_t n_orders, gpue_dbuf* lk, gpue_dbuf* ext, gpue_dbuf* disc,
                         gpue_dbuf* ship) {
    ARG_CHECK(s && lk && ext && disc && ship && n_orders > 0);
    ARG_CHECK(lk->bytes >= n * 8 && ext->bytes >= n * 8 && disc->bytes >= n * 8 &&
              ship->bytes >= n * 4);
    int rc = ensure_datekey(s);
    if (rc != GPUE_OK) return rc;
    hipLaunchKernelGGL(k_gen_lineitem_q3, dim3(grid_stream(n)), dim3(BLOCK), 0, s->stream,
                       (int64_t*)lk->ptr, (int64_t*)ext->ptr, (int64_t*)disc->ptr,
                       (int32_t*)ship->ptr, s->d_datekey, seed, row_start, n, n_orders);
    HIP_CHECK(hipGetLastError());
    return GPUE_OK;
}

int gpue_gen_orders_q3(gpue_session* s, uint64_t seed, uint64_t n_orders, uint32_t n_custs,
                       gpue_dbuf* ocust, gpue_dbuf* odate) {
    ARG_CHECK(s && ocust && odate && ocust->bytes >= n_orders * 4 &&
              odate->bytes >= n_orders * 4);
    int rc = ensure_datekey(s);
    if (rc != GPUE_OK) return rc;
    hipLaunchKernelGGL(k_gen_orders_q3, dim3(grid_stream(n_orders)), dim3(BLOCK), 0, s->stream,
                       (int32_t*)ocust->ptr, (int32_t*)odate->ptr, s->d_datekey, seed,
                       n_orders, n_custs);
    HIP_CHECK(hipGetLastError());
    return GPUE_OK;
}

int gpue_gen_cust_mkt16(gpue_session* s, uint64_t seed, uint32_t n_custs, gpue_dbuf* out) {
    ARG_CHECK(s && out && out->bytes >= (uint64_t)n_custs * 16);
    hipLaunchKernelGGL(k_gen_cust_mkt16, dim3(grid_stream(n_custs)), dim3(BLOCK), 0, s->stream,
                       (uint8_t*)out->ptr, seed, n_custs);
    HIP_CHECK(hipGetLastError());
    return GPUE_OK;
}

int gpue_bits_str16_eq(gpue_session* s, gpue_dbuf* col16, uint64_t n, const void* lit16,
                       gpue_dbuf* bits) {
    ARG_CHECK(s && col16 && lit16 && bits);
    ARG_CHECK(col16->bytes >= n * 16 && bits->bytes >= (n + 31) / 32 * 4);
    unsigned long long la, lb;
    memcpy(&la, lit16, 8);
    memcpy(&lb, (const char*)lit16 + 8, 8);
    HIP_CHECK(hipMemsetAsync(bits->ptr, 0, (n + 31) / 32 * 4, s->stream));
    hipLaunchKernelGGL(k_bits_str16_eq, dim3(grid_stream(n)), dim3(BLOCK), 0, s->stream,
                       (const ulonglong2*)col16->ptr, n, la, lb, (uint32_t*)bits->ptr);
    HIP_CHECK(hipGetLastError());
    return GPUE_OK;
}

int gpue_q3_order_bits(gpue_session* s, gpue_dbuf* ocust, gpue_dbuf* odate, uint64_t n_orders,
                       gpue_dbuf* cust_bits, int32_t cutoff, gpue_dbuf* order_bits) {
    ARG_CHECK(s && ocust && odate && cust_bits && order_bits);
    ARG_CHECK(order_bits->bytes >= (n_orders + 31) / 32 * 4);
    HIP_CHECK(hipMemsetAsync(order_bits->ptr, 0, (n_orders + 31) / 32 * 4, s->stream));
    if (env_cap("GPUE_Q3_OBITS_V3", 0) == 1)
        hipLaunchKernelGGL(k_q3_order_bits_bl, dim3(grid_for(n_orders)), dim3(BLOCK), 0,
                           s->stream, (const int32_t*)ocust->ptr, (const int32_t*)odate->ptr,
                           n_orders, (const uint32_t*)cust_bits->ptr, cutoff,
                           (uint32_t*)order_bits->ptr);
    else if (env_cap("GPUE_Q3_OBITS_V2", 0) == 1)
        hipLaunchKernelGGL(k_q3_order_bits_ballot, dim3(grid_stream(n_orders)), dim3(BLOCK),
                           0, s->stream, (const int32_t*)ocust->ptr,
                           (const int32_t*)odate->ptr, n_orders,
                           (const uint32_t*)cust_bits->ptr, cutoff,
                           (uint32_t*)order_bits->ptr);
    else
    hipLaunchKernelGGL(k_q3_order_bits, dim3(grid_for(n_orders)), dim3(BLOCK), 0, s->stream,
                       (const int32_t*)ocust->ptr, (const int32_t*)odate->ptr, n_orders,
                       (const uint32_t*)cust_bits->ptr, cutoff, (uint32_t*)order_bits->ptr);
    HIP_CHECK(hipGetLastError());
    return GPUE_OK;
}

extern "C" int gpue_q3_probe_agg_t(gpue_session* s, gpue_dbuf* lk, gpue_dbuf* ext,
                                   gpue_dbuf* disc, gpue_dbuf* ship, uint64_t n,
                                   gpue_dbuf* order_bits, int32_t ship_cutoff,
                                   gpue_agg_table* at, gpue_dbuf* out_keys,
                                   gpue_dbuf* out_sums, uint64_t max_out, uint64_t* n_groups);

// GPUE_Q3_CT=1: cached ext/disc gathers in the fused q3 kernel (A/B vs the
// NT default, which keeps L2 free for the 56 MB order bitset)
static bool q3_ct() {
    static int v = -1;
    if (v < 0) {
        const char* e = getenv("GPUE_Q3_CT");
        v = (e && atoi(e)) ? 1 : 0;
    }
    return v == 1;
}
int gpue_q3_probe_agg_t(gpue_session* s, gpue_dbuf* lk, gpue_dbuf* ext, gpue_dbuf* disc,
                        gpue_dbuf* ship, uint64_t n, gpue_dbuf* order_bits,
                        int32_t ship_cutoff, gpue_agg_table* at, gpue_dbuf* out_keys,
                        gpue_dbuf* out_sums, uint64_t max_out, uint64_t* n_groups) {
    ARG_CHECK(s && lk && ext && disc && ship && order_bits && at && out_keys && out_sums &&
              n_groups);
    int rc = agg_table_reset(at, /*with_counts=*/false);
    if (rc != GPUE_OK) return rc;
    // MEASURED NEGATIVE (r02): the two-stage wave-queue probe is 5.76 vs
    // 5.65 ms — compaction doubles lane density but drops per-lane MLP from
    // 4 in-flight bitset gathers (int4 quads) to 1 per drain, and the leg is
    // latency-bound on a 56 MB footprint (profiles/r02_bitgather_ubench.json;
    // same lesson as round 1's ballot-form rejection). Kept env-gated.
    const char* wq = getenv("GPUE_Q3_WQ");
    if (wq && atoi(wq))
        hipLaunchKernelGGL(k_q3_probe_agg_wq, dim3(grid_capped(n, env_cap("GPUE_GRID_Q3", MAX_GRID))), dim3(BLOCK), 0, s->stream,
                           (const int64_t*)lk->ptr, (const int64_t*)ext->ptr,
                           (const int64_t*)disc->ptr, (const int32_t*)ship->ptr, n,
                           (const uint32_t*)order_bits->ptr, ship_cutoff, at->slots,
                           at->sums, at->counts, at->cap - 1, s->d_agg_err, at->n_groups);
    else
        hipLaunchKernelGGL(q3_ct() ? k_q3_probe_agg<true> : k_q3_probe_agg<false>, dim3(grid_capped(n, env_cap("GPUE_GRID_Q3", MAX_GRID))), dim3(BLOCK), 0, s->stream,
                           (const int64_t*)lk->ptr, (const int64_t*)ext->ptr,
                           (const int64_t*)disc->ptr, (const int32_t*)ship->ptr, n,
                           (const uint32_t*)order_bits->ptr, ship_cutoff, at->slots,
                           at->sums, at->counts, at->cap - 1, s->d_agg_err, at->n_groups);
    hipLaunchKernelGGL(k_hash_agg_emit, dim3(grid_for(at->cap)), dim3(BLOCK), 0, s->stream,
                       at->slots, at->sums, at->counts, at->cap, at->cursor, max_out,
                       (uint64_t*)out_keys->ptr, (int64_t*)out_sums->ptr, nullptr);
    unsigned long long groups = 0;
    HIP_CHECK(hipMemcpyAsync(&groups, at->cursor, 8, hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    *n_groups = groups;
    int erc = agg_err_check(s, "q3");
    if (erc != GPUE_OK) return erc;
    if (groups > max_out) {
        snprintf(g_err, sizeof(g_err), "q3: %llu groups exceed max_out %llu", groups,
                 (unsigned long long)max_out);
        return GPUE_ERR_ARG;
    }
    return GPUE_OK;
}

// probe-accumulate only (no table reset, no emit, async): the chunked
// exchange pipeline probes each received block as it lands; callers reset
// once per step (gpue_agg_table_reset) and emit once at the end
// (gpue_hash_agg_emit_u64)
extern "C" int gpue_q3_probe_accum(gpue_session* s, gpue_dbuf* lk, gpue_dbuf* ext,
                                   gpue_dbuf* disc, gpue_dbuf* ship, uint64_t n,
                                   gpue_dbuf* order_bits, int32_t ship_cutoff,
                                   gpue_agg_table* at);
int gpue_q3_probe_accum(gpue_session* s, gpue_dbuf* lk, gpue_dbuf* ext, gpue_dbuf* disc,
                        gpue_dbuf* ship, uint64_t n, gpue_dbuf* order_bits,
                        int32_t ship_cutoff, gpue_agg_table* at) {
    ARG_CHECK(s && lk && ext && disc && ship && order_bits && at);
    hipLaunchKernelGGL(q3_ct() ? k_q3_probe_agg<true> : k_q3_probe_agg<false>, dim3(grid_capped(n, env_cap("GPUE_GRID_Q3", MAX_GRID))), dim3(BLOCK), 0, s->stream,
                       (const int64_t*)lk->ptr, (const int64_t*)ext->ptr,
                       (const int64_t*)disc->ptr, (const int32_t*)ship->ptr, n,
                       (const uint32_t*)order_bits->ptr, ship_cutoff, at->slots, at->sums,
                       at->counts, at->cap - 1, s->d_agg_err, at->n_groups);
    HIP_CHECK(hipGetLastError());
    return GPUE_OK;
}

extern "C" int gpue_q3_probe_agg_part(gpue_session* s, gpue_dbuf* lk, gpue_dbuf* ext,
                                      gpue_dbuf* disc, gpue_dbuf* ship, uint64_t n,
                                      uint64_t n_orders, gpue_dbuf* order_bits,
                                      int32_t ship_cutoff, gpue_agg_table* at,
                                      gpue_dbuf* keys_scratch, gpue_dbuf* vals_scratch,
                                      uint32_t nparts, gpue_dbuf* out_keys,
                                      gpue_dbuf* out_sums, uint64_t max_out,
                                      uint64_t* n_groups);
int gpue_q3_probe_agg_part(gpue_session* s, gpue_dbuf* lk, gpue_dbuf* ext, gpue_dbuf* disc,
                           gpue_dbuf* ship, uint64_t n, uint64_t n_orders,
                           gpue_dbuf* order_bits, int32_t ship_cutoff, gpue_agg_table* at,
                           gpue_dbuf* keys_scratch, gpue_dbuf* vals_scratch, uint32_t nparts,
                           gpue_dbuf* out_keys, gpue_dbuf* out_sums, uint64_t max_out,
                           uint64_t* n_groups) {
    ARG_CHECK(s && lk && ext && disc && ship && order_bits && at && keys_scratch &&
              vals_scratch && out_keys && out_sums && n_groups);
    ARG_CHECK(nparts >= 1 && nparts <= 256 && n_orders > 0);
    ARG_CHECK(keys_scratch->bytes >= n * 4 && vals_scratch->bytes >= n * 8);
    int rc = agg_table_reset(at, /*with_counts=*/false);
    if (rc != GPUE_OK) return rc;
    uint64_t slice = (n_orders + nparts - 1) / nparts;
    uint32_t nb = grid_capped(n, env_cap("GPUE_GRID_Q3P", 512));
    uint64_t tile = (n + nb - 1) / nb;
    const uint32_t nw = BLOCK / WAVE; // waves per block
    uint64_t hist_len = (uint64_t)nb * nw * nparts;
    uint32_t* d_hist = nullptr;
    uint64_t* d_off = nullptr;
    HIP_CHECK(hipMalloc(&d_hist, hist_len * 4));
    hipLaunchKernelGGL(k_q3_part_hist, dim3(nb), dim3(BLOCK), nw * nparts * 4, s->stream,
                       (const int64_t*)lk->ptr, (const int32_t*)ship->ptr, n, ship_cutoff,
                       slice, nparts, tile, d_hist);
    uint32_t* h_hist = (uint32_t*)malloc(hist_len * 4);
    uint64_t* h_off = (uint64_t*)malloc(hist_len * 8);
    uint64_t* pstart = (uint64_t*)malloc((nparts + 1) * 8);
    HIP_CHECK(hipMemcpyAsync(h_hist, d_hist, hist_len * 4, hipMemcpyDeviceToHost,
                             s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    uint64_t acc = 0;
    for (uint32_t p = 0; p < nparts; p++) {
        pstart[p] = acc;
        for (uint32_t b = 0; b < nb; b++)
            for (uint32_t w = 0; w < nw; w++) {
                uint64_t idx = ((uint64_t)b * nw + w) * nparts + p;
                h_off[idx] = acc;
                acc += h_hist[idx];
            }
    }
    pstart[nparts] = acc;
    HIP_CHECK(hipMalloc(&d_off, hist_len * 8));
    if (env_cap("GPUE_Q3P_STAGED", 1) == 1) {
        // per-BLOCK offsets (partition-major): the staged emit keeps one
        // cursor per (block, partition)
        uint64_t* h_boff = (uint64_t*)malloc((uint64_t)nb * nparts * 8);
        uint64_t acc2 = 0;
        for (uint32_t p = 0; p < nparts; p++)
            for (uint32_t b = 0; b < nb; b++) {
                h_boff[(uint64_t)b * nparts + p] = acc2;
                for (uint32_t w = 0; w < nw; w++)
                    acc2 += h_hist[((uint64_t)b * nw + w) * nparts + p];
            }
        HIP_CHECK(hipMemcpyAsync(d_off, h_boff, (uint64_t)nb * nparts * 8,
                                 hipMemcpyHostToDevice, s->stream));
        HIP_CHECK(hipStreamSynchronize(s->stream));
        free(h_boff);
        hipLaunchKernelGGL(k_q3_part_emit_staged, dim3(nb), dim3(BLOCK), 0, s->stream,
                           (const int64_t*)lk->ptr, (const int64_t*)ext->ptr,
                           (const int64_t*)disc->ptr, (const int32_t*)ship->ptr, n,
                           ship_cutoff, slice, nparts, tile, d_off,
                           (uint32_t*)keys_scratch->ptr, (long long*)vals_scratch->ptr);
    } else {
    HIP_CHECK(hipMemcpyAsync(d_off, h_off, hist_len * 8, hipMemcpyHostToDevice, s->stream));
    hipLaunchKernelGGL(k_q3_part_emit, dim3(nb), dim3(BLOCK), nw * nparts * 8, s->stream,
                       (const int64_t*)lk->ptr, (const int64_t*)ext->ptr,
                       (const int64_t*)disc->ptr, (const int32_t*)ship->ptr, n, ship_cutoff,
                       slice, nparts, tile, d_off, (uint32_t*)keys_scratch->ptr,
                       (long long*)vals_scratch->ptr);
    }
    // fused pass B: assign contiguous block ranges per partition, one launch
    uint32_t total_blocks = env_cap("GPUE_GRID_Q3S", 2048);
    if (total_blocks < nparts) total_blocks = nparts;
    uint32_t* bpp = (uint32_t*)malloc(nparts * 4);
    uint32_t* fbp = (uint32_t*)malloc(nparts * 4);
    uint64_t total_rows = acc ? acc : 1;
    uint32_t assigned = 0;
    for (uint32_t p = 0; p < nparts; p++) {
        uint64_t cnt = pstart[p + 1] - pstart[p];
        uint32_t want = (uint32_t)((__uint128_t)cnt * total_blocks / total_rows);
        bpp[p] = cnt ? (want ? want : 1u) : 0u;
        assigned += bpp[p];
    }
    uint32_t nb2 = 0;
    uint32_t* bpart = (uint32_t*)malloc((assigned ? assigned : 1) * 4);
    for (uint32_t p = 0; p < nparts; p++) {
        fbp[p] = nb2;
        for (uint32_t b = 0; b < bpp[p]; b++) bpart[nb2++] = p;
    }
    if (nb2 > 0) {
        uint32_t *d_bpart = nullptr, *d_bpp = nullptr, *d_fbp = nullptr;
        uint64_t* d_pstart = nullptr;
        HIP_CHECK(hipMalloc(&d_bpart, nb2 * 4));
        HIP_CHECK(hipMalloc(&d_bpp, nparts * 4));
        HIP_CHECK(hipMalloc(&d_fbp, nparts * 4));
        HIP_CHECK(hipMalloc(&d_pstart, (nparts + 1) * 8));
        HIP_CHECK(hipMemcpyAsync(d_bpart, bpart, nb2 * 4, hipMemcpyHostToDevice, s->stream));
        HIP_CHECK(hipMemcpyAsync(d_bpp, bpp, nparts * 4, hipMemcpyHostToDevice, s->stream));
        HIP_CHECK(hipMemcpyAsync(d_fbp, fbp, nparts * 4, hipMemcpyHostToDevice, s->stream));
        HIP_CHECK(hipMemcpyAsync(d_pstart, pstart, (nparts + 1) * 8, hipMemcpyHostToDevice,
                                 s->stream));
        hipLaunchKernelGGL(k_q3_probe_slices_fused, dim3(nb2), dim3(BLOCK), 0, s->stream,
                           (const uint32_t*)keys_scratch->ptr,
                           (const long long*)vals_scratch->ptr, d_bpart, d_pstart, d_bpp,
                           d_fbp, (const uint32_t*)order_bits->ptr, at->slots, at->sums,
                           at->cap - 1, s->d_agg_err, at->n_groups);
        HIP_CHECK(hipStreamSynchronize(s->stream));
        (void)hipFree(d_bpart);
        (void)hipFree(d_bpp);
        (void)hipFree(d_fbp);
        (void)hipFree(d_pstart);
    }
    hipLaunchKernelGGL(k_hash_agg_emit, dim3(grid_for(at->cap)), dim3(BLOCK), 0, s->stream,
                       at->slots, at->sums, at->counts, at->cap, at->cursor, max_out,
                       (uint64_t*)out_keys->ptr, (int64_t*)out_sums->ptr, nullptr);
    unsigned long long groups = 0;
    HIP_CHECK(hipMemcpyAsync(&groups, at->cursor, 8, hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    (void)hipFree(d_hist);
    (void)hipFree(d_off);
    free(h_hist);
    free(h_off);
    free(pstart);
    free(bpp);
    free(fbp);
    free(bpart);
    *n_groups = groups;
    if (groups > max_out) {
        snprintf(g_err, sizeof(g_err), "q3p: %llu groups exceed max_out %llu", groups,
                 (unsigned long long)max_out);
        return GPUE_ERR_ARG;
    }
    return GPUE_OK;
}

int gpue_q3_probe_agg(gpue_session* s, gpue_dbuf* lk, gpue_dbuf* ext, gpue_dbuf* disc,
                      gpue_dbuf* ship, uint64_t n, gpue_dbuf* order_bits, int32_t ship_cutoff,
                      uint64_t capacity_hint, gpue_dbuf* out_keys, gpue_dbuf* out_sums,
                      uint64_t max_out, uint64_t* n_groups) {
    ARG_CHECK(s && lk && ext && disc && ship && order_bits && out_keys && out_sums && n_groups);
    uint64_t cap = 16;
    uint64_t want = capacity_hint ? capacity_hint : n / 4;
    while (cap < want) cap <<= 1;
    unsigned long long *d_slots = nullptr, *d_sums = nullptr, *d_counts = nullptr,
                       *d_cursor = nullptr;
    HIP_CHECK(hipMalloc(&d_slots, cap * 8));
    HIP_CHECK(hipMalloc(&d_sums, cap * 8));
    HIP_CHECK(hipMalloc(&d_counts, cap * 8));
    HIP_CHECK(hipMalloc(&d_cursor, 8));
    HIP_CHECK(hipMemsetAsync(d_slots, 0xFF, cap * 8, s->stream));
    HIP_CHECK(hipMemsetAsync(d_sums, 0, cap * 8, s->stream));
    HIP_CHECK(hipMemsetAsync(d_counts, 0, cap * 8, s->stream));
    HIP_CHECK(hipMemsetAsync(d_cursor, 0, 8, s->stream));
    hipLaunchKernelGGL(q3_ct() ? k_q3_probe_agg<true> : k_q3_probe_agg<false>, dim3(grid_capped(n, env_cap("GPUE_GRID_Q3", MAX_GRID))), dim3(BLOCK), 0, s->stream,
                       (const int64_t*)lk->ptr, (const int64_t*)ext->ptr,
                       (const int64_t*)disc->ptr, (const int32_t*)ship->ptr, n,
                       (const uint32_t*)order_bits->ptr, ship_cutoff, d_slots, d_sums,
                       d_counts, cap - 1, s->d_agg_err, nullptr);
    hipLaunchKernelGGL(k_hash_agg_emit, dim3(grid_for(cap)), dim3(BLOCK), 0, s->stream,
                       d_slots, d_sums, d_counts, cap, d_cursor, max_out,
                       (uint64_t*)out_keys->ptr, (int64_t*)out_sums->ptr, nullptr);
    unsigned long long groups = 0;
    HIP_CHECK(hipMemcpyAsync(&groups, d_cursor, 8, hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    (void)hipFree(d_slots);
    (void)hipFree(d_sums);
    (void)hipFree(d_counts);
    (void)hipFree(d_cursor);
    *n_groups = groups;
    int erc = agg_err_check(s, "q3");
    if (erc != GPUE_OK) return erc;
    if (groups > max_out) {
        snprintf(g_err, sizeof(g_err), "q3: %llu groups exceed max_out %llu", groups,
                 (unsigned long long)max_out);
        return GPUE_ERR_ARG;
    }
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// row gather by index — the exchange sink's add_rows_selective analog
// (reference exchange_sink_operator.cpp:670: per-channel row slices are
// materialized by gathering source rows at the counting-sorted indexes)
// ---------------------------------------------------------------------------
__global__ void k_gather_u32(const uint32_t* __restrict__ in,
                             const uint32_t* __restrict__ idx, uint64_t n,
                             uint32_t* __restrict__ out) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
        out[i] = in[idx[i]];
}

__global__ void k_gather_u64(const uint64_t* __restrict__ in,
                             const uint32_t* __restrict__ idx, uint64_t n,
                             uint64_t* __restrict__ out) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
        out[i] = in[idx[i]];
}

extern "C" int gpue_gather_u64(gpue_session* s, gpue_dbuf* in, gpue_dbuf* idx, uint64_t n,
                               gpue_dbuf* out);
int gpue_gather_u64(gpue_session* s, gpue_dbuf* in, gpue_dbuf* idx, uint64_t n,
                    gpue_dbuf* out) {
    ARG_CHECK(s && in && idx && out);
    ARG_CHECK(idx->bytes >= n * 4 && out->bytes >= n * 8);
    hipLaunchKernelGGL(k_gather_u64, dim3(grid_for(n)), dim3(BLOCK), 0, s->stream,
                       (const uint64_t*)in->ptr, (const uint32_t*)idx->ptr, n,
                       (uint64_t*)out->ptr);
    HIP_CHECK(hipGetLastError());
    return GPUE_OK;
}

extern "C" int gpue_gather_u32(gpue_session* s, gpue_dbuf* in, gpue_dbuf* idx, uint64_t n,
                               gpue_dbuf* out);
int gpue_gather_u32(gpue_session* s, gpue_dbuf* in, gpue_dbuf* idx, uint64_t n,
                    gpue_dbuf* out) {
    ARG_CHECK(s && in && idx && out);
    ARG_CHECK(idx->bytes >= n * 4 && out->bytes >= n * 4);
    hipLaunchKernelGGL(k_gather_u32, dim3(grid_for(n)), dim3(BLOCK), 0, s->stream,
                       (const uint32_t*)in->ptr, (const uint32_t*)idx->ptr, n,
                       (uint32_t*)out->ptr);
    HIP_CHECK(hipGetLastError());
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// Storage ingress: decode the reference's numeric page
// body — bshuf_compress_lz4 framing (per block: 4-byte big-endian length +
// one LZ4 block of the bit-plane-transposed values; bitshuffle 0.5.1
// published algorithm, block = 2048 int32 elements). Three kernels:
// block-starts scan (serial walk of the lengths), per-block LZ4 decode (one
// thread per independent block — thousands of blocks in flight), and the
// bit-plane un-transpose (grid-stride per element). Formats restated in
// oracle/oracle.c; parity GPU <-> oracle (byte-compat caveat in DESIGN.md).
// ---------------------------------------------------------------------------
static constexpr uint32_t BSHUF_BLOCK_I32 = 2048;

__global__ void k_page_block_starts(const uint8_t* __restrict__ page, uint32_t n_blocks,
                                    uint32_t* __restrict__ starts) {
    if (threadIdx.x == 0 && blockIdx.x == 0) {
        uint32_t off = 0;
        for (uint32_t b = 0; b < n_blocks; b++) {
            starts[b] = off;
            uint32_t c = ((uint32_t)page[off] << 24) | ((uint32_t)page[off + 1] << 16) |
                         ((uint32_t)page[off + 2] << 8) | page[off + 3];
            off += 4 + c;
        }
        starts[n_blocks] = off;
    }
}

__global__ void k_lz4_decode_blocks(const uint8_t* __restrict__ page,
                                    const uint32_t* __restrict__ starts, uint32_t n_blocks,
                                    uint32_t n_values, uint8_t* __restrict__ scratch,
                                    unsigned long long* __restrict__ error_out) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t b = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; b < n_blocks;
         b += stride) {
        uint32_t elems = min(n_values - (uint32_t)b * BSHUF_BLOCK_I32, BSHUF_BLOCK_I32);
        uint32_t s = starts[b];
        uint32_t comp_n = ((uint32_t)page[s] << 24) | ((uint32_t)page[s + 1] << 16) |
                          ((uint32_t)page[s + 2] << 8) | page[s + 3];
        const uint8_t* src = page + s + 4;
        uint8_t* dst = scratch + (uint64_t)b * BSHUF_BLOCK_I32 * 4;
        uint64_t dst_cap = (uint64_t)elems * 4;
        uint64_t si = 0, d = 0;
        while (si < comp_n) {
            uint8_t tok = src[si++];
            uint64_t lit = tok >> 4;
            if (lit == 15) {
                uint8_t x;
                do { x = src[si++]; lit += x; } while (x == 255);
            }
            if (d + lit > dst_cap) { atomicOr(error_out, 1ull); return; }
            for (uint64_t k = 0; k < lit; k++) dst[d + k] = src[si + k];
            si += lit;
            d += lit;
            if (si >= comp_n) break;
            uint32_t off = src[si] | ((uint32_t)src[si + 1] << 8);
            si += 2;
            uint64_t mlen = tok & 0xF;
            if (mlen == 15) {
                uint8_t x;
                do { x = src[si++]; mlen += x; } while (x == 255);
            }
            mlen += 4;
            if (d + mlen > dst_cap || off > d) { atomicOr(error_out, 1ull); return; }
            for (uint64_t k = 0; k < mlen; k++) { dst[d] = dst[d - off]; d++; }
        }
        if (d != dst_cap) atomicOr(error_out, 1ull);
    }
}

__global__ void k_bshuf_untranspose_i32(const uint8_t* __restrict__ scratch, uint32_t n_values,
                                        int32_t* __restrict__ out) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t e = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; e < n_values;
         e += stride) {
        uint32_t b = (uint32_t)(e / BSHUF_BLOCK_I32);
        uint32_t le = (uint32_t)(e % BSHUF_BLOCK_I32);
        uint32_t elems = min(n_values - b * BSHUF_BLOCK_I32, BSHUF_BLOCK_I32);
        const uint8_t* blk = scratch + (uint64_t)b * BSHUF_BLOCK_I32 * 4;
        uint32_t plane_bytes = elems / 8;
        uint32_t v = 0;
        #pragma unroll
        for (uint32_t p = 0; p < 32; p++) {
            uint32_t bit = (blk[(uint64_t)p * plane_bytes + le / 8] >> (le % 8)) & 1u;
            v |= bit << ((p / 8) * 8 + (p % 8));
        }
        out[e] = (int32_t)v;
    }
}

// ---------------------------------------------------------------------------
// RLE page decode for int32 (storage ingress):
// storage/rowset/rle_page.h (4-byte LE num_elements header) over
// base/bit/rle_encoding.h's Parquet-style RLE/bit-pack hybrid at
// bit_width = 32 (rle_page.h:82) — every run is byte-aligned:
//   repeated := varint(count<<1) + 4-byte LE value
//   literal  := byte(groups<<1|1) + groups*8 LE u32 values
// Decode is two-phase: a single-wave scan walks the (compact) run headers
// into a run table — one entry per RUN, not per value — then a grid-stride
// fill kernel binary-searches the table per output tile. The oracle
// restates the reference's own encoder byte-for-byte (hand KATs in
// tests/test_page_decode.py), so GPU<->oracle parity pins the format.
// ---------------------------------------------------------------------------
__global__ void k_rle_scan_i32(const uint8_t* __restrict__ page, uint64_t page_bytes,
                               uint32_t* __restrict__ run_start,
                               uint64_t* __restrict__ run_info, // bit0: literal; >>1: byte off or value
                               uint32_t* __restrict__ n_runs_out,
                               uint32_t* __restrict__ err) {
    if (blockIdx.x != 0 || threadIdx.x != 0) return;
    uint32_t n;
    memcpy(&n, page, 4);
    uint64_t pos = 4;
    uint32_t out = 0, nruns = 0;
    while (out < n) {
        if (pos >= page_bytes) { *err = 1; break; }
        // varint indicator
        uint32_t ind = 0;
        int shift = 0;
        for (;;) {
            uint8_t b = page[pos++];
            ind |= (uint32_t)(b & 0x7F) << shift;
            if (!(b & 0x80)) break;
            shift += 7;
        }
        run_start[nruns] = out;
        if (ind & 1) {
            uint32_t cnt = (ind >> 1) * 8;
            if (pos + (uint64_t)cnt * 4 > page_bytes) { *err = 1; break; }
            run_info[nruns] = (pos << 1) | 1;
            pos += (uint64_t)cnt * 4;
            out += cnt; // may pad past n (zero-padded tail group)
        } else {
            if (pos + 4 > page_bytes) { *err = 1; break; }
            uint32_t v;
            memcpy(&v, page + pos, 4);
            pos += 4;
            run_info[nruns] = ((uint64_t)v << 1);
            out += (ind >> 1);
        }
        nruns++;
    }
    run_start[nruns] = out < n ? n : out;
    *n_runs_out = nruns;
}

__global__ void k_rle_fill_i32(const uint8_t* __restrict__ page,
                               const uint32_t* __restrict__ run_start,
                               const uint64_t* __restrict__ run_info, uint32_t n_runs,
                               uint64_t n, int32_t* __restrict__ out) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        // binary search: largest run with run_start <= i
        uint32_t lo = 0, hi = n_runs - 1;
        while (lo < hi) {
            uint32_t mid = (lo + hi + 1) >> 1;
            if ((uint64_t)run_start[mid] <= i) lo = mid;
            else hi = mid - 1;
        }
        uint64_t info = run_info[lo];
        if (info & 1) { // literal: byte-assemble the LE u32 (unaligned)
            const uint8_t* p = page + (info >> 1) + (i - run_start[lo]) * 4;
            out[i] = (int32_t)((uint32_t)p[0] | ((uint32_t)p[1] << 8) |
                               ((uint32_t)p[2] << 16) | ((uint32_t)p[3] << 24));
        } else {
            out[i] = (int32_t)(uint32_t)(info >> 1);
        }
    }
}

// ---------------------------------------------------------------------------
// BinaryPlainPage decode (PLAIN_ENCODING, binary_plain_page.h:28-46): body
// of concatenated strings + u32-LE absolute-offset trailer + u32 count.
// Decodes straight into the engine's BinaryColumn shape (bytes + u32
// offsets[n+1]) — the format a dict page's dictionary itself ships in, so
// this + RLE/bitshuffle code pages + gpue_dict_decode_binary covers the
// whole dict-encoded varchar ingress chain on device.
// ---------------------------------------------------------------------------
__global__ void k_binary_plain_offsets(const uint8_t* __restrict__ page, uint64_t body,
                                       uint32_t n, uint32_t* __restrict__ out_offsets) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i <= n; i += stride) {
        if (i == n) {
            out_offsets[n] = (uint32_t)body;
        } else {
            uint32_t v;
            memcpy(&v, page + body + i * 4, 4);
            out_offsets[i] = v;
        }
    }
}

extern "C" int gpue_page_decode_binary_plain(gpue_session* s, gpue_dbuf* page,
                                             uint64_t n_values, gpue_dbuf* out_bytes,
                                             gpue_dbuf* out_offsets);
int gpue_page_decode_binary_plain(gpue_session* s, gpue_dbuf* page, uint64_t n_values,
                                  gpue_dbuf* out_bytes, gpue_dbuf* out_offsets) {
    ARG_CHECK(s && page && out_bytes && out_offsets && page->bytes >= 4);
    ARG_CHECK(out_offsets->bytes >= (n_values + 1) * 4);
    uint32_t n = 0;
    HIP_CHECK(hipMemcpyAsync(&n, (const uint8_t*)page->ptr + page->bytes - 4, 4,
                             hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    if (n != n_values || page->bytes < 4 + (uint64_t)n * 4) {
        snprintf(g_err, sizeof(g_err), "binary_plain page: n=%u want %llu", n,
                 (unsigned long long)n_values);
        return GPUE_ERR_ARG;
    }
    uint64_t body = page->bytes - 4 - (uint64_t)n * 4;
    ARG_CHECK(out_bytes->bytes >= body || body == 0);
    if (body)
        HIP_CHECK(hipMemcpyAsync(out_bytes->ptr, page->ptr, body,
                                 hipMemcpyDeviceToDevice, s->stream));
    hipLaunchKernelGGL(k_binary_plain_offsets, dim3(grid_for(n + 1)), dim3(BLOCK), 0,
                       s->stream, (const uint8_t*)page->ptr, body, n,
                       (uint32_t*)out_offsets->ptr);
    HIP_CHECK(hipStreamSynchronize(s->stream));
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// BinaryPrefixPage decode (PREFIX_ENCODING, binary_prefix_page.{h,cpp}):
// front coding with a restart point every 16 entries — decode parallelizes
// per restart GROUP. Phase A (thread per group) parses the varint entry
// headers into per-entry {share_len, data_off} + lengths; the shared
// block-scan turns lengths into BinaryColumn offsets; phase B (thread per
// group) reconstructs bytes sequentially within its group (entry e copies
// entry e-1's materialized prefix — an intra-thread dependency only, since
// restarts reset the chain).
// ---------------------------------------------------------------------------
__device__ static inline uint64_t bp_varint(const uint8_t* p, uint64_t pos, uint32_t* v) {
    uint32_t r = 0;
    int shift = 0;
    for (;;) {
        uint8_t b = p[pos++];
        r |= (uint32_t)(b & 0x7F) << shift;
        if (!(b & 0x80)) break;
        shift += 7;
    }
    *v = r;
    return pos;
}

__global__ void k_bprefix_scan(const uint8_t* __restrict__ page, uint64_t restart_tab,
                               uint32_t nrestart, uint32_t n,
                               uint32_t* __restrict__ lens,
                               uint2* __restrict__ meta) { // {share, data_off}
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t g = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; g < nrestart;
         g += stride) {
        uint32_t ro;
        memcpy(&ro, page + restart_tab + g * 4, 4);
        uint64_t pos = ro;
        uint32_t e0 = (uint32_t)g * 16;
        uint32_t e1 = min(e0 + 16, n);
        for (uint32_t e = e0; e < e1; e++) {
            uint32_t share, non_share;
            pos = bp_varint(page, pos, &share);
            pos = bp_varint(page, pos, &non_share);
            lens[e] = share + non_share;
            meta[e] = make_uint2(share, (uint32_t)pos);
            pos += non_share;
        }
    }
}

__global__ void k_bprefix_offsets(const uint64_t* __restrict__ off64, uint32_t n,
                                  uint32_t* __restrict__ out_offsets) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t e = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; e <= n; e += stride)
        out_offsets[e] = (uint32_t)off64[e];
}

__global__ void k_bprefix_fill(const uint8_t* __restrict__ page, uint32_t nrestart,
                               uint32_t n, const uint2* __restrict__ meta,
                               const uint32_t* __restrict__ out_offsets,
                               uint8_t* __restrict__ out_bytes) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t g = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; g < nrestart;
         g += stride) {
        uint32_t e0 = (uint32_t)g * 16;
        uint32_t e1 = min(e0 + 16, n);
        for (uint32_t e = e0; e < e1; e++) {
            uint2 m = meta[e];
            uint32_t w = out_offsets[e];
            uint32_t prev = e > e0 ? out_offsets[e - 1] : 0;
            for (uint32_t j = 0; j < m.x; j++) out_bytes[w + j] = out_bytes[prev + j];
            uint32_t nsh = out_offsets[e + 1] - w - m.x;
            for (uint32_t j = 0; j < nsh; j++) out_bytes[w + m.x + j] = page[m.y + j];
        }
    }
}

extern "C" int gpue_page_decode_binary_prefix(gpue_session* s, gpue_dbuf* page,
                                              uint64_t n_values, gpue_dbuf* out_bytes,
                                              gpue_dbuf* out_offsets);
int gpue_page_decode_binary_prefix(gpue_session* s, gpue_dbuf* page, uint64_t n_values,
                                   gpue_dbuf* out_bytes, gpue_dbuf* out_offsets) {
    ARG_CHECK(s && page && out_bytes && out_offsets && page->bytes >= 13);
    ARG_CHECK(out_offsets->bytes >= (n_values + 1) * 4);
    uint32_t nrestart = 0;
    HIP_CHECK(hipMemcpyAsync(&nrestart, (const uint8_t*)page->ptr + page->bytes - 4, 4,
                             hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    uint64_t restart_tab = page->bytes - 4 - (uint64_t)nrestart * 4;
    uint64_t trailer = restart_tab - 1 - 4;
    uint32_t n = 0;
    HIP_CHECK(hipMemcpyAsync(&n, (const uint8_t*)page->ptr + trailer, 4,
                             hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    if (n != n_values || nrestart != (n + 15) / 16) {
        snprintf(g_err, sizeof(g_err), "binary_prefix page: n=%u restarts=%u want %llu",
                 n, nrestart, (unsigned long long)n_values);
        return GPUE_ERR_ARG;
    }
    if (n == 0) return GPUE_OK;
    uint32_t* d_lens = nullptr;
    uint2* d_meta = nullptr;
    uint64_t *d_bsums = nullptr, *d_off64 = nullptr;
    HIP_CHECK(hipMalloc(&d_lens, (uint64_t)n * 4));
    HIP_CHECK(hipMalloc(&d_meta, (uint64_t)n * 8));
    uint32_t nb = grid_for(n);
    uint64_t tile = ((uint64_t)n + nb - 1) / nb;
    HIP_CHECK(hipMalloc(&d_bsums, (nb + 1) * 8));
    HIP_CHECK(hipMalloc(&d_off64, ((uint64_t)n + 1) * 8));
    hipLaunchKernelGGL(k_bprefix_scan, dim3(grid_for(nrestart)), dim3(BLOCK), 0, s->stream,
                       (const uint8_t*)page->ptr, restart_tab, nrestart, n, d_lens, d_meta);
    hipLaunchKernelGGL(k_block_sums_u32, dim3(nb), dim3(BLOCK), 0, s->stream, d_lens, n,
                       tile, d_bsums);
    hipLaunchKernelGGL(k_scan_small, dim3(1), dim3(1), 0, s->stream, d_bsums, nb);
    hipLaunchKernelGGL(k_scan_offsets, dim3(nb), dim3(BLOCK), 0, s->stream, d_lens, n, tile,
                       d_bsums, d_off64);
    uint64_t total = 0;
    HIP_CHECK(hipMemcpyAsync(&total, d_bsums + nb, 8, hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    int rc = GPUE_OK;
    if (out_bytes->bytes < total && total > 0) {
        snprintf(g_err, sizeof(g_err), "binary_prefix: out_bytes %llu < total %llu",
                 (unsigned long long)out_bytes->bytes, (unsigned long long)total);
        rc = GPUE_ERR_ARG;
    } else {
        // d_off64 holds per-entry EXCLUSIVE offsets; write u32 offsets[ n+1 ]
        HIP_CHECK(hipMemcpyAsync(d_off64 + n, d_bsums + nb, 8, hipMemcpyDeviceToDevice,
                                 s->stream));
        hipLaunchKernelGGL(k_bprefix_offsets, dim3(grid_for(n + 1)), dim3(BLOCK), 0,
                           s->stream, d_off64, n, (uint32_t*)out_offsets->ptr);
        hipLaunchKernelGGL(k_bprefix_fill, dim3(grid_for(nrestart)), dim3(BLOCK), 0,
                           s->stream, (const uint8_t*)page->ptr, nrestart, n, d_meta,
                           (const uint32_t*)out_offsets->ptr, (uint8_t*)out_bytes->ptr);
        HIP_CHECK(hipStreamSynchronize(s->stream));
    }
    (void)hipFree(d_lens);
    (void)hipFree(d_meta);
    (void)hipFree(d_bsums);
    (void)hipFree(d_off64);
    return rc;
}

// ---------------------------------------------------------------------------
// Frame-of-reference page decode for int32 (FOR_ENCODING,
// frame_of_reference_page.h over frame_of_reference_coding.{h,cpp}; format
// and the decoder-authoritative frame-advance documented in oracle.c).
// Frames are independent 128-value units -> one 128-thread block per frame:
// each thread bit-unpacks its value (MSB-first), format 1 (ascending
// deltas) does an LDS inclusive scan, formats 0/2 add the frame min / pass
// through. The footer walk is a single-wave scan kernel like the RLE one.
// ---------------------------------------------------------------------------
__global__ void k_for_scan_i32(const uint8_t* __restrict__ page, uint64_t page_bytes,
                               uint64_t* __restrict__ frame_off, // byte offset of frame body
                               uint8_t* __restrict__ frame_fmt,
                               uint8_t* __restrict__ frame_bw,
                               uint32_t* __restrict__ meta) { // [nframes, err, n, fvn]
    if (blockIdx.x != 0 || threadIdx.x != 0) return;
    if (page_bytes < 5) { meta[1] = 1; return; }
    uint8_t fvn = page[page_bytes - 5];
    uint32_t n;
    memcpy(&n, page + page_bytes - 4, 4);
    meta[2] = n;
    meta[3] = fvn;
    if (n == 0 || fvn == 0) { meta[0] = 0; return; }
    uint32_t nframes = n / fvn + (n % fvn != 0);
    uint64_t footer = (uint64_t)nframes * 2 + 5;
    if (footer > page_bytes) { meta[1] = 1; return; }
    uint64_t fpos = page_bytes - footer;
    uint64_t off = 0;
    for (uint32_t f = 0; f < nframes; f++) {
        uint8_t fmt = page[fpos + f * 2];
        uint8_t bw = page[fpos + f * 2 + 1];
        frame_off[f] = off;
        frame_fmt[f] = fmt;
        frame_bw[f] = bw;
        off += 4 + (uint64_t)bw * fvn / 8;
        if (off > page_bytes) { meta[1] = 1; return; }
    }
    meta[0] = nframes;
}

__global__ void k_for_fill_i32(const uint8_t* __restrict__ page,
                               const uint64_t* __restrict__ frame_off,
                               const uint8_t* __restrict__ frame_fmt,
                               const uint8_t* __restrict__ frame_bw, uint32_t fvn,
                               uint64_t n, int32_t* __restrict__ out) {
    __shared__ uint32_t vals[256]; // fvn <= 255 by format (u8)
    uint32_t f = blockIdx.x;
    uint32_t num = min((uint64_t)fvn, n - (uint64_t)f * fvn);
    const uint8_t* body = page + frame_off[f];
    uint32_t mn;
    memcpy(&mn, body, 4);
    const uint8_t* bits = body + 4;
    uint32_t bw = frame_bw[f];
    uint32_t t = threadIdx.x;
    if (t < num) {
        // bit-unpack value t: MSB-first stream, value bits big-endian
        uint64_t bit0 = (uint64_t)t * bw;
        uint32_t v = 0;
        for (uint32_t k = 0; k < bw; k++) {
            uint64_t b = bit0 + k;
            v |= (uint32_t)((bits[b >> 3] >> (7 - (b & 7))) & 1u) << (bw - k - 1);
        }
        vals[t] = v;
    }
    __syncthreads();
    uint8_t fmt = frame_fmt[f];
    uint32_t* o = (uint32_t*)out + (uint64_t)f * fvn;
    if (fmt == 1) {
        // ascending: inclusive scan of deltas (delta[0] encoded as 0)
        for (uint32_t s = 1; s < num; s <<= 1) {
            uint32_t add = (t < num && t >= s) ? vals[t - s] : 0u;
            __syncthreads();
            if (t < num) vals[t] += add;
            __syncthreads();
        }
        if (t < num) o[t] = mn + vals[t];
    } else if (fmt == 2) {
        if (t < num) o[t] = vals[t];
    } else {
        if (t < num) o[t] = mn + vals[t];
    }
}

extern "C" int gpue_page_decode_for_i32(gpue_session* s, gpue_dbuf* page,
                                        uint64_t n_values, gpue_dbuf* out);
int gpue_page_decode_for_i32(gpue_session* s, gpue_dbuf* page, uint64_t n_values,
                             gpue_dbuf* out) {
    ARG_CHECK(s && page && out && out->bytes >= n_values * 4 && page->bytes >= 5);
    uint32_t max_frames = (uint32_t)(n_values / 1 + 2); // bounded below by meta check
    uint64_t* d_off = nullptr;
    uint8_t *d_fmt = nullptr, *d_bw = nullptr;
    uint32_t* d_meta = nullptr;
    HIP_CHECK(hipMalloc(&d_off, (uint64_t)max_frames * 8));
    HIP_CHECK(hipMalloc(&d_fmt, max_frames));
    HIP_CHECK(hipMalloc(&d_bw, max_frames));
    HIP_CHECK(hipMalloc(&d_meta, 16));
    HIP_CHECK(hipMemsetAsync(d_meta, 0, 16, s->stream));
    hipLaunchKernelGGL(k_for_scan_i32, dim3(1), dim3(64), 0, s->stream,
                       (const uint8_t*)page->ptr, page->bytes, d_off, d_fmt, d_bw, d_meta);
    uint32_t meta[4] = {0, 0, 0, 0};
    HIP_CHECK(hipMemcpyAsync(meta, d_meta, 16, hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    int rc = GPUE_OK;
    if (meta[1] || meta[2] != n_values || meta[3] == 0 || meta[3] > 255) {
        snprintf(g_err, sizeof(g_err), "for page: err=%u n=%u (want %llu) fvn=%u",
                 meta[1], meta[2], (unsigned long long)n_values, meta[3]);
        rc = GPUE_ERR_ARG;
    } else if (meta[0] > 0) {
        hipLaunchKernelGGL(k_for_fill_i32, dim3(meta[0]), dim3(256), 0, s->stream,
                           (const uint8_t*)page->ptr, d_off, d_fmt, d_bw, meta[3],
                           n_values, (int32_t*)out->ptr);
        HIP_CHECK(hipStreamSynchronize(s->stream));
    }
    (void)hipFree(d_off);
    (void)hipFree(d_fmt);
    (void)hipFree(d_bw);
    (void)hipFree(d_meta);
    return rc;
}

// BOOL variant (bit_width = 1): repeated value is one byte; literal groups
// are bit-packed LSB-first, one byte per 8 values. Output u8 per value
// (the reference decodes bool columns to u8).
__global__ void k_rle_scan_bool(const uint8_t* __restrict__ page, uint64_t page_bytes,
                                uint32_t* __restrict__ run_start,
                                uint64_t* __restrict__ run_info,
                                uint32_t* __restrict__ n_runs_out,
                                uint32_t* __restrict__ err) {
    if (blockIdx.x != 0 || threadIdx.x != 0) return;
    uint32_t n;
    memcpy(&n, page, 4);
    uint64_t pos = 4;
    uint32_t out = 0, nruns = 0;
    while (out < n) {
        if (pos >= page_bytes) { *err = 1; break; }
        uint32_t ind = 0;
        int shift = 0;
        for (;;) {
            uint8_t b = page[pos++];
            ind |= (uint32_t)(b & 0x7F) << shift;
            if (!(b & 0x80)) break;
            shift += 7;
        }
        run_start[nruns] = out;
        if (ind & 1) {
            uint32_t groups = ind >> 1;
            if (pos + groups > page_bytes) { *err = 1; break; }
            run_info[nruns] = (pos << 1) | 1;
            pos += groups;
            out += groups * 8;
        } else {
            if (pos + 1 > page_bytes) { *err = 1; break; }
            run_info[nruns] = ((uint64_t)(page[pos] & 1) << 1);
            pos += 1;
            out += (ind >> 1);
        }
        nruns++;
    }
    run_start[nruns] = out < n ? n : out;
    *n_runs_out = nruns;
}

__global__ void k_rle_fill_bool(const uint8_t* __restrict__ page,
                                const uint32_t* __restrict__ run_start,
                                const uint64_t* __restrict__ run_info, uint32_t n_runs,
                                uint64_t n, uint8_t* __restrict__ out) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        uint32_t lo = 0, hi = n_runs - 1;
        while (lo < hi) {
            uint32_t mid = (lo + hi + 1) >> 1;
            if ((uint64_t)run_start[mid] <= i) lo = mid;
            else hi = mid - 1;
        }
        uint64_t info = run_info[lo];
        if (info & 1) {
            uint64_t d = i - run_start[lo];
            out[i] = (page[(info >> 1) + (d >> 3)] >> (d & 7)) & 1;
        } else {
            out[i] = (uint8_t)(info >> 1);
        }
    }
}

extern "C" int gpue_page_decode_rle_bool(gpue_session* s, gpue_dbuf* page,
                                         uint64_t n_values, gpue_dbuf* out);
int gpue_page_decode_rle_bool(gpue_session* s, gpue_dbuf* page, uint64_t n_values,
                              gpue_dbuf* out) {
    ARG_CHECK(s && page && out && out->bytes >= n_values && page->bytes >= 4);
    uint64_t max_runs = page->bytes + 2; // a bool run is >= 2 encoded bytes
    uint32_t* d_start = nullptr;
    uint64_t* d_info = nullptr;
    uint32_t* d_meta = nullptr;
    HIP_CHECK(hipMalloc(&d_start, (max_runs + 1) * 4));
    HIP_CHECK(hipMalloc(&d_info, max_runs * 8));
    HIP_CHECK(hipMalloc(&d_meta, 8));
    HIP_CHECK(hipMemsetAsync(d_meta, 0, 8, s->stream));
    hipLaunchKernelGGL(k_rle_scan_bool, dim3(1), dim3(64), 0, s->stream,
                       (const uint8_t*)page->ptr, page->bytes, d_start, d_info, d_meta,
                       d_meta + 1);
    uint32_t meta[2] = {0, 0};
    HIP_CHECK(hipMemcpyAsync(meta, d_meta, 8, hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    if (meta[1] || meta[0] == 0) {
        (void)hipFree(d_start); (void)hipFree(d_info); (void)hipFree(d_meta);
        if (meta[1]) {
            snprintf(g_err, sizeof(g_err), "rle bool page truncated/corrupt");
            return GPUE_ERR_ARG;
        }
        return n_values == 0 ? GPUE_OK : GPUE_ERR_ARG;
    }
    hipLaunchKernelGGL(k_rle_fill_bool, dim3(grid_for(n_values)), dim3(BLOCK), 0, s->stream,
                       (const uint8_t*)page->ptr, d_start, d_info, meta[0], n_values,
                       (uint8_t*)out->ptr);
    HIP_CHECK(hipStreamSynchronize(s->stream));
    (void)hipFree(d_start);
    (void)hipFree(d_info);
    (void)hipFree(d_meta);
    return GPUE_OK;
}

extern "C" int gpue_page_decode_rle_i32(gpue_session* s, gpue_dbuf* page,
                                        uint64_t n_values, gpue_dbuf* out);
int gpue_page_decode_rle_i32(gpue_session* s, gpue_dbuf* page, uint64_t n_values,
                             gpue_dbuf* out) {
    ARG_CHECK(s && page && out && out->bytes >= n_values * 4 && page->bytes >= 4);
    uint64_t max_runs = page->bytes / 5 + 2; // a run is >= 5 encoded bytes
    uint32_t* d_start = nullptr;
    uint64_t* d_info = nullptr;
    uint32_t* d_meta = nullptr; // [n_runs, err]
    HIP_CHECK(hipMalloc(&d_start, (max_runs + 1) * 4));
    HIP_CHECK(hipMalloc(&d_info, max_runs * 8));
    HIP_CHECK(hipMalloc(&d_meta, 8));
    HIP_CHECK(hipMemsetAsync(d_meta, 0, 8, s->stream));
    hipLaunchKernelGGL(k_rle_scan_i32, dim3(1), dim3(64), 0, s->stream,
                       (const uint8_t*)page->ptr, page->bytes, d_start, d_info, d_meta,
                       d_meta + 1);
    uint32_t meta[2] = {0, 0};
    HIP_CHECK(hipMemcpyAsync(meta, d_meta, 8, hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    if (meta[1] || meta[0] == 0) {
        (void)hipFree(d_start); (void)hipFree(d_info); (void)hipFree(d_meta);
        if (meta[1]) {
            snprintf(g_err, sizeof(g_err), "rle page truncated/corrupt");
            return GPUE_ERR_ARG;
        }
        return n_values == 0 ? GPUE_OK : GPUE_ERR_ARG;
    }
    hipLaunchKernelGGL(k_rle_fill_i32, dim3(grid_for(n_values)), dim3(BLOCK), 0, s->stream,
                       (const uint8_t*)page->ptr, d_start, d_info, meta[0], n_values,
                       (int32_t*)out->ptr);
    HIP_CHECK(hipStreamSynchronize(s->stream));
    (void)hipFree(d_start);
    (void)hipFree(d_info);
    (void)hipFree(d_meta);
    return GPUE_OK;
}

extern "C" int gpue_page_decode_bshuf_lz4_i32(gpue_session* s, gpue_dbuf* page,
                                              uint32_t n_values, gpue_dbuf* out);
int gpue_page_decode_bshuf_lz4_i32(gpue_session* s, gpue_dbuf* page, uint32_t n_values,
                                   gpue_dbuf* out) {
    ARG_CHECK(s && page && out && n_values > 0 && n_values % 8 == 0);
    ARG_CHECK(out->bytes >= (uint64_t)n_values * 4);
    uint32_t n_blocks = (n_values + BSHUF_BLOCK_I32 - 1) / BSHUF_BLOCK_I32;
    uint32_t* d_starts = nullptr;
    uint8_t* d_scratch = nullptr;
    unsigned long long* d_err = nullptr;
    HIP_CHECK(hipMalloc(&d_starts, (n_blocks + 1) * 4));
    HIP_CHECK(hipMalloc(&d_scratch, (uint64_t)n_blocks * BSHUF_BLOCK_I32 * 4));
    HIP_CHECK(hipMalloc(&d_err, 8));
    HIP_CHECK(hipMemsetAsync(d_err, 0, 8, s->stream));
    hipLaunchKernelGGL(k_page_block_starts, dim3(1), dim3(1), 0, s->stream,
                       (const uint8_t*)page->ptr, n_blocks, d_starts);
    hipLaunchKernelGGL(k_lz4_decode_blocks, dim3(grid_for(n_blocks)), dim3(BLOCK), 0,
                       s->stream, (const uint8_t*)page->ptr, d_starts, n_blocks, n_values,
                       d_scratch, d_err);
    hipLaunchKernelGGL(k_bshuf_untranspose_i32, dim3(grid_for(n_values)), dim3(BLOCK), 0,
                       s->stream, d_scratch, n_values, (int32_t*)out->ptr);
    unsigned long long h_err = 0;
    HIP_CHECK(hipMemcpyAsync(&h_err, d_err, 8, hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    (void)hipFree(d_starts);
    (void)hipFree(d_scratch);
    (void)hipFree(d_err);
    if (h_err) {
        snprintf(g_err, sizeof(g_err), "page decode: malformed LZ4 block");
        return GPUE_ERR_ARG;
    }
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// TopN — ORDER BY value DESC LIMIT k (reference exec/chunks_sorter_topn.cpp;
// SSB/TPC-H result shapes end in ORDER BY + LIMIT).
// Order: (value, key) lexicographic descending — deterministic under value
// ties. Per-thread register top-k -> per-block LDS merge -> host merges the
// per-block candidates (nb × k entries, tiny).
// ---------------------------------------------------------------------------
static constexpr int TOPK_MAX = 16;

__global__ void k_topk_block(const uint64_t* __restrict__ keys,
                             const int64_t* __restrict__ vals, uint64_t n, int k,
                             uint64_t* __restrict__ blk_keys,
                             int64_t* __restrict__ blk_vals) {
    int64_t tv[TOPK_MAX];
    uint64_t tk[TOPK_MAX];
    for (int j = 0; j < k; j++) {
        tv[j] = INT64_MIN;
        tk[j] = 0;
    }
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
        int64_t v = vals[i];
        uint64_t ky = keys[i];
        if (v < tv[k - 1] || (v == tv[k - 1] && ky <= tk[k - 1])) continue;
        int j = k - 1;
        while (j > 0 && (tv[j - 1] < v || (tv[j - 1] == v && tk[j - 1] < ky))) {
            tv[j] = tv[j - 1];
            tk[j] = tk[j - 1];
            j--;
        }
        tv[j] = v;
        tk[j] = ky;
    }
    __shared__ int64_t sv[BLOCK * TOPK_MAX];
    __shared__ uint64_t sk[BLOCK * TOPK_MAX];
    for (int j = 0; j < k; j++) {
        sv[threadIdx.x * k + j] = tv[j];
        sk[threadIdx.x * k + j] = tk[j];
    }
    __syncthreads();
    if (threadIdx.x == 0) {
        // merge the block's blockDim×k candidates into k
        int64_t bv[TOPK_MAX];
        uint64_t bk[TOPK_MAX];
        for (int j = 0; j < k; j++) {
            bv[j] = INT64_MIN;
            bk[j] = 0;
        }
        for (int t = 0; t < (int)blockDim.x * k; t++) {
            int64_t v = sv[t];
            uint64_t ky = sk[t];
            if (v < bv[k - 1] || (v == bv[k - 1] && ky <= bk[k - 1])) continue;
            int j = k - 1;
            while (j > 0 && (bv[j - 1] < v || (bv[j - 1] == v && bk[j - 1] < ky))) {
                bv[j] = bv[j - 1];
                bk[j] = bk[j - 1];
                j--;
            }
            bv[j] = v;
            bk[j] = ky;
        }
        for (int j = 0; j < k; j++) {
            blk_vals[(uint64_t)blockIdx.x * k + j] = bv[j];
            blk_keys[(uint64_t)blockIdx.x * k + j] = bk[j];
        }
    }
}

extern "C" int gpue_topk_i64(gpue_session* s, gpue_dbuf* keys, gpue_dbuf* vals, uint64_t n,
                             int k, uint64_t* out_keys, int64_t* out_vals);
int gpue_topk_i64(gpue_session* s, gpue_dbuf* keys, gpue_dbuf* vals, uint64_t n, int k,
                  uint64_t* out_keys, int64_t* out_vals) {
    ARG_CHECK(s && keys && vals && out_keys && out_vals && k >= 1 && k <= TOPK_MAX);
    ARG_CHECK(keys->bytes >= n * 8 && vals->bytes >= n * 8);
    uint32_t nb = grid_for(n);
    uint64_t* d_bk = nullptr;
    int64_t* d_bv = nullptr;
    HIP_CHECK(hipMalloc(&d_bk, (uint64_t)nb * k * 8));
    HIP_CHECK(hipMalloc(&d_bv, (uint64_t)nb * k * 8));
    hipLaunchKernelGGL(k_topk_block, dim3(nb), dim3(BLOCK), 0, s->stream,
                       (const uint64_t*)keys->ptr, (const int64_t*)vals->ptr, n, k,
                       d_bk, d_bv);
    uint64_t* h_bk = (uint64_t*)malloc((uint64_t)nb * k * 8);
    int64_t* h_bv = (int64_t*)malloc((uint64_t)nb * k * 8);
    HIP_CHECK(hipMemcpyAsync(h_bk, d_bk, (uint64_t)nb * k * 8, hipMemcpyDeviceToHost,
                             s->stream));
    HIP_CHECK(hipMemcpyAsync(h_bv, d_bv, (uint64_t)nb * k * 8, hipMemcpyDeviceToHost,
                             s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    // final k-way selection over nb×k host candidates
    for (int j = 0; j < k; j++) {
        out_vals[j] = INT64_MIN;
        out_keys[j] = 0;
    }
    for (uint64_t t = 0; t < (uint64_t)nb * k; t++) {
        int64_t v = h_bv[t];
        uint64_t ky = h_bk[t];
        if (v < out_vals[k - 1] || (v == out_vals[k - 1] && ky <= out_keys[k - 1])) continue;
        int j = k - 1;
        while (j > 0 && (out_vals[j - 1] < v ||
                         (out_vals[j - 1] == v && out_keys[j - 1] < ky))) {
            out_vals[j] = out_vals[j - 1];
            out_keys[j] = out_keys[j - 1];
            j--;
        }
        out_vals[j] = v;
        out_keys[j] = ky;
    }
    free(h_bk);
    free(h_bv);
    (void)hipFree(d_bk);
    (void)hipFree(d_bv);
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// PlainPage numeric decode (plain_page.h:51,83-102,148-158): u32 LE count
// header + raw LE values — the fallback encoding every numeric type can
// take (encoding_info.cpp). Decode is a validated device copy.
// ---------------------------------------------------------------------------
__global__ void k_plain_copy_i32(const int32_t* __restrict__ body, uint64_t n,
                                 int32_t* __restrict__ out) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
        out[i] = body[i];
}

int gpue_page_decode_plain_i32(gpue_session* s, gpue_dbuf* page, uint64_t n_values,
                               gpue_dbuf* out) {
    ARG_CHECK(s && page && out && page->bytes >= 4 && out->bytes >= n_values * 4);
    uint32_t count = 0;
    HIP_CHECK(hipMemcpyAsync(&count, page->ptr, 4, hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    if (count != n_values || page->bytes < 4 + (uint64_t)count * 4) {
        snprintf(g_err, sizeof(g_err),
                 "plain page malformed: count %u, expected %llu, page %llu B", count,
                 (unsigned long long)n_values, (unsigned long long)page->bytes);
        return GPUE_ERR_ARG;
    }
    // header is 4 B so the body is not 16 B aligned in the page buffer —
    // plain i32 copy (HBM-bound either way)
    hipLaunchKernelGGL(k_plain_copy_i32, dim3(grid_stream(n_values)), dim3(BLOCK), 0,
                       s->stream, (const int32_t*)((const uint8_t*)page->ptr + 4),
                       n_values, (int32_t*)out->ptr);
    HIP_CHECK(hipGetLastError());
    HIP_CHECK(hipStreamSynchronize(s->stream));
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// stream event timing (bench roofline evidence — HIP events on the session
// stream, the stream every kernel above launches on)
// ---------------------------------------------------------------------------
int gpue_timer_start(gpue_session* s) {
    ARG_CHECK(s);
    HIP_CHECK(hipEventRecord(s->ev_start, s->stream));
    return GPUE_OK;
}

int gpue_timer_stop(gpue_session* s, float* ms_out) {
    ARG_CHECK(s && ms_out);
    HIP_CHECK(hipEventRecord(s->ev_stop, s->stream));
    HIP_CHECK(hipEventSynchronize(s->ev_stop));
    HIP_CHECK(hipEventElapsedTime(ms_out, s->ev_start, s->ev_stop));
    return GPUE_OK;
}

// ---------------------------------------------------------------------------
// ASOF join — reference LinearChainedAsofJoinHashMap (join_hash_map_method.h:
// 201-217): an equi-key hash map whose buckets each carry an AsofIndex
// (join_hash_table_descriptor.h:59-104) of {asof_value, row_index} entries,
// sorted by asof_value ascending for LT/LE and descending for GT/GE
// (is_descending = GE||GT, is_strict = LT||GT, :70-80), probed with the
// branchless lower-bound find_asof_match (:83-134).
//
// MI355X layout: the CPU chains + per-bucket std::vector
// become ONE open-addressing slot table (u64 per slot: bit63 occupancy flag |
// u32 key) and ONE contiguous (asof_value[], row[]) arena grouped by slot —
// the probe's binary search then walks a contiguous sorted run (coalescible,
// log2(len) dependent loads) instead of pointer chains. Build is four
// streaming passes (claim+count, scan, scatter, segmented sort); the
// segmented sort runs one block per distinct key: LDS bitonic up to 4096
// entries (48 KB of the 160 KB LDS), single-block global bitonic over a
// pow2-padded segment beyond that (segments get pow2 capacity in the arena
// so the network needs no virtual-padding cases).
//
// Tie pinning: the reference sorts with pdqsort (unstable) comparing ONLY
// asof_value (join_hash_table_descriptor.cpp:70-80), so which duplicate
// (key, asof) row lands at the match boundary is implementation-defined
// there. We sort by (asof_value, row) lexicographic (row ascending in both
// directions) — deterministic, and identical to the reference whenever
// (key, asof) pairs are distinct. The oracle pins the same refinement.
// ---------------------------------------------------------------------------
struct gpue_asof_table {
    gpue_session* s = nullptr;
    int opcode = 0;          // GPUE_ASOF_*
    uint64_t row_count = 0;  // build rows (1-based, row 0 sentinel)
    uint64_t n_slots = 0;    // pow2 open-addressing size (build-time table)
    uint32_t log_slots = 0;
    uint64_t* slots = nullptr;   // (1ull<<63)|key when occupied, 0 empty (freed
                                 // after build — probing uses the compact table)
    uint2* meta = nullptr;       // per slot {segment start, length} (freed too)
    int64_t* asof_vals = nullptr;  // arena: padded-pow2 segments, sorted
    uint32_t* asof_rows = nullptr; // matching 1-based build rows
    // compact probe table, rebuilt once the distinct-key count is known:
    // pow2 >= 2*n_occ slots of uint4 {key, flag, seg start, seg len} — ONE
    // 16 B load resolves key AND segment (vs key slot + meta gather), and at
    // 1 M distinct keys the footprint drops 536 MB -> 32 MB
    // (Infinity-Cache-resident). len==0 marks empty (occupied slots always
    // hold >= 1 entry); flag (bit 31 of .y) disambiguates from a live key.
    uint4* cslots = nullptr;
    uint64_t n_cslots = 0;
    uint32_t log_cslots = 0;
};

#define ASOF_OCCUPIED (1ull << 63)

// claim-or-find a slot for key; returns slot index. Winners append the slot
// to the occupied list (drives the per-segment sort grid).
__device__ static inline uint32_t asof_slot_insert(uint32_t k, uint32_t log_slots,
                                                   uint32_t mask, uint64_t* slots,
                                                   uint32_t* occ, uint32_t* occ_cursor) {
    uint64_t want = ASOF_OCCUPIED | (uint64_t)k;
    uint32_t p = join_hash_u32(k, log_slots);
    for (;;) {
        uint64_t prev = atomicCAS((unsigned long long*)&slots[p], 0ull,
                                  (unsigned long long)want);
        if (prev == 0ull) {
            occ[atomicAdd(occ_cursor, 1u)] = p;
            return p;
        }
        if (prev == want) return p;
        p = (p + 1) & mask;
    }
}

// lookup-only probe; returns slot or 0xFFFFFFFF on miss
__device__ static inline uint32_t asof_slot_find(uint32_t k, uint32_t log_slots,
                                                 uint32_t mask,
                                                 const uint64_t* __restrict__ slots) {
    uint64_t want = ASOF_OCCUPIED | (uint64_t)k;
    uint32_t p = join_hash_u32(k, log_slots);
    for (;;) {
        uint64_t v = slots[p];
        if (v == want) return p;
        if (v == 0ull) return 0xFFFFFFFFu;
        p = (p + 1) & mask;
    }
}

// nulls (1-based, may be null): rows flagged null — equi key OR temporal —
// are skipped, the reference's is_null_row (join_hash_table_descriptor.h:
// 447-456; the caller ORs the two null masks into one)
__global__ void k_asof_insert(const int32_t* __restrict__ keys,
                              const uint8_t* __restrict__ nulls, uint64_t row_count,
                              uint32_t log_slots, uint32_t mask, uint64_t* slots,
                              uint32_t* counts, uint32_t* occ, uint32_t* occ_cursor) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = 1 + (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i <= row_count;
         i += stride) {
        if (nulls && nulls[i]) continue;
        uint32_t p = asof_slot_insert((uint32_t)keys[i], log_slots, mask, slots, occ,
                                      occ_cursor);
        atomicAdd(&counts[p], 1u);
    }
}

// per-slot arena capacity = next pow2 of the entry count: pow2 segments let
// the >LDS sort run a plain bitonic network with sentinel padding in place
__global__ void k_asof_caps(const uint32_t* __restrict__ counts, uint64_t n_slots,
                            uint32_t* __restrict__ caps) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n_slots;
         i += stride) {
        uint32_t c = counts[i];
        caps[i] = c <= 1 ? c : (2u << (31 - __clz(c - 1)));
    }
}

__global__ void k_asof_pack_meta(const uint64_t* __restrict__ cap_offsets,
                                 const uint32_t* __restrict__ counts, uint64_t n_slots,
                                 uint2* __restrict__ meta) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n_slots;
         i += stride)
        meta[i] = make_uint2((uint32_t)cap_offsets[i], counts[i]);
}

__global__ void k_asof_scatter(const int32_t* __restrict__ keys,
                               const int64_t* __restrict__ asof,
                               const uint8_t* __restrict__ nulls, uint64_t row_count,
                               uint32_t log_slots, uint32_t mask,
                               const uint64_t* __restrict__ slots,
                               const uint2* __restrict__ meta, uint32_t* __restrict__ cursors,
                               int64_t* __restrict__ vals, uint32_t* __restrict__ rows) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = 1 + (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i <= row_count;
         i += stride) {
        if (nulls && nulls[i]) continue;
        uint32_t p = asof_slot_find((uint32_t)keys[i], log_slots, mask, slots);
        uint32_t pos = meta[p].x + atomicAdd(&cursors[p], 1u);
        vals[pos] = asof[i];
        rows[pos] = (uint32_t)i;
    }
}

// composite order: by asof_value in the opcode's direction, row ascending on
// ties (the pinned refinement; see the section comment)
__device__ static inline bool asof_before(int64_t av, uint32_t ar, int64_t bv, uint32_t br,
                                          bool descending) {
    if (av != bv) return descending ? (av > bv) : (av < bv);
    return ar < br;
}

// one block per occupied slot. len <= 4096: LDS bitonic (48 KB static).
// Larger: sentinel-pad the segment to its pow2 capacity in place, then the
// same bitonic network block-strided over global memory (__syncthreads is
// the stage barrier — valid because exactly one block owns the segment).
// Sentinels (+inf,row ~0u for ascending / -inf,~0u for descending) order
// strictly after every real entry, so positions [0,len) hold exactly the
// real entries post-sort and the probe never reads the pad.
#define ASOF_LDS_CAP 4096
__global__ void k_asof_sort_segments(const uint32_t* __restrict__ occ, uint32_t n_occ,
                                     const uint2* __restrict__ meta, int descending,
                                     int64_t* __restrict__ vals,
                                     uint32_t* __restrict__ rows) {
    __shared__ int64_t s_val[ASOF_LDS_CAP];
    __shared__ uint32_t s_row[ASOF_LDS_CAP];
    for (uint32_t seg = blockIdx.x; seg < n_occ; seg += gridDim.x) {
        uint2 m = meta[occ[seg]];
        uint32_t base = m.x, len = m.y;
        if (len <= 1) continue;
        uint32_t cap = 2u << (31 - __clz(len - 1)); // pow2 >= len (len >= 2)
        const int64_t pad_val = descending ? INT64_MIN : INT64_MAX;
        if (len <= ASOF_LDS_CAP) {
            for (uint32_t i = threadIdx.x; i < cap; i += blockDim.x) {
                s_val[i] = i < len ? vals[base + i] : pad_val;
                s_row[i] = i < len ? rows[base + i] : 0xFFFFFFFFu;
            }
            __syncthreads();
            for (uint32_t k = 2; k <= cap; k <<= 1) {
                for (uint32_t j = k >> 1; j > 0; j >>= 1) {
                    for (uint32_t i = threadIdx.x; i < cap; i += blockDim.x) {
                        uint32_t ij = i ^ j;
                        if (ij > i) {
                            bool up = (i & k) == 0;
                            bool sw = asof_before(s_val[ij], s_row[ij], s_val[i], s_row[i],
                                                  descending);
                            if (sw == up) {
                                int64_t tv = s_val[i]; s_val[i] = s_val[ij]; s_val[ij] = tv;
                                uint32_t tr = s_row[i]; s_row[i] = s_row[ij]; s_row[ij] = tr;
                            }
                        }
                    }
                    __syncthreads();
                }
            }
            for (uint32_t i = threadIdx.x; i < len; i += blockDim.x) {
                vals[base + i] = s_val[i];
                rows[base + i] = s_row[i];
            }
            __syncthreads();
        } else {
            for (uint32_t i = len + threadIdx.x; i < cap; i += blockDim.x) {
                vals[base + i] = pad_val;
                rows[base + i] = 0xFFFFFFFFu;
            }
            __syncthreads();
            for (uint32_t k = 2; k <= cap; k <<= 1) {
                for (uint32_t j = k >> 1; j > 0; j >>= 1) {
                    for (uint32_t i = threadIdx.x; i < cap; i += blockDim.x) {
                        uint32_t ij = i ^ j;
                        if (ij > i) {
                            bool up = (i & k) == 0;
                            bool sw = asof_before(vals[base + ij], rows[base + ij],
                                                  vals[base + i], rows[base + i],
                                                  descending);
                            if (sw == up) {
                                int64_t tv = vals[base + i];
                                vals[base + i] = vals[base + ij];
                                vals[base + ij] = tv;
                                uint32_t tr = rows[base + i];
                                rows[base + i] = rows[base + ij];
                                rows[base + ij] = tr;
                            }
                        }
                    }
                    __syncthreads();
                }
            }
        }
    }
}

// rebuild the sparse build-time table as the compact probe table (one pass
// over the occupied-slot list; claim via CAS on the slot's first 8 bytes,
// meta written non-atomically — probes only run after build completes)
__global__ void k_asof_compact(const uint32_t* __restrict__ occ, uint32_t n_occ,
                               const uint64_t* __restrict__ slots,
                               const uint2* __restrict__ meta, uint32_t log_cslots,
                               uint32_t cmask, uint4* __restrict__ cslots) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n_occ;
         i += stride) {
        uint32_t p = occ[i];
        uint32_t key = (uint32_t)slots[p];
        uint2 m = meta[p];
        uint64_t want = ASOF_OCCUPIED | (uint64_t)key;
        uint32_t q = join_hash_u32(key, log_cslots);
        for (;;) { // keys are distinct here: CAS either claims or collides
            uint64_t prev = atomicCAS((unsigned long long*)&cslots[q], 0ull,
                                      (unsigned long long)want);
            if (prev == 0ull) {
                cslots[q].z = m.x;
                cslots[q].w = m.y;
                break;
            }
            q = (q + 1) & cmask;
        }
    }
}

// find_asof_match restated (join_hash_table_descriptor.cpp:83-134): branchless
// lower bound over the sorted segment; the CPU's two while-loops (an
// unroll-hinted >=8 loop then the tail) are one loop here — the #pragma GCC
// unroll is a scheduling hint, the iteration sequence is identical.
// OP: 0 LT, 1 LE, 2 GT, 3 GE. "low = other_low" condition per variant
// (:112-131): LT probe>=entry / LE probe>entry / GT probe<=entry /
// GE probe<entry. Returns the 1-based build row, 0 on no match (:105-108).
template <int OP>
__device__ static inline uint32_t asof_find(const int64_t* __restrict__ vals,
                                            const uint32_t* __restrict__ rows,
                                            uint32_t base, uint32_t len, int64_t probe) {
    if (len == 0) return 0;
    uint32_t size = len, low = 0;
    while (size > 0) {
        uint32_t half = size >> 1;
        uint32_t probe_pos = low + half;
        uint32_t other_low = low + (size - half);
        int64_t entry = vals[base + probe_pos];
        size = half;
        bool cond;
        if (OP == 0) cond = probe >= entry;
        else if (OP == 1) cond = probe > entry;
        else if (OP == 2) cond = probe <= entry;
        else cond = probe < entry;
        low = cond ? other_low : low;
    }
    return low < len ? rows[base + low] : 0;
}

// probe kernels: 4-way interleaved rows per lane. The probe is LATENCY-bound
// (a dependent chain: slot load -> log2(len) binary-search loads -> row
// gather over a multi-hundred-MB footprint); per-lane memory-level
// parallelism is the controlling resource for such gathers (DESIGN.md §4d,
// the q3 wave-queue lesson), so each lane advances FOUR independent probe
// rows in lockstep — every level issues 4 concurrent loads instead of 1.
// Measured: 61.0 -> 13.0 ms at 600 M rows / 1 M keys (profiles/asof_bench).
template <int OP>
__device__ static inline void asof_probe4(const int32_t* __restrict__ pkeys,
                                          const int64_t* __restrict__ pasof,
                                          const uint8_t* __restrict__ pnulls, uint64_t n,
                                          uint32_t log_cslots, uint32_t cmask,
                                          const uint4* __restrict__ cslots,
                                          const int64_t* __restrict__ vals,
                                          const uint32_t* __restrict__ rows,
                                          uint64_t base, uint64_t stride,
                                          uint32_t match[4]) {
    uint32_t start[4], len[4], low[4], size[4];
    int64_t pv[4];
    uint32_t q[4];
    bool live[4];
    #pragma unroll
    for (int j = 0; j < 4; j++) {
        uint64_t i = base + (uint64_t)j * stride;
        match[j] = 0;
        live[j] = i < n && !(pnulls && pnulls[i]);
        q[j] = live[j] ? join_hash_u32((uint32_t)pkeys[i], log_cslots) : 0u;
        pv[j] = live[j] ? pasof[i] : 0;
        len[j] = 0;
    }
    // slot lookup, first probe batched (load <= 1/2 so collisions are rare)
    uint4 sv[4];
    #pragma unroll
    for (int j = 0; j < 4; j++)
        if (live[j]) sv[j] = cslots[q[j]];
    #pragma unroll
    for (int j = 0; j < 4; j++) {
        if (!live[j]) continue;
        uint64_t i = base + (uint64_t)j * stride;
        uint64_t want = ASOF_OCCUPIED | (uint64_t)(uint32_t)pkeys[i];
        uint4 v = sv[j];
        for (;;) {
            uint64_t head = ((uint64_t)v.y << 32) | v.x;
            if (head == want) {
                start[j] = v.z;
                len[j] = v.w;
                break;
            }
            if (head == 0ull) break; // miss
            q[j] = (q[j] + 1) & cmask;
            v = cslots[q[j]];
        }
        low[j] = 0;
        size[j] = len[j];
    }
    // branchless lower bound, 4 searches in lockstep (4 loads per level)
    while (size[0] | size[1] | size[2] | size[3]) {
        int64_t e[4];
        #pragma unroll
        for (int j = 0; j < 4; j++)
            if (size[j]) e[j] = vals[start[j] + low[j] + (size[j] >> 1)];
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            if (!size[j]) continue;
            uint32_t half = size[j] >> 1;
            uint32_t other_low = low[j] + (size[j] - half);
            size[j] = half;
            bool cond;
            if (OP == 0) cond = pv[j] >= e[j];
            else if (OP == 1) cond = pv[j] > e[j];
            else if (OP == 2) cond = pv[j] <= e[j];
            else cond = pv[j] < e[j];
            low[j] = cond ? other_low : low[j];
        }
    }
    #pragma unroll
    for (int j = 0; j < 4; j++)
        if (len[j] && low[j] < len[j]) match[j] = rows[start[j] + low[j]];
}

template <int OP>
__global__ void k_asof_probe_count(const int32_t* __restrict__ pkeys,
                                   const int64_t* __restrict__ pasof,
                                   const uint8_t* __restrict__ pnulls, uint64_t n,
                                   uint32_t log_cslots, uint32_t cmask,
                                   const uint4* __restrict__ cslots,
                                   const int64_t* __restrict__ vals,
                                   const uint32_t* __restrict__ rows, int mode,
                                   uint32_t* __restrict__ row_counts) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t base = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; base < n;
         base += 4 * stride) {
        uint32_t match[4];
        asof_probe4<OP>(pkeys, pasof, pnulls, n, log_cslots, cmask, cslots, vals, rows,
                        base, stride, match);
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            uint64_t i = base + (uint64_t)j * stride;
            if (i < n)
                row_counts[i] = match[j] ? 1u : (mode == GPUE_JOIN_LEFT_OUTER ? 1u : 0u);
        }
    }
}

template <int OP>
__global__ void k_asof_probe_emit(const int32_t* __restrict__ pkeys,
                                  const int64_t* __restrict__ pasof,
                                  const uint8_t* __restrict__ pnulls, uint64_t n,
                                  uint32_t log_cslots, uint32_t cmask,
                                  const uint4* __restrict__ cslots,
                                  const int64_t* __restrict__ vals,
                                  const uint32_t* __restrict__ rows, int mode,
                                  const uint32_t* __restrict__ row_counts,
                                  const uint64_t* __restrict__ row_offsets,
                                  uint32_t* __restrict__ out_probe,
                                  uint32_t* __restrict__ out_build) {
    uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    for (uint64_t base = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; base < n;
         base += 4 * stride) {
        uint32_t match[4];
        asof_probe4<OP>(pkeys, pasof, pnulls, n, log_cslots, cmask, cslots, vals, rows,
                        base, stride, match);
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            uint64_t i = base + (uint64_t)j * stride;
            if (i < n && row_counts[i] != 0) {
                uint64_t pos = row_offsets[i];
                out_probe[pos] = (uint32_t)i;
                out_build[pos] = match[j]; // 0 = LEFT_OUTER miss (reference row 0)
            }
        }
    }
}

static int asof_build_impl(gpue_session* s, gpue_dbuf* keys, gpue_dbuf* asof,
                           const uint8_t* d_nulls, uint64_t row_count, int opcode,
                           gpue_asof_table** out) {
    ARG_CHECK(s && keys && asof && out && row_count > 0 && row_count + 1 < (1ull << 31));
    ARG_CHECK(opcode >= GPUE_ASOF_LT && opcode <= GPUE_ASOF_GE);
    ARG_CHECK(keys->bytes >= (row_count + 1) * 4 && asof->bytes >= (row_count + 1) * 8);
    gpue_asof_table* t = new gpue_asof_table();
    t->s = s;
    t->opcode = opcode;
    t->row_count = row_count;
    uint64_t want = 2 * (row_count + 1);
    t->n_slots = 64;
    while (t->n_slots < want) t->n_slots <<= 1;
    t->log_slots = (uint32_t)__builtin_ctzll(t->n_slots);
    uint32_t mask = (uint32_t)(t->n_slots - 1);
    uint32_t* d_counts = nullptr;
    uint32_t* d_caps = nullptr;
    uint32_t* d_occ = nullptr;
    uint32_t* d_misc = nullptr; // occ cursor + scatter cursors
    uint64_t* d_bsums = nullptr;
    uint64_t* d_offsets = nullptr;
    HIP_CHECK(hipMalloc(&t->slots, t->n_slots * 8));
    HIP_CHECK(hipMalloc(&t->meta, t->n_slots * 8));
    HIP_CHECK(hipMalloc(&d_counts, t->n_slots * 4));
    HIP_CHECK(hipMalloc(&d_caps, t->n_slots * 4));
    HIP_CHECK(hipMalloc(&d_occ, row_count * 4));
    HIP_CHECK(hipMalloc(&d_misc, (t->n_slots + 1) * 4));
    HIP_CHECK(hipMemsetAsync(t->slots, 0, t->n_slots * 8, s->stream));
    HIP_CHECK(hipMemsetAsync(d_counts, 0, t->n_slots * 4, s->stream));
    HIP_CHECK(hipMemsetAsync(d_misc, 0, (t->n_slots + 1) * 4, s->stream));
    uint32_t nb_rows = grid_for(row_count);
    hipLaunchKernelGGL(k_asof_insert, dim3(nb_rows), dim3(BLOCK), 0, s->stream,
                       (const int32_t*)keys->ptr, d_nulls, row_count, t->log_slots, mask,
                       t->slots, d_counts, d_occ, d_misc);
    // pow2 arena capacities, block-scanned to segment starts
    uint32_t nb_slots = grid_for(t->n_slots);
    uint64_t tile = (t->n_slots + nb_slots - 1) / nb_slots;
    HIP_CHECK(hipMalloc(&d_bsums, (nb_slots + 1) * 8));
    HIP_CHECK(hipMalloc(&d_offsets, t->n_slots * 8));
    hipLaunchKernelGGL(k_asof_caps, dim3(nb_slots), dim3(BLOCK), 0, s->stream, d_counts,
                       t->n_slots, d_caps);
    hipLaunchKernelGGL(k_block_sums_u32, dim3(nb_slots), dim3(BLOCK), 0, s->stream, d_caps,
                       t->n_slots, tile, d_bsums);
    hipLaunchKernelGGL(k_scan_small, dim3(1), dim3(1), 0, s->stream, d_bsums, nb_slots);
    hipLaunchKernelGGL(k_scan_offsets, dim3(nb_slots), dim3(BLOCK), 0, s->stream, d_caps,
                       t->n_slots, tile, d_bsums, d_offsets);
    hipLaunchKernelGGL(k_asof_pack_meta, dim3(nb_slots), dim3(BLOCK), 0, s->stream,
                       d_offsets, d_counts, t->n_slots, t->meta);
    uint64_t arena = 0; // padded total <= 2*row_count
    uint32_t n_occ = 0;
    HIP_CHECK(hipMemcpyAsync(&arena, d_bsums + nb_slots, 8, hipMemcpyDeviceToHost,
                             s->stream));
    HIP_CHECK(hipMemcpyAsync(&n_occ, d_misc, 4, hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    HIP_CHECK(hipMalloc(&t->asof_vals, (arena ? arena : 1) * 8));
    HIP_CHECK(hipMalloc(&t->asof_rows, (arena ? arena : 1) * 4));
    hipLaunchKernelGGL(k_asof_scatter, dim3(nb_rows), dim3(BLOCK), 0, s->stream,
                       (const int32_t*)keys->ptr, (const int64_t*)asof->ptr, d_nulls,
                       row_count, t->log_slots, mask, t->slots, t->meta, d_misc + 1,
                       t->asof_vals, t->asof_rows);
    int descending = opcode >= GPUE_ASOF_GT; // is_descending = GE||GT (:67)
    uint32_t nb_sort = n_occ < MAX_GRID ? (n_occ ? n_occ : 1) : (uint32_t)MAX_GRID;
    hipLaunchKernelGGL(k_asof_sort_segments, dim3(nb_sort), dim3(BLOCK), 0, s->stream,
                       d_occ, n_occ, t->meta, descending, t->asof_vals, t->asof_rows);
    // compact probe table sized by the now-known distinct-key count
    t->n_cslots = 64;
    while (t->n_cslots < 2ull * (n_occ ? n_occ : 1)) t->n_cslots <<= 1;
    t->log_cslots = (uint32_t)__builtin_ctzll(t->n_cslots);
    HIP_CHECK(hipMalloc(&t->cslots, t->n_cslots * sizeof(uint4)));
    HIP_CHECK(hipMemsetAsync(t->cslots, 0, t->n_cslots * sizeof(uint4), s->stream));
    hipLaunchKernelGGL(k_asof_compact, dim3(grid_for(n_occ ? n_occ : 1)), dim3(BLOCK), 0,
                       s->stream, d_occ, n_occ, t->slots, t->meta,
                       t->log_cslots, (uint32_t)(t->n_cslots - 1), t->cslots);
    HIP_CHECK(hipGetLastError());
    HIP_CHECK(hipStreamSynchronize(s->stream));
    (void)hipFree(t->slots);
    t->slots = nullptr;
    (void)hipFree(t->meta);
    t->meta = nullptr;
    (void)hipFree(d_counts);
    (void)hipFree(d_caps);
    (void)hipFree(d_occ);
    (void)hipFree(d_misc);
    (void)hipFree(d_bsums);
    (void)hipFree(d_offsets);
    *out = t;
    return GPUE_OK;
}

int gpue_asof_build_i32(gpue_session* s, gpue_dbuf* keys, gpue_dbuf* asof,
                        uint64_t row_count, int opcode, gpue_asof_table** out) {
    return asof_build_impl(s, keys, asof, nullptr, row_count, opcode, out);
}

int gpue_asof_build_nulls_i32(gpue_session* s, gpue_dbuf* keys, gpue_dbuf* asof,
                              gpue_dbuf* is_nulls, uint64_t row_count, int opcode,
                              gpue_asof_table** out) {
    ARG_CHECK(s && is_nulls && is_nulls->bytes >= row_count + 1);
    return asof_build_impl(s, keys, asof, (const uint8_t*)is_nulls->ptr, row_count,
                           opcode, out);
}

static int asof_probe_impl(gpue_session* s, gpue_asof_table* t, gpue_dbuf* probe_keys,
                           gpue_dbuf* probe_asof, const uint8_t* d_pnulls, uint64_t n_rows,
                           int mode, gpue_dbuf* out_probe_idx, gpue_dbuf* out_build_idx,
                           uint64_t* match_count) {
    ARG_CHECK(s && t && probe_keys && probe_asof && match_count);
    ARG_CHECK(mode == GPUE_JOIN_INNER || mode == GPUE_JOIN_LEFT_OUTER);
    ARG_CHECK(probe_keys->bytes >= n_rows * 4 && probe_asof->bytes >= n_rows * 8);
    uint32_t nb = grid_for(n_rows);
    uint64_t tile = (n_rows + nb - 1) / nb;
    uint32_t cmask = (uint32_t)(t->n_cslots - 1);
    uint32_t* d_counts = nullptr;
    uint64_t* d_bsums = nullptr;
    uint64_t* d_offsets = nullptr;
    HIP_CHECK(hipMalloc(&d_counts, (n_rows ? n_rows : 1) * 4));
    HIP_CHECK(hipMalloc(&d_bsums, (nb + 1) * 8));
    switch (t->opcode) {
    case GPUE_ASOF_LT:
        hipLaunchKernelGGL(k_asof_probe_count<0>, dim3(nb), dim3(BLOCK), 0, s->stream,
                           (const int32_t*)probe_keys->ptr, (const int64_t*)probe_asof->ptr, d_pnulls,
                           n_rows, t->log_cslots, cmask, t->cslots, t->asof_vals,
                           t->asof_rows, mode, d_counts);
        break;
    case GPUE_ASOF_LE:
        hipLaunchKernelGGL(k_asof_probe_count<1>, dim3(nb), dim3(BLOCK), 0, s->stream,
                           (const int32_t*)probe_keys->ptr, (const int64_t*)probe_asof->ptr, d_pnulls,
                           n_rows, t->log_cslots, cmask, t->cslots, t->asof_vals,
                           t->asof_rows, mode, d_counts);
        break;
    case GPUE_ASOF_GT:
        hipLaunchKernelGGL(k_asof_probe_count<2>, dim3(nb), dim3(BLOCK), 0, s->stream,
                           (const int32_t*)probe_keys->ptr, (const int64_t*)probe_asof->ptr, d_pnulls,
                           n_rows, t->log_cslots, cmask, t->cslots, t->asof_vals,
                           t->asof_rows, mode, d_counts);
        break;
    default:
        hipLaunchKernelGGL(k_asof_probe_count<3>, dim3(nb), dim3(BLOCK), 0, s->stream,
                           (const int32_t*)probe_keys->ptr, (const int64_t*)probe_asof->ptr, d_pnulls,
                           n_rows, t->log_cslots, cmask, t->cslots, t->asof_vals,
                           t->asof_rows, mode, d_counts);
        break;
    }
    hipLaunchKernelGGL(k_block_sums_u32, dim3(nb), dim3(BLOCK), 0, s->stream, d_counts,
                       n_rows, tile, d_bsums);
    hipLaunchKernelGGL(k_scan_small, dim3(1), dim3(1), 0, s->stream, d_bsums, nb);
    uint64_t total = 0;
    HIP_CHECK(hipMemcpyAsync(&total, d_bsums + nb, 8, hipMemcpyDeviceToHost, s->stream));
    HIP_CHECK(hipStreamSynchronize(s->stream));
    *match_count = total;
    if (out_probe_idx && out_build_idx && total > 0) {
        ARG_CHECK(out_probe_idx->bytes >= total * 4 && out_build_idx->bytes >= total * 4);
        HIP_CHECK(hipMalloc(&d_offsets, n_rows * 8));
        hipLaunchKernelGGL(k_scan_offsets, dim3(nb), dim3(BLOCK), 0, s->stream, d_counts,
                           n_rows, tile, d_bsums, d_offsets);
        switch (t->opcode) {
        case GPUE_ASOF_LT:
            hipLaunchKernelGGL(k_asof_probe_emit<0>, dim3(nb), dim3(BLOCK), 0, s->stream,
                               (const int32_t*)probe_keys->ptr,
                               (const int64_t*)probe_asof->ptr, d_pnulls, n_rows, t->log_cslots, cmask,
                               t->cslots, t->asof_vals, t->asof_rows, mode,
                               d_counts, d_offsets, (uint32_t*)out_probe_idx->ptr,
                               (uint32_t*)out_build_idx->ptr);
            break;
        case GPUE_ASOF_LE:
            hipLaunchKernelGGL(k_asof_probe_emit<1>, dim3(nb), dim3(BLOCK), 0, s->stream,
                               (const int32_t*)probe_keys->ptr,
                               (const int64_t*)probe_asof->ptr, d_pnulls, n_rows, t->log_cslots, cmask,
                               t->cslots, t->asof_vals, t->asof_rows, mode,
                               d_counts, d_offsets, (uint32_t*)out_probe_idx->ptr,
                               (uint32_t*)out_build_idx->ptr);
            break;
        case GPUE_ASOF_GT:
            hipLaunchKernelGGL(k_asof_probe_emit<2>, dim3(nb), dim3(BLOCK), 0, s->stream,
                               (const int32_t*)probe_keys->ptr,
                               (const int64_t*)probe_asof->ptr, d_pnulls, n_rows, t->log_cslots, cmask,
                               t->cslots, t->asof_vals, t->asof_rows, mode,
                               d_counts, d_offsets, (uint32_t*)out_probe_idx->ptr,
                               (uint32_t*)out_build_idx->ptr);
            break;
        default:
            hipLaunchKernelGGL(k_asof_probe_emit<3>, dim3(nb), dim3(BLOCK), 0, s->stream,
                               (const int32_t*)probe_keys->ptr,
                               (const int64_t*)probe_asof->ptr, d_pnulls, n_rows, t->log_cslots, cmask,
                               t->cslots, t->asof_vals, t->asof_rows, mode,
                               d_counts, d_offsets, (uint32_t*)out_probe_idx->ptr,
                               (uint32_t*)out_build_idx->ptr);
            break;
        }
        HIP_CHECK(hipStreamSynchronize(s->stream));
        (void)hipFree(d_offsets);
    }
    (void)hipFree(d_counts);
    (void)hipFree(d_bsums);
    return GPUE_OK;
}

int gpue_asof_probe_emit_i32(gpue_session* s, gpue_asof_table* t, gpue_dbuf* probe_keys,
                             gpue_dbuf* probe_asof, uint64_t n_rows, int mode,
                             gpue_dbuf* out_probe_idx, gpue_dbuf* out_build_idx,
                             uint64_t* match_count) {
    return asof_probe_impl(s, t, probe_keys, probe_asof, nullptr, n_rows, mode,
                           out_probe_idx, out_build_idx, match_count);
}

int gpue_asof_probe_emit_nulls_i32(gpue_session* s, gpue_asof_table* t,
                                   gpue_dbuf* probe_keys, gpue_dbuf* probe_asof,
                                   gpue_dbuf* probe_nulls, uint64_t n_rows, int mode,
                                   gpue_dbuf* out_probe_idx, gpue_dbuf* out_build_idx,
                                   uint64_t* match_count) {
    ARG_CHECK(s && probe_nulls && probe_nulls->bytes >= n_rows);
    return asof_probe_impl(s, t, probe_keys, probe_asof,
                           (const uint8_t*)probe_nulls->ptr, n_rows, mode, out_probe_idx,
                           out_build_idx, match_count);
}

int gpue_asof_table_destroy(gpue_asof_table* t) {
    if (!t) return GPUE_OK;
    (void)hipFree(t->slots);
    (void)hipFree(t->meta);
    (void)hipFree(t->cslots);
    (void)hipFree(t->asof_vals);
    (void)hipFree(t->asof_rows);
    delete t;
    return GPUE_OK;
}
