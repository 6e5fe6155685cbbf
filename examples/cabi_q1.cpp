// Standalone C++ caller of the C-ABI drop-in boundary (include/gpue.h) —
// the same calls a pipeline::Operator subclass inside the reference BE would
// make. No Python, no torch: g++ against libgpue.so.
//
// Runs the config-2 plan: generate the lineorder shard on-device, build the
// (pre-filtered) date payload table, run the fused join+sum, print one JSON
// line. tests/test_gpu_parity.py::test_cabi_cpp_demo compares the output
// against the oracle.
//
// Build (done by __graft_entry__.build()):
//   g++ -O2 -Iinclude examples/cabi_q1.cpp -Lstarrocks_amd -lgpue \
//       -Wl,-rpath,'$ORIGIN/../starrocks_amd' -o examples/cabi_q1
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <vector>

#include "gpue.h"

#define CK(expr)                                                            \
    do {                                                                    \
        int _rc = (expr);                                                   \
        if (_rc != GPUE_OK) {                                               \
            fprintf(stderr, "%s -> %d: %s\n", #expr, _rc, gpue_last_error());\
            return 1;                                                       \
        }                                                                   \
    } while (0)

// SSB date dim from 1992-01-01, N_DAYS consecutive days (DESIGN.md §8d
// generator contract — same arrays starrocks_amd/gen.py::gen_dates builds).
static void gen_dates(int n_days, std::vector<int32_t>& datekey,
                      std::vector<int32_t>& dyear) {
    static const int md[12] = {31, 28, 31, 30, 31, 30, 31, 31, 30, 31, 30, 31};
    int y = 1992, m = 1, d = 1;
    for (int i = 0; i < n_days; i++) {
        datekey.push_back(y * 10000 + m * 100 + d);
        dyear.push_back(y);
        bool leap = (y % 4 == 0 && y % 100 != 0) || y % 400 == 0;
        int lim = md[m - 1] + ((m == 2 && leap) ? 1 : 0);
        if (++d > lim) {
            d = 1;
            if (++m > 12) { m = 1; y++; }
        }
    }
}

int main(int argc, char** argv) {
    const uint64_t seed = 42;
    const uint64_t n = (argc > 1) ? strtoull(argv[1], nullptr, 10) : 2000000ull;
    const int year = 1993, n_days = 2556;

    gpue_session* s = nullptr;
    CK(gpue_session_create(0, &s));

    // lineorder shard, generated straight into HBM
    gpue_dbuf *od, *ep, *dc;
    CK(gpue_dbuf_alloc(s, n * 4, &od));
    CK(gpue_dbuf_alloc(s, n * 4, &ep));
    CK(gpue_dbuf_alloc(s, n * 4, &dc));
    CK(gpue_gen_lineorder_q1(s, seed, 0, n, od, ep, dc));

    // date dim: payload = (d_year-1992)+1 where the year filter passes
    std::vector<int32_t> datekey, dyear;
    gen_dates(n_days, datekey, dyear);
    std::vector<uint32_t> payload(n_days);
    for (int i = 0; i < n_days; i++)
        payload[i] = (dyear[i] == year) ? (uint32_t)(dyear[i] - 1992 + 1) : 0u;
    gpue_dbuf *kb, *pb;
    CK(gpue_dbuf_alloc(s, n_days * 4, &kb));
    CK(gpue_dbuf_alloc(s, n_days * 4, &pb));
    CK(gpue_dbuf_h2d(kb, datekey.data(), n_days * 4, 0));
    CK(gpue_dbuf_h2d(pb, payload.data(), n_days * 4, 0));
    gpue_join_table* dates = nullptr;
    CK(gpue_join_build_payload_i32(s, kb, pb, n_days, &dates));

    int64_t sum = 0;
    uint64_t cnt = 0;
    CK(gpue_q1_join_sum(s, dates, od, ep, dc, n, &sum, &cnt));
    printf("{\"sum\": %lld, \"count\": %llu, \"rows\": %llu}\n",
           (long long)sum, (unsigned long long)cnt, (unsigned long long)n);

    gpue_join_table_destroy(dates);
    gpue_dbuf_free(kb);
    gpue_dbuf_free(pb);
    gpue_dbuf_free(od);
    gpue_dbuf_free(ep);
    gpue_dbuf_free(dc);
    gpue_session_destroy(s);
    return 0;
}
