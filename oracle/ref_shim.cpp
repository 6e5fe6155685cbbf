// ref_shim.cpp — compiles the REFERENCE's own hash headers IN PLACE (from
// /root/reference, read-only; no sources copied into this repo) and exports
// them over a C ABI so tests can cross-check oracle.c's restatement against
// the reference binary itself. Built only where /root/reference exists (this
// container); the resulting oracle/_ref/ref.so travels to the GPU box.
//
// TEST INFRASTRUCTURE ONLY — see oracle.h header comment.
// (phmap.h would add NormalizeCapacity but drags in generated thrift headers
// that are download-script-only — the bucket-size formula is instead pinned by
// the ported calc_bucket_size KATs in tests/test_oracle_golden.py.)
#include "base/hash/hash.h"          // crc_hash_32, phmap_mix (be/src/base/hash/hash.h)
#include "base/hash/hash_util.hpp"   // HashUtil::fnv_hash, xorshift32

extern "C" {
unsigned ref_crc_hash_32(const void* d, int n, unsigned seed) {
    return starrocks::crc_hash_32(d, n, seed);
}
unsigned ref_fnv_hash(const void* d, int n, unsigned seed) {
    return starrocks::HashUtil::fnv_hash(d, n, seed);
}
unsigned ref_xorshift32(unsigned x) {
    return starrocks::HashUtil::xorshift32(x);
}
unsigned long long ref_phmap_mix8(unsigned long long a) {
    return starrocks::phmap_mix<8>()(a);
}
}
