"""High-cardinality hash-agg insert ubench.

Times gpue_hash_agg_push_u64 (the generic CAS table) across group
cardinalities at a fixed 100 M-row push. The question the two-level
(16-way partitioned) reference design answers on CPU is cache residency
(agg_hash_variant.cpp:318); the GPU analog would be partition-by-hash then
per-XCD L2-resident sub-tables — only worth building if the insert leg
degrades materially once the table exceeds the 4 MiB per-XCD L2 /
256 MiB Infinity Cache tiers. Prints one JSON line per cardinality.
"""

import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import numpy as np

from starrocks_amd.engine import Engine


def main():
    eng = Engine()
    n = 100_000_000
    rng = np.random.default_rng(42)
    vals_h = rng.integers(0, 1000, n).astype(np.int64)
    vals = eng.alloc(n * 8)
    vals.h2d(vals_h)
    for ngroups in (10_000, 100_000, 1_000_000, 10_000_000, 50_000_000):
        keys_h = rng.integers(0, ngroups, n).astype(np.uint64)
        keys = eng.alloc(n * 8)
        keys.h2d(keys_h)
        at = eng.agg_table_create(2 * ngroups)
        eng.hash_agg_push(at, keys, vals, n)  # warm (claims all groups)
        eng.sync()
        def best_of(flags):
            b = None
            for _ in range(3):
                eng.agg_table_reset(at)
                eng.sync()
                t0 = time.perf_counter()
                eng.hash_agg_push(at, keys, vals, n, update_only=flags)
                eng.sync()
                dt = time.perf_counter() - t0
                b = dt if b is None else min(b, dt)
            return b
        best = best_of(0)
        best_so = best_of(2)  # GPUE_AGG_SUM_ONLY: skip the count atomic
        table_mb = 2 * ngroups * 24 / 1e6  # key + sum + count slots
        print(json.dumps({
            "ngroups": ngroups, "rows": n, "push_ms": round(best * 1e3, 3),
            "grows_per_s": round(n / best / 1e9, 1),
            "sum_only_ms": round(best_so * 1e3, 3),
            "sum_only_grows_per_s": round(n / best_so / 1e9, 1),
            "table_mb": round(table_mb, 1),
            "n_groups_seen": eng.agg_table_size(at),
        }), flush=True)
        eng.agg_table_destroy(at)
        keys.free()


if __name__ == "__main__":
    main()
