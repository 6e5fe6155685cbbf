"""ASOF join perf characterization (DESIGN.md §4c-r2 evidence).

Times gpue_asof_build_i32 (claim/scan/scatter/segmented-bitonic/compact) and
the probe at lineorder scale: the probe streams 12 B/row (i32 key + i64
value), then one compact-slot load + ~log2(entries-per-key) search loads +
one row gather — ~6 divergent requests/row. Measured 10.0 Grows/s at 600 M
rows / 1 M keys = 60 G random requests/s, the divergent-gather request
envelope (profiles/r02_asof_kernel_stats.txt; the bit-gather ubench ceiling
is 57.4 Gprobe/s). A compact-table fold and 4-way per-lane interleaving both
measured flat — the wall is request rate, not latency. Prints one JSON line.

Run: gpurun -- 'python tools/asof_bench.py > gpurun_out/asof_bench.json'
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
    rng = np.random.default_rng(42)
    rc = 10_000_000          # build rows
    n_keys = 1_000_000       # ~10 entries/key -> ~3-4 search steps
    n_probe = 600_000_000    # lineorder scale

    bk = np.concatenate([[0], rng.integers(1, n_keys + 1, rc)]).astype(np.int32)
    ba = np.concatenate([[0], rng.integers(-10**12, 10**12, rc)]).astype(np.int64)
    kb = eng.alloc(bk.nbytes); kb.h2d(bk)
    ab = eng.alloc(ba.nbytes); ab.h2d(ba)

    t0 = time.perf_counter()
    t = eng.asof_build(kb, ab, rc, 1)  # LE, ascending
    build_s = time.perf_counter() - t0

    pk = rng.integers(1, n_keys + 1, n_probe).astype(np.int32)
    pa = rng.integers(-10**12, 10**12, n_probe).astype(np.int64)
    pkb = eng.alloc(pk.nbytes); pkb.h2d(pk)
    pab = eng.alloc(pa.nbytes); pab.h2d(pa)

    cnt = eng.asof_probe_emit(t, pkb, pab, n_probe, 0)  # warm
    best = None
    for _ in range(5):
        t0 = time.perf_counter()
        c2 = eng.asof_probe_emit(t, pkb, pab, n_probe, 0)
        dt = time.perf_counter() - t0
        assert c2 == cnt
        best = dt if best is None else min(best, dt)

    stream_gb = n_probe * 12 / 1e9
    print(json.dumps({
        "what": "asof_probe_count (LE) + build",
        "build_rows": rc, "distinct_keys": n_keys, "probe_rows": n_probe,
        "build_s": round(build_s, 4),
        "probe_s": round(best, 4),
        "probe_grows_per_s": round(n_probe / best / 1e9, 1),
        "probe_stream_gb": round(stream_gb, 2),
        "probe_stream_gbps_floor": round(stream_gb / best, 0),
        "match_count": int(cnt),
    }))


if __name__ == "__main__":
    main()
