"""Config-1 measurement: int64 scan + predicate filter at 1 B rows.

Reports rows/s and achieved GB/s against the reference's algorithmic bytes
definition (read 8 B + write 8*s per row) and the
two-pass implementation's actual byte count (16 + 8*s) beside it.
"""

import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import numpy as np
from starrocks_amd.engine import Engine


def main():
    eng = Engine(0)
    n = 1_000_000_000
    inp = eng.alloc(n * 8)
    eng.gen_i64(inp, 42, 3, 0, n)
    out = eng.alloc(n * 8)
    eng.sync()
    for sel in (0.01, 0.1, 0.5):
        # uniform u64-as-i64 domain: theta at the requested selectivity
        theta = int((sel - 0.5) * 2**64) if True else 0
        theta = int(-(2**63) + sel * 2**64)
        cnt = eng.scan_filter_i64_lt(inp, n, theta, out)  # warm
        eng.sync()
        reps = 5
        eng.timer_start()
        for _ in range(reps):
            eng.scan_filter_i64_lt(inp, n, theta, out)
        ms = eng.timer_stop() / reps
        algo = (8 + 8 * sel) * n
        actual = (16 + 8 * sel) * n  # two-pass reads the input twice
        cnt2 = eng.scan_filter_i64_lt_sp(inp, n, theta, out)
        assert cnt2 == cnt, (cnt, cnt2)
        eng.sync()
        eng.timer_start()
        for _ in range(reps):
            eng.scan_filter_i64_lt_sp(inp, n, theta, out)
        ms_sp = eng.timer_stop() / reps
        print(json.dumps({
            "selectivity": sel, "rows": n, "matched": cnt,
            "twopass_ms": round(ms, 3),
            "twopass_algorithmic_gbps": round(algo / (ms / 1e3) / 1e9, 1),
            "singlepass_ms": round(ms_sp, 3),
            "singlepass_algorithmic_gbps": round(algo / (ms_sp / 1e3) / 1e9, 1),
            "rows_per_s_singlepass": round(n / (ms_sp / 1e3), 1)}))
    eng.close()


if __name__ == "__main__":
    main()
