"""CPU-baseline credibility diagnostics: OMP thread
scaling of the oracle q21 kernel + a numpy copy bandwidth probe, run on the
GPU box's host cores."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import numpy as np
from oracle import pyoracle as orc
from starrocks_amd import gen

n = 30_000_000
pk, sk, od, rv = orc.gen_lineorder_q21(42, 0, n)
pf = gen.build_part_dim_payload(42, gen.N_PARTS_SF100, 12)
sf = gen.build_supp_dim_payload(42, gen.N_SUPPS_SF100, 2)
mn, _, df = gen.build_date_dim_payload(None)
print("cpu_count", os.cpu_count(), "affinity", len(os.sched_getaffinity(0)),
      "OMP_NUM_THREADS", os.environ.get("OMP_NUM_THREADS"), flush=True)
for t in (1, 8, 32, 64, 128, 256):
    orc.q21_kernel(pk, sk, od, rv, pf, sf, df, mn, threads=t)  # warm
    t0 = time.perf_counter()
    p = 0
    while time.perf_counter() - t0 < 3:
        orc.q21_kernel(pk, sk, od, rv, pf, sf, df, mn, threads=t)
        p += 1
    dt = time.perf_counter() - t0
    print(f"threads={t:4d}  {p * n / dt / 1e6:8.0f} Mrows/s  "
          f"{p * n * 16 / dt / 1e9:6.1f} GB/s algorithmic", flush=True)
a = np.empty(25_000_000)
b = np.empty_like(a)
np.copyto(b, a)
t0 = time.perf_counter()
for _ in range(20):
    np.copyto(b, a)
dt = time.perf_counter() - t0
print(f"numpy 200MB copy: {20 * a.nbytes * 2 / dt / 1e9:.1f} GB/s (1 thread)")
