"""A/B: fused q21 kernel vs two-stream pipelined operator pair."""
import ctypes, os, sys
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import numpy as np
from starrocks_amd.engine import Engine
from starrocks_amd import gen

eng = Engine(0)
n = 600_000_000
cols = [eng.alloc(n * 4) for _ in range(4)]
eng.gen_lineorder_q21(42, 0, n, *cols)

def table(keys, pay):
    kb = eng.alloc(keys.nbytes); kb.h2d(keys)
    pb = eng.alloc(pay.nbytes); pb.h2d(pay)
    t = eng.join_build_payload(kb, pb, len(keys))
    kb.free(); pb.free()
    return t

parts = table(np.arange(1, gen.N_PARTS_SF100 + 1, dtype=np.int32),
              gen.build_part_dim_payload(42, gen.N_PARTS_SF100, 12))
supps = table(np.arange(1, gen.N_SUPPS_SF100 + 1, dtype=np.int32),
              gen.build_supp_dim_payload(42, gen.N_SUPPS_SF100, 2))
datekey, dyear = gen.gen_dates()
dates = table(datekey.astype(np.int32), (dyear - 1992 + 1).astype(np.uint32))
acc = eng.alloc(7000 * 8)
scratch = eng.alloc(n * 2)
eng.sync()

eng.q21_star_agg_async(parts, supps, dates, *cols, n, acc)
ref = acc.d2h(np.int64, 7000)
for chunks in (4, 8, 16, 32):
    eng.q21_star_agg_pipe(parts, supps, dates, *cols, n, scratch, acc, chunks)
    got = acc.d2h(np.int64, 7000)
    assert np.array_equal(got, ref), f"parity FAIL chunks={chunks}"
eng.sync()

R = 10
eng.timer_start()
for _ in range(R):
    eng.q21_star_agg_async(parts, supps, dates, *cols, n, acc)
ms = eng.timer_stop() / R
print(f"fused: {ms:.3f} ms -> {16*n/(ms/1e3)/1e9:.0f} GB/s")
for chunks in (4, 8, 16, 32):
    eng.timer_start()
    for _ in range(R):
        eng.q21_star_agg_pipe(parts, supps, dates, *cols, n, scratch, acc, chunks)
    ms = eng.timer_stop() / R
    print(f"pipe x{chunks}: {ms:.3f} ms -> {16*n/(ms/1e3)/1e9:.0f} GB/s (parity OK)")
eng.close()
