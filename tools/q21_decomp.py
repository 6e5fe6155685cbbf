"""Decompose the q21 star-agg kernel's cost: streams-only vs phase-1-only."""
import ctypes, os, sys
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import numpy as np
from starrocks_amd.engine import Engine, _ck, c_vp, c_u64, c_i32, c_i64
from starrocks_amd import gen

eng = Engine(0)
lib = eng._lib
lib.gpue_ubench_q21.restype = c_i32
lib.gpue_ubench_q21.argtypes = [c_vp, c_i32] + [c_vp] * 5 + [c_i64, c_u64, c_u64,
                                c_i32, c_i32, ctypes.POINTER(ctypes.c_float)]
n = 600_000_000
cols = [eng.alloc(n * 4) for _ in range(4)]
eng.gen_lineorder_q21(42, 0, n, *cols)
pfirst = gen.build_part_dim_payload(42, gen.N_PARTS_SF100, 12)
kb = eng.alloc(gen.N_PARTS_SF100 * 4); kb.h2d(np.arange(1, gen.N_PARTS_SF100 + 1, dtype=np.int32))
pb = eng.alloc(pfirst.nbytes); pb.h2d(pfirst)
t = eng.join_build_payload(kb, pb, gen.N_PARTS_SF100)
mn, mx = t.minmax
# bitset dbuf view: use join table's internal bitset via... expose through first_d2h? Instead
# rebuild a bitset dbuf directly from pfirst host-side
smin = int(np.flatnonzero(pfirst)[0]) + 1
smax = int(np.flatnonzero(pfirst)[-1]) + 1
bits = np.zeros(((smax - smin + 1) + 31)//32, np.uint32)
nz = np.flatnonzero(pfirst) + 1
idx = nz - smin
np.bitwise_or.at(bits, idx // 32, (np.uint32(1) << (idx % 32).astype(np.uint32)))
bbuf = eng.alloc(bits.nbytes); bbuf.h2d(bits)
eng.sync()
ms = ctypes.c_float()
for grid in (256, 512, 1024):
    _ck(lib, lib.gpue_ubench_q21(eng._h, 1, cols[0]._h, cols[1]._h, cols[2]._h, cols[3]._h,
                                 None, 0, 0, n, grid, 10, ctypes.byref(ms)))
    print(f"grid {grid}: streams-only (4x16B): {ms.value:.3f} ms -> {16*n/(ms.value/1e3)/1e9:.0f} GB/s")
    _ck(lib, lib.gpue_ubench_q21(eng._h, 0, cols[0]._h, None, None, None, bbuf._h,
                                 smin, smax - smin + 1, n, grid, 10, ctypes.byref(ms)))
    print(f"grid {grid}: phase1-only (pk+bitgather): {ms.value:.3f} ms -> {4*n/(ms.value/1e3)/1e9:.0f} GB/s(pk)")
eng.close()

# r02: PF-kernel geometry legs (120 KB LDS, 1024 threads, 1 block/CU).
# which=2 streams-only, which=3 + LDS prefilter test. Needs a 64 KB folded
# prefilter buffer: fold the bitset on host exactly as k_build_prefilter.
eng2 = Engine(0)
lib2 = eng2._lib
lib2.gpue_ubench_q21.restype = c_i32
lib2.gpue_ubench_q21.argtypes = [c_vp, c_i32] + [c_vp] * 5 + [c_i64, c_u64, c_u64,
                                 c_i32, c_i32, ctypes.POINTER(ctypes.c_float)]
cols2 = [eng2.alloc(n * 4) for _ in range(4)]
eng2.gen_lineorder_q21(42, 0, n, *cols2)
pf_fold = np.zeros((1 << 19) // 32, np.uint32)
# fold: bit i of `bits` -> prefilter bit (i & (2^19-1))
words = bits
for w in range(len(words)):
    v = int(words[w])
    while v:
        b = (v & -v).bit_length() - 1
        i = w * 32 + b
        f = i & ((1 << 19) - 1)
        pf_fold[f >> 5] |= np.uint32(1 << (f & 31))
        v &= v - 1
pfb = eng2.alloc(pf_fold.nbytes)
pfb.h2d(pf_fold)
eng2.sync()
ms2 = ctypes.c_float()
for which, name in ((2, "pf-geom streams-only"), (3, "pf-geom streams+LDS-test")):
    _ck(lib2, lib2.gpue_ubench_q21(eng2._h, which, cols2[0]._h, cols2[1]._h, cols2[2]._h,
                                   cols2[3]._h, pfb._h, smin, smax - smin + 1, n, 256, 10,
                                   ctypes.byref(ms2)))
    print(f"{name}: {ms2.value:.3f} ms -> {16*n/(ms2.value/1e3)/1e9:.0f} GB/s")
eng2.close()
