"""Streaming microbenchmarks on the GPU box: establishes the achievable
HBM ceilings the fused kernels are judged against.

Run via gpurun: python tools/ubench.py
"""

import ctypes
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from starrocks_amd.engine import Engine, _ck, c_vp, c_u64, c_i32


def main():
    eng = Engine(0)
    lib = eng._lib
    lib.gpue_ubench.restype = c_i32
    lib.gpue_ubench.argtypes = [c_vp, c_i32, c_vp, c_vp, c_vp, c_u64,
                                c_i32, ctypes.POINTER(ctypes.c_float)]
    for n in (59_986_052, 240_000_000):
        a, b, c = (eng.alloc(n * 4) for _ in range(3))
        for buf, tag in ((a, 1), (b, 2), (c, 3)):
            eng.gen_u32_mod(buf, 42, tag, 0, n, 0, 0)
        eng.sync()
        ms = ctypes.c_float()
        _ck(lib, lib.gpue_ubench(eng._h, 0, a._h, None, None, n, 20, ctypes.byref(ms)))
        gbps1 = n * 4 / (ms.value / 1e3) / 1e9
        print(f"n={n}: ub_sum1 (1 stream read): {ms.value:.3f} ms -> {gbps1:.0f} GB/s")
        _ck(lib, lib.gpue_ubench(eng._h, 1, a._h, b._h, c._h, n, 20, ctypes.byref(ms)))
        gbps3 = 3 * n * 4 / (ms.value / 1e3) / 1e9
        print(f"n={n}: ub_sum3 (3 stream read): {ms.value:.3f} ms -> {gbps3:.0f} GB/s")
        for x in (a, b, c):
            x.free()
    eng.close()


if __name__ == "__main__":
    main()
