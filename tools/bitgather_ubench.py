"""Raw random 1-bit gather rate vs bitset footprint:
the architectural ceiling for k_q3_probe_agg's order-bits leg. 208 M random
probes (the q3 post-filter probe count scale), footprints 1 MB..64 MB."""
import json
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import numpy as np
from starrocks_amd.engine import Engine


def main():
    eng = Engine(0)
    n = 208_000_000
    idx = eng.alloc(n * 4)
    eng.gen_u32_mod(idx, 7, 11, 0, n, 0, 0)  # uniform u32; kernel masks
    out = {"n_probes": n}
    for mb in (1, 4, 8, 16, 32, 64):
        nbits = mb * (1 << 20) * 8
        bits = eng.alloc(nbits // 8)
        eng.gen_u32_mod(bits, 9, 12, 0, nbits // 32, 0, 0)
        ms = eng.ubench_bitgather(idx, n, bits, nbits, reps=5)
        out[f"{mb}mb"] = {"ms": round(ms, 3),
                          "gprobe_per_s": round(n / ms / 1e6, 1)}
        bits.free()
    print(json.dumps(out))
    eng.close()


if __name__ == "__main__":
    main()
