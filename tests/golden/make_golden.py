"""Generates tests/golden/hash_kats.json from the reference's OWN compiled
hash code (oracle/_ref/ref.so, built against /root/reference headers in
place). Run in the survey container only; the committed JSON is what the GPU
box (no /root/reference) checks against.

Usage: python tests/golden/make_golden.py
"""

import json
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

import numpy as np

from oracle import pyoracle as orc


def main():
    ref = orc.load_ref()
    assert ref is not None, "build oracle/_ref first (make -C oracle)"
    rng = np.random.default_rng(20260915)
    crc, fnv = [], []
    for _ in range(64):
        n = int(rng.integers(0, 64))
        data = rng.integers(0, 256, n, dtype=np.uint8).tobytes()
        seed = int(rng.integers(0, 2**32, dtype=np.uint64))
        crc.append({"data_hex": data.hex(), "seed": seed,
                    "expect": int(ref.ref_crc_hash_32(data, n, seed))})
        fnv.append({"data_hex": data.hex(), "seed": seed,
                    "expect": int(ref.ref_fnv_hash(data, n, seed))})
    # the reference test-file KATs verbatim (join_hash_map_test.cpp:1009)
    crc.append({"data_hex": b"abcd".hex(), "seed": 0x811C9DC5,
                "expect": int(ref.ref_crc_hash_32(b"abcd", 4, 0x811C9DC5))})
    out = {"source": "reference be/src/base/hash via oracle/_ref/ref.so",
           "crc_hash_32": crc, "fnv_hash": fnv}
    path = os.path.join(os.path.dirname(__file__), "hash_kats.json")
    with open(path, "w") as f:
        json.dump(out, f, indent=1)
    print(f"wrote {path}: {len(crc)} crc + {len(fnv)} fnv vectors")


if __name__ == "__main__":
    main()
