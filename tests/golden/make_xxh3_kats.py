#!/usr/bin/env python3
"""Generates xxh3_kats.json — golden vectors for the XXH3-64 4-to-8-byte
input path, produced by the PUBLISHED python `xxhash` module (the binding of
the upstream xxHash library whose algorithm oracle.c/gpue.hip restate; the
C library itself is absent from the offline reference checkout).

Pins the version-1 exchange hash (exchange_sink_operator.cpp:604-610):
per-row XXH3_64bits_withSeed over each 4/8-byte key value, chained per
column from HashUtil::XXH3_SEED_32 = 0x9E3779B1 and truncated to u32
between columns (column_hash.cpp:65-68, hash_util.hpp:126).

Run in a container with the xxhash wheel:  python tests/golden/make_xxh3_kats.py
"""

import json
import os
import random
import struct

import xxhash

XXH3_SEED_32 = 0x9E3779B1


def main():
    random.seed(20260915)
    raw = []
    for _ in range(64):
        ln = random.choice([4, 5, 6, 7, 8])
        data = bytes(random.randrange(256) for _ in range(ln))
        seed = random.choice([0, XXH3_SEED_32, random.randrange(1 << 32),
                              random.randrange(1 << 64)])
        raw.append({"data_hex": data.hex(), "len": ln, "seed": seed,
                    "xxh3_64": xxhash.xxh3_64_intdigest(data, seed)})

    # chained exchange-hash vectors: two i32 columns, per-row chain
    cols = [[random.randrange(-2**31, 2**31) for _ in range(16)] for _ in range(2)]
    chained = []
    for r in range(16):
        h = XXH3_SEED_32
        for col in cols:
            h = xxhash.xxh3_64_intdigest(struct.pack("<i", col[r]), h) & 0xFFFFFFFF
        chained.append(h)

    out = {"single": raw, "exchange_cols_i32": cols, "exchange_chained_u32": chained}
    path = os.path.join(os.path.dirname(__file__), "xxh3_kats.json")
    with open(path, "w") as f:
        json.dump(out, f, indent=1)
    print(f"wrote {path}: {len(raw)} single + {len(chained)} chained vectors")


if __name__ == "__main__":
    main()
