"""Pinned double-buffered H2D ingest measurement:
streams an 8 GB host buffer to HBM through gpue_ingest (2 pinned staging
buffers, copy stream) across chunk sizes, vs the pageable single-shot
gpue_dbuf_h2d (56.5 GB/s in r01)."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import numpy as np
from starrocks_amd.engine import Engine


def main():
    eng = Engine(0)
    n_bytes = 8 << 30
    host = np.random.default_rng(1).integers(0, 2**63, n_bytes // 8,
                                             dtype=np.int64)
    dst = eng.alloc(n_bytes)
    out = {"total_gib": 8}
    # pageable baseline
    t0 = time.perf_counter()
    dst.h2d(host)
    eng.sync()
    out["pageable_gbps"] = round(n_bytes / (time.perf_counter() - t0) / 1e9, 1)
    # pure pinned-DMA rate: producer writes directly into page-locked memory
    # (gpue_pinned_alloc), no staging copy at all — the zero-copy io-task form
    import ctypes
    pin_bytes = 1 << 30
    pp = eng.pinned_alloc(pin_bytes)
    ctypes.memmove(pp, host.ctypes.data, pin_bytes)
    pin_arr = np.ctypeslib.as_array((ctypes.c_int64 * (pin_bytes // 8)).from_address(pp))
    dst.h2d(pin_arr[:1 << 20])  # warm
    eng.sync()
    t0 = time.perf_counter()
    for rep in range(8):
        dst.h2d(pin_arr)
    eng.sync()
    out["pinned_zero_copy_gbps"] = round(8 * pin_bytes / (time.perf_counter() - t0) / 1e9, 1)
    eng.pinned_free(pp)
    for mb in (8, 16, 32, 64, 128):
        ing = eng.ingest_create(mb << 20)
        eng.ingest_push(ing, host[:1 << 20], dst)  # warm
        eng.ingest_sync(ing)
        t0 = time.perf_counter()
        eng.ingest_push(ing, host, dst)
        eng.ingest_sync(ing)
        out[f"pinned_{mb}mb_gbps"] = round(n_bytes / (time.perf_counter() - t0) / 1e9, 1)
        eng.ingest_destroy(ing)
    print(json.dumps(out))
    eng.close()


if __name__ == "__main__":
    main()
