"""Leg decomposition of the config-5 q3 probe+agg kernel:
times the kernel with successive legs enabled — 1: ship stream, 3: +lk read
and order-bits gather, 7: +ext/disc product, 15: full (hash insert). Run on
the GPU box; writes gpurun_out/q3_decomp.txt.

Leg-15 iterations after the first re-add into claimed slots (steady-state
adds, like bench steps without the per-pass reset), so its delta is the
insert/add cost, not first-touch claiming.
"""

import sys

sys.path.insert(0, ".")
import numpy as np  # noqa: E402

from starrocks_amd.engine import Engine  # noqa: E402
from oracle import pyoracle as orc  # noqa: E402

SEED = 42
N = 225_000_000
N_ORDERS = 450_000_000
N_CUSTS = 45_000_000
CUTOFF = 19950315
WARMUP, STEPS = 2, 5


def log(m):
    print(m, flush=True)


def main():
    import time
    t0 = time.time()
    global N, N_ORDERS, N_CUSTS
    if len(sys.argv) > 1 and sys.argv[1] == "small":
        N, N_ORDERS, N_CUSTS = 20_000_000, 40_000_000, 4_000_000
    e = Engine(0)
    log(f"engine up {time.time()-t0:.1f}s")
    mkt = e.alloc(N_CUSTS * 16)
    e.gen_cust_mkt16(SEED, N_CUSTS, mkt)
    cbits = e.alloc((N_CUSTS + 31) // 32 * 4)
    e.bits_str16_eq(mkt, N_CUSTS, orc.mkt_literal(1), cbits)
    mkt.free()
    oc, od = e.alloc(N_ORDERS * 4), e.alloc(N_ORDERS * 4)
    e.gen_orders_q3(SEED, N_ORDERS, N_CUSTS, oc, od)
    obits = e.alloc((N_ORDERS + 31) // 32 * 4)
    e.q3_order_bits(oc, od, N_ORDERS, cbits, CUTOFF, obits)
    for b in (cbits, oc, od):
        b.free()
    lk, ext, disc = (e.alloc(N * 8) for _ in range(3))
    ship = e.alloc(N * 4)
    e.gen_lineitem_q3(SEED, 0, N, N_ORDERS, lk, ext, disc, ship)
    e.sync()
    log(f"data generated {time.time()-t0:.1f}s")
    at = e.agg_table_create(64_000_000)
    e.sync()
    log(f"agg table ready {time.time()-t0:.1f}s")
    sink = e.alloc(8)
    sink.h2d(np.zeros(1, np.uint64))

    lines = []
    prev = None
    names = {1: "ship stream", 3: "+lk & obits gather", 7: "+ext*disc", 15: "+hash insert",
             17: "NT ship stream", 19: "NT +lk & obits", 23: "NT +ext*disc", 31: "NT full"}
    for legs in (1, 3, 7, 15, 17, 19, 23, 31):
        for _ in range(WARMUP):
            if legs & 8:
                e.agg_table_reset(at)
            e.q3_decomp(lk, ext, disc, ship, N, obits, CUTOFF, legs, at, sink)
        e.sync()
        log(f"legs={legs} warm {time.time()-t0:.1f}s")
        e.timer_start()
        for _ in range(STEPS):
            if legs & 8:  # the real step resets the table per pass too
                e.agg_table_reset(at)
            e.q3_decomp(lk, ext, disc, ship, N, obits, CUTOFF, legs, at, sink)
        ms = e.timer_stop() / STEPS
        if legs == 17:
            prev = None
        delta = "" if prev is None else f"  (+{ms - prev:.3f} ms)"
        lines.append(f"legs={legs:2d} {names[legs]:<20s} {ms:8.3f} ms/pass{delta}")
        prev = ms
    out = "\n".join(lines) + "\n"
    print(out)
    with open("gpurun_out/q3_decomp.txt", "w") as f:
        f.write(f"q3 leg decomposition, N={N}, {STEPS} steps after {WARMUP} warmup\n")
        f.write(out)


if __name__ == "__main__":
    main()
