#!/usr/bin/env python3
"""bench.py — measures BASELINE.json's metric: SSB rows/sec through the
join+agg pipeline on MI355X, N=1 workload = configs[1] (SSB SF10
lineorder⋈date + SUM — the single-GPU quoted config).

Contract (driver): `python bench.py --gpus N --steps K --warmup W`; for N>1
launched via torch.distributed.run with one rank per GPU over RCCL. W untimed
warmup steps, then EXACTLY K timed steps bracketed by barrier +
synchronize on both sides; MAX elapsed over ranks; rank 0 prints ONE JSON
line. A step = one pass of the fused join+aggregate over the rank's resident
shard (inputs already in HBM) + the cross-rank partial-aggregate merge
 + the 16-byte result read.

--workload q21 measures config 3 (SF100 Q2.1 star join) instead; the default
(and what the driver records) is the q1 config-2 line.
"""

import argparse
import json
import os
import sys
import time

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

import numpy as np

SEED = 42
SF10_ROWS = 59_986_052
SF100_ROWS = 600_000_000     # config 3
Q1_YEAR = 1993
Q21_CATEGORY, Q21_REGION = 12, 2
Q1_BYTES_PER_ROW = 12        # 3 × int32 read (algorithmic, §8d)
Q21_BYTES_PER_ROW = 16       # 4 × int32 read
HBM_PEAK_GBPS = 8000.0       # 8 TB/s spec peak (MI355X_MICROARCH.md)


def log(msg):
    if int(os.environ.get("RANK", 0)) == 0:
        print(msg, file=sys.stderr, flush=True)


def build_dim_tables(eng, workload):
    from starrocks_amd import gen
    tables = {}
    datekey, dyear = gen.gen_dates()
    keys = eng.alloc(datekey.nbytes)
    keys.h2d(datekey.astype(np.int32))
    if workload == "q1":
        payload = np.where(dyear == Q1_YEAR, dyear - 1992 + 1, 0).astype(np.uint32)
    else:
        payload = (dyear - 1992 + 1).astype(np.uint32)
    pay = eng.alloc(payload.nbytes)
    pay.h2d(payload)
    tables["dates"] = eng.join_build_payload(keys, pay, len(datekey))
    keys.free(); pay.free()
    if workload == "q21":
        pfirst = gen.build_part_dim_payload(SEED, gen.N_PARTS_SF100, Q21_CATEGORY)
        k = eng.alloc(gen.N_PARTS_SF100 * 4)
        k.h2d(np.arange(1, gen.N_PARTS_SF100 + 1, dtype=np.int32))
        p = eng.alloc(pfirst.nbytes); p.h2d(pfirst)
        tables["parts"] = eng.join_build_payload(k, p, gen.N_PARTS_SF100)
        k.free(); p.free()
        sfirst = gen.build_supp_dim_payload(SEED, gen.N_SUPPS_SF100, Q21_REGION)
        k = eng.alloc(gen.N_SUPPS_SF100 * 4)
        k.h2d(np.arange(1, gen.N_SUPPS_SF100 + 1, dtype=np.int32))
        p = eng.alloc(sfirst.nbytes); p.h2d(sfirst)
        tables["supps"] = eng.join_build_payload(k, p, gen.N_SUPPS_SF100)
        k.free(); p.free()
    return tables


def cpu_baseline(workload, rows_full):
    """Oracle (kind 'port') on this box's host cores; bounded sample
    (~10-30 s of CPU work), compute-only region timed (generation excluded,
    matching the GPU timed region)."""
    from oracle import pyoracle as orc
    from starrocks_amd import gen
    cores = os.cpu_count()
    sample = min(rows_full, 60_000_000)
    if workload == "q1":
        od, ep, dc = orc.gen_lineorder_q1(SEED, 0, sample)
        mn, mx, dfirst = gen.build_date_dim_payload(Q1_YEAR)
        t0 = time.perf_counter()
        passes = 0
        while time.perf_counter() - t0 < 10.0:
            orc.q1_kernel(od, ep, dc, dfirst, mn, mx)
            passes += 1
        dt = time.perf_counter() - t0
    else:
        sample = min(rows_full, 30_000_000)
        pk, sk, od, rv = orc.gen_lineorder_q21(SEED, 0, sample)
        pfirst = gen.build_part_dim_payload(SEED, gen.N_PARTS_SF100, Q21_CATEGORY)
        sfirst = gen.build_supp_dim_payload(SEED, gen.N_SUPPS_SF100, Q21_REGION)
        mn, _, dfirst = gen.build_date_dim_payload(None)
        t0 = time.perf_counter()
        passes = 0
        while time.perf_counter() - t0 < 10.0:
            orc.q21_kernel(pk, sk, od, rv, pfirst, sfirst, dfirst, mn)
            passes += 1
        dt = time.perf_counter() - t0
    rate = passes * sample / dt
    return {"value": round(rate, 1), "unit": "rows/s", "cores": cores, "kind": "port",
            "sample": f"{passes} passes over {sample} rows ({dt:.1f}s, oracle -O3 -fopenmp, "
                      f"OMP over {cores} cores, generation untimed)"}


def read_pmc_traffic(workload):
    p = os.path.join(REPO, "profiles", f"pmc_{workload}.json")
    if os.path.exists(p):
        with open(p) as f:
            d = json.load(f)
        return d.get("traffic_bytes_per_launch")
    return None


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=50)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--workload", choices=["q1", "q21"], default="q1")
    ap.add_argument("--rows", type=int, default=0, help="override rows per GPU")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", args.gpus))
    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))

    from starrocks_amd.engine import Engine

    dist = None
    if world > 1:
        import torch
        import torch.distributed as tdist
        torch.cuda.set_device(local_rank)
        tdist.init_process_group("nccl")
        dist = tdist

    eng = Engine(local_rank)
    rows = args.rows or (SF10_ROWS if args.workload == "q1" else SF100_ROWS)
    row_start = rank * rows  # weak scaling: each rank owns its shard

    log(f"[bench] workload={args.workload} rows/gpu={rows} world={world} "
        f"steps={args.steps} warmup={args.warmup}")

    # ---- untimed setup: generate shard on device, build dim tables ----
    t_setup = time.perf_counter()
    if args.workload == "q1":
        cols = [eng.alloc(rows * 4) for _ in range(3)]
        eng.gen_lineorder_q1(SEED, row_start, rows, *cols)
        tables = build_dim_tables(eng, "q1")
        acc = eng.alloc(16)

        def step():
            eng.q1_join_sum_async(tables["dates"], cols[0], cols[1], cols[2], rows, acc)
            vals = acc.d2h(np.int64, 2)  # result read (syncs stream)
            return vals
    else:
        cols = [eng.alloc(rows * 4) for _ in range(4)]
        eng.gen_lineorder_q21(SEED, row_start, rows, *cols)
        tables = build_dim_tables(eng, "q21")
        acc = eng.alloc(7000 * 8)

        def step():
            eng.q21_star_agg_async(tables["parts"], tables["supps"], tables["dates"],
                                   cols[0], cols[1], cols[2], cols[3], rows, acc)
            return acc.d2h(np.int64, 7000)

    eng.sync()
    log(f"[bench] setup {time.perf_counter()-t_setup:.1f}s")

    merge_buf = None
    if world > 1:
        import torch
        merge_buf = torch.zeros(2 if args.workload == "q1" else 7000,
                                dtype=torch.int64, device="cuda")

    def run_step():
        vals = step()
        if dist is not None:
            merge_buf.copy_(torch.from_numpy(np.asarray(vals)))
            dist.all_reduce(merge_buf)  # phase1→phase2 aggregate merge (RCCL)
            return merge_buf
        return vals

    for _ in range(args.warmup):
        run_step()

    # ---- timed region ----
    if dist is not None:
        import torch
        dist.barrier()
        torch.cuda.synchronize()
    eng.sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        run_step()
    eng.sync()
    if dist is not None:
        import torch
        torch.cuda.synchronize()
        dist.barrier()
    elapsed = time.perf_counter() - t0

    if dist is not None:
        import torch
        t = torch.tensor([elapsed], device="cuda")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    value = rows * world * args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1000

    # ---- roofline evidence: HIP events around kernel-only launches ----
    bytes_per_row = Q1_BYTES_PER_ROW if args.workload == "q1" else Q21_BYTES_PER_ROW
    R = 20
    eng.sync()
    eng.timer_start()
    for _ in range(R):
        if args.workload == "q1":
            eng.q1_join_sum_async(tables["dates"], cols[0], cols[1], cols[2], rows, acc)
        else:
            eng.q21_star_agg_async(tables["parts"], tables["supps"], tables["dates"],
                                   cols[0], cols[1], cols[2], cols[3], rows, acc)
    kernel_ms = eng.timer_stop() / R
    algo_bytes = bytes_per_row * rows
    achieved_gbps = algo_bytes / (kernel_ms / 1e3) / 1e9
    roofline = {"bound": "hbm", "achieved": round(achieved_gbps, 1), "peak": HBM_PEAK_GBPS,
                "unit": "GB/s", "frac": round(achieved_gbps / HBM_PEAK_GBPS, 4),
                "traffic": read_pmc_traffic(args.workload)}

    result = None
    if rank == 0:
        cb = cpu_baseline(args.workload, rows) if world == 1 else None
        result = {
            "metric": "ssb_join_agg_rows_per_sec",
            "value": round(value, 1),
            "unit": "rows/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 4),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # no published number for this metric
            "dtype": "int64",
            "data": "synthetic",
            "config": {
                "workload": ("ssb_sf10_q1_join_sum" if args.workload == "q1"
                             else "ssb_sf100_q21_star_groupby"),
                "rows_per_gpu": rows,
                "seed": SEED,
                "parallelism": f"dp{world}-replicated-dims",
            },
            "roofline": roofline,
            "cpu_baseline": cb,
        }
        print(json.dumps(result), flush=True)

    for c in cols:
        c.free()
    acc.free()
    for t in tables.values():
        t.destroy()
    eng.close()
    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
