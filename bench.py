#!/usr/bin/env python3
"""bench.py — measures BASELINE.json's metric: SSB rows/sec through the
join+agg pipeline on MI355X. N=1 workload = q21 (configs[2]: SSB SF100 Q2.1,
600 M rows — the largest single-GPU BASELINE config, the one the "SSB SF100
rows/sec" metric is quoted on). q1 (config 1, SF10) stays
available via --workload q1.

Contract (driver): `python bench.py --gpus N --steps K --warmup W`; for N>1
launched via torch.distributed.run with one rank per GPU over RCCL. W untimed
warmup steps, then EXACTLY K timed steps bracketed by barrier + synchronize on
both sides; MAX elapsed over ranks; rank 0 prints ONE JSON line. A step = one
pass of the fused join+aggregate over the rank's resident shard (inputs
already in HBM) + the cross-rank partial-aggregate merge + the
result read.

Workloads: q21 (default, config 3), q1 (config 2), q43 (config 4 — at N>1 it
runs the MANDATED hash-partitioned mode: per-step partition kernel + gather +
RCCL all-to-all of the probe columns on lo_custkey, then local probe+agg).
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
SF100_ROWS = 600_000_000     # configs 3-4
Q1_YEAR = 1993
Q21_CATEGORY, Q21_REGION = 12, 2
Q43_REGION, Q43_NATION, Q43_CATEGORY = 1, 7, 12
BYTES_PER_ROW = {"q1": 12, "q21": 16, "q43": 24, "q3": 28}  # algorithmic (§8d)
Q3_N_ORDERS, Q3_N_CUSTS = 450_000_000, 45_000_000  # SF300
Q3_ROWS_PER_GPU = 225_000_000  # 8 GPUs × 225 M = the §8d 1.8 B-row dataset
Q3_CUTOFF = 19950315
HBM_PEAK_GBPS = 8000.0       # 8 TB/s spec peak (MI355X_MICROARCH.md)


def log(msg):
    if int(os.environ.get("RANK", 0)) == 0:
        print(msg, file=sys.stderr, flush=True)


def build_payload_table(eng, keys_i32, payload_u32):
    # Plan step: run the reference's JoinHashMapSelector on this build side
    # (join_hash_table.cpp:164-344, restated as gpue_join_select_method). The
    # SSB dims are dense unique int keys, so the decision must land in the
    # RANGE_DIRECT family — whose MI355X fused form is the payload table
    # (first[key-min] holds payload+1). Any other decision
    # means the workload changed and the fused kernel no longer applies.
    m = eng.join_select_method(0, 1, len(keys_i32),  # ONE_KEY constructor, LT_INT
                               int(keys_i32.min()), int(keys_i32.max()))
    if eng.JM_NAMES[m] not in ("RANGE_DIRECT", "RANGE_DIRECT_SET",
                               "DENSE_RANGE_DIRECT", "DIRECT"):
        raise RuntimeError(f"selector chose {eng.JM_NAMES[m]}; the fused "
                           "payload-table path requires a direct-mapped plan")
    k = eng.alloc(keys_i32.nbytes)
    k.h2d(keys_i32)
    p = eng.alloc(payload_u32.nbytes)
    p.h2d(payload_u32)
    t = eng.join_build_payload(k, p, len(keys_i32))
    k.free()
    p.free()
    return t


def build_dim_tables(eng, workload, rank=0, world=1):
    from starrocks_amd import gen
    tables = {}
    datekey, dyear = gen.gen_dates()
    dkeys = datekey.astype(np.int32)
    if workload == "q1":
        dpay = np.where(dyear == Q1_YEAR, dyear - 1992 + 1, 0).astype(np.uint32)
    elif workload == "q21":
        dpay = (dyear - 1992 + 1).astype(np.uint32)
    else:  # q43: 1997 -> 1, 1998 -> 2
        dpay = np.where(dyear == 1997, 1, np.where(dyear == 1998, 2, 0)).astype(np.uint32)
    tables["dates"] = build_payload_table(eng, dkeys, dpay)
    if workload == "q21":
        tables["parts"] = build_payload_table(
            eng, np.arange(1, gen.N_PARTS_SF100 + 1, dtype=np.int32),
            gen.build_part_dim_payload(SEED, gen.N_PARTS_SF100, Q21_CATEGORY))
        tables["supps"] = build_payload_table(
            eng, np.arange(1, gen.N_SUPPS_SF100 + 1, dtype=np.int32),
            gen.build_supp_dim_payload(SEED, gen.N_SUPPS_SF100, Q21_REGION))
    elif workload == "q43":
        cpay = gen.build_cust_dim_q43(SEED, gen.N_CUSTS_SF100, Q43_REGION)
        if world > 1:
            # hash-partitioned join on lo_custkey: this rank owns only the
            # customers the partition function routes to it
            ckeys = np.arange(1, gen.N_CUSTS_SF100 + 1, dtype=np.uint32)
            owned = gen.partition_channels(ckeys, world) == rank
            cpay = np.where(owned, cpay, 0).astype(np.uint32)
        tables["custs"] = build_payload_table(
            eng, np.arange(1, gen.N_CUSTS_SF100 + 1, dtype=np.int32), cpay)
        tables["supps"] = build_payload_table(
            eng, np.arange(1, gen.N_SUPPS_SF100 + 1, dtype=np.int32),
            gen.build_supp_dim_q43(SEED, gen.N_SUPPS_SF100, Q43_NATION))
        tables["parts"] = build_payload_table(
            eng, np.arange(1, gen.N_PARTS_SF100 + 1, dtype=np.int32),
            gen.build_part_dim_q43(SEED, gen.N_PARTS_SF100, Q43_CATEGORY))
    return tables


def cpu_affinity_cores():
    """The CPU budget this process actually has. os.cpu_count() and even
    sched_getaffinity lie under container CPU QUOTAS (cgroup cpu.max): the
    GPU box reports 256 CPUs with a full affinity mask, but the oracle's
    thread sweep (profiles/r02_cpu_diag.txt) peaks at 32 threads
    (5.8 Grows/s) and collapses 26x at 256 — a >8x-oversubscribed OMP team
    spends its time preempted at barriers. Use the cgroup quota when one is
    set; otherwise the affinity count."""
    try:
        cores = len(os.sched_getaffinity(0))
    except AttributeError:
        cores = os.cpu_count()
    try:  # cgroup v2
        with open("/sys/fs/cgroup/cpu.max") as f:
            quota, period = f.read().split()
            if quota != "max":
                cores = min(cores, max(1, int(int(quota) / int(period))))
    except OSError:
        try:  # cgroup v1
            with open("/sys/fs/cgroup/cpu/cpu.cfs_quota_us") as f:
                quota = int(f.read())
            with open("/sys/fs/cgroup/cpu/cpu.cfs_period_us") as f:
                period = int(f.read())
            if quota > 0:
                cores = min(cores, max(1, quota // period))
        except OSError:
            pass
    return cores


def cpu_baseline(workload, rows_full):
    """Oracle (kind 'port') on this box's host cores; bounded sample
    (~10 s of CPU work), compute-only region timed (generation excluded,
    matching the GPU timed region). cores = sched_getaffinity (the real
    budget), OMP explicitly set to it; achieved CPU GB/s stated so the
    number is checkable against the box's memory bandwidth."""
    from oracle import pyoracle as orc
    from starrocks_amd import gen
    cores = cpu_affinity_cores()
    threads = int(os.environ.get("OMP_NUM_THREADS") or 0) or cores
    if workload == "q1":
        sample = min(rows_full, 60_000_000)
        od, ep, dc = orc.gen_lineorder_q1(SEED, 0, sample)
        mn, mx, dfirst = gen.build_date_dim_payload(Q1_YEAR)
        run = lambda: orc.q1_kernel(od, ep, dc, dfirst, mn, mx, threads=threads)
    elif workload == "q21":
        sample = min(rows_full, 30_000_000)
        pk, sk, od, rv = orc.gen_lineorder_q21(SEED, 0, sample)
        pfirst = gen.build_part_dim_payload(SEED, gen.N_PARTS_SF100, Q21_CATEGORY)
        sfirst = gen.build_supp_dim_payload(SEED, gen.N_SUPPS_SF100, Q21_REGION)
        mn, _, dfirst = gen.build_date_dim_payload(None)
        run = lambda: orc.q21_kernel(pk, sk, od, rv, pfirst, sfirst, dfirst, mn,
                                     threads=threads)
    else:
        sample = min(rows_full, 30_000_000)
        ck, sk, pk, od, rv, sc = orc.gen_lineorder_q43(SEED, 0, sample)
        cfirst = gen.build_cust_dim_q43(SEED, gen.N_CUSTS_SF100, Q43_REGION)
        sfirst = gen.build_supp_dim_q43(SEED, gen.N_SUPPS_SF100, Q43_NATION)
        pfirst = gen.build_part_dim_q43(SEED, gen.N_PARTS_SF100, Q43_CATEGORY)
        mn, _, dfirst = gen.build_date_dim_q43()
        run = lambda: orc.q43_kernel(ck, sk, pk, od, rv, sc, cfirst, sfirst,
                                     pfirst, dfirst, mn, threads=threads)
    # Thread-count calibration (profiles/r02_cpu_diag.txt): the box reports
    # 256 CPUs but quota/contention makes big OMP teams collapse (peak at 32
    # threads, 26x slower at 256). Probe a few candidates briefly and keep
    # the fastest — the reported `cores` is what was actually used.
    def rate_at(t, seconds):
        nonlocal threads
        threads = t
        run()  # warm team (untimed)
        t0 = time.perf_counter()
        p = 0
        while time.perf_counter() - t0 < seconds:
            run()
            p += 1
        return p * sample / (time.perf_counter() - t0)

    candidates = sorted({cores, 128, 64, 32, 16}, reverse=True)
    candidates = [t for t in candidates if t <= cores] or [cores]
    best = max(candidates, key=lambda t: rate_at(t, 0.8))
    rate = rate_at(best, 8.0)
    gbps = rate * BYTES_PER_ROW[workload] / 1e9
    return {"value": round(rate, 1), "unit": "rows/s", "cores": best, "kind": "port",
            "sample": f"oracle -O3 -fopenmp over {sample} rows, OMP threads "
                      f"calibrated over {candidates} -> {best} "
                      f"(sched_getaffinity={cores}, cpu_count={os.cpu_count()}; "
                      f"large teams collapse under the box's CPU quota — "
                      f"profiles/r02_cpu_diag.txt), achieved {gbps:.1f} GB/s "
                      f"algorithmic, generation untimed"}


def cpu_baseline_q3(rows_full):
    """Oracle Q3 compute leg (orders-bits build + probe + hash agg) on this
    box's cores; generation untimed, matching the GPU timed region."""
    from oracle import pyoracle as orc
    sample = min(rows_full, 20_000_000)
    n_orders, n_custs = 8_000_000, 800_000  # dims scaled with the sample
    mkt = orc.gen_cust_mkt16(SEED, n_custs)
    cbits = np.zeros((n_custs + 7) // 8, np.uint8)
    orc.load().orc_q3_build_cust_bits(orc._p(mkt), n_custs, orc.mkt_literal(1),
                                      orc._p(cbits))
    oc, od_ = orc.gen_orders_q3(SEED, n_orders, n_custs)
    obits = np.zeros((n_orders + 7) // 8, np.uint8)
    lk, ext, disc, ship = orc.gen_lineitem_q3(SEED, 0, sample, n_orders)
    ok = np.empty(sample, np.uint64)
    os_ = np.empty(sample, np.int64)
    cores = cpu_affinity_cores()
    # same quota-collapse mitigation as cpu_baseline (profiles/r02_cpu_diag.txt)
    threads = min(cores, 32)
    orc.load().orc_set_threads(threads)

    def run():
        orc.load().orc_q3_build_order_bits(orc._p(oc), orc._p(od_), n_orders,
                                           orc._p(cbits), Q3_CUTOFF, orc._p(obits))
        orc.load().orc_q3_probe_agg(orc._p(lk), orc._p(ext), orc._p(disc), orc._p(ship),
                                    sample, orc._p(obits), Q3_CUTOFF, orc._p(ok),
                                    orc._p(os_), sample)

    run()  # warmup (untimed)
    t0 = time.perf_counter()
    passes = 0
    while time.perf_counter() - t0 < 10.0:
        run()
        passes += 1
    dt = time.perf_counter() - t0
    rate = passes * sample / dt
    gbps = rate * BYTES_PER_ROW["q3"] / 1e9
    return {"value": round(rate, 1), "unit": "rows/s", "cores": threads, "kind": "port",
            "sample": f"{passes} passes over {sample} rows, dims scaled to "
                      f"{n_orders}/{n_custs} ({dt:.1f}s, OMP partition-scatter oracle, "
                      f"{threads} threads of sched_getaffinity={cores}, "
                      f"cpu_count={os.cpu_count()} — large teams collapse under "
                      f"the box's CPU quota, profiles/r02_cpu_diag.txt; achieved "
                      f"{gbps:.1f} GB/s algorithmic; generation untimed)"}


def read_pmc_traffic(workload, rows, world):
    """Committed PMC-counted bytes per launch — only valid for the standard
    N=1 config the counters were collected on."""
    if world != 1 or rows != {"q1": SF10_ROWS, "q21": SF100_ROWS,
                              "q43": SF100_ROWS, "q3": Q3_ROWS_PER_GPU}[workload]:
        return None
    p = os.path.join(REPO, "profiles", f"pmc_{workload}.json")
    if os.path.exists(p):
        with open(p) as f:
            return json.load(f).get("traffic_bytes_per_launch")
    return None


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=50)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--workload", choices=["q1", "q21", "q43", "q3"], default="q21")
    ap.add_argument("--rows", type=int, default=0, help="override rows per GPU")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", args.gpus))
    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    wl = args.workload

    from starrocks_amd.engine import Engine

    dist = None
    torch = None
    # GPUE_DIST_BACKEND=gloo lets the N>1 code path be smoke-tested with two
    # ranks sharing one GPU (gloo supports cuda-tensor all_reduce; the real
    # runs use RCCL). Device = local_rank modulo visible devices — identical
    # to local_rank on a full node.
    backend = os.environ.get("GPUE_DIST_BACKEND", "nccl")
    if world > 1:
        # torch's HIP context must initialize BEFORE the ctypes engine loads
        # the HIP runtime (the reverse order leaves torch.cuda device-less
        # under torchrun)
        import torch as _torch
        import torch.distributed as tdist
        torch = _torch
        dev = local_rank % max(torch.cuda.device_count(), 1)
        torch.cuda.set_device(dev)
        tdist.init_process_group(backend)
        dist = tdist
    else:
        dev = local_rank % max(Engine.device_count(), 1)

    eng = Engine(dev)

    def a2a(out_t, in_t, out_list=None, in_list=None):
        """all_to_all_single, with a CPU object-gather route under the gloo
        smoke backend (gloo has no all_to_all; correctness-only, tiny rows)."""
        if backend != "gloo":
            if out_list is None:
                dist.all_to_all_single(out_t, in_t)
            else:
                dist.all_to_all_single(out_t, in_t, out_list, in_list)
            return
        il = in_list or [len(in_t) // world] * world
        chunks = [c.clone() for c in torch.split(in_t.cpu(), il)]
        gathered = [None] * world
        dist.all_gather_object(gathered, chunks)
        parts = [gathered[r][rank] for r in range(world)]
        out_t.copy_(torch.cat(parts).to(out_t.device))

    rows = args.rows or {"q1": SF10_ROWS, "q21": SF100_ROWS, "q43": SF100_ROWS,
                         "q3": Q3_ROWS_PER_GPU}[wl]
    row_start = rank * rows  # weak scaling: each rank owns its shard

    log(f"[bench] workload={wl} rows/gpu={rows} world={world} "
        f"steps={args.steps} warmup={args.warmup}")
    t_setup = time.perf_counter()
    tables = build_dim_tables(eng, wl, rank, world) if wl != "q3" else None
    n_merge = {"q1": 2, "q21": 7000, "q43": 800, "q3": 0}[wl]

    # ---- untimed setup: generate shard on device, build step closure ----
    # At N>1 the accumulator is torch-backed (wrap_ptr) so the phase1→phase2
    # merge is ONE RCCL all-reduce on the kernel's own output buffer — no
    # host round trip.
    def make_acc(n_items):
        if world > 1:
            t = torch.zeros(n_items, dtype=torch.int64, device="cuda")
            return eng.wrap_ptr(t.data_ptr(), n_items * 8), t
        return eng.alloc(n_items * 8), None

    if wl == "q1":
        cols = [eng.alloc(rows * 4) for _ in range(3)]
        eng.gen_lineorder_q1(SEED, row_start, rows, *cols)
        acc, acc_t = make_acc(2)

        if world == 1 and os.environ.get("GPUE_Q1_NO_ACCUM") != "1":
            # accumulating step (gpue_q1_join_sum_accum): the persistent acc
            # is never reset — each step is one kernel launch, the per-step
            # value is the host-side difference of successive readbacks
            acc.h2d(np.zeros(2, np.int64))
            prev = np.zeros(2, np.int64)

            def kernel_only():
                eng.q1_join_sum_accum(tables["dates"], cols[0], cols[1], cols[2], rows, acc)

            def step():
                nonlocal prev
                kernel_only()
                cur = acc.d2h(np.int64, 2)
                out = cur - prev
                prev = cur
                return out
        else:
            def kernel_only():
                eng.q1_join_sum_async(tables["dates"], cols[0], cols[1], cols[2], rows, acc)

            def step():
                kernel_only()
                return acc.d2h(np.int64, 2)
    elif wl == "q21":
        cols = [eng.alloc(rows * 4) for _ in range(4)]
        eng.gen_lineorder_q21(SEED, row_start, rows, *cols)
        acc, acc_t = make_acc(7000)

        def kernel_only():
            eng.q21_star_agg_async(tables["parts"], tables["supps"], tables["dates"],
                                   cols[0], cols[1], cols[2], cols[3], rows, acc)

        def step():
            kernel_only()
            return acc.d2h(np.int64, 7000)
    elif wl == "q3":
        from starrocks_amd import gen
        lit = gen.mkt_literal(1)
        mkt = eng.alloc(Q3_N_CUSTS * 16)
        eng.gen_cust_mkt16(SEED, Q3_N_CUSTS, mkt)
        cbits = eng.alloc((Q3_N_CUSTS + 31) // 32 * 4)
        eng.bits_str16_eq(mkt, Q3_N_CUSTS, lit, cbits)
        mkt.free()
        oc, od_ = eng.alloc(Q3_N_ORDERS * 4), eng.alloc(Q3_N_ORDERS * 4)
        eng.gen_orders_q3(SEED, Q3_N_ORDERS, Q3_N_CUSTS, oc, od_)
        obits = eng.alloc((Q3_N_ORDERS + 31) // 32 * 4)
        if world == 1:
            lk, ext, disc = (eng.alloc(rows * 8) for _ in range(3))
            ship = eng.alloc(rows * 4)
            eng.gen_lineitem_q3(SEED, row_start, rows, Q3_N_ORDERS, lk, ext, disc, ship)
        max_out = 64_000_000
        ok_b, os_b = eng.alloc(max_out * 8), eng.alloc(max_out * 8)
        agg_tab = eng.agg_table_create(64_000_000)  # persistent; reset per pass

        nparts = int(os.environ.get("GPUE_Q3_PART", "0"))
        if nparts and world == 1:
            # range-partitioned probe (DESIGN.md §4b L2-locality lever):
            # pass A partitions ship-passing rows by orderkey range, pass B
            # probes one L2-resident bitset window per partition
            ks_scr, vs_scr = eng.alloc(rows * 4), eng.alloc(rows * 8)

            def kernel_only():
                return eng.q3_probe_agg_part(lk, ext, disc, ship, rows, Q3_N_ORDERS,
                                             obits, Q3_CUTOFF, agg_tab, ks_scr, vs_scr,
                                             nparts, ok_b, os_b, max_out)
        else:
            def kernel_only():
                # fused lineitem filter + orders semi-probe + hash aggregate
                # (per pass: agg-table reset + probe + emit — the whole
                # probe-side query pass; the table itself persists like the
                # CPU's reused chunk allocations)
                return eng.q3_probe_agg_t(lk, ext, disc, ship, rows, obits, Q3_CUTOFF,
                                          agg_tab, ok_b, os_b, max_out)

        if world == 1:
            def step():
                # orders build pass (the hash-join BUILD phase: scan orders,
                # apply date + customer filters) + probe
                eng.q3_order_bits(oc, od_, Q3_N_ORDERS, cbits, Q3_CUTOFF, obits)
                g = kernel_only()
                return np.array([g], np.int64)
        else:
            # hash-partitioned mode on l_orderkey (config 5
            # "genuinely requires partitioning"): after the all-to-all every
            # rank's received rows carry orderkeys the partition function
            # routes to it, so the HIGH-CARDINALITY group space is DISJOINT
            # across ranks — local aggregation is final, no merge.
            # Columns move between gpue kernels and RCCL in the same torch
            # tensors (wrap_ptr).
            # CHUNKED EXCHANGE/COMPUTE OVERLAP (SURVEY.md §7 hard part (d),
            # VERDICT r01 next #5): the shard is split into S row-blocks;
            # block b's RCCL all-to-all (comm stream) runs while the engine
            # stream partitions/gathers block b+1 and probes block b-1 into
            # the persistent agg table — the analog of the reference's
            # overlapped SinkBuffer (sink_buffer.cpp:533-536). Ordering is by
            # CUDA events between the comm stream and the engine's own HIP
            # stream (wrapped as a torch ExternalStream); no device-wide
            # syncs inside the block loop.
            S = max(1, int(os.environ.get("GPUE_A2A_SLICES", "4")))
            blk = [(b * rows // S, (b + 1) * rows // S) for b in range(S)]
            widths = [8, 8, 8, 4]
            dtypes = [torch.int64, torch.int64, torch.int64, torch.int32]
            cols_t = [torch.empty(rows, dtype=dt, device="cuda") for dt in dtypes]
            lk, ext, disc, ship = (eng.wrap_ptr(t.data_ptr(), rows * w)
                                   for t, w in zip(cols_t, widths))
            eng.gen_lineitem_q3(SEED, row_start, rows, Q3_N_ORDERS, lk, ext, disc, ship)
            eng.sync()
            ridx = eng.alloc(rows * 4)  # per-block scratch (blocks run in order)
            col_views, send_t, send, recv_t, recv = [], [], [], [], []
            in_lists, out_lists, n_recvs = [], [], []
            for lo, hi in blk:
                nb = hi - lo
                col_views.append([eng.wrap_ptr_offset(c, lo * w, nb * w)
                                  for c, w in zip((lk, ext, disc, ship), widths)])
                st = [torch.empty(nb, dtype=dt, device="cuda") for dt in dtypes]
                send_t.append(st)
                send.append([eng.wrap_ptr(t.data_ptr(), nb * w)
                             for t, w in zip(st, widths)])
                # block splits are static (static data): discover once, untimed
                sp = eng.partition_i64(col_views[-1][0], nb, world, ridx)
                ins = np.diff(sp).astype(np.int64)
                out_sz = torch.empty(world, dtype=torch.int64, device="cuda")
                a2a(out_sz, torch.from_numpy(ins).cuda())
                outs = out_sz.cpu().numpy()
                nr = max(int(outs.sum()), 1)
                rt = [torch.empty(nr, dtype=dt, device="cuda") for dt in dtypes]
                recv_t.append(rt)
                recv.append([eng.wrap_ptr(t.data_ptr(), nr * w)
                             for t, w in zip(rt, widths)])
                in_lists.append([int(x) for x in ins])
                out_lists.append([int(x) for x in outs])
                n_recvs.append(int(outs.sum()))
            ext_ts = torch.cuda.ExternalStream(eng.stream_ptr())
            comm_s = torch.cuda.Stream()
            ev_g = [torch.cuda.Event() for _ in range(S)]
            ev_a = [torch.cuda.Event() for _ in range(S)]

            part_scr = eng.alloc(eng.partition_scratch_bytes(rows, world))

            def _gather(b):
                lo, hi = blk[b]
                nb = hi - lo
                eng.partition_i64_async(col_views[b][0], nb, world, ridx, part_scr)
                for j, (c, s_) in enumerate(zip(col_views[b], send[b])):
                    if j < 3:
                        eng.gather_u64(c, ridx, nb, s_)
                    else:
                        eng.gather_u32(c, ridx, nb, s_)

            def _exchange(b):
                ev_g[b].record(ext_ts)
                with torch.cuda.stream(comm_s):
                    comm_s.wait_event(ev_g[b])
                    for st, rt in zip(send_t[b], recv_t[b]):
                        a2a(rt[:max(n_recvs[b], 1)], st, out_lists[b], in_lists[b])
                    ev_a[b].record(comm_s)

            def _probe(b):
                if n_recvs[b] == 0:
                    return
                ext_ts.wait_event(ev_a[b])
                eng.q3_probe_accum(recv[b][0], recv[b][1], recv[b][2], recv[b][3],
                                   n_recvs[b], obits, Q3_CUTOFF, agg_tab)

            n_recv = sum(n_recvs)  # total received rows (roofline denominator)

            def kernel_only():
                # roofline instrumentation: the compute leg only (probe of
                # every received block into the persistent table)
                for b in range(S):
                    if n_recvs[b]:
                        eng.q3_probe_accum(recv[b][0], recv[b][1], recv[b][2],
                                           recv[b][3], n_recvs[b], obits, Q3_CUTOFF,
                                           agg_tab)

            def step():
                # orders build pass + table reset run on the engine stream
                # while block 0's exchange is prepared
                eng.agg_table_reset(agg_tab)
                eng.q3_order_bits(oc, od_, Q3_N_ORDERS, cbits, Q3_CUTOFF, obits)
                for b in range(S):
                    _gather(b)
                    _exchange(b)
                    if b > 0:
                        _probe(b - 1)
                _probe(S - 1)
                g = eng.hash_agg_emit(agg_tab, ok_b, os_b, max_out)
                return np.array([g], np.int64)
    else:  # q43
        acc, acc_t = make_acc(800)
        if world == 1:
            cols = [eng.alloc(rows * 4) for _ in range(6)]
            eng.gen_lineorder_q43(SEED, row_start, rows, *cols)

            def kernel_only():
                eng.q43_star_agg_async(tables["custs"], tables["supps"], tables["parts"],
                                       tables["dates"], *cols, rows, acc)

            def step():
                kernel_only()
                return acc.d2h(np.int64, 800)
        else:
            # hash-partitioned mode with CHUNKED EXCHANGE/COMPUTE OVERLAP —
            # same pipeline shape as the q3 branch above (S row-blocks,
            # events between the comm stream and the engine's HIP stream;
            # sink_buffer.cpp:533-536 analog). The probe accumulates per
            # received block (gpue_q43_star_agg_accum_async); acc zeroed once
            # per step on the engine stream.
            S = max(1, int(os.environ.get("GPUE_A2A_SLICES", "4")))
            blk = [(b * rows // S, (b + 1) * rows // S) for b in range(S)]
            cols_t = [torch.empty(rows, dtype=torch.int32, device="cuda")
                      for _ in range(6)]
            cols = [eng.wrap_ptr(t.data_ptr(), rows * 4) for t in cols_t]
            eng.gen_lineorder_q43(SEED, row_start, rows, *cols)
            eng.sync()
            ridx = eng.alloc(rows * 4)
            col_views, send_t, send, recv_t, recv = [], [], [], [], []
            in_lists, out_lists, n_recvs = [], [], []
            for lo, hi in blk:
                nb = hi - lo
                col_views.append([eng.wrap_ptr_offset(c, lo * 4, nb * 4) for c in cols])
                st = [torch.empty(nb, dtype=torch.int32, device="cuda") for _ in range(6)]
                send_t.append(st)
                send.append([eng.wrap_ptr(t.data_ptr(), nb * 4) for t in st])
                # block splits are static (static data): discover once, untimed
                sp = eng.partition(col_views[-1][0], nb, world, ridx)
                ins = np.diff(sp).astype(np.int64)
                out_sz = torch.empty(world, dtype=torch.int64, device="cuda")
                a2a(out_sz, torch.from_numpy(ins).cuda())
                outs = out_sz.cpu().numpy()
                nr = max(int(outs.sum()), 1)
                rt = [torch.empty(nr, dtype=torch.int32, device="cuda") for _ in range(6)]
                recv_t.append(rt)
                recv.append([eng.wrap_ptr(t.data_ptr(), nr * 4) for t in rt])
                in_lists.append([int(x) for x in ins])
                out_lists.append([int(x) for x in outs])
                n_recvs.append(int(outs.sum()))
            ext_ts = torch.cuda.ExternalStream(eng.stream_ptr())
            comm_s = torch.cuda.Stream()
            ev_g = [torch.cuda.Event() for _ in range(S)]
            ev_a = [torch.cuda.Event() for _ in range(S)]

            part_scr = eng.alloc(eng.partition_scratch_bytes(rows, world))

            def _gather(b):
                lo, hi = blk[b]
                nb = hi - lo
                # partition (fnv->channel + counting sort) + gather: the
                # exchange sink stage (exchange_sink_operator.cpp:611-660).
                # Async form: splits are static (discovered at setup), so the
                # timed step never host-syncs here — the comm stream's
                # all-to-all of block b-1 keeps flowing
                eng.partition_async(col_views[b][0], nb, world, ridx, part_scr)
                for c, s_ in zip(col_views[b], send[b]):
                    eng.gather_u32(c, ridx, nb, s_)

            def _exchange(b):
                # the brpc transmit_chunk leg -> RCCL all-to-all over xGMI
                ev_g[b].record(ext_ts)
                with torch.cuda.stream(comm_s):
                    comm_s.wait_event(ev_g[b])
                    for st, rt in zip(send_t[b], recv_t[b]):
                        a2a(rt[:max(n_recvs[b], 1)], st, out_lists[b], in_lists[b])
                    ev_a[b].record(comm_s)

            def _probe(b):
                if n_recvs[b] == 0:
                    return
                ext_ts.wait_event(ev_a[b])
                eng.q43_star_agg_accum_async(tables["custs"], tables["supps"],
                                             tables["parts"], tables["dates"],
                                             *recv[b], n_recvs[b], acc)

            n_recv = sum(n_recvs)  # total received rows (roofline denominator)

            def step_pipeline():
                with torch.cuda.stream(ext_ts):
                    acc_t.zero_()
                for b in range(S):
                    _gather(b)
                    _exchange(b)
                    if b > 0:
                        _probe(b - 1)
                _probe(S - 1)

            def kernel_only():
                # roofline instrumentation: the compute leg only (probe of
                # every received block)
                for b in range(S):
                    if n_recvs[b]:
                        eng.q43_star_agg_accum_async(tables["custs"], tables["supps"],
                                                     tables["parts"], tables["dates"],
                                                     *recv[b], n_recvs[b], acc)

            def step():
                step_pipeline()
                eng.sync()
                return acc.d2h(np.int64, 800)

    eng.sync()
    log(f"[bench] setup {time.perf_counter()-t_setup:.1f}s")

    # hipGraph step replay (include/gpue.h graph API) — measured NEGATIVE for
    # these 2-node steps (memset+kernel): 0.160 vs 0.154 ms/step on the SF10
    # line; hipGraphLaunch costs more than the two direct async launches it
    # replaces. Kept opt-in (GPUE_GRAPH=1) for multi-kernel
    # steps where the node count amortizes it. q3 is excluded: its step does
    # a device->host count readback inside q3_probe_agg_t.
    if wl != "q3" and os.environ.get("GPUE_GRAPH") == "1":
        _graph = eng.graph_capture(kernel_only)

        def kernel_only():
            eng.graph_launch(_graph)

    def run_step():
        if dist is not None:
            if wl == "q3":
                # partitioned high-cardinality agg: group spaces are disjoint
                # across ranks — local aggregation IS the final result
                return step()
            if wl == "q43":
                step_pipeline()  # chunked exchange/compute overlap
            else:
                kernel_only()
            eng.sync()
            # phase1 -> phase2 aggregate merge (agg_hash_variant.h merge_batch
            # semantics): one RCCL all-reduce on the kernel's output buffer
            dist.all_reduce(acc_t)
            return acc_t.cpu().numpy()  # per-step result read
        return step()

    for _ in range(args.warmup):
        run_step()

    # ---- timed region ----
    if dist is not None:
        dist.barrier()
        torch.cuda.synchronize()
    eng.sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        run_step()
    eng.sync()
    if dist is not None:
        torch.cuda.synchronize()
        dist.barrier()
    elapsed = time.perf_counter() - t0

    if dist is not None:
        t = torch.tensor([elapsed], device="cuda")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    value = rows * world * args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1000

    # ---- roofline evidence: HIP events around kernel-only launches ----
    R = 20
    eng.sync()
    eng.timer_start()
    for _ in range(R):
        kernel_only()
    kernel_ms = eng.timer_stop() / R
    kernel_rows = rows if not (wl in ("q43", "q3") and world > 1) else max(n_recv, 1)
    algo_bytes = BYTES_PER_ROW[wl] * kernel_rows
    achieved_gbps = algo_bytes / (kernel_ms / 1e3) / 1e9
    traffic = read_pmc_traffic(wl, rows, world)
    # Honest presentation: `achieved`/`frac` are
    # kernel-only HIP-event timing of ALGORITHMIC bytes; `achieved_step`/
    # `frac_step` divide the same bytes by the full step (incl. result read);
    # `frac_physical` uses the PMC-counted bytes actually moved (q43's
    # deferred loads fetch FEWER than algorithmic; q3 fetches more).
    step_gbps = BYTES_PER_ROW[wl] * rows / (ms_per_step / 1e3) / 1e9
    roofline = {"bound": "hbm", "achieved": round(achieved_gbps, 1), "peak": HBM_PEAK_GBPS,
                "unit": "GB/s", "frac": round(achieved_gbps / HBM_PEAK_GBPS, 4),
                "achieved_step": round(step_gbps, 1),
                "frac_step": round(step_gbps / HBM_PEAK_GBPS, 4),
                "frac_physical": (round(traffic / (kernel_ms / 1e3) / 1e9
                                        / HBM_PEAK_GBPS, 4) if traffic else None),
                "traffic": traffic}

    if rank == 0:
        # GPUE_SKIP_CPU_BASELINE=1: omit the ~10 s oracle leg during kernel
        # sweeps; the driver's default run always measures it
        if os.environ.get("GPUE_SKIP_CPU_BASELINE") == "1":
            cb = None
        else:
            cb = cpu_baseline(wl, rows) if world == 1 and wl != "q3" else (
                cpu_baseline_q3(rows) if world == 1 else None)
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
                "workload": {"q1": "ssb_sf10_q1_join_sum",
                             "q21": "ssb_sf100_q21_star_groupby",
                             "q43": "ssb_sf100_q43_4way_star_groupby",
                             "q3": "tpch_sf300_q3_join_highcard_agg"}[wl],
                "rows_per_gpu": rows,
                "seed": SEED,
                "parallelism": (f"dp{world}-hash-partitioned-alltoall"
                                if wl in ("q43", "q3") and world > 1
                                else f"dp{world}-replicated-dims"),
            },
            "roofline": roofline,
            "cpu_baseline": cb,
        }
        print(json.dumps(result), flush=True)

    eng.close()
    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
