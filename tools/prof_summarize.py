"""Summarize rocprofv3 result databases (gpurun_out/*) into committed
artifacts under profiles/.

Usage:
  python tools/prof_summarize.py <results.db> --out profiles/<name>.txt
  python tools/prof_summarize.py <pmc.db> --pmc FETCH_SIZE --kernel k_q1_join_sum \
      --json profiles/pmc_q1.json [--correction 2.0] [--streaming-bytes N]

PMC notes (MI355X_MICROARCH.md §HBM): FETCH_SIZE derives from TCC_EA0_RDREQ
x 64 B and on gfx950 reports HALF the bytes of a wide (16 B/lane) coalesced
streaming read — hence --correction 2.0 for pure-streaming kernels. Counter
value unit observed: KB. For mixed streaming+gather kernels pass
--streaming-bytes (the algorithmic streamed bytes): traffic is then estimated
as streaming_bytes + (counted*1024 - streaming_bytes/2), i.e. the gather
component counted 1:1 plus the corrected streaming component.
"""

import argparse
import json
import re
import sqlite3


def prefix(con):
    for (name,) in con.execute("SELECT name FROM sqlite_master WHERE type='table'"):
        m = re.match(r"rocpd_kernel_dispatch_(.*)", name)
        if m:
            return m.group(1)
    raise SystemExit("no kernel dispatch table found")


def kernel_rows(con, pre):
    q = f"""
    SELECT ks.display_name, COUNT(*), AVG(k.end-k.start)/1e6, SUM(k.end-k.start)/1e6,
           MIN(k.end-k.start)/1e6, MAX(k.end-k.start)/1e6
    FROM rocpd_kernel_dispatch_{pre} k
    JOIN rocpd_info_kernel_symbol_{pre} ks ON k.kernel_id = ks.id
    GROUP BY ks.display_name ORDER BY 4 DESC"""
    return list(con.execute(q))


def pmc_rows(con, pre):
    q = f"""
    SELECT ks.display_name, COUNT(*), AVG(p.value)
    FROM rocpd_pmc_event_{pre} p
    JOIN rocpd_kernel_dispatch_{pre} k ON p.event_id = k.event_id
    JOIN rocpd_info_kernel_symbol_{pre} ks ON k.kernel_id = ks.id
    GROUP BY ks.display_name ORDER BY 3 DESC"""
    return list(con.execute(q))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("db")
    ap.add_argument("--out")
    ap.add_argument("--pmc")
    ap.add_argument("--kernel")
    ap.add_argument("--json")
    ap.add_argument("--correction", type=float, default=1.0)
    ap.add_argument("--streaming-bytes", type=float, default=None)
    args = ap.parse_args()

    con = sqlite3.connect(args.db)
    pre = prefix(con)
    lines = [f"# rocprofv3 summary of {args.db}", "# kernel, n, avg_ms, total_ms, min_ms, max_ms"]
    for name, n, avg, tot, mn, mx in kernel_rows(con, pre):
        lines.append(f"{name}, {n}, {avg:.4f}, {tot:.2f}, {mn:.4f}, {mx:.4f}")
    if args.pmc:
        lines.append(f"# PMC {args.pmc} (unit: KB as observed; see MI355X_MICROARCH.md corrections)")
        for name, n, avg in pmc_rows(con, pre):
            lines.append(f"PMC {args.pmc}: {name}, n={n}, avg={avg:.0f}")
    text = "\n".join(lines) + "\n"
    if args.out:
        with open(args.out, "w") as f:
            f.write(text)
        print("wrote", args.out)
    else:
        print(text)

    if args.json and args.kernel and args.pmc:
        val = None
        for name, n, avg in pmc_rows(con, pre):
            if args.kernel in name:
                val = avg
                break
        assert val is not None, f"kernel {args.kernel} not in PMC rows"
        counted = val * 1024.0
        if args.streaming_bytes is not None:
            traffic = args.streaming_bytes + max(0.0, counted - args.streaming_bytes / 2.0)
            method = ("streaming_bytes + (counted - streaming_bytes/2): gather component "
                      "counted 1:1, 16B/lane streaming component halved per gfx950 FETCH_SIZE")
        else:
            traffic = counted * args.correction
            method = f"counted KB*1024*{args.correction} (gfx950 wide-read halving correction)"
        out = {"kernel": args.kernel, "pmc": args.pmc, "counted_kb_avg": val,
               "traffic_bytes_per_launch": traffic, "method": method, "source_db": args.db}
        with open(args.json, "w") as f:
            json.dump(out, f, indent=1)
        print("wrote", args.json, "traffic", traffic / 1e9, "GB/launch")


if __name__ == "__main__":
    main()
