"""Multi-rank merge semantics on CPU (gloo, world_size 2) — DESIGN.md §6.

Configs 2-3 shard lineorder rows across ranks with replicated dim tables and
merge partial aggregates with one all-reduce (phase1→phase2 merge semantics,
reference agg_hash_variant.h phase2 + merge_batch): the sharded+merged result
must equal the single-rank result bit-exactly.
"""

import multiprocessing as mp
import os

import numpy as np


def _rank_main(rank, world, q, seed, n_rows, year):
    import torch
    import torch.distributed as dist

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29471")
    dist.init_process_group("gloo", rank=rank, world_size=world)

    import sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    from oracle import pyoracle as orc

    # weak-scaling shard: rank r owns rows [r*per, (r+1)*per)
    per = n_rows // world
    start = rank * per
    n = per if rank < world - 1 else n_rows - start
    s, cnt = orc.q1_pipeline(seed, start, n, year)
    t = torch.tensor([s, cnt], dtype=torch.int64)
    dist.all_reduce(t)
    if rank == 0:
        q.put((int(t[0]), int(t[1])))
    dist.destroy_process_group()


def test_sharded_q1_equals_single_rank():
    seed, n_rows, year = 42, 400_000, 1993
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_rank_main, args=(r, 2, q, seed, n_rows, year))
             for r in range(2)]
    for p in procs:
        p.start()
    merged = q.get(timeout=180)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    from oracle import pyoracle as orc
    s, cnt = orc.q1_pipeline(seed, 0, n_rows, year)
    assert merged == (s, cnt)


def test_sharded_q21_group_merge():
    """Same for the grouped aggregate, merged in-process (numpy) — the gloo
    transport is covered above; this pins the group-wise merge math."""
    import sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    from oracle import pyoracle as orc

    seed, n_rows, cat, reg = 42, 300_000, 12, 2
    whole = orc.q21_pipeline(seed, 0, n_rows, cat, reg)
    half0 = orc.q21_pipeline(seed, 0, n_rows // 2, cat, reg)
    half1 = orc.q21_pipeline(seed, n_rows // 2, n_rows - n_rows // 2, cat, reg)
    assert np.array_equal(half0 + half1, whole)
