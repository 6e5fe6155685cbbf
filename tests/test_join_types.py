"""Per-join-type probe semantics (reference join_hash_map.h:228-333) —
oracle vs brute force on CPU; GPU parity in test_gpu_parity."""

import numpy as np

from oracle import pyoracle as orc


def setup_tables(seed=3, nbuild=500, nprobe=800, keyspace=300):
    rng = np.random.default_rng(seed)
    build_keys = np.concatenate([[0], rng.integers(0, keyspace, nbuild)]).astype(np.uint32)
    probe_keys = rng.integers(0, keyspace + 100, nprobe).astype(np.uint32)
    first, nxt, bs, log = orc.bucket_chained_build(build_keys)
    heads = orc.bucket_chained_lookup(probe_keys, first, bs, log)
    return build_keys, probe_keys, nxt, heads


def brute(build_keys, probe_keys, mode):
    out = []
    for i, k in enumerate(probe_keys):
        matches = [j for j in range(1, len(build_keys)) if build_keys[j] == k]
        if mode == 0:
            out += [(i, j) for j in matches]
        elif mode == 1 and matches:
            out.append((i, "any"))
        elif mode == 2 and not matches:
            out.append((i, 0))
        elif mode == 3:
            out += [(i, j) for j in matches] if matches else [(i, 0)]
    return out


def test_left_modes_vs_brute():
    bk, pk, nxt, heads = setup_tables()
    for mode in (0, 1, 2, 3):
        op, ob = orc.probe_emit_mode(bk, nxt, pk, heads, mode)
        exp = brute(bk, pk, mode)
        if mode == 1:
            # SEMI: the emitted build row is chain-order-dependent; the probe
            # row set and key-match property are the contract
            assert sorted(op.tolist()) == sorted(i for i, _ in exp)
            assert (bk[ob] == pk[op]).all()
        else:
            got = sorted(zip(op.tolist(), ob.tolist()))
            assert got == sorted(exp), mode


def test_right_semi_anti_vs_brute():
    bk, pk, nxt, heads = setup_tables(seed=9)
    probe_set = set(pk.tolist())
    for anti in (0, 1):
        got = sorted(orc.probe_right(bk, nxt, pk, heads, anti).tolist())
        exp = sorted(j for j in range(1, len(bk))
                     if (bk[j] in probe_set) != bool(anti))
        assert got == exp
