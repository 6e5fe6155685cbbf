"""Per-join-type probe semantics (reference join_hash_map.h:228-333) —
oracle vs brute force on CPU; GPU parity in test_gpu_parity."""

import numpy as np
import pytest

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


@pytest.mark.gpu
def test_u128_key_join_parity(engine):
    """16-byte key joins (JoinKeyHash<T,16> = crc32 over the key bytes,
    join_hash_map_helper.h:23-30): the dedicated fixed-16 path must produce
    the same multisets as the varchar/Slice path over the identical 16-byte
    slices (which the oracle already pins), for every probe mode."""
    rng = np.random.default_rng(55)
    nbuild, nprobe = 20_000, 80_000
    pool = rng.integers(0, 2**63, (3000, 2), dtype=np.uint64)
    bkeys = np.zeros((nbuild + 1, 2), np.uint64)
    bkeys[1:] = pool[rng.integers(0, len(pool), nbuild)]
    pkeys = pool[rng.integers(0, len(pool), nprobe)].copy()
    pkeys[::7] = rng.integers(0, 2**63, (len(pkeys[::7]), 2), dtype=np.uint64)

    kb = engine.alloc(bkeys.nbytes)
    kb.h2d(bkeys)
    t128 = engine.join_build_bucket_chained_u128(kb, nbuild)
    pb = engine.alloc(pkeys.nbytes)
    pb.h2d(pkeys)

    # varchar reference: same bytes as fixed-16 slices
    boff = (np.arange(nbuild + 2, dtype=np.uint32) * 16)
    poff = (np.arange(nprobe + 1, dtype=np.uint32) * 16)
    bb = engine.alloc(bkeys.nbytes)
    bb.h2d(bkeys.view(np.uint8).reshape(-1))
    bo = engine.alloc(boff.nbytes)
    bo.h2d(boff)
    tvc = engine.join_build_varchar(bb, bo, nbuild)
    pvb = engine.alloc(pkeys.nbytes)
    pvb.h2d(pkeys.view(np.uint8).reshape(-1))
    po = engine.alloc(poff.nbytes)
    po.h2d(poff)

    for mode in (0, 1, 2, 3):
        c1 = engine.join_probe_emit_mode_u128(t128, pb, nprobe, mode)
        c2 = engine.join_probe_emit_varchar_mode(tvc, pvb, po, nprobe, mode)
        assert c1 == c2, mode
        o1p, o1b = engine.alloc(max(c1, 1) * 4), engine.alloc(max(c1, 1) * 4)
        o2p, o2b = engine.alloc(max(c2, 1) * 4), engine.alloc(max(c2, 1) * 4)
        engine.join_probe_emit_mode_u128(t128, pb, nprobe, mode, o1p, o1b)
        engine.join_probe_emit_varchar_mode(tvc, pvb, po, nprobe, mode, o2p, o2b)
        if mode == 1:
            assert np.array_equal(np.sort(o1p.d2h(np.uint32, c1)),
                                  np.sort(o2p.d2h(np.uint32, c2)))
        else:
            pk_ = lambda a, b: np.sort(a.astype(np.uint64) << np.uint64(32) | b)
            assert np.array_equal(pk_(o1p.d2h(np.uint32, c1), o1b.d2h(np.uint32, c1)),
                                  pk_(o2p.d2h(np.uint32, c2), o2b.d2h(np.uint32, c2))), mode
        for x in (o1p, o1b, o2p, o2b):
            x.free()
    t128.destroy()
    tvc.destroy()
    for x in (kb, pb, bb, bo, pvb, po):
        x.free()


@pytest.mark.gpu
def test_2xi64_packed_join_gpu(engine):
    """Multi-column (2 x i64) join keys through the 16-byte constructor:
    pack_keys_2xi64 + the u128 bucket-chained table vs a numpy pair-set
    reference — the SERIALIZED_FIXED_SIZE_LARGEINT end-to-end path."""
    rng = np.random.default_rng(56)
    nbuild, nprobe = 10_000, 50_000
    ba = np.concatenate([[0], rng.integers(0, 500, nbuild)]).astype(np.uint64)
    bb_ = np.concatenate([[0], rng.integers(0, 500, nbuild)]).astype(np.uint64)
    pa = rng.integers(0, 600, nprobe).astype(np.uint64)
    pb_ = rng.integers(0, 600, nprobe).astype(np.uint64)
    da, db_ = engine.alloc(ba.nbytes), engine.alloc(bb_.nbytes)
    da.h2d(ba)
    db_.h2d(bb_)
    bkeys = engine.alloc((nbuild + 1) * 16)
    engine.pack_keys_2xi64(da, db_, nbuild + 1, bkeys)
    t = engine.join_build_bucket_chained_u128(bkeys, nbuild)
    qa, qb = engine.alloc(pa.nbytes), engine.alloc(pb_.nbytes)
    qa.h2d(pa)
    qb.h2d(pb_)
    pkeys = engine.alloc(nprobe * 16)
    engine.pack_keys_2xi64(qa, qb, nprobe, pkeys)
    cnt = engine.join_probe_emit_mode_u128(t, pkeys, nprobe, 1)  # LEFT SEMI
    present = set(zip(ba[1:].tolist(), bb_[1:].tolist()))
    expect = sum((a, b) in present for a, b in zip(pa.tolist(), pb_.tolist()))
    assert cnt == expect
    t.destroy()
    for x in (da, db_, bkeys, qa, qb, pkeys):
        x.free()
