"""Storage ingress (§8f row 4): bitshuffle+LZ4 page decode — CPU roundtrip
properties here; GPU parity in test_gpu_parity."""

import numpy as np
import pytest

from oracle import pyoracle as orc

SEED = 42


def test_roundtrip_shapes():
    rng = np.random.default_rng(7)
    for vals in [rng.integers(0, 100, 8192).astype(np.int32),       # compressible
                 rng.integers(0, 2**31, 6152).astype(np.int32),     # incompressible
                 np.zeros(8, np.int32),
                 np.arange(4096, dtype=np.int32),
                 np.full(2048 * 5, -7, np.int32)]:
        page = orc.bshuf_lz4_encode_i32(vals)
        got = orc.bshuf_lz4_decode_i32(page, len(vals))
        assert np.array_equal(got, vals)


def test_compression_ratio_on_low_cardinality():
    rng = np.random.default_rng(8)
    vals = rng.integers(0, 16, 2048 * 64).astype(np.int32)  # 4-bit values
    page = orc.bshuf_lz4_encode_i32(vals)
    # bit-plane transpose concentrates the 28 zero planes -> LZ4 collapses them
    assert len(page) < len(vals) * 4 * 0.25


def _make_dict(strings):
    off = np.zeros(len(strings) + 1, np.uint32)
    np.cumsum([len(s) for s in strings], out=off[1:])
    return np.frombuffer(b"".join(strings), np.uint8).copy(), off


def test_dict_decode_oracle_vs_python():
    """binary_dict_page decode restatement: codes -> dict strings, exact
    BinaryColumn bytes+offsets."""
    rng = np.random.default_rng(47)
    words = [f"city{i}".encode() + b"z" * int(rng.integers(0, 6)) for i in range(64)]
    db, do = _make_dict(words)
    codes = rng.integers(0, 64, 5000).astype(np.int32)
    ob, oo = orc.dict_decode_binary(db, do, codes)
    expect = b"".join(words[c] for c in codes)
    assert ob.tobytes() == expect
    assert oo[-1] == len(expect)
    for i in (0, 1, 4999):
        assert ob[oo[i]:oo[i + 1]].tobytes() == words[codes[i]]


def test_rle_page_oracle_roundtrip_and_kats():
    """RLE page codec (rle_page.h + rle_encoding.h at bit_width 32): oracle
    encode->decode roundtrips on run-heavy / literal-heavy / mixed / edge
    patterns, plus hand-derived byte KATs from the published format
    (repeated := varint(count<<1) + LE value; literal := byte(groups<<1|1)
    + groups*8 LE u32; 4-byte LE num_elements page header)."""
    rng = np.random.default_rng(3)
    pats = [np.full(1000, 7, np.int32),
            rng.integers(-100, 100, 1000).astype(np.int32),
            np.repeat(rng.integers(0, 5, 50),
                      rng.integers(1, 40, 50)).astype(np.int32),
            np.arange(17, dtype=np.int32),
            np.array([3] * 8 + list(range(1, 9)) + [9] * 100, np.int32),
            np.array([42], np.int32),
            rng.integers(0, 2**31, 600).astype(np.int32),  # >63 literal groups
            np.repeat(rng.integers(0, 3, 200),
                      rng.integers(1, 2000, 200)).astype(np.int32)]
    for pat in pats:
        page = orc.rle_page_encode_i32(pat)
        assert np.array_equal(orc.rle_page_decode_i32(page, len(pat)), pat)
    p = orc.rle_page_encode_i32(np.full(100, 7, np.int32))
    assert p.tobytes() == bytes([100, 0, 0, 0, 0xC8, 0x01, 7, 0, 0, 0])
    p = orc.rle_page_encode_i32(np.arange(1, 9, dtype=np.int32))
    assert p.tobytes() == bytes([8, 0, 0, 0, 0x03]) + b"".join(
        int(i).to_bytes(4, "little") for i in range(1, 9))


@pytest.mark.gpu
def test_rle_page_decode_gpu_parity(engine):
    """GPU two-phase RLE decode (run-table scan + binary-search fill) must
    reproduce the oracle decode bit-exactly, including the dict-code-shaped
    low-cardinality pattern the scan path would feed to a join."""
    rng = np.random.default_rng(8)
    pats = [np.repeat(rng.integers(0, 40, 3000),
                      rng.integers(1, 300, 3000)).astype(np.int32),
            rng.integers(-2**31, 2**31 - 1, 100_000).astype(np.int32),
            np.full(1_000_000, -5, np.int32),
            np.arange(23, dtype=np.int32)]
    for pat in pats:
        page = orc.rle_page_encode_i32(pat)
        pb = engine.alloc(max(page.nbytes, 4))
        pb.h2d(page)
        ob = engine.alloc(max(len(pat), 1) * 4)
        engine.page_decode_rle_i32(pb, len(pat), ob)
        got = ob.d2h(np.int32, len(pat))
        assert np.array_equal(got, pat), (len(pat), pat[:10])
        pb.free()
        ob.free()
    # corrupt page (truncated) must error, not hang
    from starrocks_amd.engine import GpueError
    page = orc.rle_page_encode_i32(np.full(1000, 3, np.int32))
    bad = page[:6].copy()
    bad[:4] = np.frombuffer(np.uint32(1000).tobytes(), np.uint8)
    pb = engine.alloc(max(bad.nbytes, 4))
    pb.h2d(bad)
    ob = engine.alloc(1000 * 4)
    with pytest.raises(GpueError):
        engine.page_decode_rle_i32(pb, 1000, ob)
    pb.free()
    ob.free()


def test_rle_bool_page_oracle_roundtrip():
    """BOOL RLE (bit_width 1, rle_page.h:82): bit-packed literal groups
    LSB-first, one-byte repeated values. Hand KAT: 100x true -> header +
    varint(200) + 0x01."""
    rng = np.random.default_rng(4)
    pats = [np.ones(100, np.uint8), np.zeros(77, np.uint8),
            (rng.random(1000) < 0.5).astype(np.uint8),
            np.repeat((rng.random(60) < 0.5).astype(np.uint8),
                      rng.integers(1, 50, 60)),
            np.array([1], np.uint8), np.array([1, 0, 1], np.uint8)]
    for pat in pats:
        page = orc.rle_page_encode_bool(pat)
        assert np.array_equal(orc.rle_page_decode_bool(page, len(pat)), pat)
    p = orc.rle_page_encode_bool(np.ones(100, np.uint8))
    assert p.tobytes() == bytes([100, 0, 0, 0, 0xC8, 0x01, 0x01])
    # 8 literals 1,0,1,1,0,0,1,0 -> indicator 0x03 + LSB-first byte 0x4D
    p = orc.rle_page_encode_bool(np.array([1, 0, 1, 1, 0, 0, 1, 0], np.uint8))
    assert p.tobytes() == bytes([8, 0, 0, 0, 0x03, 0x4D])


@pytest.mark.gpu
def test_rle_bool_page_decode_gpu_parity(engine):
    rng = np.random.default_rng(9)
    pats = [np.repeat((rng.random(500) < 0.5).astype(np.uint8),
                      rng.integers(1, 400, 500)),
            (rng.random(200_000) < 0.3).astype(np.uint8),
            np.ones(1_000_000, np.uint8)]
    for pat in pats:
        page = orc.rle_page_encode_bool(pat)
        pb = engine.alloc(max(page.nbytes, 4))
        pb.h2d(page)
        ob = engine.alloc(max(len(pat), 1))
        engine.page_decode_rle_bool(pb, len(pat), ob)
        assert np.array_equal(ob.d2h(np.uint8, len(pat)), pat)
        pb.free()
        ob.free()


def test_for_page_oracle_roundtrip_and_kat():
    """Frame-of-reference codec (FOR_ENCODING, frame_of_reference_coding):
    formats 0 (min+delta), 1 (ascending prefix deltas), 2 (raw on range
    overflow); MSB-first bit packing (hand KAT derived from the reference's
    own bit_pack example, coding.cpp:93-95)."""
    rng = np.random.default_rng(5)
    pats = [np.arange(1000, dtype=np.int32),
            rng.integers(1000, 2000, 1000).astype(np.int32),
            rng.integers(-2**31, 2**31 - 1, 1000).astype(np.int32),
            np.sort(rng.integers(0, 10**9, 777)).astype(np.int32),
            np.array([5], np.int32), np.full(300, 7, np.int32),
            np.concatenate([np.arange(128), rng.integers(0, 50, 128),
                            rng.integers(-2**31, 2**31 - 1, 130)]).astype(np.int32)]
    for pat in pats:
        page = orc.for_page_encode_i32(pat)
        assert np.array_equal(orc.for_page_decode_i32(page, len(pat)), pat)
    # 8,4,2,1: descending -> format 0, min 1, deltas 7,3,1,0 at bw 3
    # MSB-first: 111 011 001 000 -> 0xEC 0x80
    p = orc.for_page_encode_i32(np.array([8, 4, 2, 1], np.int32))
    assert p[:6].tobytes() == bytes([1, 0, 0, 0, 0xEC, 0x80])


@pytest.mark.gpu
def test_for_page_decode_gpu_parity(engine):
    rng = np.random.default_rng(6)
    pats = [np.sort(rng.integers(0, 10**9, 500_000)).astype(np.int32),
            rng.integers(-50, 50, 300_000).astype(np.int32),
            rng.integers(-2**31, 2**31 - 1, 100_000).astype(np.int32),
            np.arange(129, dtype=np.int32),
            np.array([42], np.int32)]
    for pat in pats:
        page = orc.for_page_encode_i32(pat)
        pb = engine.alloc(max(page.nbytes, 5))
        pb.h2d(page)
        ob = engine.alloc(max(len(pat), 1) * 4)
        engine.page_decode_for_i32(pb, len(pat), ob)
        assert np.array_equal(ob.d2h(np.int32, len(pat)), pat), len(pat)
        pb.free()
        ob.free()


def _bin_col(rng, n, card):
    pool = [f"seg_{i:03d}".encode() + b"z" * int(rng.integers(0, 7))
            for i in range(card)]
    rows = [pool[int(i)] for i in rng.integers(0, card, n)]
    offsets = np.zeros(n + 1, np.uint32)
    np.cumsum([len(r) for r in rows], out=offsets[1:])
    return np.frombuffer(b"".join(rows), np.uint8).copy(), offsets, rows


def test_binary_plain_page_oracle_roundtrip():
    """BinaryPlainPage (binary_plain_page.h:28-46): string body + u32
    absolute-offset trailer + count — the dict page's dictionary format."""
    rng = np.random.default_rng(7)
    b, o, rows = _bin_col(rng, 500, 40)
    page = orc.binary_plain_encode(b, o)
    db, do = orc.binary_plain_decode(page, 500)
    assert np.array_equal(db, b) and np.array_equal(do, o)
    # empty strings edge
    o2 = np.zeros(4, np.uint32)
    page2 = orc.binary_plain_encode(np.zeros(0, np.uint8), o2)
    db2, do2 = orc.binary_plain_decode(page2, 3)
    assert len(db2) == 0 and np.array_equal(do2, o2)


@pytest.mark.gpu
def test_dict_varchar_full_chain_gpu(engine):
    """The complete dict-encoded varchar ingress chain on device: the
    DICTIONARY ships as a BinaryPlainPage and the CODES as an RLE page
    (binary_dict_page.cpp); decode both on GPU, materialize the
    BinaryColumn (gpue_dict_decode_binary), and probe a varchar join —
    all compared against the oracle restatements."""
    rng = np.random.default_rng(13)
    card, n = 60, 200_000
    # a DISTINCT dictionary: one row per pool entry (unique seg_NNN prefixes)
    pool_rows = [f"seg_{i:03d}".encode() + b"z" * int(rng.integers(0, 7))
                 for i in range(card)]
    dict_o = np.zeros(card + 1, np.uint32)
    np.cumsum([len(r) for r in pool_rows], out=dict_o[1:])
    dict_b = np.frombuffer(b"".join(pool_rows), np.uint8).copy()
    codes = rng.integers(0, card, n).astype(np.int32)
    dict_page = orc.binary_plain_encode(dict_b, dict_o)
    codes_page = orc.rle_page_encode_i32(codes)

    dp = engine.alloc(dict_page.nbytes)
    dp.h2d(dict_page)
    db = engine.alloc(max(int(dict_o[-1]), 1))
    do = engine.alloc((card + 1) * 4)
    engine.page_decode_binary_plain(dp, card, db, do)
    cp = engine.alloc(codes_page.nbytes)
    cp.h2d(codes_page)
    cb = engine.alloc(n * 4)
    engine.page_decode_rle_i32(cp, n, cb)
    # materialize the BinaryColumn from dict + codes
    total = int(np.array([len(r) for r in pool_rows])[codes].sum())
    ob = engine.alloc(max(total, 1))
    oo = engine.alloc((n + 1) * 4)
    engine.dict_decode_binary(db, do, cb, n, ob, oo)
    # oracle materialization
    exp_rows = [pool_rows[c] for c in codes.tolist()]
    exp_off = np.zeros(n + 1, np.uint32)
    np.cumsum([len(r) for r in exp_rows], out=exp_off[1:])
    exp_bytes = np.frombuffer(b"".join(exp_rows), np.uint8)
    assert np.array_equal(oo.d2h(np.uint32, n + 1), exp_off)
    assert np.array_equal(ob.d2h(np.uint8, total), exp_bytes)
    # probe a varchar join against a build side of the pool (1-based)
    build_rows = [b""] + pool_rows[: card // 2]
    boff = np.zeros(len(build_rows) + 1, np.uint32)
    np.cumsum([len(r) for r in build_rows], out=boff[1:])
    bbytes = np.frombuffer(b"".join(build_rows), np.uint8).copy()
    hb = engine.alloc(max(bbytes.nbytes, 1))
    hb.h2d(bbytes)
    ho = engine.alloc(boff.nbytes)
    ho.h2d(boff.astype(np.uint32))
    t = engine.join_build_varchar(hb, ho, len(build_rows) - 1)
    cnt = engine.join_probe_emit_varchar_mode(t, ob, oo, n, 1)  # LEFT_SEMI
    expect = int(np.isin(codes, np.arange(card // 2)).sum())
    assert cnt == expect
    t.destroy()
    for x in (dp, db, do, cp, cb, ob, oo, hb, ho):
        x.free()


def test_lz4_block_vs_pyarrow():
    """Pin the LZ4 BLOCK layer of the bitshuffle+LZ4 page against a
    PUBLISHED implementation (pyarrow's bundled lz4_raw codec) in both
    directions — closing the compression half of the format's
    byte-compatibility gap (the bitshuffle bit-transpose half remains a
    spec restatement; the library is absent offline)."""
    pa = pytest.importorskip("pyarrow")
    import ctypes
    lib = orc.load()
    lib.orc_lz4_compress_block.restype = np.uint64 and ctypes.c_uint64
    lib.orc_lz4_compress_block.argtypes = [ctypes.c_void_p, ctypes.c_uint64,
                                           ctypes.c_void_p]
    lib.orc_lz4_decompress_block.restype = ctypes.c_uint64
    lib.orc_lz4_decompress_block.argtypes = [ctypes.c_void_p, ctypes.c_uint64,
                                             ctypes.c_void_p, ctypes.c_uint64]
    codec = pa.Codec("lz4_raw")
    rng = np.random.default_rng(31)
    cases = [b"", b"a", b"ab" * 500,
             bytes(rng.integers(0, 4, 5000, dtype=np.uint8)),   # compressible
             bytes(rng.integers(0, 256, 5000, dtype=np.uint8)), # incompressible
             bytes(rng.integers(0, 256, 17, dtype=np.uint8)),
             b"\x00" * 8192]
    for data in cases:
        if len(data) == 0:
            continue
        # ours -> pyarrow
        out = np.zeros(len(data) * 2 + 64, np.uint8)
        nb = lib.orc_lz4_compress_block(data, len(data), out.ctypes.data)
        assert nb > 0
        got = bytes(codec.decompress(out[:nb].tobytes(), len(data)))
        assert got == data, (len(data), "ours->pyarrow")
        # pyarrow -> ours
        comp = bytes(codec.compress(data))
        dec = np.zeros(len(data) + 8, np.uint8)
        m = lib.orc_lz4_decompress_block(comp, len(comp), dec.ctypes.data, len(dec))
        assert m == len(data) and dec[:m].tobytes() == data, (len(data), "pyarrow->ours")


def test_bitshuffle_transpose_vs_independent_numpy():
    """Two INDEPENDENT restatements of the published bitshuffle bit-plane
    transpose (bitshuffle 0.5.1's bshuf_trans_bit_elem for elem_size 4) must
    agree: the C oracle (loop form) vs a from-the-spec numpy form
    (bit-plane-major, LSB-first within plane bytes). With the library absent
    offline this is the strongest available pin for the one remaining
    spec-restated surface."""
    import ctypes
    lib = orc.load()
    lib.orc_bshuf_transpose_i32.restype = ctypes.c_uint64
    lib.orc_bshuf_transpose_i32.argtypes = [ctypes.c_void_p, ctypes.c_uint32,
                                            ctypes.c_void_p]

    def numpy_bshuf_i32(vals):
        v = vals.view(np.uint32)
        bits = ((v[:, None] >> np.arange(32, dtype=np.uint32)[None, :]) & 1)
        planes = bits.T.astype(np.uint8)  # [32 bit-planes, elems]
        return np.packbits(planes.reshape(32, -1, 8), axis=-1,
                           bitorder="little").reshape(-1)

    rng = np.random.default_rng(17)
    for elems in (8, 64, 4096, 8192):
        vals = rng.integers(-2**31, 2**31 - 1, elems).astype(np.int32)
        out = np.zeros(elems * 4, np.uint8)
        nb = lib.orc_bshuf_transpose_i32(vals.ctypes.data, elems, out.ctypes.data)
        assert nb == elems * 4
        assert np.array_equal(out, numpy_bshuf_i32(vals)), elems


def test_binary_prefix_page_oracle_roundtrip():
    """BinaryPrefixPage (front coding, restart every 16 entries;
    binary_prefix_page.{h,cpp}): sorted dictionary keys compress well and
    roundtrip exactly, including empty strings and non-multiple-of-16
    counts."""
    rng = np.random.default_rng(19)
    rows = sorted(f"prefix_{i:06d}_{rng.integers(100)}".encode()
                  for i in range(1000)) + [b"", b"x"]
    off = np.zeros(len(rows) + 1, np.uint32)
    np.cumsum([len(r) for r in rows], out=off[1:])
    bts = np.frombuffer(b"".join(rows), np.uint8).copy()
    page = orc.binary_prefix_encode(bts, off)
    assert len(page) < bts.nbytes // 2  # front coding pays on sorted keys
    db, do = orc.binary_prefix_decode(page, len(rows), bts.nbytes)
    assert np.array_equal(do, off) and np.array_equal(db, bts)


@pytest.mark.gpu
def test_binary_prefix_page_decode_gpu_parity(engine):
    rng = np.random.default_rng(20)
    cases = []
    rows = sorted(f"key_{i:07d}".encode() + b"s" * int(rng.integers(0, 5))
                  for i in range(50_000))
    cases.append(rows)
    cases.append([bytes(rng.integers(97, 123, int(rng.integers(0, 30)),
                                     dtype=np.uint8)) for _ in range(997)])
    for rows in cases:
        off = np.zeros(len(rows) + 1, np.uint32)
        np.cumsum([len(r) for r in rows], out=off[1:])
        bts = np.frombuffer(b"".join(rows), np.uint8).copy()
        page = orc.binary_prefix_encode(bts, off)
        pb = engine.alloc(page.nbytes)
        pb.h2d(page)
        ob = engine.alloc(max(bts.nbytes, 1))
        oo = engine.alloc((len(rows) + 1) * 4)
        engine.page_decode_binary_prefix(pb, len(rows), ob, oo)
        assert np.array_equal(oo.d2h(np.uint32, len(rows) + 1), off)
        if bts.nbytes:
            assert np.array_equal(ob.d2h(np.uint8, bts.nbytes), bts)
        pb.free()
        ob.free()
        oo.free()


@pytest.mark.gpu
def test_for_page_feeds_q1_join_gpu(engine):
    """Storage-ingress chain for the metric path: lo_orderdate stored as a
    frame-of-reference page, decoded on device, then the q1 date-join+SUM
    runs over the decoded column — results bit-exact vs the oracle pipeline
    (the same chain test shape as the bitshuffle page in round 1)."""
    from starrocks_amd import gen
    n, year = 2_000_000, 1993
    od, ep, dc = orc.gen_lineorder_q1(42, 0, n)
    page = orc.for_page_encode_i32(od)
    pb = engine.alloc(page.nbytes)
    pb.h2d(page)
    od_dev = engine.alloc(n * 4)
    engine.page_decode_for_i32(pb, n, od_dev)
    ep_dev = engine.alloc(n * 4)
    ep_dev.h2d(ep)
    dc_dev = engine.alloc(n * 4)
    dc_dev.h2d(dc)
    datekey, dyear = gen.gen_dates()
    kb = engine.alloc(datekey.nbytes)
    kb.h2d(datekey.astype(np.int32))
    pay = np.where(dyear == year, dyear - 1992 + 1, 0).astype(np.uint32)
    payb = engine.alloc(pay.nbytes)
    payb.h2d(pay)
    dates = engine.join_build_payload(kb, payb, len(datekey))
    s, cnt = engine.q1_join_sum(dates, od_dev, ep_dev, dc_dev, n)
    es, ecnt = orc.q1_pipeline(42, 0, n, year)
    assert (s, cnt) == (es, ecnt)
    dates.destroy()
    for b in (pb, od_dev, ep_dev, dc_dev, kb, payb):
        b.free()


def test_plain_page_roundtrip_cpu():
    """PlainPage numeric codec (plain_page.h:51,83-102,148-158): u32 LE count
    header + raw LE values; hand-checked layout + roundtrips."""
    rng = np.random.default_rng(8)
    for n in (0, 1, 100, 4096):
        v = rng.integers(-2**31, 2**31, n).astype(np.int32)
        page = orc.plain_page_encode_i32(v)
        assert len(page) == 4 + 4 * n
        assert int.from_bytes(page[:4].tobytes(), "little") == n
        assert np.array_equal(orc.plain_page_decode_i32(page, n), v)
    # hand KAT
    page = orc.plain_page_encode_i32(np.array([1, -1], np.int32))
    assert page.tobytes() == (b"\x02\x00\x00\x00" +
                              b"\x01\x00\x00\x00" + b"\xff\xff\xff\xff")


@pytest.mark.gpu
def test_plain_page_decode_gpu(engine):
    rng = np.random.default_rng(9)
    n = 1_000_000
    v = rng.integers(-2**31, 2**31, n).astype(np.int32)
    page = orc.plain_page_encode_i32(v)
    pg = engine.alloc(len(page)); pg.h2d(page)
    out = engine.alloc(n * 4)
    engine.page_decode_plain_i32(pg, n, out)
    assert np.array_equal(out.d2h(np.int32, n), v)
    # malformed: wrong expected count must error, not read garbage
    import pytest as _pytest
    from starrocks_amd.engine import GpueError
    with _pytest.raises(GpueError):
        engine.page_decode_plain_i32(pg, n + 1, out)
    pg.free(); out.free()
