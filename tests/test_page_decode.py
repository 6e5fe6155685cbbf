"""Storage ingress (§8f row 4): bitshuffle+LZ4 page decode — CPU roundtrip
properties here; GPU parity in test_gpu_parity."""

import numpy as np

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
