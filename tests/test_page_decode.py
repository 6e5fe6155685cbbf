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
