"""CPU oracle tests for CRC32-IEEE + crc32block framing, including the
GF(2) fold math the GPU kernel uses (pinned here before any GPU run)."""
import os
import zlib

import numpy as np
import pytest


def test_crc32_kat(oracle):
    # universal known-answer test + zlib cross-check
    assert oracle.crc32(b"123456789") == 0xCBF43926
    rng = np.random.default_rng(0)
    for n in (0, 1, 3, 64, 1000, 65532):
        buf = rng.integers(0, 256, n, dtype=np.uint8)
        assert oracle.crc32(buf) == zlib.crc32(buf.tobytes())


def test_crc32_running_update(oracle):
    """Go crc32.Update semantics: orc_crc32 chains on finalized values."""
    rng = np.random.default_rng(1)
    buf = rng.integers(0, 256, 5000, dtype=np.uint8)
    c = oracle.crc32(buf[:2000])
    c = oracle.crc32(buf[2000:], crc=c)
    assert c == oracle.crc32(buf)


def test_crc32_combine(oracle):
    rng = np.random.default_rng(2)
    buf = rng.integers(0, 256, 30000, dtype=np.uint8)
    whole = oracle.crc32(buf)
    for cut in (0, 1, 4, 13, 15000, 29999, 30000):
        c1, c2 = oracle.crc32(buf[:cut]), oracle.crc32(buf[cut:])
        assert oracle.crc32_combine(c1, c2, 30000 - cut) == whole


def test_gpu_fold_formula(oracle):
    """Pins the parallel decomposition used by crc32b_k on device:
    crc(M) = ~( x^(8|M|)·(~0) ^ XOR_t x^(8·suffix_t)·raw(chunk_t) )
    where raw(chunk) = table update from state 0, no complements.
    raw(chunk) relates to the API as raw = ~orc_crc32(0xFFFFFFFF, chunk)."""
    rng = np.random.default_rng(3)
    for total, chunk in [(65532, 256), (1000, 256), (255, 64), (65532, 1024)]:
        buf = rng.integers(0, 256, total, dtype=np.uint8)
        fold = 0
        for c0 in range(0, total, chunk):
            piece = buf[c0:c0 + chunk]
            raw = 0xFFFFFFFF ^ oracle.crc32(piece, crc=0xFFFFFFFF)
            suffix = total - (c0 + len(piece))
            fold ^= oracle.crc32_shift(raw, suffix)
        init_term = oracle.crc32_shift(0xFFFFFFFF, total)
        crc = 0xFFFFFFFF ^ (init_term ^ fold)
        assert crc == oracle.crc32(buf), (total, chunk)


def test_encode_size_math(oracle):
    """util_test.go:26-110 semantics (EncodeSize/DecodeSize inverses)."""
    bl = 64 * 1024
    for size in (1, 100, 65531, 65532, 65533, 131064, 4 << 20):
        enc = oracle.crc32b_encode_size(size, bl)
        blocks = -(-size // (bl - 4))
        assert enc == size + 4 * blocks
        assert oracle.crc32b_decode_size(enc, bl) == size
    # invalid block length (must be positive multiple of 4096, util.go:40)
    assert oracle.crc32b_encode_size(100, 1000) == -10
    assert oracle.crc32b_encode_size(100, 0) == -10


def test_frame_roundtrip_and_golden(oracle, golden_dir):
    z = np.load(os.path.join(golden_dir, "rs_vectors.npz"))
    for name in [f for f in z.files if f.startswith("crc") and f.endswith("/raw")]:
        raw = z[name]
        framed = z[name.replace("/raw", "/framed")]
        got = oracle.crc32b_encode(raw.copy())
        assert np.array_equal(got, framed), name
        assert oracle.crc32b_verify(framed.copy()) == -1
        back = oracle.crc32b_decode(framed.copy())
        assert np.array_equal(back, raw)


def test_frame_structure(oracle):
    """Frame layout: 4 B LE crc ‖ payload per 64 KiB block (block.go:22-49)."""
    rng = np.random.default_rng(4)
    raw = rng.integers(0, 256, 70000, dtype=np.uint8)
    framed = oracle.crc32b_encode(raw)
    # block 0: payload 65532
    hdr = int.from_bytes(framed[:4].tobytes(), "little")
    assert hdr == oracle.crc32(raw[:65532])
    assert np.array_equal(framed[4:65536], raw[:65532])
    # block 1 (tail): remaining 4468 bytes
    hdr2 = int.from_bytes(framed[65536:65540].tobytes(), "little")
    assert hdr2 == oracle.crc32(raw[65532:])
    assert np.array_equal(framed[65540:], raw[65532:])


def test_corruption_detection(oracle):
    rng = np.random.default_rng(5)
    raw = rng.integers(0, 256, 200000, dtype=np.uint8)
    framed = oracle.crc32b_encode(raw)
    framed[65536 + 4 + 17] ^= 1  # corrupt payload of block 1
    assert oracle.crc32b_verify(framed) == 1
    with pytest.raises(ValueError):
        oracle.crc32b_decode(framed)


def test_zero_crc(oracle):
    """ConstZeroCrc(n) == ChecksumIEEE(zeros) (util_test.go:84-95)."""
    for n in (0, 5, 16384, 65532):
        assert oracle.crc32(np.zeros(n, np.uint8)) == zlib.crc32(bytes(n))
