"""sized coder (crc32block/sized_coder.go — the rpc2 body framing:
payload ‖ CRC32 big-endian per block, 512-B tail alignment)."""
import numpy as np
import pytest


def test_oracle_sized_structure(oracle):
    rng = np.random.default_rng(21)
    n = 200000
    raw = rng.integers(0, 256, n, dtype=np.uint8)
    framed, tail = oracle.sized_encode(raw)
    total, t2 = oracle.partial_encode_size(n)
    assert framed.size == total and tail == t2
    # frame 0: payload 65532 then BE crc
    crc0 = int.from_bytes(framed[65532:65536].tobytes(), "big")
    assert crc0 == oracle.crc32(raw[:65532])
    assert np.array_equal(framed[:65532], raw[:65532])
    # total is 512-aligned with zero pad
    assert total % 512 == 0
    assert not framed[total - tail:].any()
    assert oracle.sized_verify(framed, tail) == -1
    back = oracle.sized_decode(framed, tail)
    assert np.array_equal(back, raw)
    framed[65536 + 100] ^= 1  # corrupt payload of frame 1
    assert oracle.sized_verify(framed, tail) == 1
    with pytest.raises(ValueError):
        oracle.sized_decode(framed, tail)


def test_partial_size_math(oracle):
    """PartialEncodeSize/PartialDecodeSize inverses (util.go:73-94),
    including nonzero stableSize (append after stable bytes)."""
    for stable in (0, 512, 4096, 65532, 70000, 131064):
        for actual in (1, 100, 65532, 200000):
            total, tail = oracle.partial_encode_size(actual, stable)
            assert total % 512 == 0
            assert oracle.partial_decode_size(total, tail, stable) == actual


def test_sizes_match_product(oracle):
    from cubefs_amd import crc32block as cb
    for n in (1, 511, 512, 65532, 200000, 4 << 20):
        total, tail = oracle.partial_encode_size(n)
        assert cb.sized_encode_size(n) == (total, tail)
        assert cb.sized_decode_size(total, tail) == n


@pytest.mark.gpu
def test_gpu_sized_roundtrip(oracle):
    torch = pytest.importorskip("torch")
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from cubefs_amd import crc32block as cb
    codec = cb.Codec()
    for n in (100, 65532, 65533, 300000, 4 << 20):
        rng = np.random.default_rng(n)
        raw = rng.integers(0, 256, n, dtype=np.uint8)
        want, tail = oracle.sized_encode(raw)
        src = torch.from_numpy(raw).to("cuda:0")
        dst = torch.zeros(want.size, dtype=torch.uint8, device="cuda:0")
        w = codec.sized_encode(dst, src)
        assert w == want.size
        assert np.array_equal(dst.cpu().numpy(), want), n
        assert codec.sized_verify(dst, tail) == -1
        back = torch.zeros(n, dtype=torch.uint8, device="cuda:0")
        assert codec.sized_decode(back, dst, tail) == n
        assert np.array_equal(back.cpu().numpy(), raw)
        # corruption localized
        dst[50] ^= 4
        assert codec.sized_verify(dst, tail) == 0
