"""GPU crc32block tests: HIP framing kernel vs the CPU oracle, bit-exact,
including ragged sizes, corruption localization and batch mode."""
import os

import numpy as np
import pytest

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def dev():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    return torch.device("cuda:0")


@pytest.fixture(scope="module")
def codec():
    from cubefs_amd import crc32block
    return crc32block.Codec()


@pytest.mark.parametrize("size", [1, 3, 100, 4096, 65531, 65532, 65533,
                                  131064, 200000, 4 << 20])
def test_encode_matches_oracle(oracle, dev, codec, size):
    from cubefs_amd import crc32block
    rng = np.random.default_rng(size)
    raw = rng.integers(0, 256, size, dtype=np.uint8)
    want = oracle.crc32b_encode(raw)
    src = torch.from_numpy(raw).to(dev)
    dst = torch.zeros(crc32block.encode_size(size), dtype=torch.uint8, device=dev)
    n = codec.encode(dst, src)
    assert n == want.size
    assert np.array_equal(dst.cpu().numpy(), want), size
    # verify passes
    assert codec.verify(dst) == -1
    # decode round-trips
    back = torch.zeros(size, dtype=torch.uint8, device=dev)
    assert codec.decode(back, dst) == size
    assert np.array_equal(back.cpu().numpy(), raw)


def test_golden_vectors(dev, codec, golden_dir):
    from cubefs_amd import crc32block
    z = np.load(os.path.join(golden_dir, "rs_vectors.npz"))
    for name in [f for f in z.files if f.startswith("crc") and f.endswith("/raw")]:
        raw = z[name]
        framed = z[name.replace("/raw", "/framed")]
        src = torch.from_numpy(raw.copy()).to(dev)
        dst = torch.zeros(framed.size, dtype=torch.uint8, device=dev)
        codec.encode(dst, src)
        assert np.array_equal(dst.cpu().numpy(), framed), name


def test_corruption_localized(dev, codec):
    rng = np.random.default_rng(7)
    raw = rng.integers(0, 256, 500000, dtype=np.uint8)
    from cubefs_amd import crc32block
    src = torch.from_numpy(raw).to(dev)
    dst = torch.zeros(crc32block.encode_size(raw.size), dtype=torch.uint8, device=dev)
    codec.encode(dst, src)
    # corrupt payload byte in block 3
    dst[3 * 65536 + 4 + 1000] ^= 0x80
    assert codec.verify(dst) == 3
    # first bad block wins
    dst[1 * 65536 + 4 + 5] ^= 1
    assert codec.verify(dst) == 1
    from cubefs_amd.runtime import GfrsError
    back = torch.zeros(raw.size, dtype=torch.uint8, device=dev)
    with pytest.raises(GfrsError) as ei:
        codec.decode(back, dst)
    assert ei.value.code == -9  # ErrMismatchedCrc


def test_batch(oracle, dev, codec):
    from cubefs_amd import crc32block
    ns, n = 9, 300000
    rng = np.random.default_rng(8)
    raw = rng.integers(0, 256, (ns, n), dtype=np.uint8)
    enc_sz = crc32block.encode_size(n)
    src = torch.from_numpy(raw).to(dev)
    dst = torch.zeros((ns, enc_sz), dtype=torch.uint8, device=dev)
    codec.encode_batch(dst, src)
    codec.synchronize()
    got = dst.cpu().numpy()
    for s in range(ns):
        assert np.array_equal(got[s], oracle.crc32b_encode(raw[s].copy())), s
    assert codec.verify_batch(dst) == [-1] * ns
    dst[4, 2 * 65536 + 4] ^= 1
    bad = codec.verify_batch(dst)
    assert bad[4] == 2 and all(b == -1 for i, b in enumerate(bad) if i != 4)


def test_other_block_lengths(oracle, dev, codec):
    """block_len is configurable (SetBlockSize, util.go:49-54): any positive
    multiple of 4096."""
    for bl in (4096, 8192, 1 << 20):
        rng = np.random.default_rng(bl)
        size = 3 * bl + 1234
        raw = rng.integers(0, 256, size, dtype=np.uint8)
        want = oracle.crc32b_encode(raw, block_len=bl)
        from cubefs_amd import crc32block
        src = torch.from_numpy(raw).to(dev)
        dst = torch.zeros(crc32block.encode_size(size, bl), dtype=torch.uint8,
                          device=dev)
        codec.encode(dst, src, block_len=bl)
        assert np.array_equal(dst.cpu().numpy(), want), bl
        assert codec.verify(dst, block_len=bl) == -1


def test_fused_encode_frame(oracle, dev):
    """Fused encode+frame: framed images bit-identical to
    oracle-encode followed by oracle-framing, across ragged sizes and
    multi-frame shards; fallback path (EC15P12) agrees too."""
    from cubefs_amd import codemode, crc32block, ec
    codemode.extend(240, "LRC12P2L2", codemode.Tactic(12, 2, 2, 2, 14, 0, 2048))
    for name, slen in [("EC6P3", 300000), ("EC6P3", 65532), ("EC6P3", 100),
                       ("EC6P3", 1 << 20), ("EC12P4", 200000),
                       ("EC15P12", 100000),   # fallback (m > 4)
                       ("LRC12P2L2", 300000),  # fused composed-LRC plan
                       ("EC6P10L2", 300000),   # LRC fallback (m+l > 4)
                       # wave-per-stripe small kernel (<= 4096) across
                       # NI boundaries, odd sizes, and the composed-LRC
                       # plan; 5000 lands in the two-kernel gap
                       ("EC6P3", 2048), ("EC6P3", 1024), ("EC6P3", 1040),
                       ("EC6P3", 17), ("EC6P3", 2049), ("EC6P3", 4095),
                       ("EC6P3", 4096), ("EC12P4", 3000),
                       ("LRC12P2L2", 2048), ("EC6P3", 5000),
                       # NI 5-8 wave-per-stripe forms (4-8 KiB, gm<=3)
                       ("EC6P3", 6000), ("EC6P3", 8192), ("EC6P3", 7169),
                       ("EC12P4", 5000),
                       # tiny-last-frame fold (trailing frame <= 64 B is
                       # emitted by the preceding frame's workgroup)
                       ("EC6P3", 65536), ("EC6P3", 65533),
                       ("EC6P3", 65532 + 64), ("EC6P3", 65532 + 65),
                       ("EC6P3", 2 * 65532 + 16), ("EC12P4", 65536),
                       ("LRC12P2L2", 131072), ("EC6P3", 262144)]:
        t = codemode.get_tactic(name)
        ns = 3
        rng = np.random.default_rng(slen ^ t.N)
        arr = rng.integers(0, 256, (ns, t.total, slen), dtype=np.uint8)
        batch = torch.from_numpy(arr.copy()).to(dev)
        enc_sz = crc32block.encode_size(slen)
        framed = torch.zeros((ns * t.total, enc_sz), dtype=torch.uint8,
                             device=dev)
        enc = ec.Encoder(t)
        enc.encode_frame_batch(framed, batch)
        enc.synchronize()
        got = framed.cpu().numpy()
        for s in range(ns):
            sh = [arr[s, i].copy() for i in range(t.total)]
            oracle.lrc_encode(t.N, t.M, t.L, t.AZCount, sh)
            for j in range(t.total):
                want = oracle.crc32b_encode(sh[j])
                assert np.array_equal(got[s * t.total + j], want), \
                    (name, slen, s, j)
