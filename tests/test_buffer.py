"""ec.Buffer semantics (buf.go:36-208, mirrors buf_test.go:45-174)."""
import numpy as np
import pytest

from cubefs_amd import codemode
from cubefs_amd.buffer import Buffer, MemPool
from cubefs_amd.runtime import GfrsError


def test_sizes_cpu(oracle):
    for name in ("EC6P3", "EC12P4", "EC6P10L2", "EC6P6Align0"):
        t = codemode.get_tactic(name)
        for ds in (1, 100, 12288, 1 << 20):
            b = Buffer(ds, t, device="cpu")
            ss, eds, es = oracle.buffer_sizes(t.N, t.M, t.L, t.MinShardSize, ds)
            assert (b.shard_size, b.ec_data_size, b.ec_size) == (ss, eds, es)
            assert b.data_buf.numel() == ds
            assert b.ec_data_buf.numel() == eds
            sh = b.shards()
            assert len(sh) == t.total
            assert all(s.numel() == ss for s in sh)


def test_padding_zeroed():
    t = codemode.get_tactic("EC6P3")
    b = Buffer(100, t, device="cpu")
    b._buf[:].fill_(0xAB)
    b2 = Buffer(100, t, device="cpu")
    # fresh buffer zeroes only the pad region; simulate reuse via resize
    b.resize(200)
    assert not b._buf[b.data_size:b.ec_data_size].any()
    assert b2.shard_size == 2048  # MinShardSize floor (buf.go:79-83)


def test_resize_reuse_and_grow():
    t = codemode.get_tactic("EC6P3")
    b = Buffer(1 << 20, t, device="cpu")
    cap = b._buf.numel()
    b.resize(1000)  # shrink: same backing tensor
    assert b._buf.numel() == cap
    assert b.shard_size == 2048
    b.resize(8 << 20)  # grow: reallocates
    assert b.ec_size == b.shard_size * t.total
    assert b.shard_size == -(-(8 << 20) // t.N)


def test_pool_recycle():
    pool = MemPool(device="cpu")
    t = codemode.get_tactic("EC6P3")
    b = Buffer(5000, t, pool=pool, device="cpu")
    backing = b._buf
    b.release()
    b2 = Buffer(4000, t, pool=pool, device="cpu")
    assert b2._buf is backing  # recycled from the pool
    with pytest.raises(GfrsError):
        Buffer(0, t, pool=pool, device="cpu")


def test_shards_are_views():
    t = codemode.get_tactic("EC4P4L2")
    b = Buffer(4096, t, device="cpu")
    sh = b.shards()
    sh[0].fill_(7)
    assert (b._buf[:b.shard_size] == 7).all()
    assert b.batch_view().shape == (1, t.total, b.shard_size)


@pytest.mark.gpu
def test_buffer_feeds_encoder(oracle):
    """PUT flow: NewBuffer -> fill data -> encode over the shard views
    (stream_put.go:120-146), bit-exact vs oracle."""
    import torch
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from cubefs_amd import ec
    t = codemode.get_tactic("EC6P3")
    ds = 100000
    b = Buffer(ds, t, device="cuda")
    rng = np.random.default_rng(3)
    data = rng.integers(0, 256, ds, dtype=np.uint8)
    b.data_buf.copy_(torch.from_numpy(data).cuda())
    enc = ec.Encoder(t)
    sh = b.shards()
    enc.encode(sh)
    ref = [s.cpu().numpy().copy() for s in sh]
    # oracle on the same padded layout
    osh = [np.zeros(b.shard_size, np.uint8) for _ in range(t.total)]
    flat = np.zeros(b.ec_data_size, np.uint8)
    flat[:ds] = data
    for i in range(t.N):
        osh[i][:] = flat[i * b.shard_size:(i + 1) * b.shard_size]
    oracle.rs_encode(t.N, t.M, osh)
    for i in range(t.total):
        assert np.array_equal(ref[i], osh[i]), i
