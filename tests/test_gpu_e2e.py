"""End-to-end blobstore data-path flow on GPU, mirroring the reference's
service composition:

  PUT:    split into an ec.Buffer -> fused encode+frame -> on-disk images
          (access/stream PUT + blobnode datafile write)
  GET:    parse + body-verify an image, strip frames, join
          (datafile read + degraded-path plumbing)
  REPAIR: lose shards, fused reconstruct+verify, write repaired images
          (scheduler -> blobnode repair tasklet)

Everything below the boundary is the HIP engine; the oracle only checks
final bytes.
"""
import io

import numpy as np
import pytest

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu


def test_put_get_repair_cycle(oracle):
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from cubefs_amd import codemode, crc32block, ec, shard
    from cubefs_amd.buffer import Buffer

    t = codemode.get_tactic("EC6P3")
    enc = ec.Encoder(t)
    scodec = shard.ShardCodec()
    blob = np.random.default_rng(99).integers(0, 256, 3 * (1 << 20) + 777,
                                              dtype=np.uint8)

    # ---- PUT: ec.Buffer + encode (stream_put.go:96-169) ----
    buf = Buffer(blob.size, t, device="cuda")
    buf.data_buf.copy_(torch.from_numpy(blob).cuda())
    shards = buf.shards()
    enc.encode(shards)
    assert enc.verify(shards)

    # blobnode write: full disk images for every shard (datafile.go:342)
    S = buf.shard_size
    dsz = shard.disk_size(S)
    imgs = torch.zeros((t.total, dsz), dtype=torch.uint8, device="cuda")
    src = buf.batch_view()[0]  # [total, S]
    scodec.write_batch(imgs, src, bids=[5000 + j for j in range(t.total)],
                       vuids=[42] * t.total)
    scodec.synchronize()

    # ---- GET: parse + verify one image, strip, join ----
    metas = scodec.parse_batch(imgs, S)
    assert all(m["err"] == 0 and m["bad_block"] == -1 for m in metas)
    assert [m["bid"] for m in metas] == [5000 + j for j in range(t.total)]
    # strip the body of each data shard and join the blob
    codec = crc32block.Codec()
    body_len = crc32block.encode_size(S)
    out = io.BytesIO()
    restored = []
    for j in range(t.N):
        raw = torch.zeros(S, dtype=torch.uint8, device="cuda")
        codec.decode(raw, imgs[j, 32:32 + body_len])
        restored.append(raw)
    enc.join(out, restored, blob.size)
    assert out.getvalue() == blob.tobytes()

    # ---- corruption is caught at the disk layer ----
    imgs[2, 32 + 4 + 1234] ^= 1
    metas = scodec.parse_batch(imgs, S)
    assert metas[2]["err"] == -9 and metas[2]["bad_block"] == 0
    assert all(m["err"] == 0 for j, m in enumerate(metas) if j != 2)

    # ---- REPAIR: lose two shards, fused tasklet, check images ----
    batch = src.clone().unsqueeze(0).contiguous()  # [1, total, S]
    bad = [2, 7]
    for i in bad:
        batch[:, i].zero_()
    rimgs = torch.zeros((len(bad), dsz), dtype=torch.uint8, device="cuda")
    fails = enc.repair_batch(batch, bad, rimgs,
                             bids=[5002, 5007], vuids=[42, 42])
    enc.synchronize()
    assert fails == [False]
    # repaired images byte-identical to the originals written at PUT
    assert torch.equal(rimgs[0], imgs_orig(imgs, 2, S, scodec, oracle, src))
    assert torch.equal(rimgs[1], imgs_orig(imgs, 7, S, scodec, oracle, src))


def imgs_orig(imgs, j, S, scodec, oracle, src):
    """Reference image for shard j rebuilt from the intact source."""
    raw = src[j].cpu().numpy()
    want = oracle.shard_write(raw.copy(), bid=5000 + j, vuid=42)
    return torch.from_numpy(want).cuda()
