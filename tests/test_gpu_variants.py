"""Variant parity: every GFRS_EF form of the fused encode+frame kernel
(register-CRC pipelines, staged fallback, dual-aligned rotation kernel)
produces bit-identical framed images.  The env var is read once per
process, so each variant runs in a subprocess.

Reference semantics: encoder.go:114 Encode + crc32block encode.go:48-109
framing, checked against the CPU oracle.
"""
import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

_REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

_SCRIPT = r"""
import numpy as np
import torch
from cubefs_amd import codemode, crc32block, ec
from oracle import pyoracle as oracle

codemode.extend(240, "LRC12P2L2", codemode.Tactic(12, 2, 2, 2, 14, 0, 2048))
# sizes cover: multi-frame + partial last frame, exactly one frame,
# prologue-only last frame (17), sub-frame odd sizes, 8 MiB-class
cases = [("EC6P3", 300000), ("EC6P3", 65532), ("EC6P3", 17),
         ("EC6P3", 100), ("EC6P3", 5000), ("EC6P3", 1 << 20),
         ("EC6P3", 65536), ("EC6P3", 2 * 65532 + 16),
         ("EC12P4", 200000), ("LRC12P2L2", 130000)]
for name, slen in cases:
    t = codemode.get_tactic(name)
    ns = 3
    rng = np.random.default_rng(slen ^ t.N)
    arr = rng.integers(0, 256, (ns, t.total, slen), dtype=np.uint8)
    batch = torch.from_numpy(arr.copy()).to("cuda:0")
    enc_sz = crc32block.encode_size(slen)
    for pad in (0, 1):
        # pad=1: 256-B image stride (the aligned production layout);
        # pad=0: tight stride, which can put frame bases at +4 mod 16 —
        # the kernels must stay correct (just slower) there
        stride = (enc_sz + 255) // 256 * 256 if pad else enc_sz
        framed = torch.zeros((ns * t.total, stride), dtype=torch.uint8,
                             device="cuda:0")
        enc = ec.Encoder(t)
        enc.encode_frame_batch(framed, batch)
        enc.synchronize()
        got = framed.cpu().numpy()
        for s in range(ns):
            sh = [arr[s, i].copy() for i in range(t.total)]
            oracle.lrc_encode(t.N, t.M, t.L, t.AZCount, sh)
            for j in range(t.total):
                want = oracle.crc32b_encode(sh[j])
                assert np.array_equal(got[s * t.total + j, :enc_sz], want), \
                    (name, slen, pad, s, j)
print("variant parity OK")
"""


@pytest.mark.parametrize("ef", ["76", "77", "87", "73", "14",
                                "103", "104", "113", "114"])
def test_encode_frame_variant_parity(ef):
    env = dict(os.environ, GFRS_EF=ef)
    r = subprocess.run([sys.executable, "-c", _SCRIPT], env=env, cwd=_REPO,
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, (ef, r.stdout[-2000:], r.stderr[-2000:])


_VRFY_SCRIPT = r"""
import numpy as np
import torch
from cubefs_amd import crc32block
from cubefs_amd.runtime import GfrsError
from oracle import pyoracle as oracle

codec = crc32block.Codec()
rng = np.random.default_rng(0x5EC1FF)
for slen in (100000, 65532, 2 * 65532 + 16, 1 << 20, 9000):
    src = rng.integers(0, 256, slen, dtype=np.uint8)
    framed_np = oracle.crc32b_encode(src.copy())
    framed = torch.from_numpy(framed_np.copy()).to("cuda:0")
    assert codec.verify(framed) == -1, slen
    # decode (framed -> raw + verify) round-trips
    out = torch.zeros(slen, dtype=torch.uint8, device="cuda:0")
    n = codec.decode(out, framed)
    assert n == slen
    assert np.array_equal(out.cpu().numpy(), src), slen
    # single flipped bit in each block position band is caught
    for pos in (5, len(framed_np) // 2, len(framed_np) - 1):
        bad = framed_np.copy()
        bad[pos] ^= 1
        fb = torch.from_numpy(bad).to("cuda:0")
        assert codec.verify(fb) == pos // 65536, (slen, pos)
print("vrfy variant parity OK")
"""


@pytest.mark.parametrize("v", ["10", "16", "14", "18"])
def test_verify_variant_parity(v):
    """GFRS_VRFY load-lookahead variants of crc32b_verify_reg_k are
    bit-identical to the default (kept in-tree as measured variants)."""
    env = dict(os.environ, GFRS_VRFY=v)
    r = subprocess.run([sys.executable, "-c", _VRFY_SCRIPT], env=env,
                       cwd=_REPO, capture_output=True, text=True,
                       timeout=300)
    assert r.returncode == 0, (v, r.stdout[-2000:], r.stderr[-2000:])
