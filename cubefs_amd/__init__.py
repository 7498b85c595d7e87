"""cubefs_amd — MI355X-native erasure-coding + checksum engine for the
CubeFS blobstore data path.

Scope (BASELINE.json north_star / SURVEY.md §8): the Reed-Solomon GF(2^8)
encode/reconstruct used by blobstore access/blobnode (vendored
klauspost/reedsolomon v1.11.7) and the crc32block shard framing, rebuilt
from scratch as hand-written HIP/CDNA4 kernels behind the reference's own
blobstore/common/ec.Encoder boundary (C ABI: include/gfrs.h).

Modules:
  codemode     — CodeMode/Tactic registry (blobstore/common/codemode)
  ec           — Encoder surface (blobstore/common/ec)
  crc32block   — shard-frame checksum codec + ranged journal Decoder
                 (blobstore/common/crc32block block/encode/decode/util.go)
  sized_stream — streaming sized/partial rpc2-body coder
                 (sized_coder.go / sized_coder_block.go)
  request_body — streaming HTTP-body encoder/decoder (request_body.go)
  buffer       — ec.Buffer size math (buf.go)
  shard        — blobnode on-disk shard image codec (core/shard.go)
  dist         — stripe-queue partitioning + peer shard staging
  runtime      — ctypes binding of libgfrs.so (no CPU fallback)
"""
from . import codemode  # noqa: F401

__version__ = "0.1.0"


def build(verbose=False):
    """Compile the HIP engine in-tree (hipcc --offload-arch=gfx950)."""
    import os
    import subprocess
    here = os.path.dirname(os.path.abspath(__file__))
    subprocess.check_call(["make", "-C", here] + ([] if verbose else ["-s"]))
