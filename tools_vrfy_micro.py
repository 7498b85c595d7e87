# verify-leg microbench: times codec.verify_batch alone.
# env: VM_SHARDS (default 512 stripes worth = 4608 images), VM_SHARD_MIB (4)
import os
import time

import torch

from cubefs_amd import crc32block
from cubefs_amd.runtime import lib

dev = torch.device("cuda:0")
codec = crc32block.Codec()
s = torch.cuda.Stream(dev)
lib().gfrs_set_stream(codec._ctx, s.cuda_stream)
ns = int(os.environ.get("VM_SHARDS", "4608"))
S = int(os.environ.get("VM_SHARD_MIB", "4")) << 20
enc_sz = crc32block.encode_size(S)
stride = (enc_sz + 255) // 256 * 256
framed = torch.zeros((ns, stride), dtype=torch.uint8, device=dev)
framed.random_(0, 256)
torch.cuda.synchronize()
for _ in range(2):
    codec.verify_batch(framed[:, :enc_sz])
t0 = time.perf_counter()
K = 6
for _ in range(K):
    codec.verify_batch(framed[:, :enc_sz])
el = (time.perf_counter() - t0) / K
rb = ns * enc_sz
print("verify leg: ns=%d %.3f ms  %.2f TB/s read" % (ns, el * 1e3,
                                                     rb / el / 1e12))
