"""Foreground single-stripe latency probe: ec.Encoder.encode on one stripe
at small shard sizes (the access PUT path, MinShardSize=2KB)."""
import json, time, sys, os
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
import torch
from cubefs_amd import codemode, ec
t = codemode.get_tactic("EC6P3")
enc = ec.Encoder(t)
out = {}
for slen in (2048, 65536, 1 << 20, 8 << 20):
    sh = [torch.randint(0, 256, (slen,), dtype=torch.uint8, device="cuda")
          for _ in range(t.N)] + \
         [torch.zeros(slen, dtype=torch.uint8, device="cuda") for _ in range(t.M)]
    for _ in range(5):
        enc.encode(sh)
    torch.cuda.synchronize()
    t0 = time.perf_counter(); N = 50
    for _ in range(N):
        enc.encode(sh)  # includes ptr upload + launch + sync per call
    el = (time.perf_counter() - t0) / N
    out["encode_1stripe_%dB_us" % slen] = round(el * 1e6, 1)
print(json.dumps(out))
