"""Calibration: practical HBM bandwidth of this box via torch copies."""
import time, torch, json
torch.cuda.init()
n = 8 << 30  # 8 GiB each
a = torch.empty(n, dtype=torch.uint8, device="cuda")
b = torch.empty(n, dtype=torch.uint8, device="cuda")
a.random_(0, 256)
for _ in range(3):
    b.copy_(a)
torch.cuda.synchronize()
t0 = time.perf_counter(); REP=10
for _ in range(REP):
    b.copy_(a)
torch.cuda.synchronize()
el = time.perf_counter() - t0
print(json.dumps({"copy_GBps_moved": round(2*n*REP/el/1e9, 1),
                  "note": "read+write bytes counted; torch copy_ kernel"}))
# write-only (fill) and read-heavy (sum) probes
t0 = time.perf_counter()
for _ in range(REP): a.fill_(7)
torch.cuda.synchronize(); el = time.perf_counter()-t0
print(json.dumps({"fill_GBps_written": round(n*REP/el/1e9, 1)}))
c = a.view(torch.int64)
for _ in range(2): c.sum()
torch.cuda.synchronize(); t0=time.perf_counter()
for _ in range(REP): c.sum()
torch.cuda.synchronize(); el=time.perf_counter()-t0
print(json.dumps({"sum_GBps_read": round(n*REP/el/1e9, 1)}))
