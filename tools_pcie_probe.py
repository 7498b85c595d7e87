#!/usr/bin/env python3
"""PCIe ceiling calibration for the current box (BASELINE config 4's
host-streamed mode is PCIe-bound by design; this probe gives the ceiling
the streamed numbers are quoted against — SURVEY.md §8d hard part (e)).

Measures pinned-host <-> HBM bandwidth with 256 MiB copies:
  h2d, d2h, and full-duplex (both directions on separate streams).
Spec: PCIe Gen5 x16 = 63 GB/s per direction.
"""
import json
import time

import torch


def run(n=256 << 20, reps=8):
    assert torch.cuda.is_available()
    dev = torch.device("cuda:0")
    hsrc = torch.empty(n, dtype=torch.uint8, pin_memory=True)
    hdst = torch.empty(n, dtype=torch.uint8, pin_memory=True)
    d1 = torch.empty(n, dtype=torch.uint8, device=dev)
    d2 = torch.empty(n, dtype=torch.uint8, device=dev)
    s1, s2 = torch.cuda.Stream(), torch.cuda.Stream()

    def timed(fn):
        fn()  # warm
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(reps):
            fn()
        torch.cuda.synchronize()
        return n * reps / (time.perf_counter() - t0) / 1e9

    def h2d():
        with torch.cuda.stream(s1):
            d1.copy_(hsrc, non_blocking=True)

    def d2h():
        with torch.cuda.stream(s1):
            hdst.copy_(d2, non_blocking=True)

    def duplex():
        with torch.cuda.stream(s1):
            d1.copy_(hsrc, non_blocking=True)
        with torch.cuda.stream(s2):
            hdst.copy_(d2, non_blocking=True)

    out = {
        "h2d_GBps": round(timed(h2d), 2),
        "d2h_GBps": round(timed(d2h), 2),
        # duplex reports per-direction rate (each direction moved n bytes)
        "duplex_per_dir_GBps": round(timed(duplex), 2),
        "spec_per_dir_GBps": 63.0,
        "copy_MiB": n >> 20,
    }
    print(json.dumps(out))
    return out


if __name__ == "__main__":
    run()
