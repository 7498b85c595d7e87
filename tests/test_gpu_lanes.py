"""Concurrent foreground callers on ONE context: the stream-lane pool.

The reference shares a single ec.Encoder across ~100 goroutines behind a
counting semaphore (encoder.go:29,115).  The C engine mirrors that shape
with a per-context pool of stream lanes (own stream + own scratch), so
single-stripe Encode/Verify/Reconstruct calls from concurrent threads
issue to the GPU in parallel instead of serializing on one stream.
"""
import threading
import time

import numpy as np
import pytest

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu


import os


def _mk_encoder(lanes):
    from cubefs_amd import codemode, ec
    if not torch.cuda.is_available():
        pytest.skip("needs GPU")
    os.environ["GFRS_LANES"] = str(lanes)
    try:
        return ec.Encoder(codemode.get_tactic("EC6P3"))
    finally:
        os.environ.pop("GFRS_LANES", None)


@pytest.fixture(scope="module")
def enc():
    return _mk_encoder(8)


def _run_threads(enc, nthreads, iters, slen):
    """Each thread owns its shards and loops single-stripe encodes."""
    t = enc.tactic
    dev = torch.device("cuda:0")
    stripes = []
    for i in range(nthreads):
        rng = np.random.default_rng(1000 + i)
        data = rng.integers(0, 256, (t.N, slen), dtype=np.uint8)
        sh = [torch.from_numpy(data[j].copy()).to(dev) for j in range(t.N)]
        sh += [torch.zeros(slen, dtype=torch.uint8, device=dev)
               for _ in range(t.M)]
        # non-contiguous layout forces the pointer-table (lane) path
        stripes.append(sh)
    errs = []

    def worker(sh):
        try:
            for _ in range(iters):
                enc.encode(sh)
        except Exception as e:  # pragma: no cover
            errs.append(e)

    ths = [threading.Thread(target=worker, args=(stripes[i],))
           for i in range(nthreads)]
    t0 = time.perf_counter()
    for th in ths:
        th.start()
    for th in ths:
        th.join()
    el = time.perf_counter() - t0
    assert not errs, errs
    return nthreads * iters / el, stripes


def test_lane_pool_concurrent_throughput(enc, oracle):
    # launch-latency-bound foreground shape: at 64 KiB the kernel is a
    # few microseconds, so a single stream is gap-bound and the lane
    # pool's overlap is what shows; larger shards saturate the device
    # from one stream and lanes cannot multiply anything
    slen = 64 << 10
    ser = _mk_encoder(0)  # round-1 shape: every call on one mutex+stream
    _run_threads(ser, 2, 4, slen)
    r_ser, _ = _run_threads(ser, 8, 40, slen)
    del ser
    # warm (plans, lanes, allocator)
    _run_threads(enc, 2, 4, slen)
    r1, _ = _run_threads(enc, 1, 60, slen)
    r8, stripes = _run_threads(enc, 8, 60, slen)
    print("lane pool: serialized-8t %.0f enc/s, pooled 1t %.0f, "
          "pooled 8t %.0f, pool/serial %.2fx"
          % (r_ser, r1, r8, r8 / r_ser))
    # correctness under concurrency: every thread's parity is bit-exact
    t = enc.tactic
    for sh in stripes:
        ref = [s.cpu().numpy().copy() for s in sh]
        for i in range(t.N + t.M, len(ref)):
            ref[i][:] = 0
        want = [r.copy() for r in ref]
        oracle.rs_encode(t.N, t.M, want)
        for i in range(t.N + t.M):
            assert np.array_equal(sh[i].cpu().numpy(), want[i])
    # Measured (r02, profiles/r02_foreground.txt): the single-stripe call
    # is HOST-submission-bound (~27 us of HIP enqueue+sync per call), so
    # stream fan-out alone moves aggregate throughput only a few percent;
    # the engine answer for concurrent foreground load is small-batch
    # aggregation (INTEGRATION.md "Foreground policy": one
    # encode_frame_batch of N stripes runs ~60x more stripes/s than N
    # single calls).  The pool's job here is correctness + isolation
    # under concurrency, and overlap for host-staged callers; assert it
    # never regresses the serialized context.
    assert r8 >= 0.9 * r_ser, (r_ser, r8)


def test_lane_pool_mixed_ops(enc, oracle):
    """Concurrent encode+verify+reconstruct on one context stay correct."""
    t = enc.tactic
    dev = torch.device("cuda:0")
    slen = 64 << 10
    errs = []

    def worker(seed):
        try:
            rng = np.random.default_rng(seed)
            data = rng.integers(0, 256, (t.N, slen), dtype=np.uint8)
            sh = [torch.from_numpy(data[j].copy()).to(dev)
                  for j in range(t.N)]
            sh += [torch.zeros(slen, dtype=torch.uint8, device=dev)
                   for _ in range(t.M)]
            for _ in range(10):
                enc.encode(sh)
                assert enc.verify(sh)
                keep = sh[2].cpu().numpy().copy()
                sh[2].zero_()
                enc.reconstruct(sh, [2])
                assert np.array_equal(sh[2].cpu().numpy(), keep)
        except Exception as e:  # pragma: no cover
            errs.append(e)

    ths = [threading.Thread(target=worker, args=(7000 + i,))
           for i in range(6)]
    for th in ths:
        th.start()
    for th in ths:
        th.join()
    assert not errs, errs


def test_lane_pool_host_memory(enc, oracle):
    """Concurrent HOST-memory callers: each lane's pinned staging buffers
    must be isolated (a shared staging buffer would interleave stripes)."""
    t = enc.tactic
    slen = 32 << 10
    errs = []

    def worker(seed):
        try:
            rng = np.random.default_rng(seed)
            data = rng.integers(0, 256, (t.N, slen), dtype=np.uint8)
            sh = [data[j].copy() for j in range(t.N)]
            sh += [np.zeros(slen, np.uint8) for _ in range(t.M)]
            want = [x.copy() for x in sh]
            oracle.rs_encode(t.N, t.M, want)
            for _ in range(8):
                for i in range(t.N, t.N + t.M):
                    sh[i][:] = 0
                enc.encode(sh)  # numpy -> host-memory staging path
                for i in range(t.N + t.M):
                    assert np.array_equal(sh[i], want[i]), i
        except Exception as e:  # pragma: no cover
            errs.append(e)

    ths = [threading.Thread(target=worker, args=(8000 + i,))
           for i in range(6)]
    for th in ths:
        th.start()
    for th in ths:
        th.join()
    assert not errs, errs
