#!/usr/bin/env python3
"""bench.py — BASELINE.json's headline benchmark on MI355X.

Metric: "GiB/s encode+CRC throughput, RS(k+m) per stripe, 1/2/4/8 MI355X
vs CPU".  A step is one pass of the hot path over one resident stripe
batch: RS encode of every stripe (one fused HIP kernel over the batch)
followed by crc32block framing of all k+m shards (the blobnode write path,
datafile.go:342).  Default N=1 workload is BASELINE configs[1]:
RS(6+3), 8 MiB shards, 1024 stripes, plus the CRC leg the metric names.

`value` = whole-job SOURCE throughput: k·S·stripes·ranks / elapsed (GiB/s).
The klauspost "total shard size" convention (vendor README.md:443) is
reported alongside in config.total_shard_gib_s.

Multi-GPU: stripes are independent; each rank owns its own
batch (weak scaling), RCCL is used only for the start/stop barrier and the
result allgather.  Inputs are synthetic seeded uniform bytes resident in
HBM before the timed region.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

GIB = float(1 << 30)
HBM_PEAK = 8.0e12  # B/s, MI355X spec (MI355X_MICROARCH.md chip parameters)


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=4)
    p.add_argument("--warmup", type=int, default=2)
    p.add_argument("--stripes", type=int, default=1024)
    p.add_argument("--shard-mib", type=int, default=8)
    p.add_argument("--shard-kib", type=int, default=0,
                   help="overrides --shard-mib when > 0 (small-blob shapes)")
    p.add_argument("--codemode", default="EC6P3",
                   help="EC6P3/EC12P4/... or LRC12P2L2 (registered via Extend)")
    p.add_argument("--workload", default="encode",
                   choices=["encode", "reconstruct", "repair"],
                   help="encode = BASELINE configs[1]+CRC; reconstruct = "
                        "configs[2]: 1-shard reconstruct + crc32block "
                        "verify; repair = fused repair tasklet (2 lost "
                        "shards -> verified framed disk images)")
    p.add_argument("--bad-idx", type=int, default=2,
                   help="shard index reconstructed in --workload reconstruct")
    p.add_argument("--no-crc", action="store_true",
                   help="EC encode only (config label form)")
    p.add_argument("--no-fused", action="store_true",
                   help="two-kernel encode-then-frame instead of the fused "
                        "single-pass kernel (the fused path is faster and "
                        "is the default)")
    p.add_argument("--cpu-sample-stripes", type=int, default=24)
    p.add_argument("--host-streamed", action="store_true",
                   help="end-to-end repair-queue mode: source stripes live "
                        "in pinned HOST memory and stream H2D ahead of the "
                        "fused compute, framed images stream back D2H "
                        "(BASELINE config 4's 'end-to-end incl. H2D' "
                        "figure; PCIe-bound by design, never `value`)")
    p.add_argument("--skip-cpu-baseline", action="store_true")
    return p.parse_args()


def hbm_traffic_lookup(workload):
    """Per-launch HBM bytes measured by rocprofv3 --pmc (separate passes,
    gfx950 FETCH_SIZE correction per MI355X_MICROARCH.md §HBM), committed
    under profiles/.  Null when no measurement exists for this workload."""
    path = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                        "profiles", "hbm_traffic.json")
    try:
        with open(path) as f:
            d = json.load(f)
        entry = d.get(workload)
        return entry.get("bytes_per_launch") if isinstance(entry, dict) \
            else None
    except (OSError, ValueError):
        return None


def _cpu_quota_cores():
    """Effective CPU budget: cgroup v2 quota if set (GPU boxes report 256
    CPUs but are quota-limited; oversubscribing collapses OpenMP), else
    the OpenMP default."""
    try:
        q, p = open("/sys/fs/cgroup/cpu.max").read().split()
        if q != "max":
            return max(1, int(int(q) / int(p)))
    except (OSError, ValueError):
        pass
    return None


def cpu_baseline_leg(t, shard_len, nstripes_sample, with_crc):
    """Oracle (nibble-table algorithm, OpenMP across the host's effective
    CPU budget) timed on the GPU box's CPUs — the reported baseline,
    kind='port'.  Thread count is calibrated: the cgroup quota, 2x the
    quota and the OpenMP default are each timed once and the fastest
    wins (the box advertises 256 CPUs but enforces a ~16-CPU quota)."""
    import numpy as np
    from oracle import pyoracle as po
    cores = po.threads_avail()
    rng = np.random.default_rng(1)
    stripes = []
    for _ in range(nstripes_sample):
        st = [rng.integers(0, 256, shard_len, dtype=np.uint8) for _ in range(t.N)]
        st += [np.zeros(shard_len, np.uint8) for _ in range(t.M + t.L)]
        stripes.append(st)
    if with_crc:
        enc_sz = po.crc32b_encode_size(shard_len, 65536)
        dsts = [[np.zeros(enc_sz, np.uint8) for _ in st] for st in stripes]

    quota = _cpu_quota_cores()
    cands = sorted({c for c in (quota, 2 * quota if quota else None, cores)
                    if c})
    best_nt, best_dt = cores, None
    for nt in cands:
        po.rs_encode_mt(t.N, t.M, stripes, nthreads=nt)  # warm
        t0 = time.perf_counter()
        po.rs_encode_mt(t.N, t.M, stripes, nthreads=nt)
        dt = time.perf_counter() - t0
        if best_dt is None or dt < best_dt:
            best_nt, best_dt = nt, dt
    cores = best_nt

    def one_pass():
        po.rs_encode_mt(t.N, t.M, stripes, nthreads=cores)
        if with_crc:
            import ctypes
            L = po.lib()
            flatd = [d for ds in dsts for d in ds]
            flats = [s for st in stripes for s in st]
            pd = (ctypes.POINTER(ctypes.c_uint8) * len(flatd))(
                *[d.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)) for d in flatd])
            ps = (ctypes.POINTER(ctypes.c_uint8) * len(flats))(
                *[s.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)) for s in flats])
            L.orc_crc32b_encode_mt(pd, ps, shard_len, 65536, len(flats),
                                   cores)

    one_pass()  # warm caches/threads
    # repeat the bounded sample until ~10s of CPU work (driver contract)
    t0 = time.perf_counter()
    reps = 0
    el = 0.0
    while el < 10.0 and reps < 1000:
        one_pass()
        reps += 1
        el = time.perf_counter() - t0
    src_gib = t.N * shard_len * nstripes_sample * reps / GIB
    return {
        "value": round(src_gib / el, 3),
        "unit": "GiB/s",
        "cores": cores,
        "kind": "port",
        "sample": "%d passes x %d stripes RS(%d+%d) %d MiB shards%s, oracle "
                  "AVX2 nibble-table + OpenMP, %.1fs" % (
                      reps, nstripes_sample, t.N, t.M, shard_len >> 20,
                      "" if not with_crc else " + crc32block framing", el),
    }


def run_host_streamed(args, t, S, ns, enc, rank, world, d):
    """Config-4 style streamed queue: batches of stripes cross PCIe in,
    framed images cross back, double-buffered so H2D/compute/D2H overlap.
    Reported separately from the resident-workload `value` (SURVEY §8d
    hard part (e))."""
    import numpy as np
    import torch
    from cubefs_amd import crc32block, dist

    B = 64  # stripes per streamed batch
    nbatches = max(1, ns // B)
    enc_sz = crc32block.encode_size(S)
    host_src = torch.empty((B, t.N, S), dtype=torch.uint8,
                           pin_memory=True)
    host_src.random_(0, 256)
    host_out = torch.empty((2, B * t.total, enc_sz), dtype=torch.uint8,
                           pin_memory=True)
    dev_in = [torch.empty((B, t.total, S), dtype=torch.uint8, device="cuda")
              for _ in range(2)]
    dev_out = [torch.empty((B * t.total, enc_sz), dtype=torch.uint8,
                           device="cuda") for _ in range(2)]
    copy_s = torch.cuda.Stream()   # H2D
    d2h_s = torch.cuda.Stream()    # D2H (own direction, own engine)
    comp_s = torch.cuda.Stream()
    ev_in = [torch.cuda.Event() for _ in range(2)]
    ev_comp = [torch.cuda.Event() for _ in range(2)]
    from cubefs_amd.runtime import lib
    lib().gfrs_set_stream(enc._ctx, comp_s.cuda_stream)

    ev_d2h = [torch.cuda.Event() for _ in range(2)]

    def stream_queue():
        for i in range(nbatches):
            b = i % 2
            with torch.cuda.stream(d2h_s):
                if i >= 2:
                    ev_comp[b].wait(d2h_s)
                    host_out[b].copy_(dev_out[b], non_blocking=True)
                    ev_d2h[b].record(d2h_s)
            with torch.cuda.stream(copy_s):
                if i >= 2:
                    ev_d2h[b].wait(copy_s)
                # only the data shards cross H2D (parity is computed);
                # per-stripe contiguous copies keep the DMA fast path
                for st in range(B):
                    dev_in[b][st, :t.N].copy_(host_src[st],
                                              non_blocking=True)
                ev_in[b].record(copy_s)
            with torch.cuda.stream(comp_s):
                ev_in[b].wait(comp_s)
                enc.encode_frame_batch(dev_out[b], dev_in[b])
                ev_comp[b].record(comp_s)
        with torch.cuda.stream(d2h_s):
            for b in range(2):
                ev_comp[b].wait(d2h_s)
                host_out[b].copy_(dev_out[b], non_blocking=True)
        torch.cuda.synchronize()

    stream_queue()  # warmup
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        stream_queue()
    torch.cuda.synchronize()
    el = time.perf_counter() - t0
    dist.barrier()
    if d is not None:
        import torch.distributed as td
        tdev = "cuda" if td.get_backend() == "nccl" else "cpu"
        te = torch.tensor([el], device=tdev)
        td.all_reduce(te, op=td.ReduceOp.MAX)
        el = float(te.item())
    src_b = t.N * S * B * nbatches * args.steps * world
    if rank == 0:
        print(json.dumps({
            "metric": "GiB/s encode+CRC throughput, RS(k+m) per stripe, "
                      "1/2/4/8 MI355X vs CPU",
            "value": round(src_b / GIB / el, 2),
            "unit": "GiB/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": 1,
            "ms_per_step": round(el * 1e3 / args.steps, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "u8",
            "data": "synthetic",
            "config": {
                "workload": "HOST-STREAMED RS(%d+%d) fused encode+frame, "
                            "%d MiB shards, %d stripes in %d-stripe batches"
                            % (t.N, t.M, S >> 20, B * nbatches, B),
                "note": "end-to-end including H2D of sources and D2H of "
                        "framed images over PCIe (the repair-queue shape); "
                        "the resident-workload headline is the default "
                        "bench run",
            },
            "roofline": None,
            "cpu_baseline": None,
        }))


def main():
    args = parse_args()
    import numpy as np
    import torch

    from cubefs_amd import codemode, crc32block, dist, ec
    from cubefs_amd.runtime import lib

    rank, world = dist.env_rank_world()
    d = dist.init_process_group()
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    # modulo only matters under GFRS_DIST_BACKEND=gloo smoke runs where
    # several ranks share one visible GPU; on a real node it is identity
    local_dev = local_rank % max(1, torch.cuda.device_count())
    torch.cuda.set_device(local_dev)
    dev = torch.device("cuda", local_dev)

    if args.codemode == "LRC12P2L2":
        codemode.extend(240, "LRC12P2L2",
                        codemode.Tactic(12, 2, 2, 2, 14, 0, 2048))
    t = codemode.get_tactic(args.codemode)
    S = (args.shard_kib << 10) if args.shard_kib else (args.shard_mib << 20)
    ns = args.stripes
    with_crc = not args.no_crc

    enc = ec.Encoder(t, device=local_dev)
    # run everything on torch's current stream so HIP events bracket the
    # kernels (torch.cuda.Event sees only torch's stream)
    cur = torch.cuda.current_stream(dev).cuda_stream
    lib().gfrs_set_stream(enc._ctx, cur)
    codec = None
    if with_crc:
        codec = crc32block.Codec(device=local_dev)
        lib().gfrs_set_stream(codec._ctx, cur)

    if args.host_streamed:
        # config-4 streamed queue: sources live in pinned HOST memory and
        # stream in batches — no resident ns-sized device buffers
        run_host_streamed(args, t, S, ns, enc, rank, world, d)
        return

    # synthetic input resident in HBM (seeded per SURVEY.md §8d)
    g = torch.Generator(device=dev)
    g.manual_seed(0xB10B5703 ^ rank)
    batch = torch.empty((ns, t.total, S), dtype=torch.uint8, device=dev)
    batch[:, :t.N].random_(0, 256, generator=g)
    framed = None
    enc_sz = crc32block.encode_size(S)
    if with_crc:
        # pad the per-image stride to 256 B (encode_size(8 MiB) is ≡4
        # mod 16): cache-line-aligned image bases; production image
        # buffers would be pool-allocated the same way
        enc_stride = (enc_sz + 255) // 256 * 256
        framed = torch.empty((ns * t.total, enc_stride), dtype=torch.uint8,
                             device=dev)

    flat = batch.view(ns * t.total, S)

    ev_stream = None  # non-default stream the timing events live on, if any

    if args.workload == "reconstruct":
        # BASELINE configs[2]: parity must exist, frames pre-built; the
        # step recomputes shard bad_idx for every stripe and CRC-verifies
        # the framed shards (the repair read path)
        enc.encode_batch(batch)
        enc.synchronize()
        if with_crc:
            codec.encode_batch(framed, flat)
            codec.synchronize()
            # the repair read path CRC-checks the fetched frames while the
            # EC rebuild runs — disjoint buffers, no data dependence (the
            # reference does the same with per-shard goroutines during the
            # download phase, worker_slice_recover.go:127-210) — so each
            # leg gets its own NON-NULL HIP stream (the torch default
            # stream is the legacy null stream, which serializes against
            # every other stream) and the two grids co-schedule;
            # torch.cuda.synchronize at the timed-region edges covers both
            rec_stream = torch.cuda.Stream(dev)
            ver_stream = torch.cuda.Stream(dev)
            lib().gfrs_set_stream(enc._ctx, rec_stream.cuda_stream)
            lib().gfrs_set_stream(codec._ctx, ver_stream.cuda_stream)
            ev_stream = rec_stream  # HIP events must sit on the kernel's stream
        bad = [args.bad_idx]

    if args.workload == "repair":
        # blobnode repair tasklet: lose a data and a parity shard, rebuild
        # them as verified, framed, pwrite-able disk images in one call
        from cubefs_amd import shard as shardmod
        enc.encode_batch(batch)
        enc.synchronize()
        bad = [args.bad_idx, t.N] if t.M >= 2 else [args.bad_idx]
        for i in bad:
            batch[:, i].zero_()
        dsz = shardmod.disk_size(S)
        imgs = torch.empty((ns * len(bad), dsz), dtype=torch.uint8,
                           device=dev)
        rbids = np.arange(ns * len(bad), dtype=np.uint64)
        rvuids = np.ones(ns * len(bad), dtype=np.uint64)
        # non-null stream so the library's finalize-pipeline aux stream
        # can overlap (the legacy null stream serializes against it)
        torch.cuda.synchronize(dev)
        rep_stream = torch.cuda.Stream(dev)
        lib().gfrs_set_stream(enc._ctx, rep_stream.cuda_stream)
        ev_stream = rep_stream

    fused = (args.workload == "encode" and with_crc and not args.no_fused)

    def _record(ev):
        if ev_stream is not None:
            ev.record(ev_stream)
        else:
            ev.record()

    def step(events=None):
        if events:
            _record(events[0])
        if fused:
            # single-pass: parity + framed images, data read once
            enc.encode_frame_batch(framed, batch)
            if events:
                _record(events[1])
        elif args.workload == "encode":
            enc.encode_batch(batch)
            if events:
                _record(events[1])
            if with_crc:
                codec.encode_batch(framed, flat)
        elif args.workload == "repair":
            enc.repair_batch(batch, bad, imgs, rbids, rvuids)
            if events:
                _record(events[1])
        else:
            # optional split of the rebuild into several queued launches
            # (measured: within noise of one launch once both legs sit on
            # non-null streams — co-residence comes from the two HW
            # queues, not from chunking)
            ch = int(os.environ.get("GFRS_BENCH_RECON_CHUNK", "0")) or ns
            for lo in range(0, ns, ch):
                enc.reconstruct_batch(batch[lo:lo + ch], bad)
            if with_crc:
                # blocks until the verify stream drains; the trailing
                # event on the rebuild stream then closes the window over
                # BOTH overlapped legs (max of the two streams' ends)
                codec.verify_batch(framed[:, :enc_sz])
            if events:
                _record(events[1])

    # warmup
    for _ in range(args.warmup):
        step()
    torch.cuda.synchronize(dev)
    dist.barrier()

    evs = [(torch.cuda.Event(enable_timing=True),
            torch.cuda.Event(enable_timing=True)) for _ in range(args.steps)]
    t0 = time.perf_counter()
    for i in range(args.steps):
        step(evs[i])
    torch.cuda.synchronize(dev)
    elapsed = time.perf_counter() - t0
    dist.barrier()

    # MAX over ranks
    if d is not None:
        import torch.distributed as td
        # gloo (CPU smoke of the multi-rank path) needs host tensors
        tdev = dev if td.get_backend() == "nccl" else "cpu"
        te = torch.tensor([elapsed], device=tdev)
        td.all_reduce(te, op=td.ReduceOp.MAX)
        elapsed = float(te.item())

    enc_ms = [a.elapsed_time(b) for a, b in evs]
    rec = {"rank": rank, "stripes": ns, "src_bytes": t.N * S * ns * args.steps}
    recs = dist.allgather_records(rec)

    if rank == 0:
        total_src = sum(r["src_bytes"] for r in recs)
        value = total_src / GIB / elapsed
        total_shard = total_src / t.N * t.total / GIB / elapsed
        if args.workload == "repair":
            kind = "repair[%s]" % ",".join(str(b) for b in bad)
        elif args.workload != "encode":
            kind = "reconstruct[%d]" % args.bad_idx
        elif fused:
            kind = "fused encode+frame"
        else:
            kind = "encode"
        ssz = ("%d KiB" % args.shard_kib) if args.shard_kib else \
            ("%d MiB" % args.shard_mib)
        workload = "RS(%d+%d%s) %s%s, %s shards, %d stripes/GPU" % (
            t.N, t.M, "+L%d" % t.L if t.L else "", kind,
            "+crc32block" if (with_crc and not fused
                              and args.workload != "repair") else "",
            ssz, ns)
        # roofline of the dominant kernel, one launch per step:
        #   encode: read k·S, write (m+l)·S per stripe
        #   fused encode+frame: read k·S, write (k+m)·(S+4·fps) per stripe
        #   1-shard reconstruct: read k·S, write 1·S
        fps = -(-S // 65532)
        if args.workload == "repair":
            # read k inputs + (m-nbad) check shards, write nbad framed
            alg_bytes = float((t.N + t.M - len(bad)) * S +
                              len(bad) * (S + 4 * fps)) * ns
        elif args.workload != "encode":
            # reconstruct kernel reads k shards + writes nbad; the CRC
            # verify leg (overlapped on its own stream) reads all k+m
            # framed images — the events window spans both legs, so the
            # roofline covers both legs' algorithmic bytes
            alg_bytes = float((t.N + len(bad)) * S * ns)
            if with_crc:
                alg_bytes += float(t.total * (S + 4 * fps) * ns)
        elif fused:
            alg_bytes = float((t.N + t.total) * S + 4 * fps * t.total) * ns
        else:
            alg_bytes = float(t.total * S * ns)
        avg_enc_s = (sum(enc_ms) / len(enc_ms)) / 1e3
        achieved = alg_bytes / avg_enc_s
        traffic = hbm_traffic_lookup(workload)
        cpu_base = None
        if (not args.skip_cpu_baseline and world == 1
                and args.workload == "encode"):
            cpu_base = cpu_baseline_leg(t, S, args.cpu_sample_stripes, with_crc)
        out = {
            "metric": "GiB/s encode+CRC throughput, RS(k+m) per stripe, "
                      "1/2/4/8 MI355X vs CPU",
            "value": round(value, 2),
            "unit": "GiB/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed * 1e3 / args.steps, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "u8",
            "data": "synthetic",
            "config": {
                "workload": workload,
                "codemode": "EC%dP%d" % (t.N, t.M),
                "shard_size": S,
                "stripes_per_gpu": ns,
                "crc_included": with_crc,
                "block_len": 65536,
                "value_convention": "source GiB/s = k*S*stripes/t",
                "total_shard_gib_s": round(total_shard, 2),
                "rs_kernel_ms": round(sum(enc_ms) / len(enc_ms), 3),
            },
            "roofline": {
                "bound": "hbm",
                "achieved": round(achieved / 1e9, 1),
                "peak": round(HBM_PEAK / 1e9, 1),
                "unit": "GB/s",
                "frac": round(achieved / HBM_PEAK, 4),
                "traffic": traffic,
            },
            "cpu_baseline": cpu_base,
        }
        print(json.dumps(out))

    if d is not None:
        import torch.distributed as td
        td.destroy_process_group()


if __name__ == "__main__":
    main()
