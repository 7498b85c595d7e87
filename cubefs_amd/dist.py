"""cubefs_amd.dist — multi-GPU stripe-queue partitioning.

Stripes are fully independent (no inter-stripe data dependence anywhere in
the RS path; blobnode already exploits this per-tasklet,
worker_slice_recover.go:567-568,822).  Partitioning is therefore pure
sharding: round-robin contiguous ranges across ranks, no data-path
collective.  torch.distributed (RCCL over xGMI on the GPU box, gloo in CPU
tests) is used only for the start/stop barrier and a tiny allgather of
per-rank result records — see SURVEY.md §8e.
"""
import os


def env_rank_world():
    """torchrun env (RANK/WORLD_SIZE), defaulting to single-process."""
    return int(os.environ.get("RANK", "0")), int(os.environ.get("WORLD_SIZE", "1"))


def shard_range(nstripes, rank, world):
    """Contiguous stripe range [lo, hi) for this rank; remainder spread over
    the first ranks so sizes differ by at most one."""
    base, rem = divmod(nstripes, world)
    lo = rank * base + min(rank, rem)
    hi = lo + base + (1 if rank < rem else 0)
    return lo, hi


def init_process_group(backend=None):
    import torch
    import torch.distributed as dist
    rank, world = env_rank_world()
    if world == 1:
        return None
    if backend is None:
        # GFRS_DIST_BACKEND=gloo lets the multi-rank path be exercised
        # with several ranks sharing one GPU (RCCL forbids that)
        backend = os.environ.get("GFRS_DIST_BACKEND") or (
            "nccl" if torch.cuda.is_available() else "gloo")
    if not dist.is_initialized():
        dist.init_process_group(backend=backend)
    if backend == "nccl":
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))
    return dist


def allgather_records(record):
    """Gather per-rank result dicts {stripes, bytes, ns, crc_fail...}; the
    only cross-GPU exchange on this path (<1 KiB per rank)."""
    import torch.distributed as dist
    if not dist.is_initialized():
        return [record]
    world = dist.get_world_size()
    out = [None] * world
    dist.all_gather_object(out, record)
    return out


def barrier():
    import torch.distributed as dist
    if dist.is_initialized():
        dist.barrier()


# ---------------------------------------------------------------------------
# Peer shard staging for reconstruct batches (the genDownloadPlans analog,
# worker_slice_recover.go:127-210): a repair tasklet needs k source shards;
# when the repair queue was staged across the node's GPUs, some of a rank's
# tasklets reference shards resident in a PEER GPU's HBM.  Those move with
# RCCL point-to-point sends over single xGMI links (~153 GB/s per link) —
# deliberately not ring collectives, which would be per-link bound and move
# every rank's data.
# ---------------------------------------------------------------------------

def stripe_rank(nstripes, stripe, world):
    """Inverse of shard_range: which rank repairs this stripe."""
    base, rem = divmod(nstripes, world)
    cut = rem * (base + 1)  # first `rem` ranks own base+1 stripes
    if stripe < cut:
        return stripe // (base + 1)
    return rem + (stripe - cut) // base if base else rem


def peer_fetch_plan(nstripes, world, rank, needed):
    """Fetch/send plan for a stripe-range partition.

    needed: iterable of (stripe, shard_idx, owner_rank) listing, for every
    reconstruct tasklet in the queue, each source shard and the rank whose
    HBM holds it (the staging layout the scheduler produced) — identical
    on every rank, like the task list itself.  Returns:
      recv: [(peer, stripe, shard_idx)] — shards this rank must pull for
            the reconstructs it owns (its shard_range);
      send: [(peer, stripe, shard_idx)] — shards this rank holds that a
            peer's reconstructs need.
    Both sides iterate `needed` in the same order, so the per-peer
    isend/irecv queues pair up without a negotiation round-trip.
    """
    lo, hi = shard_range(nstripes, rank, world)
    recv, send = [], []
    for s, i, owner in needed:
        if lo <= s < hi and owner != rank:
            recv.append((owner, s, i))
        elif not (lo <= s < hi) and owner == rank:
            send.append((stripe_rank(nstripes, s, world), s, i))
    return recv, send


def exchange_peer_shards(recv, send, get_local, get_dst, group=None):
    """Run the paired point-to-point exchange.

    get_local(stripe, shard_idx) -> tensor this rank holds (send side);
    get_dst(stripe, shard_idx)   -> tensor to receive into (recv side).
    Uses batched isend/irecv: on the GPU box each pair maps to an RCCL
    point-to-point transfer over the direct xGMI link; in CPU tests the
    same code runs over gloo.
    """
    import torch.distributed as dist
    if not dist.is_initialized() or (not recv and not send):
        return
    ops = []
    for peer, s, i in send:
        ops.append(dist.P2POp(dist.isend, get_local(s, i), peer, group))
    for peer, s, i in recv:
        ops.append(dist.P2POp(dist.irecv, get_dst(s, i), peer, group))
    if ops:
        for w in dist.batch_isend_irecv(ops):
            w.wait()
