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
        backend = "nccl" if torch.cuda.is_available() else "gloo"
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
