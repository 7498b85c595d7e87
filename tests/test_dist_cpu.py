"""Multi-process CPU coverage of the distributed path: stripe sharding
arithmetic plus the barrier + allgather-of-records exchange bench.py uses,
over gloo with world_size=2 (runs in the no-GPU container)."""
import os
import subprocess
import sys

import pytest

from cubefs_amd.dist import shard_range


def test_shard_range_covers_all():
    for nstripes in (1, 7, 1024, 65536):
        for world in (1, 2, 3, 8):
            seen = []
            for r in range(world):
                lo, hi = shard_range(nstripes, r, world)
                assert 0 <= lo <= hi <= nstripes
                seen.extend(range(lo, hi))
            assert seen == list(range(nstripes))
            sizes = [shard_range(nstripes, r, world) for r in range(world)]
            widths = [h - l for l, h in sizes]
            assert max(widths) - min(widths) <= 1


_WORKER = r"""
import os, sys
import torch.distributed as dist
from cubefs_amd.dist import allgather_records, barrier, env_rank_world, init_process_group, shard_range

init_process_group("gloo")
rank, world = env_rank_world()
lo, hi = shard_range(100, rank, world)
barrier()
recs = allgather_records({"rank": rank, "stripes": hi - lo, "bytes": (hi - lo) * 7})
assert len(recs) == world, recs
assert sum(r["stripes"] for r in recs) == 100, recs
assert [r["rank"] for r in recs] == list(range(world)), recs
barrier()
dist.destroy_process_group()
print("RANK_OK", rank)
"""


def _run_world(tmp_path, script_text, world, port):
    script = tmp_path / "worker.py"
    script.write_text(script_text)
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    env["MASTER_PORT"] = str(port)
    env["PYTHONPATH"] = repo + os.pathsep + env.get("PYTHONPATH", "")
    procs = []
    for r in range(world):
        e = dict(env, RANK=str(r), WORLD_SIZE=str(world), LOCAL_RANK=str(r))
        procs.append(subprocess.Popen([sys.executable, str(script)], env=e,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT,
                                      cwd=repo))
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=180)
        outs.append(out.decode())
    return procs, outs


def test_gloo_allgather_ws2(tmp_path):
    procs, outs = _run_world(tmp_path, _WORKER, 2, 29511)
    for r, (p, out) in enumerate(zip(procs, outs)):
        assert p.returncode == 0, out
        assert "RANK_OK %d" % r in out, out


def test_peer_fetch_plan_pairs():
    """send/recv plans pair up exactly across ranks (plan math only)."""
    from cubefs_amd.dist import peer_fetch_plan, stripe_rank
    nstripes, world = 37, 4
    # synthetic staging layout: shard (s, i) parked on rank (s + i) % world
    needed = [(s, i, (s + i) % world)
              for s in range(nstripes) for i in range(6)]
    # stripe_rank must invert shard_range
    for s in range(nstripes):
        r = stripe_rank(nstripes, s, world)
        lo, hi = shard_range(nstripes, r, world)
        assert lo <= s < hi, (s, r)
    plans = [peer_fetch_plan(nstripes, world, r, needed)
             for r in range(world)]
    # every recv on rank r from peer p matches a send on p to r, in order
    for r in range(world):
        recv, _ = plans[r]
        for p in range(world):
            want = [(s, i) for peer, s, i in recv if peer == p]
            have = [(s, i) for peer, s, i in plans[p][1] if peer == r]
            assert want == have, (r, p)
    # locally-held shards are never exchanged
    for r in range(world):
        recv, send = plans[r]
        assert all(peer != r for peer, _, _ in recv + send)


_P2P_WORKER = r"""
import numpy as np
import torch
import torch.distributed as dist
from cubefs_amd.dist import (barrier, env_rank_world, exchange_peer_shards,
                             init_process_group, peer_fetch_plan, shard_range)

init_process_group("gloo")
rank, world = env_rank_world()
NS, K, SLEN = 13, 6, 4096
# staging layout: shard (s, i) lives on rank (s + i) % world; its bytes are
# a deterministic function of (s, i) so the receiver can verify provenance
def shard_bytes(s, i):
    rng = np.random.default_rng(s * 100 + i)
    return torch.from_numpy(rng.integers(0, 256, SLEN, dtype=np.uint8))

needed = [(s, i, (s + i) % world) for s in range(NS) for i in range(K)]
local = {(s, i): shard_bytes(s, i) for s, i, o in needed if o == rank}
lo, hi = shard_range(NS, rank, world)
dst = {(s, i): torch.zeros(SLEN, dtype=torch.uint8)
       for s, i, o in needed if lo <= s < hi and o != rank}
recv, send = peer_fetch_plan(NS, world, rank, needed)
barrier()
exchange_peer_shards(recv, send,
                     get_local=lambda s, i: local[(s, i)],
                     get_dst=lambda s, i: dst[(s, i)])
barrier()
# every fetched shard must be bit-identical to its origin bytes
for (s, i), t in dst.items():
    assert torch.equal(t, shard_bytes(s, i)), (rank, s, i)
# and the rank can now assemble every tasklet's full source set
for s in range(lo, hi):
    for i in range(K):
        t = local.get((s, i)) if (s + i) % world == rank else dst[(s, i)]
        assert t is not None and torch.equal(t, shard_bytes(s, i))
barrier()
dist.destroy_process_group()
print("P2P_OK", rank)
"""


def test_gloo_peer_exchange_ws4(tmp_path):
    """World-size-4 staging exchange: the xGMI p2p path's plan + pairing
    over gloo (the GPU box runs the identical code over RCCL)."""
    procs, outs = _run_world(tmp_path, _P2P_WORKER, 4, 29517)
    for r, (p, out) in enumerate(zip(procs, outs)):
        assert p.returncode == 0, out
        assert "P2P_OK %d" % r in out, out
