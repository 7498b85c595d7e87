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


def test_gloo_allgather_ws2(tmp_path):
    script = tmp_path / "worker.py"
    script.write_text(_WORKER)
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    env["MASTER_PORT"] = "29511"
    env["PYTHONPATH"] = repo + os.pathsep + env.get("PYTHONPATH", "")
    procs = []
    for r in range(2):
        e = dict(env, RANK=str(r), WORLD_SIZE="2", LOCAL_RANK=str(r))
        procs.append(subprocess.Popen([sys.executable, str(script)], env=e,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT,
                                      cwd=os.path.dirname(os.path.dirname(__file__))))
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=120)
        outs.append(out.decode())
    for r, (p, out) in enumerate(zip(procs, outs)):
        assert p.returncode == 0, out
        assert "RANK_OK %d" % r in out, out
