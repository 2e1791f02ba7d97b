"""Multi-process collectives over gloo (world_size 2, CPU) — validates the
RCCL code path by construction (same torch.distributed calls; backend swaps
to nccl on GPU). Reference analog: the Redis pub/sub fabric
(session_affinity / broadcast / metrics) replaced by collectives."""

import json
import os
import subprocess
import sys

WORKER = r"""
import json, os, sys
import torch.distributed as dist
from mcp_context_forge_amd.parallel import collectives

rank, world = collectives.init_from_env()
assert world == 2

# all_to_all_bytes: rank r sends b"r->d" to each d
buckets = [f"{rank}->{d}".encode() * (d + 1) for d in range(world)]
got = collectives.all_to_all_bytes(buckets)
assert got[0] == f"0->{rank}".encode() * (rank + 1), got
assert got[1] == f"1->{rank}".encode() * (rank + 1), got

# all_to_all_objects
objs = [[{"from": rank, "to": d, "i": i} for i in range(3)] for d in range(world)]
got = collectives.all_to_all_objects(objs)
for src in range(world):
    assert [o["from"] for o in got[src]] == [src] * 3
    assert all(o["to"] == rank for o in got[src])

# empty buckets: zero-byte exchanges must not wedge the collective
empty = collectives.all_to_all_bytes([b"", b""])
assert empty == [b"", b""], empty
# rank 0 sends 5 bytes to itself only; rank 1 sends nothing
one_way = collectives.all_to_all_bytes([b"x" * 5 if rank == 0 else b"", b""])
if rank == 0:
    assert one_way == [b"x" * 5, b""], one_way
else:
    assert one_way == [b"", b""], one_way

# broadcast (registry invalidation analog)
cfg = {"plugins_enabled": True, "gen": 42} if rank == 0 else None
out = collectives.broadcast_object(cfg, src=0)
assert out == {"plugins_enabled": True, "gen": 42}

# all_reduce counters (metric aggregation analog)
counters = {"tool_invocations_total": 10.0 * (rank + 1), "tool_errors_total": float(rank)}
agg = collectives.all_reduce_counters(counters)
assert agg["tool_invocations_total"] == 30.0
assert agg["tool_errors_total"] == 1.0

collectives.barrier()
if rank == 0:
    print("COLLECTIVES_OK")
dist.destroy_process_group()
"""


def test_collectives_gloo_world2(tmp_path):
    script = tmp_path / "worker.py"
    script.write_text(WORKER)
    env = dict(os.environ, MASTER_ADDR="127.0.0.1", MASTER_PORT="29611", PYTHONPATH="/root/repo")
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1", "--nproc-per-node", "2",
         "--master-addr", "127.0.0.1", "--master-port", "29611", str(script)],
        capture_output=True, text=True, timeout=240, env=env, cwd="/root/repo")
    assert out.returncode == 0, out.stdout + out.stderr
    assert "COLLECTIVES_OK" in out.stdout


def test_bench_cpu_world2_smoke():
    """The bench's sharded step over gloo world 2 (CPU fallback path)."""
    env = dict(os.environ, PYTHONPATH="/root/repo")
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1", "--nproc-per-node", "2",
         "--master-addr", "127.0.0.1", "--master-port", "29613", "bench.py",
         "--no-gpu", "--steps", "2", "--warmup", "1", "--requests-per-step", "32",
         "--upstreams", "4"],
        capture_output=True, text=True, timeout=300, env=env, cwd="/root/repo")
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    line = [ln for ln in out.stdout.splitlines() if ln.startswith("{")][-1]
    d = json.loads(line)
    assert d["n_gpus"] == 2 and d["value"] > 0
    assert d["config"]["parallelism"] == "shard2"
