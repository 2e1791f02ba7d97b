"""RCCL-on-silicon smoke (VERDICT round-1 item 2): run the distributed
gateway fabric as a REAL torchrun world on the GPU box.

On a multi-GPU node the world uses the nccl backend (= RCCL on ROCm): the
bus pump's all_to_all rounds, broadcast invalidation and cross-rank batch
forwarding all ride RCCL. On a single-GPU box RCCL refuses two ranks on
one device ("Duplicate GPU detected"), so the worker falls back to gloo
for the COLLECTIVES while both ranks still run the full GPU pipeline on
cuda:0 — proving the multi-process GPU gateway path end to end; the test
asserts nccl was used whenever >=2 devices exist.

The driver's round-end 8-GPU run executes this same test with 8 visible
devices, where the nccl branch is mandatory.
"""

import json
import os
import subprocess
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

WORKER = r"""
import asyncio, json, os, sys
import torch
import torch.distributed as dist

rank = int(os.environ["RANK"]); world = int(os.environ["WORLD_SIZE"])
ndev = torch.cuda.device_count()
backend = "nccl" if ndev >= world else os.environ.get("FALLBACK_BACKEND", "gloo")
os.environ["FORGE_DIST_BACKEND"] = backend
dev = rank % max(ndev, 1)
torch.cuda.set_device(dev)

from mcp_context_forge_amd.config import Settings
from mcp_context_forge_amd.engine import GatewayEngine
from mcp_context_forge_amd.parallel import collectives
from mcp_context_forge_amd.parallel.runtime import DistributedGateway
from mcp_context_forge_amd.services.upstream import NativeInProcUpstream

r, w = collectives.init_from_env()
assert (r, w) == (rank, world)

async def main():
    # raw-fabric proof first: all_to_all_bytes over the real backend
    buckets = [f"from{rank}to{d}".encode() * 64 for d in range(world)]
    got = collectives.all_to_all_bytes(buckets)
    for src in range(world):
        assert got[src] == f"from{src}to{rank}".encode() * 64, (rank, src)

    engine = GatewayEngine(Settings(database_url="sqlite://", federation_enabled=False,
                                    auth_required=False, gpu_enabled=True),
                           rank=rank, world_size=world)
    dg = DistributedGateway(engine, cadence_us=2000)
    up = NativeInProcUpstream(name=f"up{rank}")
    await engine.gateway_service.register_gateway(
        name=f"up{rank}", url=f"inproc://up{rank}", client=up, owner_rank=rank)
    await dg.sync_tool_ownership()
    await dg.start()
    assert engine.enable_gpu(), "GPU pipeline must attach on every rank"
    try:
        other = (rank + 1) % world
        raws = []
        for i in range(256):
            target = rank if i % 2 == 0 else other
            raws.append(json.dumps({
                "jsonrpc": "2.0", "id": i, "method": "tools/call",
                "params": {"name": f"up{target}-echo", "arguments": {"m": f"r{rank}i{i}"}},
            }).encode())
        import time
        t0 = time.monotonic()
        outs = await engine.process_rpc_batch(raws, users=[f"u{rank}"] * len(raws))
        dt = time.monotonic() - t0
        assert len(outs) == 256
        for i, o in enumerate(outs):
            res = json.loads(o)
            assert "result" in res, (i, res)
            assert f"r{rank}i{i}" in res["result"]["content"][0]["text"], (i, res)
        # publish/invalidation channel over the same fabric
        engine.invalidate_peers("registry")
        await asyncio.to_thread(collectives.barrier)
        if rank == 0:
            print(json.dumps({"ok": True, "backend": dist.get_backend(), "world": world,
                              "devices": ndev, "batch_ms": round(dt * 1000, 2),
                              "bus": dg.bus.stats() if dg.bus else None,
                              "gpu_batches": engine.gpu_pipeline.batches}))
    finally:
        await dg.stop()

asyncio.run(main())
dist.destroy_process_group()
"""


def test_world2_fabric_on_gpu(tmp_path):
    if not torch.cuda.is_available():
        pytest.skip("no ROCm device")
    script = tmp_path / "rccl_worker.py"
    script.write_text(WORKER)
    env = dict(os.environ, PYTHONPATH="/root/repo")
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1", "--nproc-per-node", "2",
         "--master-addr", "127.0.0.1", "--master-port", "29677", str(script)],
        capture_output=True, text=True, timeout=420, env=env)
    assert out.returncode == 0, out.stdout[-3000:] + out.stderr[-3000:]
    line = [ln for ln in out.stdout.splitlines() if ln.startswith('{"ok"')]
    assert line, out.stdout[-2000:]
    res = json.loads(line[0])
    assert res["ok"] and res["world"] == 2
    assert res["gpu_batches"] >= 1
    if res["devices"] >= 2:
        # multi-GPU node: the fabric MUST be RCCL
        assert res["backend"] == "nccl", res
    print("FABRIC:", json.dumps(res))
