"""Cross-GPU gateway fabric: RCCL over xGMI.

Reference analog: the reference distributes work across
gunicorn workers with Redis pub/sub — session-affinity RPC
(services/session_affinity.py:747 forward_request_to_owner), broadcast
invalidation (plugins/__init__.py:46), leader election
(gateway_service.py:1254) and metric aggregation. Here the fabric is
torch.distributed over RCCL (backend "nccl" IS RCCL on ROCm): one gateway
rank per GPU, request fan-out via all_to_all over the 7 point-to-point xGMI
links, registry/plugin invalidation via broadcast, metric aggregation via
all_reduce. Leader = rank 0 by construction. gloo backend keeps every path
testable on CPU (multi-process, world_size>1, no GPU).
"""

from __future__ import annotations

import os
import pickle
from typing import Any, List, Optional

import torch
import torch.distributed as dist


def init_from_env(device: Optional[str] = None) -> tuple[int, int]:
    """Initialize torch.distributed from torchrun env; returns (rank, world)."""
    if dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        return 0, 1
    # FORGE_DIST_BACKEND forces gloo/nccl (e.g. world-2 on a single-GPU box:
    # RCCL refuses two ranks on one device, gloo keeps the GPU pipeline
    # testable with the bus pump live)
    backend = os.environ.get("FORGE_DIST_BACKEND") or \
        ("nccl" if torch.cuda.is_available() else "gloo")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29517")
    dist.init_process_group(backend=backend)
    rank = dist.get_rank()
    if backend == "nccl":
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank % torch.cuda.device_count())))
    return rank, world


def _device() -> torch.device:
    if dist.is_initialized() and dist.get_backend() == "nccl":
        return torch.device("cuda")
    return torch.device("cpu")


def new_bus_group():
    """Dedicated communicator for the affinity bus so its rounds never
    interleave with application collectives (ordering rule)."""
    if not dist.is_initialized():
        return None
    return dist.new_group(backend=dist.get_backend())


def all_to_all_bytes(buckets: List[bytes], group=None) -> List[bytes]:
    """Exchange byte buffers: buckets[d] goes to rank d; returns what each
    rank sent to us (index = source rank).

    RCCL path: two all_to_all_single calls (sizes, then payload) on device
    tensors — the federation fan-out of BASELINE.json config 3. The xGMI
    topology note favors direct all-to-all over rings for
    this small-message latency-sensitive traffic.
    gloo path (CPU tests): all_to_all_single on CPU tensors when supported,
    else an all_gather fallback.
    """
    world = dist.get_world_size()
    assert len(buckets) == world
    dev = _device()

    send_sizes = torch.tensor([len(b) for b in buckets], dtype=torch.int64, device=dev)
    recv_sizes = torch.empty(world, dtype=torch.int64, device=dev)
    try:
        dist.all_to_all_single(recv_sizes, send_sizes, group=group)
    except RuntimeError:
        # gloo without alltoall support: all_gather the full matrix
        gathered: List[Any] = [None] * world
        dist.all_gather_object(gathered, buckets, group=group)
        me = dist.get_rank()
        return [gathered[src][me] for src in range(world)]

    send_buf = torch.frombuffer(bytearray(b"".join(buckets)) or bytearray(1), dtype=torch.uint8).to(dev)
    if sum(len(b) for b in buckets) == 0:
        send_buf = torch.zeros(0, dtype=torch.uint8, device=dev)
    recv_total = int(recv_sizes.sum().item())
    recv_buf = torch.empty(recv_total, dtype=torch.uint8, device=dev)
    in_splits = [len(b) for b in buckets]
    out_splits = [int(x) for x in recv_sizes.tolist()]
    dist.all_to_all_single(recv_buf, send_buf, out_splits, in_splits, group=group)
    flat = recv_buf.cpu().numpy().tobytes()
    out: List[bytes] = []
    off = 0
    for s in out_splits:
        out.append(flat[off:off + s])
        off += s
    return out


def all_to_all_objects(buckets: List[List[Any]], group=None) -> List[List[Any]]:
    """Object-level fan-out: buckets[d] (a list) is delivered to rank d."""
    payloads = [pickle.dumps(b) for b in buckets]
    received = all_to_all_bytes(payloads, group=group)
    return [pickle.loads(p) if p else [] for p in received]


def broadcast_object(obj: Any, src: int = 0) -> Any:
    """Registry/plugin-config invalidation channel (reference: signed Redis
    pub/sub, plugins/__init__.py:103-120)."""
    box = [obj if dist.get_rank() == src else None]
    dist.broadcast_object_list(box, src=src)
    return box[0]


def all_reduce_counters(counters: dict[str, float]) -> dict[str, float]:
    """Aggregate metric counters across ranks (reference: per-worker metric
    rows flushed to a shared DB; here one all_reduce)."""
    keys = sorted(counters.keys())
    t = torch.tensor([counters[k] for k in keys], dtype=torch.float64, device=_device())
    dist.all_reduce(t)
    return {k: float(v) for k, v in zip(keys, t.tolist())}


def barrier() -> None:
    if dist.is_initialized():
        dist.barrier()


def rank_world() -> tuple[int, int]:
    if dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    return 0, 1
