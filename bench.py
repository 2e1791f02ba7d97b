#!/usr/bin/env python3
"""Flagship benchmark: JSON-RPC tools/call throughput (BASELINE.json metric:
tool-call req/s + p50 latency, 1/2/4/8 MI355X).

DEFAULT MODE (--mode http) measures the number the reference publishes:
requests over REAL TCP sockets. Per rank, a native C++ edge
(ops/csrc/edge.cpp) serves HTTP/1.1 on 127.0.0.1 and a native closed-loop
load generator (ops/csrc/forge_hey.cpp) drives it at --connections
concurrency (default 1000 — the reference's sustained-load knee,
crates/mcp_runtime/STATUS.md:166). Every request carries an Authorization
header and is authenticated (reference numbers pay auth middleware too);
bodies run the full gateway pipeline: parse -> GPU plugin chain (json
guard, deny/pii/regex/harm DFA scans, hashed featurize, bf16-MFMA
moderation classifier) -> federation fan-out to 64 in-proc MCP upstreams
per rank -> post chain -> serialize. With N>1 ranks, requests targeting
tools owned by other GPUs ride the RCCL bus over xGMI (BASELINE config 3)
and their responses ride back.

--mode engine measures the in-process engine ceiling (no sockets) — the
pipeline metric, clearly labeled as such in the output config.

Contract: W untimed warmup steps, then EXACTLY K timed steps bracketed by
barrier + torch.cuda.synchronize on both sides (the load generator holds at
a WARM/GO handshake over stdin so the brackets are real); elapsed = MAX
over ranks; rank 0 prints one JSON line; value = whole-job req/s across all
N GPUs.
"""

from __future__ import annotations

import argparse
import asyncio
import json
import os
import random
import statistics
import sys
import tempfile
import time
from typing import List, Optional

REFERENCE_BEST_RPS = 10454.16  # MCP tools-only 60s/1000u, Rust full mode


def make_request(rng: random.Random, tool_names: List[str], rid: int, flagged_frac: float,
                 pad_bytes: int = 0, unknown_frac: float = 0.0,
                 nonascii_frac: float = 0.0) -> bytes:
    name = tool_names[rng.randrange(len(tool_names))]
    if unknown_frac and rng.random() < unknown_frac:
        name = f"no-such-tool-{rng.randrange(64)}"
    r = rng.random()
    if name.endswith("convert_time"):
        args = {
            "time": f"2026-01-{rng.randrange(1,29):02d}T{rng.randrange(24):02d}:{rng.randrange(60):02d}:{rng.randrange(60):02d}Z",
            "source_timezone": rng.choice(["UTC", "America/New_York", "Europe/London", "Asia/Tokyo"]),
            "target_timezone": rng.choice(["UTC", "America/Chicago", "Europe/Berlin", "Asia/Kolkata"]),
        }
    elif name.endswith("get_system_time"):
        args = {"timezone": rng.choice(["UTC", "America/New_York", "Europe/Paris"])}
    else:
        args = {"msg": f"payload {rng.randrange(1 << 30)} lorem ipsum dolor sit amet", "n": rng.randrange(100)}
    if r < flagged_frac * 0.5:
        args["note"] = f"contact me at user{rng.randrange(1000)}@example.com"  # PII slow path
    if pad_bytes:
        args["payload"] = "x" * pad_bytes  # size probe: larger real-world bodies
    if nonascii_frac and rng.random() < nonascii_frac:
        args["intl"] = "café ünïcode 日本語 №" + str(rid % 97)  # punt-path pressure
    elif r < flagged_frac:
        args["note"] = "this is   spaced\ttext"  # normalizer slow path
    return json.dumps(
        {"jsonrpc": "2.0", "id": rid, "method": "tools/call", "params": {"name": name, "arguments": args}},
        separators=(",", ":"),
    ).encode()


def local_tool_names(rank: int, upstreams: int) -> List[str]:
    names = []
    for u in range(upstreams):
        for t in ("convert_time", "get_system_time", "echo"):
            names.append(f"up{rank}-{u}-{t}")
    return names


async def build_engine(rank: int, world: int, upstreams: int, use_gpu: bool,
                       semcache: bool, auth_required: bool):
    from mcp_context_forge_amd.config import Settings
    from mcp_context_forge_amd.engine import GatewayEngine
    from mcp_context_forge_amd.plugins.loader import default_chain_specs, load_plugin_manager
    from mcp_context_forge_amd.services.upstream import NativeInProcUpstream

    settings = Settings(
        database_url="sqlite://",
        federation_enabled=False,  # no background health loop during timing
        auth_required=auth_required,
        gpu_enabled=use_gpu,
        rank=rank,
        world_size=world,
    )
    specs = default_chain_specs()
    if semcache:
        # the semantic cache substitutes results ONLY for explicitly
        # allowlisted tools (deterministic given args here); the default
        # bench keeps the allowlist empty like the reference's default
        all_tools = [n for r in range(world) for n in local_tool_names(r, upstreams)]
        for s in specs:
            if s["name"] == "response_cache_by_prompt":
                s["config"] = {"cacheable_tools": all_tools}
    pm = load_plugin_manager(specs=specs)
    engine = GatewayEngine(settings, plugin_manager=pm, rank=rank, world_size=world)
    dg = None
    if world > 1:
        from mcp_context_forge_amd.parallel.runtime import DistributedGateway

        dg = DistributedGateway(engine, cadence_us=1000)
    # 64 federated upstreams per rank (BASELINE config 2); each exposes 3
    # tools. Native C++ upstreams — the reference's benchmark upstream
    # (fast_time_server) is a native Go binary; this is its in-proc analog.
    for u in range(upstreams):
        up = NativeInProcUpstream(name=f"up{rank}-{u}")
        await engine.gateway_service.register_gateway(
            name=f"up{rank}-{u}", url=f"inproc://up{rank}-{u}", client=up, owner_rank=rank)
    if dg is not None:
        # collective on the default group; MUST run before the bus pump starts
        await dg.sync_tool_ownership()
        await dg.start()
    else:
        await engine.startup()
    if use_gpu:
        if not engine.enable_gpu():
            raise RuntimeError("GPU requested but pipeline unavailable")
    return engine, dg


async def sync(collectives, use_gpu: bool):
    """Timed-region bracket: barrier + device sync. The barrier runs in a
    thread — a rank that arrives early must keep its event loop alive to
    serve bus-forwarded requests from ranks still finishing their steps
    (documented pitfall, parallel/runtime.py)."""
    import torch

    await asyncio.to_thread(collectives.barrier)
    if use_gpu:
        torch.cuda.synchronize()


def max_over_ranks(value: float, world: int, use_gpu: bool) -> float:
    if world <= 1:
        return value
    import torch
    import torch.distributed as dist

    t = torch.tensor([value], dtype=torch.float64,
                     device="cuda" if (use_gpu and dist.get_backend() == "nccl") else "cpu")
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return float(t.item())


def gather_objects(obj, world: int):
    if world <= 1:
        return [obj]
    import torch.distributed as dist

    box = [None] * world
    dist.all_gather_object(box, obj)
    return box


# ---------------------------------------------------------------------------
# HTTP mode (headline): native edge + native load generator, real sockets
# ---------------------------------------------------------------------------


async def run_http(args, rank: int, world: int, use_gpu: bool, R: int):
    import torch

    from mcp_context_forge_amd.auth.service import AuthService
    from mcp_context_forge_amd.parallel import collectives
    from mcp_context_forge_amd.transports.native_edge import NativeEdge
    from mcp_context_forge_amd.ops.build import HEY, build_hey

    build_hey(verbose=False)
    engine, dg = await build_engine(rank, world, args.upstreams, use_gpu,
                                    args.semcache, auth_required=True)
    auth = AuthService(engine.db, engine.settings, token_blocklist=engine.token_blocklist)
    auth.bootstrap_admin()
    token = auth.create_api_token("admin@example.com", "bench")

    port = args.port_base + rank
    threads = args.edge_threads or min(8, max(2, (os.cpu_count() or 16) // (2 * max(world, 1))))
    edge = NativeEdge(engine, app=None, auth=auth, port=port, threads=threads,
                      depth=args.edge_depth)
    await edge.start()

    # payload corpus: P distinct requests, rotated by the generator; targets
    # are uniform over ALL ranks' tools so world>1 drives the RCCL fan-out
    all_names = [n for r in range(world) for n in local_tool_names(r, args.upstreams)]
    rng = random.Random(1000003 * (rank + 1))
    with tempfile.NamedTemporaryFile("wb", suffix=".jsonl", delete=False) as f:
        payload_file = f.name
        for i in range(args.payloads):
            f.write(make_request(rng, all_names, i, args.flagged_frac, args.payload_bytes,
                                 args.unknown_frac, args.nonascii_frac) + b"\n")

    hey_threads = max(2, min(6, (os.cpu_count() or 16) // (2 * max(world, 1))))
    proc = await asyncio.create_subprocess_exec(
        str(HEY), "--host", "127.0.0.1", "--port", str(port), "--path", "/rpc",
        "--connections", str(args.connections), "--threads", str(hey_threads),
        "--requests-per-step", str(R), "--warmup", str(args.warmup),
        "--steps", str(args.steps), "--payload-file", payload_file,
        "--auth", f"Bearer {token}",
        stdin=asyncio.subprocess.PIPE, stdout=asyncio.subprocess.PIPE)

    line = await proc.stdout.readline()  # warmup steps run now
    assert line.strip() == b"WARM", line
    await sync(collectives, use_gpu)     # timed-region opening bracket
    t_start = time.monotonic()
    proc.stdin.write(b"GO\n")
    await proc.stdin.drain()
    out_line = await proc.stdout.readline()
    await proc.wait()
    res = json.loads(out_line)
    await sync(collectives, use_gpu)     # closing bracket
    elapsed_here = time.monotonic() - t_start
    os.unlink(payload_file)

    elapsed = max_over_ranks(res["elapsed_s"], world, use_gpu)
    all_res = gather_objects(res, world)

    if rank == 0:
        total = R * args.steps * world
        value = total / elapsed
        stats = engine.gpu_pipeline.stats() if engine.gpu_pipeline else {}
        out = {
            "metric": "tool_call_req_per_s",
            "value": round(value, 2),
            "unit": "req/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000.0, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": round(value / REFERENCE_BEST_RPS, 3),
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": "gateway-pipeline(deny+regex+pii+schema+moderation-mlp-4096x1024x8"
                         + ("+semcache-65536x4096" if args.semcache else "") + "+toon)",
                "global_batch": R * world,
                "seq_len": 0,
                "parallelism": f"shard{world}" if world > 1 else "single",
                "transport": "http/1.1 real TCP sockets (native C++ edge), auth enforced per request",
                "load_shape": f"{args.connections} concurrent connections per rank (reference knee), "
                              f"closed-loop native generator",
                "p50_ms": all_res[0]["p50_ms"],
                "p90_ms": all_res[0]["p90_ms"],
                "p99_ms": round(max(r["p99_ms"] for r in all_res), 3),
                "errors": sum(r["errors"] for r in all_res),
                "non200": sum(r["non200"] for r in all_res),
                "upstreams": args.upstreams * world,
                "requests_per_step_per_rank": R,
                "flagged_frac": args.flagged_frac,
                "gpu_path": bool(engine.gpu_pipeline),
                "semcache": bool(args.semcache),
                "edge": edge.stats(),
                "pipeline_stats": {k: v for k, v in stats.items() if k != "banks"},
                "elapsed_wallclock_s": round(elapsed_here, 3),
                "reference_metric": "MCP tools/call RPS over HTTP (BASELINE.md: 10454.16 burst / "
                                    "6350.12 sustained, x86 CPU compose, no plugin chain)",
            },
        }
        print(json.dumps(out))
    await edge.stop()
    if dg is not None:
        await dg.stop()
    else:
        await engine.shutdown()


# ---------------------------------------------------------------------------
# engine mode: in-process batches, no sockets (pipeline ceiling, secondary)
# ---------------------------------------------------------------------------


async def run_engine_only(args, rank: int, world: int, use_gpu: bool, R: int):
    import torch

    from mcp_context_forge_amd.parallel import collectives

    engine, dg = await build_engine(rank, world, args.upstreams, use_gpu,
                                    args.semcache, auth_required=False)
    all_names = [local_tool_names(r, args.upstreams) for r in range(world)]

    def gen_step(step: int):
        rng = random.Random(1000003 * (rank + 1) + step)
        raws = []
        for i in range(R):
            d = rng.randrange(world)
            raws.append(make_request(rng, all_names[d], step * R + i, args.flagged_frac,
                                      args.payload_bytes, args.unknown_frac, args.nonascii_frac))
        return raws

    warm_data = [gen_step(s) for s in range(args.warmup)]
    step_data = [gen_step(10_000 + s) for s in range(args.steps)]

    for raws in warm_data:
        out = await engine.process_rpc_batch(raws)
        assert len(out) == len(raws)
    await sync(collectives, use_gpu)

    prof = None
    if os.environ.get("FORGE_CPROFILE"):  # timed-loop-only python profile
        import cProfile

        prof = cProfile.Profile()
        prof.enable()
    conc = int(os.environ.get("FORGE_ENGINE_CONC", "1"))
    step_times: List[float] = []
    t_start = time.monotonic()
    for s in range(args.steps):
        t0 = time.monotonic()
        if conc > 1:
            k = (len(step_data[s]) + conc - 1) // conc
            parts = [step_data[s][i:i + k] for i in range(0, len(step_data[s]), k)]
            await asyncio.gather(*(engine.process_rpc_batch(p) for p in parts))
        else:
            await engine.process_rpc_batch(step_data[s])
        if use_gpu:
            torch.cuda.synchronize()
        step_times.append(time.monotonic() - t0)
    if prof is not None:
        prof.disable()
        prof.dump_stats(os.environ["FORGE_CPROFILE"])
    await sync(collectives, use_gpu)
    elapsed = max_over_ranks(time.monotonic() - t_start, world, use_gpu)

    if rank == 0:
        value = R * args.steps * world / elapsed
        stats = engine.gpu_pipeline.stats() if engine.gpu_pipeline else {}
        out = {
            "metric": "tool_call_req_per_s",
            "value": round(value, 2),
            "unit": "req/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000.0, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": round(value / REFERENCE_BEST_RPS, 3),
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": "gateway-pipeline(deny+regex+pii+schema+moderation-mlp-4096x1024x8"
                         + ("+semcache-65536x4096" if args.semcache else "") + "+toon)",
                "global_batch": R * world,
                "seq_len": 0,
                "parallelism": f"shard{world}" if world > 1 else "single",
                "transport": "ENGINE-ONLY (in-process batches, no sockets, no auth) — "
                             "pipeline ceiling, NOT comparable to the reference's HTTP RPS; "
                             "the http mode (default) is the like-for-like number",
                "p50_batch_ms": round(statistics.median(step_times) * 1000.0, 3),
                "upstreams": args.upstreams * world,
                "requests_per_step_per_rank": R,
                "flagged_frac": args.flagged_frac,
                "gpu_path": bool(engine.gpu_pipeline),
                "semcache": bool(args.semcache),
                "pipeline_stats": {k: v for k, v in stats.items() if k != "banks"},
            },
        }
        print(json.dumps(out))
    if dg is not None:
        await dg.stop()
    else:
        await engine.shutdown()


async def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--requests-per-step", type=int, default=0, help="per rank; 0 = auto")
    ap.add_argument("--mode", choices=["http", "engine"], default="http")
    ap.add_argument("--connections", type=int, default=1000, help="HTTP concurrency per rank")
    ap.add_argument("--payloads", type=int, default=2048, help="distinct payloads in the corpus")
    ap.add_argument("--port-base", type=int, default=18400)
    ap.add_argument("--edge-threads", type=int, default=0)
    ap.add_argument("--edge-depth", type=int, default=2,
                    help="batches in flight in the edge loop (host/GPU overlap)")
    ap.add_argument("--upstreams", type=int, default=64)
    ap.add_argument("--flagged-frac", type=float, default=0.02)
    ap.add_argument("--payload-bytes", type=int, default=0,
                    help="pad every request's arguments by N bytes (size probe)")
    ap.add_argument("--unknown-frac", type=float, default=0.0,
                    help="fraction of requests targeting unknown tools (error-path probe)")
    ap.add_argument("--nonascii-frac", type=float, default=0.0,
                    help="fraction of requests with non-ASCII args (punt-path probe)")
    ap.add_argument("--semcache", action="store_true",
                    help="allowlist the bench tools in the semantic cache (labeled in output)")
    ap.add_argument("--no-gpu", action="store_true")
    ap.add_argument("--engine-only", action="store_true", help="alias for --mode engine")
    args = ap.parse_args()
    if args.engine_only:
        args.mode = "engine"

    import torch

    from mcp_context_forge_amd.parallel import collectives

    rank, world = collectives.init_from_env()
    use_gpu = torch.cuda.is_available() and not args.no_gpu
    if args.mode == "http":
        R = args.requests_per_step or (100_000 if use_gpu else 2_000)
        await run_http(args, rank, world, use_gpu, R)
    else:
        R = args.requests_per_step or (8192 if use_gpu else 256)
        await run_engine_only(args, rank, world, use_gpu, R)
    if world > 1:
        import torch.distributed as dist

        dist.destroy_process_group()


if __name__ == "__main__":
    sys.exit(asyncio.run(main()))
