#!/usr/bin/env python3
"""BASELINE config 5: gRPC→MCP translate path — 10k-method reflection +
schema_guard batch validate + toon_encoder.

Phase 1 (reported as discovery_s): native reflection against an in-proc
gRPC server exposing 10,000 methods (100 services × 100 methods across 20
descriptor files), schema synthesis from descriptors, and registration of
all 10k tools into the registry (exercises the native toolmap at 10k).

Phase 2 (the metric): R tools/call per step against the translated tools
through the batched GPU pipeline — schema-shape validation on device for
the flat request schemas, invalid-args mix rejected by schema_guard, large
responses (repeated fields) TOON-encoded on the post chain, real gRPC
hops to the in-proc server for dispatched rows.

Output: one bench-style JSON line (metric grpc_translate_req_per_s).
Run (GPU box):  python loadtest/bench_grpc.py --steps 5 --warmup 2
"""

from __future__ import annotations

import argparse
import asyncio
import json
import os
import random
import socket
import statistics
import sys
import time
from concurrent import futures

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

N_FILES = 20
SVC_PER_FILE = 5
METH_PER_SVC = 100  # 20*5*100 = 10,000 methods


def build_files():
    from google.protobuf import descriptor_pb2

    files = []
    for f in range(N_FILES):
        fdp = descriptor_pb2.FileDescriptorProto()
        fdp.name = f"bench/svc{f}.proto"
        fdp.package = f"bench.f{f}"
        fdp.syntax = "proto3"
        req = fdp.message_type.add()
        req.name = "Req"
        fld = req.field.add(); fld.name = "name"; fld.number = 1; fld.type = fld.TYPE_STRING; fld.label = fld.LABEL_OPTIONAL
        fld = req.field.add(); fld.name = "count"; fld.number = 2; fld.type = fld.TYPE_INT32; fld.label = fld.LABEL_OPTIONAL
        rep = fdp.message_type.add()
        rep.name = "Rep"
        fld = rep.field.add(); fld.name = "message"; fld.number = 1; fld.type = fld.TYPE_STRING; fld.label = fld.LABEL_OPTIONAL
        fld = rep.field.add(); fld.name = "echoes"; fld.number = 2; fld.type = fld.TYPE_STRING; fld.label = fld.LABEL_REPEATED
        for s in range(SVC_PER_FILE):
            svc = fdp.service.add()
            svc.name = f"Svc{s}"
            for m in range(METH_PER_SVC):
                meth = svc.method.add()
                meth.name = f"M{m}"
                meth.input_type = f".bench.f{f}.Req"
                meth.output_type = f".bench.f{f}.Rep"
        files.append(fdp)
    return files


def start_server(files):
    import grpc
    from google.protobuf import descriptor_pool, message_factory

    from mcp_context_forge_amd.services.grpc_translate import ReflectionServicer

    pool = descriptor_pool.DescriptorPool()
    for fdp in files:
        pool.Add(fdp)
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=16))
    service_names = []
    for fdp in files:
        req_cls = message_factory.GetMessageClass(pool.FindMessageTypeByName(f"{fdp.package}.Req"))
        rep_cls = message_factory.GetMessageClass(pool.FindMessageTypeByName(f"{fdp.package}.Rep"))

        def handler(request, context, _rep=rep_cls):
            reply = _rep()
            reply.message = f"ok:{request.name}"
            for i in range(request.count):
                reply.echoes.append(f"{request.name}-{i}")
            return reply

        h = grpc.unary_unary_rpc_method_handler(
            handler, request_deserializer=req_cls.FromString,
            response_serializer=lambda m: m.SerializeToString())
        for s in range(SVC_PER_FILE):
            full = f"{fdp.package}.Svc{s}"
            service_names.append(full)
            server.add_generic_rpc_handlers((grpc.method_handlers_generic_handler(
                full, {f"M{m}": h for m in range(METH_PER_SVC)}),))
    ReflectionServicer(pool, service_names, {f.name: f for f in files}).add_to_server(server)
    with socket.socket() as sk:
        sk.bind(("127.0.0.1", 0))
        port = sk.getsockname()[1]
    server.add_insecure_port(f"127.0.0.1:{port}")
    server.start()
    return server, f"127.0.0.1:{port}"


async def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--requests-per-step", type=int, default=0)
    ap.add_argument("--invalid-frac", type=float, default=0.1)
    ap.add_argument("--big-frac", type=float, default=0.2, help="large responses → toon path")
    ap.add_argument("--no-gpu", action="store_true")
    args = ap.parse_args()

    import torch

    from mcp_context_forge_amd.config import Settings
    from mcp_context_forge_amd.engine import GatewayEngine
    from mcp_context_forge_amd.services.grpc_translate import GrpcToMcpTranslator

    use_gpu = torch.cuda.is_available() and not args.no_gpu
    R = args.requests_per_step or (4096 if use_gpu else 128)

    server, target = start_server(build_files())
    engine = GatewayEngine(Settings(database_url="sqlite://", federation_enabled=False,
                                    auth_required=False, gpu_enabled=use_gpu))

    t_d = time.monotonic()
    tr = GrpcToMcpTranslator(target, prefix="g")
    tools = tr.register_into(engine.tool_service)
    discovery_s = time.monotonic() - t_d
    assert len(tools) == N_FILES * SVC_PER_FILE * METH_PER_SVC, len(tools)
    names = [t["name"] for t in tools]
    if use_gpu:
        assert engine.enable_gpu(), "GPU pipeline must attach"

    def gen(step: int):
        rng = random.Random(9000 + step)
        out = []
        for i in range(R):
            name = names[rng.randrange(len(names))]
            r = rng.random()
            if r < args.invalid_frac:
                arguments = {"name": rng.randrange(5), "count": "not-an-int"}  # schema reject
            elif r < args.invalid_frac + args.big_frac:
                arguments = {"name": f"row{i}", "count": 40}                   # toon-sized reply
            else:
                arguments = {"name": f"row{i}", "count": rng.randrange(3)}
            out.append(json.dumps({"jsonrpc": "2.0", "id": step * R + i, "method": "tools/call",
                                   "params": {"name": name, "arguments": arguments}},
                                  separators=(",", ":")).encode())
        return out

    for s in range(args.warmup):
        outs = await engine.process_rpc_batch(gen(s))
        assert all(o is not None for o in outs)
    if use_gpu:
        torch.cuda.synchronize()

    step_times = []
    data = [gen(1000 + s) for s in range(args.steps)]
    rejected = served = 0
    t0 = time.monotonic()
    for s in range(args.steps):
        ts = time.monotonic()
        outs = await engine.process_rpc_batch(data[s])
        if use_gpu:
            torch.cuda.synchronize()
        step_times.append(time.monotonic() - ts)
        for o in outs:
            if o and b'"error"' in o:
                rejected += 1
            elif o:
                served += 1
    elapsed = time.monotonic() - t0
    assert served + rejected == R * args.steps

    stats = engine.gpu_pipeline.stats() if engine.gpu_pipeline else {}
    print(json.dumps({
        "metric": "grpc_translate_req_per_s",
        "value": round(R * args.steps / elapsed, 2),
        "unit": "req/s", "n_gpus": 1, "steps": args.steps, "warmup": args.warmup,
        "ms_per_step": round(elapsed / args.steps * 1000, 3),
        "higher_is_better": True, "scaling": "weak", "vs_baseline": None,
        "dtype": "bf16", "data": "synthetic",
        "config": {"model": "grpc-translate(10k-method-reflection+schema_guard+toon_encoder)",
                   "global_batch": R, "seq_len": 0, "parallelism": "single",
                   "methods": len(names), "discovery_s": round(discovery_s, 3),
                   "invalid_frac": args.invalid_frac, "big_frac": args.big_frac,
                   "schema_rejected": rejected, "served": served,
                   "p50_batch_ms": round(statistics.median(step_times) * 1000, 3),
                   "gpu_path": bool(engine.gpu_pipeline),
                   "pipeline_stats": {k: v for k, v in stats.items() if k != "banks"},
                   "baseline_config": "BASELINE.json config 5"},
    }))
    await engine.shutdown()
    tr.close()
    server.stop(grace=None)


if __name__ == "__main__":
    sys.exit(asyncio.run(main()))
