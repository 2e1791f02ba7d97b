#!/usr/bin/env python3
"""Fixed per-batch cost probe: time process_rpc_batch at small batch sizes
with per-stage accounting. The low-load edge p50 (~60 ms) is set by this
fixed cost (latency ≈ 2-3 batch cycles), so stage times at B=64..1024 show
where to cut.

    FORGE_PIPELINE_TIMING=1 python loadtest/batch_latency.py
"""
import asyncio
import json
import time


async def main():
    from mcp_context_forge_amd.config import Settings
    from mcp_context_forge_amd.engine import GatewayEngine
    from mcp_context_forge_amd.services.upstream import NativeInProcUpstream

    settings = Settings(database_url="sqlite://", federation_enabled=False, auth_required=False)
    engine = GatewayEngine(settings)
    for u in range(8):
        await engine.gateway_service.register_gateway(
            name=f"up-{u}", url=f"inproc://up-{u}", client=NativeInProcUpstream(name=f"up-{u}"))
    assert engine.enable_gpu()
    pipe = engine.gpu_pipeline

    def mk(i):
        return json.dumps({"jsonrpc": "2.0", "id": i, "method": "tools/call",
                           "params": {"name": f"up-{i % 8}-convert_time",
                                      "arguments": {"time": f"2026-01-01T00:{i % 60:02d}:{(i * 7) % 60:02d}Z",
                                                    "source_timezone": "UTC", "target_timezone": "Asia/Tokyo"}}},
                          separators=(",", ":")).encode()

    for B in (64, 256, 1024, 4096):
        raws = [mk(i + B * 1000) for i in range(B)]
        for _ in range(3):
            await engine.process_rpc_batch(raws)  # warmup (incl. semcache fill)
        pipe.timing.clear()
        iters = 30 if B <= 1024 else 10
        t0 = time.monotonic()
        for _ in range(iters):
            await engine.process_rpc_batch(raws)
        dt = (time.monotonic() - t0) / iters
        stages = {k: round(v / iters * 1e3, 3) for k, v in sorted(
            pipe.timing.items(), key=lambda kv: -kv[1])}
        print(json.dumps({"batch": B, "ms_per_batch": round(dt * 1e3, 2),
                          "req_per_s": round(B / dt), "stages_ms": stages}))
    await engine.shutdown()


if __name__ == "__main__":
    asyncio.run(main())
