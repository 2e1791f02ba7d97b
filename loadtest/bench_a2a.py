#!/usr/bin/env python3
"""BASELINE config 4: A2A agent gateway — OpenAI-compat routing with the
content_moderation / PII-mask HIP classifiers (bf16 MFMA) on the agent path.

Per step, R `tools/call` requests targeting A2A-integrated tools flow
through the batched GPU pipeline (scan banks + MFMA moderation classifier
over the message text, PII mask on flagged rows, agent_pre/post hooks in
the A2A service) and route to 16 in-proc agents whose handler is an
OpenAI-compatible chat completion shape (the llm_proxy payload contract).

Output: one bench-style JSON line (metric a2a_invoke_req_per_s).
Run (GPU box):  python loadtest/bench_a2a.py --steps 10 --warmup 3
"""

from __future__ import annotations

import argparse
import asyncio
import json
import os
import random
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def make_message(rng: random.Random, flagged_frac: float) -> str:
    base = rng.choice([
        "summarize the quarterly report for region %d" % rng.randrange(50),
        "draft a reply to customer ticket #%d about shipping delays" % rng.randrange(10000),
        "translate the release notes %d to French" % rng.randrange(1000),
        "plan a three-step rollout for feature flag f%d" % rng.randrange(300),
    ])
    r = rng.random()
    if r < flagged_frac * 0.5:
        base += " and cc user%d@example.com" % rng.randrange(1000)   # PII mask path
    elif r < flagged_frac:
        base += "  with   spaced\ttext"                              # normalizer path
    return base


async def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--requests-per-step", type=int, default=0)
    ap.add_argument("--agents", type=int, default=16)
    ap.add_argument("--flagged-frac", type=float, default=0.02)
    ap.add_argument("--no-gpu", action="store_true")
    args = ap.parse_args()

    import torch

    from mcp_context_forge_amd.config import Settings
    from mcp_context_forge_amd.engine import GatewayEngine

    use_gpu = torch.cuda.is_available() and not args.no_gpu
    R = args.requests_per_step or (8192 if use_gpu else 256)

    engine = GatewayEngine(Settings(database_url="sqlite://", federation_enabled=False,
                                    auth_required=False, gpu_enabled=use_gpu))

    # 16 OpenAI-compat routing agents: each "routes" to a model id and
    # returns a chat.completion-shaped reply (llm_proxy payload contract)
    for a in range(args.agents):
        model = f"forge-model-{a}"

        async def handler(message, context, _m=model):
            # OpenAI-compat completion payload, serialized (the agent reply
            # is text per the A2A message contract)
            return json.dumps({
                "id": "chatcmpl-bench", "object": "chat.completion", "model": _m,
                "choices": [{"index": 0, "finish_reason": "stop",
                             "message": {"role": "assistant",
                                         "content": f"[{_m}] routed reply to: {message[:96]}"}}],
            }, separators=(",", ":"))

        engine.a2a_service.register_local_agent(f"agent{a}", handler,
                                                description="OpenAI-compat router",
                                                agent_type="openai")
        engine.registry.create("tool", name=f"agent{a}-chat", original_name=f"agent{a}",
                               integration_type="A2A", description="A2A chat routing",
                               input_schema={"type": "object",
                                             "properties": {"message": {"type": "string"}},
                                             "required": ["message"]})
    if use_gpu:
        assert engine.enable_gpu(), "GPU pipeline must attach"

    def gen(step: int):
        rng = random.Random(7000 + step)
        return [json.dumps({"jsonrpc": "2.0", "id": step * R + i, "method": "tools/call",
                            "params": {"name": f"agent{rng.randrange(args.agents)}-chat",
                                       "arguments": {"message": make_message(rng, args.flagged_frac)}}},
                           separators=(",", ":")).encode()
                for i in range(R)]

    for s in range(args.warmup):
        outs = await engine.process_rpc_batch(gen(s))
        assert all(o is not None for o in outs)
    if use_gpu:
        torch.cuda.synchronize()

    step_times = []
    data = [gen(1000 + s) for s in range(args.steps)]
    t0 = time.monotonic()
    for s in range(args.steps):
        ts = time.monotonic()
        outs = await engine.process_rpc_batch(data[s])
        if use_gpu:
            torch.cuda.synchronize()
        step_times.append(time.monotonic() - ts)
        ok = sum(1 for o in outs if o and b'"result"' in o)
        blocked = sum(1 for o in outs if o and b'"error"' in o)
        assert ok + blocked == R, (ok, blocked, R)
    elapsed = time.monotonic() - t0

    stats = engine.gpu_pipeline.stats() if engine.gpu_pipeline else {}
    print(json.dumps({
        "metric": "a2a_invoke_req_per_s",
        "value": round(R * args.steps / elapsed, 2),
        "unit": "req/s", "n_gpus": 1, "steps": args.steps, "warmup": args.warmup,
        "ms_per_step": round(elapsed / args.steps * 1000, 3),
        "higher_is_better": True, "scaling": "weak", "vs_baseline": None,
        "dtype": "bf16", "data": "synthetic",
        "config": {"model": "a2a-agent-gateway(openai-compat-routing+moderation-mlp-bf16-mfma+pii-mask)",
                   "global_batch": R, "seq_len": 0, "parallelism": "single",
                   "agents": args.agents, "flagged_frac": args.flagged_frac,
                   "p50_batch_ms": round(statistics.median(step_times) * 1000, 3),
                   "gpu_path": bool(engine.gpu_pipeline),
                   "pipeline_stats": {k: v for k, v in stats.items() if k != "banks"},
                   "baseline_config": "BASELINE.json config 4"},
    }))
    await engine.shutdown()


if __name__ == "__main__":
    sys.exit(asyncio.run(main()))
