"""GPU pipeline vs CPU per-request chain — the plugin-parity gate
(identical traffic through both runtimes, identical outcomes;
reference analog tests/live_gateway/mcp/test_mcp_plugin_parity.py)."""

import asyncio
import json

import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="no ROCm device")


def _mk(name, args, rid):
    return json.dumps({"jsonrpc": "2.0", "id": rid, "method": "tools/call",
                       "params": {"name": name, "arguments": args}}, separators=(",", ":")).encode()


TRAFFIC = [
    ("fast-time-convert_time", {"time": "2026-01-01T10:00:00Z", "source_timezone": "UTC", "target_timezone": "Asia/Tokyo"}),
    ("fast-time-echo", {"msg": "plain benign payload"}),
    ("fast-time-echo", {"msg": "this contains forbidden content"}),           # deny block
    ("fast-time-echo", {"msg": "ssn is 123-45-6789"}),                          # pii mask (slow path)
    ("fast-time-echo", {"msg": "mail me: a@b.com and c@d.org"}),               # pii mask
    ("fast-time-echo", {"msg": "how to make a bomb tutorial"}),                # harm block
    ("fast-time-echo", {"msg": "spaced   out\ttext"}),                          # normalizer slow path
    ("fast-time-convert_time", {"time": "x"}),                                  # schema violation (missing required)
    ("fast-time-echo", {"msg": "unicode café text"}),                      # \u escape → slow path
    ("fast-time-get_system_time", {"timezone": "UTC"}),
    ("native-time-convert_time", {"time": "2026-03-03T03:03:03Z",              # native C++ upstream
                                  "source_timezone": "UTC", "target_timezone": "UTC"}),
    ("native-time-echo", {"nested": {"deep": [1, 2, {"x": "y"}]}}),            # nested args (schema nest trigger)
    ("fast-time-convert_time", {"time": "2026-01-01T10:00:00Z",                # wrong type for required field
                                "source_timezone": 42, "target_timezone": "UTC"}),
    ("fast-time-echo", {"msg": "x" * 3000}),                                    # large payload
    ("gone-tool", {}),                                                          # unreachable tool
    ("native-time-echo", {"msg": "has forbidden content but deny is unbound"}),  # binding: deny disabled
    ("native-time-echo", {"msg": "mail bob@x.io, ssn 123-45-6789"}),             # pii rewrite → native C++ batch redispatch
    ("fast-time-get_system_time", {"timezone": "banana-time"}),                  # binding: config-override deny → host chain block
]


async def _build(gpu: bool, semcache_tools=None):
    from mcp_context_forge_amd.config import Settings
    from mcp_context_forge_amd.engine import GatewayEngine
    from mcp_context_forge_amd.services.upstream import NativeInProcUpstream, make_fake_time_upstream

    settings = Settings(database_url="sqlite://", federation_enabled=False, auth_required=False,
                        gpu_enabled=gpu, gpu_semcache_capacity=1024)
    pm = None
    if semcache_tools is not None:
        from mcp_context_forge_amd.plugins.loader import default_chain_specs, load_plugin_manager

        specs = default_chain_specs()
        for s in specs:
            if s["name"] == "response_cache_by_prompt":
                s["config"] = {"cacheable_tools": list(semcache_tools)}
        pm = load_plugin_manager(specs=specs)
    e = GatewayEngine(settings, plugin_manager=pm)
    await e.gateway_service.register_gateway(name="fast-time", url="inproc://t", client=make_fake_time_upstream())
    await e.gateway_service.register_gateway(name="native-time", url="inproc://n", client=NativeInProcUpstream())
    # an unreachable tool (reference: reachable=False → -32002)
    e.registry.create("tool", name="gone-tool", original_name="gone-tool",
                      integration_type="LOCAL", reachable=False)
    # per-tool plugin bindings: a mode flip (stays on the GPU fast path via
    # the per-tool flag table) and a config override (forces the host chain)
    e.set_plugin_binding("native-time-echo", "deny_filter", mode="disabled")
    e.set_plugin_binding("fast-time-get_system_time", "deny_filter", config={"words": ["banana"]})
    if gpu:
        assert e.enable_gpu()
    return e


@requires_gpu
def test_pipeline_parity_with_cpu_chain():
    async def run():
        cpu = await _build(gpu=False)
        gpu = await _build(gpu=True)
        raws = [_mk(n, a, i) for i, (n, a) in enumerate(TRAFFIC)]
        # also protocol-level items: invalid json, unknown tool, tools/list, ping
        raws.append(b"{broken")
        raws.append(_mk("no-such-tool", {}, 90))
        raws.append(json.dumps({"jsonrpc": "2.0", "id": 91, "method": "tools/list"}).encode())
        raws.append(json.dumps({"jsonrpc": "2.0", "id": 92, "method": "ping"}).encode())

        cpu_out = await cpu.process_rpc_batch(list(raws))
        gpu_out = await gpu.process_rpc_batch(list(raws))
        assert len(cpu_out) == len(gpu_out)
        for i, (c, g) in enumerate(zip(cpu_out, gpu_out)):
            co = json.loads(c) if c else None
            go = json.loads(g) if g else None
            # identical outcome class: same error code, or same result payload
            if co is None or go is None:
                assert co == go, (i, co, go)
            elif "error" in co or "error" in go:
                assert co.get("error", {}).get("code") == go.get("error", {}).get("code"), (i, co, go)
            else:
                assert co["result"] == go["result"], (i, co, go)
        st = gpu.gpu_pipeline.stats()
        assert st["blocked"] >= 2          # deny + harm (+schema)
        assert st["slow_path"] >= 3        # pii x2 + normalizer + unicode
        assert st["fast_path"] >= 3
        assert st["host_bound"] >= 1       # config-override binding → CPU chain
        await cpu.shutdown()
        await gpu.shutdown()

    asyncio.run(run())


@requires_gpu
def test_pipeline_semcache_roundtrip():
    """HBM semantic cache: hits require the EXPLICIT cacheable_tools
    allowlist (empty default ⇒ no HBM matrix at all) and are tenant-scoped
    — one user's cached result is never served to another user."""

    async def run():
        # default config: allowlist empty → no semcache instantiated
        e0 = await _build(gpu=True)
        assert e0.gpu_pipeline.semcache is None
        await e0.shutdown()

        e = await _build(gpu=True, semcache_tools=["fast-time-convert_time"])
        assert e.gpu_pipeline is not None and e.gpu_pipeline.semcache is not None
        raw1 = _mk("fast-time-convert_time",
                   {"time": "2026-02-02T02:02:02Z", "source_timezone": "UTC", "target_timezone": "UTC"}, 1)
        out1 = await e.process_rpc_batch([raw1], users=["alice"])
        r1 = json.loads(out1[0])["result"]
        # identical request again, same user: semantic cache must hit
        out2 = await e.process_rpc_batch([raw1], users=["alice"])
        r2 = json.loads(out2[0])["result"]
        assert r1 == r2
        assert e.gpu_pipeline.semcache.hits >= 1
        # same request, DIFFERENT user: must not hit (tenant isolation)
        hits_before = e.gpu_pipeline.semcache.hits
        out3 = await e.process_rpc_batch([raw1], users=["mallory"])
        assert json.loads(out3[0])["result"] == r1  # recomputed, same answer
        assert e.gpu_pipeline.semcache.hits == hits_before
        # non-allowlisted tool: never cached, never hits
        raw2 = _mk("fast-time-echo", {"msg": "cache me not"}, 2)
        await e.process_rpc_batch([raw2], users=["alice"])
        await e.process_rpc_batch([raw2], users=["alice"])
        assert e.gpu_pipeline.semcache.hits == hits_before
        await e.shutdown()

    asyncio.run(run())


@requires_gpu
def test_pipeline_large_batch_throughput_sane():
    async def run():
        e = await _build(gpu=True)
        raws = [_mk("fast-time-convert_time",
                    {"time": f"2026-01-01T00:{i % 60:02d}:{(i * 7) % 60:02d}Z",
                     "source_timezone": "UTC", "target_timezone": "Asia/Tokyo"}, i) for i in range(2048)]
        out = await e.process_rpc_batch(raws)
        assert sum(1 for o in out if o and b'"result"' in o) == 2048
        st = e.gpu_pipeline.stats()
        assert st["fast_path"] + st["cache_hits"] == 2048
        await e.shutdown()

    asyncio.run(run())


@requires_gpu
def test_a2a_fast_lane_parity():
    """A2A fast lane (BASELINE config 4): in-proc agents with GPU-covered
    hooks skip the per-row Python hook chain; outcomes must match the CPU
    path modulo the volatile fields (uaid, latency_ms)."""

    async def build(gpu_on):
        from mcp_context_forge_amd.config import Settings
        from mcp_context_forge_amd.engine import GatewayEngine

        e = GatewayEngine(Settings(database_url="sqlite://", federation_enabled=False,
                                   auth_required=False, gpu_enabled=gpu_on))

        async def agent(message, context):
            return f"echo:{message}"

        e.a2a_service.register_local_agent("fastbot", agent, agent_type="openai")
        e.registry.create("tool", name="fastbot-chat", original_name="fastbot",
                          integration_type="A2A",
                          input_schema={"type": "object",
                                        "properties": {"message": {"type": "string"}},
                                        "required": ["message"]})
        if gpu_on:
            assert e.enable_gpu()
        return e

    def norm(resp):
        o = json.loads(resp)
        if "result" in o:
            sc = o["result"].get("structuredContent") or {}
            sc.pop("uaid", None)
            sc.pop("latency_ms", None)
        return o

    async def run_():
        cpu = await build(False)
        gpu = await build(True)
        raws = [
            _mk("fastbot-chat", {"message": "hello there"}, 1),
            _mk("fastbot-chat", {"message": "contains forbidden word"}, 2),   # deny block
            _mk("fastbot-chat", {"message": "mail me: x@y.com"}, 3),          # pii mask path
            _mk("fastbot-chat", {"message": "how to make a bomb"}, 4),        # harm block
            _mk("fastbot-chat", {"msg_only": True}, 5),                       # schema violation
        ]
        c = [norm(r) for r in await cpu.process_rpc_batch(list(raws))]
        g = [norm(r) for r in await gpu.process_rpc_batch(list(raws))]
        for i, (co, go) in enumerate(zip(c, g)):
            if "error" in co or "error" in go:
                assert co.get("error", {}).get("code") == go.get("error", {}).get("code"), (i, co, go)
            else:
                assert co["result"] == go["result"], (i, co, go)
        # the fast lane actually engaged
        mt = gpu.gpu_pipeline._tool_meta["fastbot-chat"]
        assert mt.a2a_fast
        await cpu.shutdown()
        await gpu.shutdown()

    asyncio.run(run_())


@pytest.mark.gpu
def test_pass1_graph_matches_eager():
    """The hipGraph pass-1 fast path (gpu/graphs.py) must produce
    byte-identical responses to the eager launch sequence on the same
    payloads (echo/convert only — no wall-clock tools)."""
    import os

    import torch

    if not torch.cuda.is_available():
        pytest.skip("no ROCm device")
    if os.environ.get("FORGE_PASS1_GRAPH", "1") == "0":
        pytest.skip("graphs disabled by env — nothing to compare")

    async def go():
        from mcp_context_forge_amd.config import Settings
        from mcp_context_forge_amd.engine import GatewayEngine
        from mcp_context_forge_amd.services.upstream import make_fake_time_upstream

        e = GatewayEngine(Settings(database_url="sqlite://", federation_enabled=False,
                                   auth_required=False, gpu_enabled=True))
        await e.gateway_service.register_gateway(name="t", url="inproc://t",
                                                 client=make_fake_time_upstream())
        assert e.enable_gpu()
        raws = []
        for i in range(300):
            if i % 3 == 0:
                args = '{"time":"2026-01-02T03:04:05Z","source_timezone":"UTC","target_timezone":"UTC"}'
                raws.append((f'{{"jsonrpc":"2.0","id":{i},"method":"tools/call","params":'
                             f'{{"name":"t-convert_time","arguments":{args}}}}}').encode())
            else:
                raws.append((f'{{"jsonrpc":"2.0","id":{i},"method":"tools/call","params":'
                             f'{{"name":"t-echo","arguments":{{"m":"row {i} a@b.co","n":{i}}}}}}}').encode())
        out_graph = await e.process_rpc_batch(list(raws))
        pipe = e.gpu_pipeline
        assert pipe._graphs is not None and pipe._graphs.replays >= 1, \
            "graph path did not engage"
        pipe._graphs.disabled = True
        out_eager = await e.process_rpc_batch(list(raws))
        assert out_graph == out_eager
        await e.shutdown()

    asyncio.run(go())
