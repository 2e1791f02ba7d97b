"""Cancellation + elicitation (reference: routers/cancellation_router.py,
session elicitation service; MCP notifications/cancelled + elicitation/create)."""

import asyncio
import json

from mcp_context_forge_amd.config import Settings
from mcp_context_forge_amd.engine import GatewayEngine


def _engine():
    return GatewayEngine(Settings(database_url="sqlite://", federation_enabled=False,
                                  auth_required=False, plugins_enabled=False))


def _rpc(method, params, rid=1):
    return json.dumps({"jsonrpc": "2.0", "id": rid, "method": method, "params": params}).encode()


def test_cancellation_kills_inflight_call(run):
    async def go():
        e = _engine()
        started = asyncio.Event()

        async def slow(args):
            started.set()
            await asyncio.sleep(30)
            return {"done": True}

        e.tool_service.register_local_tool("slow", slow)
        sess = e.sessions.create()

        call = asyncio.create_task(e.handle_rpc_bytes(
            _rpc("tools/call", {"name": "slow", "arguments": {}}, rid=7), session=sess))
        await asyncio.wait_for(started.wait(), 5)
        assert e.cancellations.inflight_count() == 1
        # client sends notifications/cancelled → in-flight task cancelled,
        # request produces NO response (MCP semantics)
        out = await e.handle_rpc_bytes(
            json.dumps({"jsonrpc": "2.0", "method": "notifications/cancelled",
                        "params": {"requestId": 7}}).encode(), session=sess)
        assert out is None
        assert await asyncio.wait_for(call, 5) is None
        assert e.cancellations.cancelled == 1
        assert e.cancellations.inflight_count() == 0
        await e.shutdown()

    run(go())


def test_cancel_unknown_request_is_noop(run):
    async def go():
        e = _engine()
        sess = e.sessions.create()
        assert e.cancellations.cancel(sess.session_id, 99) is False
        await e.shutdown()

    run(go())


def test_elicitation_roundtrip(run):
    async def go():
        e = _engine()
        sess = e.sessions.create()

        async def client():
            # client side: receives elicitation/create on the session stream,
            # replies with a JSON-RPC response over the POST channel
            msg = await asyncio.wait_for(sess.queue.get(), 5)
            assert msg["method"] == "elicitation/create"
            assert msg["params"]["message"] == "Which region?"
            await e.handle_rpc_bytes(json.dumps(
                {"jsonrpc": "2.0", "id": msg["id"], "result": {"region": "eu"}}).encode())

        ct = asyncio.create_task(client())
        result = await e.elicitation.elicit(sess.session_id, "Which region?",
                                            {"type": "object", "properties": {"region": {"type": "string"}}})
        assert result == {"region": "eu"}
        await ct
        assert e.elicitation.pending_count() == 0
        await e.shutdown()

    run(go())


def test_elicitation_no_session(run):
    async def go():
        e = _engine()
        try:
            await e.elicitation.elicit("nope", "hello?")
            assert False, "expected RuntimeError"
        except RuntimeError:
            pass
        await e.shutdown()

    run(go())
