"""Distributed gateway runtime: one engine per GPU rank, wired together.

Reference analog: the reference's multi-worker deployment =
gunicorn workers + Redis (session-affinity RPC, invalidation pub/sub, leader
election). Here: torchrun launches one process per GPU; this module connects
the GatewayEngine to the RcclBus so that

  * cross-rank request forwarding works (session/tool ownership),
  * registry mutations broadcast an invalidation that makes peer ranks
    reload from the shared database,
  * rank 0 is the leader for background singletons (health loop).

Pitfall: torch.distributed collectives BLOCK the calling thread. Never call
`collectives.barrier()` (or any default-group collective) directly on the
asyncio event loop of a rank that may concurrently serve forwarded bus
requests — wrap it in `asyncio.to_thread(...)` so the loop keeps running.
"""

from __future__ import annotations

import asyncio
import json
from typing import Any, Optional

from ..engine import GatewayEngine
from . import collectives
from .bus import RcclBus


class DistributedGateway:
    def __init__(self, engine: GatewayEngine, cadence_us: int = 1000):
        self.engine = engine
        self.rank, self.world = collectives.rank_world()
        self.bus: Optional[RcclBus] = None
        if self.world > 1:
            self.bus = RcclBus(self._handle_forward, cadence_us=cadence_us)
            self.bus.on_publish = self._on_publish
            engine.bus = self.bus
            engine.forward_rpc = self.forward_rpc
            engine.forward_rpc_batch = self.forward_rpc_batch

    async def start(self) -> None:
        await self.engine.startup()
        if self.bus is not None:
            self.bus.loop = asyncio.get_running_loop()
            self.bus.start()

    async def stop(self) -> None:
        if self.bus is not None:
            self.bus.stop()
        await self.engine.shutdown()

    # -- forwarded execution (reference: _execute_forwarded_request :950) ----
    async def _handle_forward(self, payload: Any) -> Any:
        if isinstance(payload, dict) and payload.get("kind") == "rpc":
            out = await self.engine.handle_rpc_bytes(payload["raw"].encode()
                                                     if isinstance(payload["raw"], str) else payload["raw"],
                                                     user=payload.get("user"),
                                                     server_id=payload.get("server_id"))
            return {"raw": out}
        if isinstance(payload, dict) and payload.get("kind") == "rpc_batch":
            outs = await self.engine.process_rpc_batch(payload["raws"], users=payload.get("users"))
            return {"raws": outs}
        return {"error": "unknown payload kind"}

    async def forward_rpc(self, dest_rank: int, raw: bytes, user: Optional[str] = None,
                          server_id: Optional[str] = None) -> Optional[bytes]:
        """Execute a raw JSON-RPC request on `dest_rank` and return its bytes."""
        assert self.bus is not None
        out = await self.bus.submit(dest_rank, {"kind": "rpc", "raw": raw, "user": user,
                                                "server_id": server_id})
        return out.get("raw")

    async def forward_rpc_batch(self, dest_rank: int, raws: list,
                                users: Optional[list] = None) -> list:
        """Batch variant: one bus message carries a whole group of requests
        destined for `dest_rank` (the pipeline ships foreign-tool rows this
        way so a micro-batch costs O(dest ranks) bus messages, not O(rows))."""
        assert self.bus is not None
        out = await self.bus.submit(dest_rank, {"kind": "rpc_batch", "raws": raws, "users": users})
        if "raws" not in out:
            raise RuntimeError(f"bus forward failed: {out}")
        return out["raws"]

    async def sync_tool_ownership(self) -> None:
        """COLLECTIVE: every rank publishes the tool names it owns locally;
        peers record name -> owner_rank so requests route over the bus.
        Call after registering upstreams (and after registry changes, from
        all ranks together). Reference analog: the shared registry DB that
        lets any gunicorn worker resolve any tool."""
        if self.world <= 1:
            return
        local = [t["name"] for t in self.engine.registry.list("tool", include_disabled=False)]

        def gather():
            import torch.distributed as dist

            box = [None] * self.world
            dist.all_gather_object(box, local)
            return box

        box = await asyncio.to_thread(gather)
        fmap = {}
        for r, names in enumerate(box):
            if r == self.rank:
                continue
            for n in names:
                fmap[n] = r
        self.engine.foreign_tools = fmap

    def owner_of_tool(self, tool_name: str) -> int:
        tool = self.engine.registry.lookup_tool(tool_name)
        if tool is None:
            return self.rank
        gw_id = tool.get("gateway_id")
        if gw_id:
            try:
                return int(self.engine.registry.get("gateway", gw_id).get("owner_rank", 0)) % self.world
            except Exception:
                return self.rank
        return self.rank

    # -- invalidation broadcast (reference: Redis pub/sub invalidation) ------
    def broadcast_invalidation(self, kind: str = "registry") -> None:
        if self.bus is not None:
            self.bus.publish({"kind": "invalidate", "what": kind})

    def _on_publish(self, src: int, payload: Any) -> None:
        if isinstance(payload, dict) and payload.get("kind") == "invalidate":
            # peer mutated the shared DB — refresh the hot caches; bindings
            # feed the plugin manager (and bump its version, so an attached
            # GPU pipeline recompiles its per-tool flag tables)
            self.engine.registry.load_all()
            self.engine.sync_plugin_bindings()

    @property
    def is_leader(self) -> bool:
        return self.rank == 0
