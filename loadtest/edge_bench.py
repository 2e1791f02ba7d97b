#!/usr/bin/env python3
"""Owner-side of the multi-worker edge benchmark: full gateway + GPU
pipeline + 64 native upstreams, edge socket enabled, private HTTP port."""
import asyncio
import sys

import uvicorn

from mcp_context_forge_amd.config import Settings
from mcp_context_forge_amd.engine import GatewayEngine
from mcp_context_forge_amd.services.upstream import NativeInProcUpstream
from mcp_context_forge_amd.transports.http_app import build_app


async def main():
    private_port = int(sys.argv[1])
    sock = sys.argv[2]
    settings = Settings(database_url="sqlite://", federation_enabled=False, auth_required=True,
                        gpu_batch_window_us=300, gpu_batch_max_requests=8192, edge_socket=sock)
    engine = GatewayEngine(settings)
    for u in range(64):
        await engine.gateway_service.register_gateway(
            name=f"up-{u}", url=f"inproc://up-{u}", client=NativeInProcUpstream(name=f"up-{u}"))
    app = build_app(engine)
    config = uvicorn.Config(app, host="127.0.0.1", port=private_port, log_level="warning", lifespan="on")
    await uvicorn.Server(config).serve()


if __name__ == "__main__":
    asyncio.run(main())
