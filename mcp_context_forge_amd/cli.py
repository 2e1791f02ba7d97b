"""Command-line interface.

Reference analogs: mcpgateway/cli.py (uvicorn wrapper), wrapper.py (stdio
bridge), translate.py (protocol bridge), tools/cli.py (`cforge`).

  python -m mcp_context_forge_amd serve                  # HTTP gateway
  python -m mcp_context_forge_amd stdio                  # engine as stdio MCP server
  python -m mcp_context_forge_amd wrapper --url URL      # stdio ↔ remote gateway
  python -m mcp_context_forge_amd translate --stdio CMD --port P  # stdio server → HTTP
  python -m mcp_context_forge_amd token --user EMAIL     # mint a JWT
  python -m mcp_context_forge_amd export / import FILE   # config dump/load
  python -m mcp_context_forge_amd bench ...              # flagship benchmark
"""

from __future__ import annotations

import argparse
import asyncio
import json
import shlex
import sys


def main(argv=None) -> int:
    ap = argparse.ArgumentParser(prog="mcp-context-forge-amd")
    sub = ap.add_subparsers(dest="cmd", required=True)

    p = sub.add_parser("serve", help="run the HTTP gateway")
    p.add_argument("--host", default=None)
    p.add_argument("--port", type=int, default=None)
    p.add_argument("--workers", type=int, default=1)
    p.add_argument("--native-edge-port", type=int, default=None,
                   help="serve POST /rpc on this port with the C++ epoll edge "
                        "(the flagship hot lane; control plane stays on --port)")
    p.add_argument("--native-edge-threads", type=int, default=None)

    sub.add_parser("stdio", help="run the gateway engine as a stdio MCP server")

    p = sub.add_parser("edge-worker", help="internal: HTTP worker forwarding to a GPU owner")
    p.add_argument("--host", default="0.0.0.0")
    p.add_argument("--port", type=int, required=True)
    p.add_argument("--owner-sock", required=True)
    p.add_argument("--owner-http", required=True)

    p = sub.add_parser("wrapper", help="stdio bridge to a remote gateway (reference: wrapper.py)")
    p.add_argument("--url", required=True)
    p.add_argument("--token", default=None)
    p.add_argument("--rpc", action="store_true", help="use /rpc instead of /mcp")

    p = sub.add_parser("translate", help="expose a stdio MCP server over HTTP (reference: translate.py)")
    p.add_argument("--stdio", required=True, help="command to launch, e.g. 'python server.py'")
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=9000)

    p = sub.add_parser("reverse-proxy", help="tunnel a local stdio MCP server out to a remote gateway")
    p.add_argument("--url", required=True, help="remote gateway base URL")
    p.add_argument("--stdio", required=True, help="local server command, e.g. 'python server.py'")
    p.add_argument("--name", default=None)
    p.add_argument("--token", default=None)

    p = sub.add_parser("plugin-scaffold", help="generate a plugin skeleton "
                       "(reference: cforge plugin bootstrap, tools/cli.py)")
    p.add_argument("name", help="plugin name, e.g. my_policy")
    p.add_argument("--dir", default="plugins", help="output directory")
    p.add_argument("--hooks", default="tool_pre_invoke,tool_post_invoke")
    p.add_argument("--external", action="store_true",
                   help="scaffold an external-process HTTP service instead of an in-proc plugin")

    p = sub.add_parser("token", help="mint an HS256 JWT for the gateway")
    p.add_argument("--user", default="admin@example.com")
    p.add_argument("--admin", action="store_true")
    p.add_argument("--expires-minutes", type=int, default=10080)

    p = sub.add_parser("export", help="dump registry configuration as JSON")
    p.add_argument("file", nargs="?", default="-")

    p = sub.add_parser("import", help="load registry configuration from JSON")
    p.add_argument("file")

    args = ap.parse_args(argv)

    if args.cmd == "serve":
        from .config import get_settings
        from .engine import GatewayEngine
        from .transports.http_app import build_app

        settings = get_settings()
        if args.host:
            settings.host = args.host
        if args.port:
            settings.port = args.port
        if args.native_edge_port:
            settings.native_edge_port = args.native_edge_port
        if args.native_edge_threads:
            settings.native_edge_threads = args.native_edge_threads
        import uvicorn

        if args.workers and args.workers > 1:
            # multi-worker edge: owner on a private port + N public workers
            import os as _os
            import subprocess as _sp

            public_port = settings.port
            private_port = public_port + 1
            settings.edge_socket = f"/tmp/forge-edge-{_os.getpid()}.sock"
            procs = []
            env = dict(_os.environ)
            for _ in range(args.workers):
                procs.append(_sp.Popen([sys.executable, "-m", "mcp_context_forge_amd", "edge-worker",
                                        "--host", settings.host, "--port", str(public_port),
                                        "--owner-sock", settings.edge_socket,
                                        "--owner-http", f"http://127.0.0.1:{private_port}"], env=env))
            try:
                uvicorn.run(build_app(GatewayEngine(settings)), host="127.0.0.1", port=private_port)
            finally:
                for p_ in procs:
                    p_.terminate()
            return 0
        uvicorn.run(build_app(GatewayEngine(settings)), host=settings.host, port=settings.port)
        return 0

    if args.cmd == "edge-worker":
        from .transports.edge import run_worker

        import time as _time

        for _ in range(120):  # wait for the owner socket
            import os as _os

            if _os.path.exists(args.owner_sock):
                break
            _time.sleep(0.5)
        run_worker(args.host, args.port, args.owner_sock, args.owner_http)
        return 0

    if args.cmd == "stdio":
        from .config import get_settings
        from .engine import GatewayEngine
        from .transports.stdio import StdioServer

        engine = GatewayEngine(get_settings())

        async def run():
            await engine.startup()
            try:
                await StdioServer(engine).serve()
            finally:
                await engine.shutdown()

        asyncio.run(run())
        return 0

    if args.cmd == "wrapper":
        from .transports.stdio import GatewayWrapper

        asyncio.run(GatewayWrapper(args.url, token=args.token, use_mcp=not args.rpc).serve())
        return 0

    if args.cmd == "translate":
        from .transports.stdio import StdioSubprocess, build_translate_app

        async def run():
            sp = StdioSubprocess(shlex.split(args.stdio))
            await sp.start()
            import uvicorn

            app = build_translate_app(sp)
            config = uvicorn.Config(app, host=args.host, port=args.port, log_level="info")
            try:
                await uvicorn.Server(config).serve()
            finally:
                await sp.stop()

        asyncio.run(run())
        return 0

    if args.cmd == "reverse-proxy":
        from .transports.reverse_proxy import ReverseProxyClient
        from .transports.stdio import StdioSubprocess

        async def run():
            sp = StdioSubprocess(shlex.split(args.stdio))
            await sp.start()

            async def forward(raw: bytes):
                try:
                    want_id = json.loads(raw).get("id")
                except ValueError:
                    want_id = None
                return await sp.send_raw(raw, want_id)

            rp = ReverseProxyClient(args.url, args.name or "reverse-proxy", forward, token=args.token)
            try:
                await rp.register()
                await rp.serve()
            finally:
                await rp.aclose()
                await sp.stop()

        asyncio.run(run())
        return 0

    if args.cmd == "plugin-scaffold":
        from pathlib import Path

        name = args.name.replace("-", "_")
        hooks = [h.strip() for h in args.hooks.split(",") if h.strip()]
        out = Path(args.dir)
        out.mkdir(parents=True, exist_ok=True)
        if args.external:
            (out / f"{name}_service.py").write_text(f'''"""External plugin service `{name}` — runs as its own process.

Start:  uvicorn {name}_service:app --port 9901
Wire into plugins/config.yaml:
    - name: {name}
      kind: external
      mode: enforce
      config: {{url: "http://127.0.0.1:9901", hooks: {hooks!r}}}
"""

from mcp_context_forge_amd.plugins.external import build_external_service_app


def policy(req: dict) -> dict:
    """req: {{hook, plugin, name, args, user, server_id}} ->
    {{action: allow|block|transform, reason?, payload?, metadata?}}"""
    # TODO: implement your policy
    return {{"action": "allow"}}


app = build_external_service_app(policy, name="{name}")
''')
            print(out / f"{name}_service.py")
        else:
            hook_methods = "\n\n".join(
                f"    async def {h}(self, ctx: PluginContext) -> PluginResult:\n"
                f"        # TODO: inspect ctx.args / ctx.name / ctx.user\n"
                f"        return PluginResult.ok()" for h in hooks)
            (out / f"{name}.py").write_text(f'''"""Plugin `{name}` (scaffolded).

Wire into plugins/config.yaml:
    - name: {name}
      kind: {name}.{name.title().replace("_", "")}Plugin
      mode: enforce
      priority: 100
"""

from mcp_context_forge_amd.plugins.framework import (HookType, Plugin, PluginContext,
                                                     PluginResult)


class {name.title().replace("_", "")}Plugin(Plugin):
    name = "{name}"
    hooks = ({", ".join(f"HookType.{h.upper()}" for h in hooks)},)
    priority = 100

{hook_methods}
''')
            print(out / f"{name}.py")
        return

    if args.cmd == "token":
        from .auth import jwt as jwt_mod
        from .config import get_settings

        s = get_settings()
        tok = jwt_mod.create_token({"sub": args.user, "admin": args.admin}, s.jwt_secret_key,
                                   expires_minutes=args.expires_minutes,
                                   audience=s.jwt_audience, issuer=s.jwt_issuer)
        print(tok)
        return 0

    if args.cmd in ("export", "import"):
        from .config import get_settings
        from .engine import GatewayEngine

        engine = GatewayEngine(get_settings())
        if args.cmd == "export":
            payload = json.dumps(engine.registry.export_configuration(), indent=2, default=str)
            if args.file == "-":
                print(payload)
            else:
                open(args.file, "w").write(payload)
        else:
            counts = engine.registry.import_configuration(json.load(open(args.file)))
            engine.sync_plugin_bindings()
            print(json.dumps(counts))
        asyncio.run(engine.shutdown())
        return 0

    return 1


if __name__ == "__main__":
    sys.exit(main())
