#!/usr/bin/env python3
"""Sequential (c=1) latency segmentation for the edge stack: times each hop
(worker /healthz = client+uvicorn only, owner /rpc = collector+pipeline,
worker /rpc = full path) to locate fixed per-request delay.

    python loadtest/latency_probe.py WORKER_URL OWNER_URL
"""
import asyncio
import base64
import json
import statistics
import sys
import time

import aiohttp

AUTH = "Basic " + base64.b64encode(b"admin:changeme").decode()
PAYLOAD = json.dumps({"jsonrpc": "2.0", "id": 1, "method": "tools/call",
                      "params": {"name": "up-0-convert_time",
                                 "arguments": {"time": "2026-01-01T10:00:00Z",
                                               "source_timezone": "UTC",
                                               "target_timezone": "Asia/Tokyo"}}}).encode()


async def probe(session, name, method, url, data=None, n=200):
    lat = []
    for _ in range(n):
        t0 = time.monotonic()
        async with session.request(method, url, data=data,
                                   headers={"Authorization": AUTH,
                                            "content-type": "application/json"}) as r:
            await r.read()
            assert r.status in (200, 202), (name, r.status)
        lat.append((time.monotonic() - t0) * 1e3)
    lat.sort()
    print(json.dumps({"probe": name, "p50_ms": round(statistics.median(lat), 3),
                      "p90_ms": round(lat[int(len(lat) * 0.9)], 3),
                      "min_ms": round(lat[0], 3)}))


async def main():
    worker, owner = sys.argv[1], sys.argv[2]
    timeout = aiohttp.ClientTimeout(total=30)
    async with aiohttp.ClientSession(timeout=timeout) as s:
        await probe(s, "worker /healthz (no owner hop)", "GET", worker + "/healthz")
        await probe(s, "owner /version (fastapi)", "GET", owner + "/version")
        await probe(s, "owner /rpc (collector+pipeline)", "POST", owner + "/rpc", PAYLOAD)
        await probe(s, "worker /rpc (full edge path)", "POST", worker + "/rpc", PAYLOAD)


if __name__ == "__main__":
    asyncio.run(main())
