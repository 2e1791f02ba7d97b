#!/usr/bin/env python3
"""HTTP load rig for POST /rpc (reference analog: tests/hey/hey.sh —
10k requests / 200 concurrency / tools/call payload). aiohttp-based: httpx
caps out near ~100 RPS at high concurrency and would measure the client.

    python loadtest/load_rpc.py --url http://localhost:4444 --n 10000 --c 200
"""
import argparse
import asyncio
import base64
import json
import statistics
import time

import aiohttp


async def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--url", default="http://127.0.0.1:4444")
    ap.add_argument("--n", type=int, default=10000)
    ap.add_argument("--c", type=int, default=200)
    ap.add_argument("--tool", default="fast-time-convert_time")
    ap.add_argument("--user", default="admin")
    ap.add_argument("--password", default="changeme")
    args = ap.parse_args()

    auth = "Basic " + base64.b64encode(f"{args.user}:{args.password}".encode()).decode()
    payload = json.dumps({
        "jsonrpc": "2.0", "id": 1, "method": "tools/call",
        "params": {"name": args.tool,
                   "arguments": {"time": "2026-01-01T10:00:00Z",
                                 "source_timezone": "UTC", "target_timezone": "Asia/Tokyo"}},
    }).encode()
    headers = {"Authorization": auth, "content-type": "application/json"}
    lat: list = []
    errors = 0
    per_worker = [args.n // args.c] * args.c
    for i in range(args.n % args.c):
        per_worker[i] += 1

    connector = aiohttp.TCPConnector(limit=args.c, limit_per_host=args.c)
    timeout = aiohttp.ClientTimeout(total=60)
    async with aiohttp.ClientSession(connector=connector, timeout=timeout) as session:

        async def worker(count: int):
            nonlocal errors
            for _ in range(count):
                t0 = time.monotonic()
                try:
                    async with session.post(args.url + "/rpc", data=payload, headers=headers) as r:
                        body = await r.read()
                        if r.status != 200 or b'"error"' in body[:60]:
                            errors += 1
                except Exception:
                    errors += 1
                lat.append(time.monotonic() - t0)

        t_start = time.monotonic()
        await asyncio.gather(*(worker(c) for c in per_worker))
        elapsed = time.monotonic() - t_start

    lat.sort()
    q = lambda p: lat[min(len(lat) - 1, int(p * len(lat)))] * 1000  # noqa: E731
    print(json.dumps({
        "requests": args.n, "concurrency": args.c, "elapsed_s": round(elapsed, 3),
        "rps": round(args.n / elapsed, 2), "errors": errors,
        "p50_ms": round(q(0.50), 2), "p90_ms": round(q(0.90), 2),
        "p99_ms": round(q(0.99), 2), "avg_ms": round(statistics.mean(lat) * 1000, 2),
    }))


if __name__ == "__main__":
    asyncio.run(main())
