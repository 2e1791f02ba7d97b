"""Large-scale GPU↔CPU parity fuzz campaign (beyond the pytest tier's
2,100 cases). Streams batches of adversarial requests through BOTH
runtimes and diffs outcome classes; prints one JSON line at the end.

Usage (on a GPU box):  PYTHONPATH=.:tests python tools/fuzz_campaign.py \
    --cases 30000 --seed 20260912 [--moderation]
"""

import argparse
import asyncio
import json
import random
import sys
import time


async def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--cases", type=int, default=30000)
    ap.add_argument("--seed", type=int, default=20260912)
    ap.add_argument("--batch", type=int, default=512)
    ap.add_argument("--moderation", action="store_true")
    args = ap.parse_args()

    sys.path.insert(0, "tests")
    from test_parity_fuzz import TOOLS, _outcome, build_pair, gen_adversarial

    cpu = await build_pair(gpu_mod=False, moderation=args.moderation)
    gpu = await build_pair(gpu_mod=True, moderation=args.moderation)
    rng = random.Random(args.seed)
    t0 = time.monotonic()
    n_done = 0
    divergences = []
    while n_done < args.cases:
        k = min(args.batch, args.cases - n_done)
        if args.moderation:
            raws = []
            for _ in range(k):
                obj = json.loads(gen_adversarial(rng, TOOLS))
                raws.append(json.dumps(obj, separators=(",", ":"), sort_keys=True,
                                       ensure_ascii=True).encode())
        else:
            raws = [gen_adversarial(rng, TOOLS) for _ in range(k)]
        c_out = await cpu.process_rpc_batch(list(raws))
        g_out = await gpu.process_rpc_batch(list(raws))
        for i, (c, g) in enumerate(zip(c_out, g_out)):
            co, go = _outcome(c), _outcome(g)
            if co != go:
                divergences.append({"case": n_done + i, "raw": raws[i][:200].decode("utf-8", "replace"),
                                    "cpu": co, "gpu": go})
        n_done += k
        if n_done % 5000 < args.batch:
            print(f"# {n_done}/{args.cases} cases, {len(divergences)} divergences, "
                  f"{time.monotonic() - t0:.1f}s", file=sys.stderr)
    await cpu.shutdown()
    await gpu.shutdown()
    print(json.dumps({"cases": n_done, "seed": args.seed, "moderation": args.moderation,
                      "divergences": len(divergences), "elapsed_s": round(time.monotonic() - t0, 1),
                      "first_divergences": divergences[:5]}))
    return 1 if divergences else 0


if __name__ == "__main__":
    raise SystemExit(asyncio.run(main()))
