#!/usr/bin/env python3
"""Mixed-fleet routing validation on ONE GPU (BASELINE config 5, single
lease): a server + two workers serving DIFFERENT models (llama3.2:3b and
llama3.2:1b) on the same device. Drives both models concurrently over
/api/generate, asserts model-aware routing (each job lands on the worker
advertising the model), then SIGKILLs the 1b worker mid-load and asserts
its in-flight jobs orphan-requeue... to nowhere (no other 1b worker) ->
they surface as failures after retries, while 3b traffic is untouched.
Prints one JSON line.

Usage: python scripts/mixed_fleet.py [--requests N] [--num-predict K]
"""

import argparse
import asyncio
import json
import os
import signal
import statistics
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

from scripts.loadtest import SERVER_SRC, WORKER_SRC, _free_port, _spawn  # noqa: E402


async def drive(base, model, n, conc, npredict, seed0):
    import aiohttp
    sem = asyncio.Semaphore(conc)
    tokens, ttfts, errors = [], [], [0]
    conn = aiohttp.TCPConnector(limit=conc + 8)
    async with aiohttp.ClientSession(base_url=base, connector=conn) as sess:
        async def one(i):
            async with sem:
                t0 = time.monotonic()
                first = None
                got = 0
                try:
                    async with sess.post("/api/generate", json={
                            "model": model,
                            "prompt": f"mixed {model} {i} " + "lorem " * 16,
                            "stream": True,
                            "options": {"temperature": 0.8,
                                        "seed": seed0 + i,
                                        "num_predict": npredict}},
                            timeout=aiohttp.ClientTimeout(total=600)) as r:
                        async for raw in r.content:
                            if not raw.strip():
                                continue
                            c = json.loads(raw)
                            if first is None and c.get("response"):
                                first = time.monotonic() - t0
                            if c.get("done"):
                                got = c.get("eval_count", 0)
                except Exception:  # noqa: BLE001
                    errors[0] += 1
                    return
                tokens.append(got)
                if first is not None:
                    ttfts.append(first * 1000)

        t0 = time.monotonic()
        await asyncio.gather(*(one(i) for i in range(n)))
        wall = time.monotonic() - t0
    return wall, tokens, ttfts, errors[0]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--requests", type=int, default=256)
    ap.add_argument("--concurrency", type=int, default=64)
    ap.add_argument("--num-predict", type=int, default=48)
    ap.add_argument("--device", default="cuda:0")
    args = ap.parse_args()

    import torch
    models = ("llama3.2:3b", "llama3.2:1b")
    if not torch.cuda.is_available():
        args.device = "cpu"
        models = ("tiny", "tiny2")
        args.requests, args.concurrency = 16, 4

    bus_port, http_port = _free_port(), _free_port()
    env = {"BUS_PORT": str(bus_port), "HTTP_PORT": str(http_port),
           "LT_CONC": str(args.concurrency),
           "LT_DEVICE": args.device,
           "LT_DTYPE": "bf16" if args.device.startswith("cuda") else "fp32",
           "LT_MAXSEQ": "1024"}
    server = _spawn(SERVER_SRC, env, "SERVER_READY")
    workers = {}
    for i, m in enumerate(models):
        wenv = {**env, "LT_MODEL": m, "LT_WID": f"mixed-{m}"}
        workers[m] = _spawn(WORKER_SRC, wenv, "WORKER_READY")
    try:
        base = f"http://127.0.0.1:{http_port}"
        import httpx
        with httpx.Client(base_url=base, timeout=30) as c:
            for _ in range(300):
                r = c.get("/api/tags")
                names = {m0["name"] for m0 in r.json().get("models", [])} \
                    if r.status_code == 200 else set()
                if set(models) <= names:
                    break
                time.sleep(0.2)
            else:
                raise TimeoutError(f"models never all advertised: {names}")

        async def both():
            return await asyncio.gather(
                drive(base, models[0], args.requests, args.concurrency,
                      args.num_predict, 0),
                drive(base, models[1], args.requests, args.concurrency,
                      args.num_predict, 10_000))

        r3b, r1b = asyncio.run(both())
        out = {
            "metric": "mixed-fleet /api/generate streaming (2 models, "
                      "1 GPU)",
            "models": {
                models[0]: {"tok_s": round(sum(r3b[1]) / r3b[0], 1),
                            "errors": r3b[3],
                            "ttft_p50_ms": round(statistics.median(r3b[2]), 1)
                            if r3b[2] else None},
                models[1]: {"tok_s": round(sum(r1b[1]) / r1b[0], 1),
                            "errors": r1b[3],
                            "ttft_p50_ms": round(statistics.median(r1b[2]), 1)
                            if r1b[2] else None},
            },
            "requests_per_model": args.requests,
            "concurrency_per_model": args.concurrency,
        }
        # failover leg: kill the second worker mid-new-load; the FIRST
        # model's traffic must be unaffected
        async def failover():
            t = asyncio.get_running_loop().create_task(
                drive(base, models[0], args.requests // 2,
                      args.concurrency, args.num_predict, 50_000))
            await asyncio.sleep(1.0)
            workers[models[1]].send_signal(signal.SIGKILL)
            return await t

        fr = asyncio.run(failover())
        out["after_kill_" + models[1]] = {
            models[0] + "_tok_s": round(sum(fr[1]) / fr[0], 1),
            "errors": fr[3]}
        print(json.dumps(out))
    finally:
        for p in list(workers.values()) + [server]:
            if p.poll() is None:
                p.kill()


if __name__ == "__main__":
    main()
