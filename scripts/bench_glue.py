#!/usr/bin/env python3
"""Serving-glue cost isolation (CPU-runnable).

Measures the same request mix twice:
  1. engine+batcher only (in-process, on_token -> lock+append like the pump)
  2. is loadtest.py's job (full HTTP/bus path)
The delta between this number and scripts/loadtest.py at the same settings is
the control-plane glue cost (bus frames, pump, gateway NDJSON, GIL mixing).

Usage: python scripts/bench_glue.py [--concurrency 64] [--requests 192]
         [--num-predict 64] [--profile]
"""

import argparse
import asyncio
import os
import sys
import threading
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


async def run(args):
    from gridllm_amd.config import EngineConfig
    from gridllm_amd.engine import InferenceEngine
    from gridllm_amd.engine.batcher import Batcher

    ecfg = EngineConfig(model="tiny", device="cpu", dtype="fp32",
                        max_seq_len=256, max_batch_size=args.concurrency,
                        use_hipgraph=False)
    engine = InferenceEngine(ecfg)
    batcher = Batcher(engine)

    total = [0]
    lock = threading.Lock()
    loop = asyncio.get_running_loop()

    async def one(i):
        buf = []

        def on_token(piece, done):
            with lock:                     # what the agent's pump does
                buf.append(piece)

        fut = batcher.submit_generate(
            f"request {i}: " + "lorem " * 24,
            {"temperature": 0.8, "seed": i,
             "num_predict": args.num_predict},
            on_token=on_token)
        res = await asyncio.wrap_future(fut, loop=loop)
        with lock:
            total[0] += res.eval_count

    sem = asyncio.Semaphore(args.concurrency)

    async def gated(i):
        async with sem:
            await one(i)

    # warmup
    await asyncio.gather(*(gated(i) for i in range(args.concurrency)))
    total[0] = 0
    t0 = time.monotonic()
    await asyncio.gather(*(gated(i) for i in range(args.requests)))
    wall = time.monotonic() - t0
    batcher.shutdown()
    print({"engine_batcher_tokens_per_s": round(total[0] / wall, 1),
           "wall_s": round(wall, 2), "tokens": total[0],
           "concurrency": args.concurrency, "requests": args.requests})


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--concurrency", type=int, default=64)
    ap.add_argument("--requests", type=int, default=192)
    ap.add_argument("--num-predict", type=int, default=64)
    ap.add_argument("--profile", action="store_true")
    args = ap.parse_args()
    if args.profile:
        # the work lives in the batcher THREAD — profile that, not the loop
        import cProfile
        import pstats
        from gridllm_amd.engine.batcher import Batcher
        orig = Batcher._loop_continuous

        def prof_loop(self):
            pr = cProfile.Profile()
            pr.enable()
            try:
                orig(self)
            finally:
                pr.disable()
                st = pstats.Stats(pr)
                st.sort_stats("tottime")
                st.dump_stats("gpurun_out/glue_profile.pstats")
                st.print_stats(25)

        Batcher._loop_continuous = prof_loop
        asyncio.run(run(args))
    else:
        asyncio.run(run(args))


if __name__ == "__main__":
    main()
