#!/usr/bin/env python3
"""Flagship serving benchmark — the BASELINE.json headline metric.

Default mode times the FULL serving path: HTTP /api/generate streaming
through gateway -> scheduler -> bus -> worker -> engine -> direct stream
back (the reference contract at server/src/routes/ollama.ts:161-319).
One rank per GPU hosts a worker (engine pinned to cuda:LOCAL_RANK); rank 0
additionally spawns the scheduler server, N gateway replicas and N async
load generators. Each load generator keeps a CONTINUOUSLY-REPLENISHED
closed loop of `--concurrency` streaming requests in flight (a finished
request is replaced at once); a "step" is the next `--concurrency`
streamed completions per GPU, i.e. concurrency x num_predict output
tokens — identical work per step, but without the drained-wave artifact
that under-measured steady-state capacity. K steps are timed between
barriers and the whole-node tokens/s printed as ONE JSON line.

`--mode engine` preserves the raw engine-decode benchmark (hipGraph decode
+ batched sampling, no serving glue) for kernel-path comparisons.

Usage: python bench.py [--gpus N] [--steps K] [--warmup W] [--mode M]
Launched for N>1 via torch.distributed.run (reads RANK/WORLD_SIZE env).
"""

from __future__ import annotations

import argparse
import asyncio
import json
import os
import statistics
import subprocess
import sys
import time

import torch

REPO = os.path.dirname(os.path.abspath(__file__))

SERVER_SRC = r"""
import asyncio, os
import uvicorn
from gridllm_amd.config import ServerConfig
from gridllm_amd.server.app import GridLLMServer

cfg = ServerConfig(host="127.0.0.1", port=int(os.environ["HTTP_PORT"]),
                   bus_port=int(os.environ["BUS_PORT"]),
                   max_concurrent_jobs_per_worker=4096)
server = GridLLMServer(cfg)
app = server.app

@app.on_event("startup")
async def _s():
    await server.start()
    print("SERVER_READY", flush=True)

uvicorn.run(app, host="127.0.0.1", port=cfg.port, log_level="error",
            timeout_keep_alive=120)
"""

LOADGEN_SRC = r"""
# Continuously-replenished closed loop at concurrency CONC: a finished
# request is immediately replaced, so the server sees a steady CONC in
# flight across step boundaries (no inter-wave drain — the drained-wave
# design under-measured steady-state capacity by ~5%). One "STEP" =
# the next CONC streamed completions (CONC * NPRED output tokens).
import asyncio, json, os, statistics, sys, time
import aiohttp

BASE = os.environ["LG_BASE"]
CONC = int(os.environ["LG_CONC"])
NPRED = int(os.environ["LG_NPREDICT"])
MODEL = os.environ["LG_MODEL"]
PROMPT = "benchmark prompt " + "lorem ipsum dolor " * 8

async def main():
    conn = aiohttp.TCPConnector(limit=CONC + 16)
    seed0 = int(os.environ.get("LG_SEED", "0")) * 10_000_000
    completions = []   # (eval_count, ttft_ms|None, errored) per finished req
    consumed = [0]
    cond = asyncio.Condition()
    sem = asyncio.Semaphore(CONC)

    async with aiohttp.ClientSession(base_url=BASE, connector=conn) as sess:

        async def one(i):
            t0 = time.monotonic()
            first = None
            n = 0
            err = False
            try:
                async with sess.post("/api/generate", json={
                        "model": MODEL,
                        "prompt": f"req {i}: " + PROMPT,
                        "stream": True,
                        "options": {"temperature": 0.8, "seed": i,
                                    "num_predict": NPRED}},
                        timeout=aiohttp.ClientTimeout(total=600)) as resp:
                    async for raw in resp.content:
                        if not raw.strip():
                            continue
                        c = json.loads(raw)
                        if first is None and c.get("response"):
                            first = time.monotonic() - t0
                        if c.get("done"):
                            n = int(c.get("eval_count", 0))
            except Exception:
                err = True
            async with cond:
                completions.append(
                    (n, None if first is None else first * 1000.0, err))
                cond.notify_all()
            sem.release()

        launched = [0]

        async def pump():
            i = seed0
            while True:
                await sem.acquire()
                i += 1
                launched[0] += 1
                asyncio.get_running_loop().create_task(one(i))

        pump_task = asyncio.get_running_loop().create_task(pump())
        print("LOADGEN_READY", flush=True)
        loop = asyncio.get_running_loop()
        while True:
            line = await loop.run_in_executor(None, sys.stdin.readline)
            if not line or line.strip() == "QUIT":
                # drain in-flight requests before exiting: tearing the
                # in-process worker down mid-generation aborts at exit
                pump_task.cancel()
                try:
                    async with cond:
                        await asyncio.wait_for(
                            cond.wait_for(
                                lambda: len(completions) >= launched[0]),
                            timeout=120)
                except asyncio.TimeoutError:
                    pass
                print("DRAINED", flush=True)
                return
            target = consumed[0] + CONC
            async with cond:
                await cond.wait_for(lambda: len(completions) >= target)
            batch = completions[consumed[0]:target]
            consumed[0] = target
            toks = sum(n for n, _, _ in batch)
            ttfts = [t for _, t, _ in batch if t is not None]
            errs = sum(1 for _, _, e in batch if e)
            p50 = statistics.median(ttfts) if ttfts else 0.0
            print(f"DONE {toks} {p50:.2f} {errs}", flush=True)

asyncio.run(main())
"""


def _spawn(argv, env, ready_tag, timeout=420):
    proc = subprocess.Popen(argv, env={**os.environ, **env,
                                       "PYTHONPATH": REPO},
                            stdin=subprocess.PIPE, stdout=subprocess.PIPE,
                            stderr=subprocess.STDOUT, text=True, cwd=REPO)
    t0 = time.time()
    while time.time() - t0 < timeout:
        line = proc.stdout.readline()
        if ready_tag in line:
            return proc
        if proc.poll() is not None:
            raise RuntimeError(f"{ready_tag} process died:\n"
                               f"{proc.stdout.read()[-4000:]}")
    proc.kill()
    raise TimeoutError(f"no {ready_tag} within {timeout}s")


def _worker_main(args, device, bus_port, worker_id):
    """Run engine + worker agent in THIS rank process (asyncio loop);
    returns (engine, loop_thread, ready_event)."""
    import threading

    from gridllm_amd.bus.tcp import TcpBus
    from gridllm_amd.config import EngineConfig, WorkerConfig
    from gridllm_amd.engine import InferenceEngine
    from gridllm_amd.worker.agent import WorkerAgent

    on_gpu = device.startswith("cuda")
    ecfg = EngineConfig(
        model=args.model if on_gpu else "tiny",
        device=device, dtype="bf16" if on_gpu else "fp32",
        max_seq_len=args.max_seq,
        max_batch_size=args.concurrency + 8,
        use_hipgraph=on_gpu)
    engine = InferenceEngine(ecfg)
    wcfg = WorkerConfig(worker_id=worker_id, bus_port=bus_port,
                        max_concurrent_tasks=args.concurrency + 8,
                        stream_flush_ms=args.flush_ms)
    ready = threading.Event()

    def run():
        async def amain():
            bus = TcpBus("127.0.0.1", bus_port)
            for _ in range(120):
                try:
                    await bus.connect()
                    break
                except OSError:
                    await asyncio.sleep(0.5)
            agent = WorkerAgent(bus, engine, wcfg)
            await agent.start()
            ready.set()
            await asyncio.Event().wait()

        asyncio.run(amain())

    t = threading.Thread(target=run, daemon=True, name="bench-worker")
    t.start()
    return engine, t, ready


def serving_mode(args, rank, world, local_rank, dist, device, on_gpu):
    import httpx

    n_gpus = world if world > 1 else args.gpus
    master_port = int(os.environ.get("MASTER_PORT", "29500"))
    bus_port = master_port + 101
    http_port = master_port + 102
    gw_port0 = master_port + 110

    procs = []
    try:
        if rank == 0:
            env = {"BUS_PORT": str(bus_port), "HTTP_PORT": str(http_port)}
            procs.append(_spawn([sys.executable, "-u", "-c", SERVER_SRC],
                                env, "SERVER_READY"))
        if dist is not None:
            dist.barrier()          # bus up before workers join

        _engine, _t, ready = _worker_main(
            args, device, bus_port, f"bench-w{rank}")
        if not ready.wait(timeout=600):
            raise TimeoutError("worker never started")

        gateways = []
        loadgens = []
        if rank == 0:
            # every worker registered?
            with httpx.Client(timeout=10) as c:
                for _ in range(600):
                    try:
                        r = c.get(f"http://127.0.0.1:{http_port}"
                                  "/health/workers")
                        if r.status_code == 200 and \
                                r.json()["counts"]["total"] >= world:
                            break
                    except Exception:  # noqa: BLE001
                        pass
                    time.sleep(0.25)
                else:
                    raise TimeoutError("workers never all registered")
            # one gateway replica + one load generator per GPU (or
            # --gateways replicas, load spread round-robin)
            n_gw = args.gateways or n_gpus
            for i in range(n_gw):
                gp = gw_port0 + i
                gateways.append(_spawn(
                    [sys.executable, "-m", "gridllm_amd.gateway",
                     "--port", str(gp), "--host", "127.0.0.1",
                     "--bus", f"127.0.0.1:{bus_port}"],
                    {}, "GATEWAY_READY"))
            for i in range(n_gpus):
                gp = gw_port0 + (i % n_gw)
                loadgens.append(_spawn(
                    [sys.executable, "-u", "-c", LOADGEN_SRC],
                    {"LG_BASE": f"http://127.0.0.1:{gp}",
                     "LG_CONC": str(args.concurrency),
                     "LG_NPREDICT": str(args.num_predict),
                     "LG_MODEL": args.model if on_gpu else "tiny",
                     "LG_SEED": str(i)},
                    "LOADGEN_READY"))
            procs.extend(gateways)
            procs.extend(loadgens)

        def run_step():
            """One wave on every load generator; returns (tokens, ttft)."""
            for lg in loadgens:
                lg.stdin.write("STEP\n")
                lg.stdin.flush()
            toks = 0
            ttfts = []
            errs = 0
            for lg in loadgens:
                line = lg.stdout.readline()
                while line and not line.startswith("DONE"):
                    line = lg.stdout.readline()
                if not line:
                    raise RuntimeError("load generator died")
                _, t, p50, e = line.split()
                toks += int(t)
                ttfts.append(float(p50))
                errs += int(e)
            if errs:
                print(f"WARNING: {errs} request errors", file=sys.stderr)
            return toks, statistics.median(ttfts) if ttfts else 0.0

        def barrier_sync():
            if on_gpu:
                torch.cuda.synchronize()
            if dist is not None:
                dist.barrier()
            if on_gpu:
                torch.cuda.synchronize()

        # warmup (captures hipGraphs, warms gateways/conn pools)
        for _ in range(args.warmup):
            if rank == 0:
                run_step()
        barrier_sync()
        t0 = time.monotonic()
        total_tokens = 0
        ttft_p50s = []
        for _ in range(args.steps):
            if rank == 0:
                tok, p50 = run_step()
                total_tokens += tok
                ttft_p50s.append(p50)
        barrier_sync()
        t1 = time.monotonic()

        elapsed = t1 - t0
        if dist is not None:
            e = torch.tensor([elapsed], dtype=torch.float64)
            if on_gpu:
                e = e.to(device)
            dist.all_reduce(e, op=dist.ReduceOp.MAX)
            elapsed = float(e.item())

        if rank == 0:
            toks_per_sec = total_tokens / elapsed
            out = {
                "metric": "output tokens/sec (node), llama3.2 "
                          "/api/generate streaming",
                "value": round(toks_per_sec, 1),
                "unit": "tokens/s",
                "n_gpus": n_gpus,
                "steps": args.steps,
                "warmup": args.warmup,
                "ms_per_step": round(elapsed * 1000.0 / args.steps, 3),
                "higher_is_better": True,
                "scaling": "weak",
                "vs_baseline": None,
                "dtype": "bf16" if on_gpu else "fp32",
                "data": "synthetic",
                "config": {
                    "model": args.model if on_gpu else "tiny",
                    "global_batch": args.concurrency * n_gpus,
                    "seq_len": args.prompt_tokens,
                    "num_predict": args.num_predict,
                    "parallelism": f"dp{n_gpus}",
                    "mode": "serving",
                    "gateways": args.gateways or n_gpus,
                    "ttft_p50_ms": round(statistics.median(ttft_p50s), 2)
                    if ttft_p50s else None,
                    "tokens_timed": total_tokens,
                },
            }
            print(json.dumps(out))
            for lg in loadgens:
                try:
                    lg.stdin.write("QUIT\n")
                    lg.stdin.flush()
                except Exception:  # noqa: BLE001
                    pass
            for lg in loadgens:   # wait for in-flight drain (see loadgen)
                try:
                    line = lg.stdout.readline()
                    while line and "DRAINED" not in line:
                        line = lg.stdout.readline()
                except Exception:  # noqa: BLE001
                    pass
        # all ranks hold their in-process workers up until rank 0's
        # loadgens have drained — exiting mid-generation aborts at exit
        if dist is not None:
            dist.barrier()
    finally:
        for p in procs:
            if p.poll() is None:
                p.kill()
        for p in procs:
            try:
                p.wait(timeout=10)
            except Exception:  # noqa: BLE001
                pass


def engine_mode(args, rank, world, local_rank, dist, device, on_gpu):
    from gridllm_amd.config import EngineConfig
    from gridllm_amd.engine import InferenceEngine
    from gridllm_amd.engine.sampler import SamplerState
    from gridllm_amd.types import SamplingOptions

    n_gpus = world if world > 1 else args.gpus
    model = args.model if on_gpu else "tiny"
    batch = args.batch if on_gpu else 4
    cfg = EngineConfig(model=model, device=device,
                       dtype="bf16" if on_gpu else "fp32",
                       max_seq_len=args.max_seq, max_batch_size=batch + 1,
                       use_hipgraph=on_gpu)
    eng = InferenceEngine(cfg)

    torch.manual_seed(1000 + rank)
    slots = [eng.model.cache.alloc() for _ in range(batch)]
    vocab = eng.spec.vocab
    for i in range(0, batch, 64):
        chunk = slots[i:i + 64]
        toks = [torch.randint(0, vocab, (args.prompt_tokens,),
                              device=eng.device) for _ in chunk]
        if len(chunk) > 1:
            eng.model.forward_prefill_batch(toks, chunk)
        else:
            eng.model.forward_prefill(toks[0], chunk[0])

    opts = SamplingOptions.from_dict({"temperature": 0.0})

    class _R:  # minimal row shims for the engine's batched sampler
        def __init__(self):
            self.state = SamplerState.create(opts, eng.device)
            self.constraint = None
    rows = [_R() for _ in range(batch)]
    tokens = [int(torch.randint(0, vocab, (1,)).item())
              for _ in range(batch)]

    def step():
        logits = eng._decode_step(tokens, slots)
        return eng._sample_batch(logits, rows)

    for _ in range(args.warmup):
        tokens = step()

    def barrier_sync():
        if on_gpu:
            torch.cuda.synchronize()
        if dist is not None:
            dist.barrier()
        if on_gpu:
            torch.cuda.synchronize()

    barrier_sync()
    t0 = time.monotonic()
    for _ in range(args.steps):
        tokens = step()
    barrier_sync()
    t1 = time.monotonic()

    elapsed = t1 - t0
    if dist is not None:
        e = torch.tensor([elapsed], dtype=torch.float64)
        if on_gpu:
            e = e.to(device)
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        elapsed = float(e.item())

    # p50 TTFT (prefill + first token), outside the timed region
    ttfts = []
    for _ in range(5):
        s = eng.model.cache.alloc()
        ids = torch.randint(0, vocab, (args.prompt_tokens,),
                            device=eng.device)
        if on_gpu:
            torch.cuda.synchronize()
        tt = time.monotonic()
        logits = eng.model.forward_prefill(ids, s)
        _ = int(logits.argmax().item())
        ttfts.append((time.monotonic() - tt) * 1000.0)
        eng.model.cache.free(s)

    if rank == 0:
        total_tokens = args.steps * batch * n_gpus
        out = {
            "metric": "output tokens/sec (node), llama3.2 decode",
            "value": round(total_tokens / elapsed, 1),
            "unit": "tokens/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed * 1000.0 / args.steps, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": cfg.dtype,
            "data": "synthetic",
            "config": {
                "model": model,
                "global_batch": batch * n_gpus,
                "seq_len": args.prompt_tokens,
                "max_seq": args.max_seq,
                "parallelism": f"dp{n_gpus}",
                "mode": "engine",
                "ttft_p50_ms": round(statistics.median(ttfts), 2),
            },
        }
        print(json.dumps(out))


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--mode", choices=["serving", "engine"],
                    default="serving")
    ap.add_argument("--model", default="llama3.2:3b")
    ap.add_argument("--concurrency", type=int, default=1024,
                    help="serving: concurrent streams per GPU")
    ap.add_argument("--num-predict", type=int, default=128,
                    help="serving: tokens generated per request")
    ap.add_argument("--flush-ms", type=int, default=25,
                    help="serving: worker stream flush interval")
    ap.add_argument("--gateways", type=int, default=0,
                    help="serving: gateway replicas (0 = one per GPU)")
    ap.add_argument("--batch", type=int, default=1024,
                    help="engine: sequences per GPU")
    ap.add_argument("--prompt-tokens", type=int, default=128)
    ap.add_argument("--prompt-len", type=int, dest="prompt_tokens",
                    help=argparse.SUPPRESS)   # back-compat alias
    ap.add_argument("--max-seq", type=int, default=1024)
    ap.add_argument("--device", default="")
    args = ap.parse_args()

    # keep stdout clean for the one result JSON line (worker/engine logs
    # go to stderr-level warn only)
    os.environ.setdefault("LOG_LEVEL", "warn")

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    on_gpu = torch.cuda.is_available()
    device = args.device or (f"cuda:{local_rank}" if on_gpu else "cpu")
    if on_gpu:
        torch.cuda.set_device(device)
    if not on_gpu:
        # CPU smoke defaults: tiny waves so the no-GPU path finishes fast
        args.concurrency = min(args.concurrency, 8)
        args.num_predict = min(args.num_predict, 8)
        args.prompt_tokens = min(args.prompt_tokens, 16)

    dist = None
    if world > 1:
        import torch.distributed as tdist
        dist = tdist
        dist.init_process_group(backend="nccl" if on_gpu else "gloo")

    if args.mode == "serving":
        serving_mode(args, rank, world, local_rank, dist, device, on_gpu)
    else:
        engine_mode(args, rank, world, local_rank, dist, device, on_gpu)

    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
