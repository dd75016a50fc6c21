#!/usr/bin/env python3
"""Serving-path load test: real server process + real worker process, driven
through HTTP /api/generate streaming — measures the BASELINE metric on the
FULL path (gateway -> scheduler -> bus -> worker -> engine -> stream back),
not just the engine loop.

Usage: python scripts/loadtest.py [--model llama3.2:3b] [--concurrency 128]
         [--requests 256] [--num-predict 64] [--device cuda:0]
Prints one JSON line: tokens/s (node), TTFT p50/p95, request stats.
"""

import argparse
import asyncio
import json
import os
import signal
import socket
import statistics
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


SERVER_SRC = r"""
import asyncio, os
import uvicorn
from gridllm_amd.config import ServerConfig
from gridllm_amd.server.app import GridLLMServer

cfg = ServerConfig(port=int(os.environ["HTTP_PORT"]),
                   bus_port=int(os.environ["BUS_PORT"]))
server = GridLLMServer(cfg)
app = server.app

@app.on_event("startup")
async def _s():
    await server.start()
    if os.environ.get("LT_PROFILE_SERVER"):
        import cProfile, pstats, signal as _sig
        pr = cProfile.Profile(); pr.enable()
        def dump(*a):
            pr.disable()
            st = pstats.Stats(pr); st.sort_stats("tottime")
            with open("gpurun_out/lt_server_loop.prof.txt", "w") as f:
                st.stream = f
                st.print_stats(30)
            os._exit(0)
        _sig.signal(_sig.SIGTERM, dump)
    print("SERVER_READY", flush=True)

uvicorn.run(app, host="127.0.0.1", port=cfg.port, log_level="error",
            timeout_keep_alive=120)
"""

WORKER_SRC = r"""
import asyncio, os
from gridllm_amd.bus.tcp import TcpBus
from gridllm_amd.config import EngineConfig, WorkerConfig
from gridllm_amd.engine import InferenceEngine
from gridllm_amd.worker.agent import WorkerAgent

async def main():
    conc = int(os.environ["LT_CONC"])
    wid = os.environ.get("LT_WID", "lt-worker")
    profs = []
    if os.environ.get("LT_PROFILE"):
        # profile BOTH worker threads: asyncio loop (this thread) and the
        # batcher decode thread; dump on SIGTERM
        import cProfile, pstats, signal
        from gridllm_amd.engine.batcher import Batcher
        pr_loop = cProfile.Profile(); pr_loop.enable()
        profs.append(("loop", pr_loop))
        orig = Batcher._loop_continuous
        def prof_batcher(self):
            pr = cProfile.Profile(); pr.enable()
            profs.append(("batcher", pr))
            orig(self)
        Batcher._loop_continuous = prof_batcher
        def dump(*a):
            for tag, pr in profs:
                pr.disable()
                st = pstats.Stats(pr); st.sort_stats("tottime")
                with open(f"gpurun_out/lt_worker_{tag}.prof.txt", "w") as f:
                    st.stream = f
                    st.print_stats(30)
            os._exit(0)
        signal.signal(signal.SIGTERM, dump)
    wcfg = WorkerConfig(worker_id=wid,
                        bus_port=int(os.environ["BUS_PORT"]),
                        max_concurrent_tasks=conc)
    ecfg = EngineConfig(model=os.environ["LT_MODEL"],
                        device=os.environ["LT_DEVICE"],
                        dtype=os.environ["LT_DTYPE"],
                        max_seq_len=int(os.environ.get("LT_MAXSEQ", "1024")),
                        max_batch_size=conc,
                        use_hipgraph=os.environ["LT_DEVICE"].startswith("cuda"))
    engine = InferenceEngine(ecfg)
    if os.environ.get("LT_SLOW"):
        # artificial per-step delay: reproduces slow-host timing (chaos
        # wedge hunting) without a GPU
        import time as _t
        _slow = float(os.environ["LT_SLOW"])
        _orig = engine._decode_step
        def _slow_step(tokens, slots, **kw):
            _t.sleep(_slow)
            return _orig(tokens, slots, **kw)
        engine._decode_step = _slow_step
        _origsub = engine.submit_step
        def _slow_submit(rows, prev=None):
            _t.sleep(_slow)
            return _origsub(rows, prev)
        engine.submit_step = _slow_submit
    if os.environ.get("LT_FAKE"):
        # stub the model: measures the serving GLUE ceiling (bus, pump,
        # gateway, GIL) with near-zero compute per token
        import torch
        m, V = engine.model, engine.spec.vocab
        eos = engine.tokenizer.eos_id
        fake = torch.randn(conc + 1, V)
        fake[:, eos] = -100.0

        def fake_prefill_batch(token_lists, slots, return_hidden=False):
            for t, s in zip(token_lists, slots):
                m.cache.seq_lens[s] = len(t)
            return fake[:len(slots)].clone()   # sampler may scatter in place

        def fake_decode(tokens, slot_map):
            m.cache.seq_lens[slot_map.long()] += 1
            return fake[:tokens.shape[0]].clone()

        m.forward_prefill_batch = fake_prefill_batch
        m.forward_decode = fake_decode
    bus = TcpBus("127.0.0.1", wcfg.bus_port)
    await bus.connect()
    agent = WorkerAgent(bus, engine, wcfg)
    await agent.start()
    print("WORKER_READY", flush=True)
    await asyncio.Event().wait()

asyncio.run(main())
"""


def _spawn(src, env, tag, timeout=300):
    logf = None
    if os.environ.get("LT_DEBUG"):
        logf = open(f"gpurun_out/lt_{tag.lower()}.log", "w")
    proc = subprocess.Popen([sys.executable, "-u", "-c", src],
                            env={**os.environ, **env, "PYTHONPATH": REPO},
                            stdout=subprocess.PIPE,
                            stderr=subprocess.STDOUT, text=True, cwd=REPO)
    t0 = time.time()
    while time.time() - t0 < timeout:
        line = proc.stdout.readline()
        if logf and line:
            logf.write(line)
            logf.flush()
        if tag in line:
            import threading

            def _drain(p=proc, lf=logf):
                try:
                    for ln in p.stdout:
                        if lf:
                            lf.write(ln)
                            lf.flush()
                except Exception:  # noqa: BLE001
                    pass
            threading.Thread(target=_drain, daemon=True).start()
            return proc
        if proc.poll() is not None:
            raise RuntimeError(f"process died:\n{proc.stdout.read()}")
    proc.kill()
    raise TimeoutError(f"no {tag} within {timeout}s")


async def run_clients(base, args):
    import aiohttp
    sem = asyncio.Semaphore(args.concurrency)
    ttfts, tokens, errors = [], [], [0]
    conn = aiohttp.TCPConnector(limit=args.concurrency + 16)
    async with aiohttp.ClientSession(base_url=base, connector=conn) as sess:

        async def one(i):
            async with sem:
                t0 = time.monotonic()
                first = None
                n = 0
                try:
                    async with sess.post("/api/generate", json={
                            "model": args.model,
                            "prompt": f"request {i}: " + "lorem " * args.prompt_words,
                            "stream": True,
                            "options": {"temperature": 0.8, "seed": i,
                                        "num_predict": args.num_predict}},
                            timeout=aiohttp.ClientTimeout(total=600)) as resp:
                        async for raw in resp.content:
                            if not raw.strip():
                                continue
                            c = json.loads(raw)
                            if first is None and c.get("response"):
                                first = time.monotonic() - t0
                            if c.get("done"):
                                n = c.get("eval_count", 0)
                except Exception:  # noqa: BLE001
                    errors[0] += 1
                    return
                if first is not None:
                    ttfts.append(first * 1000)
                tokens.append(n)
                if len(tokens) % 64 == 0:
                    print(f"  {len(tokens)} requests done",
                          file=sys.stderr, flush=True)

        t0 = time.monotonic()
        await asyncio.gather(*(one(i) for i in range(args.requests)))
        wall = time.monotonic() - t0
    return wall, ttfts, tokens, errors[0]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama3.2:3b")
    ap.add_argument("--device", default="cuda:0")
    ap.add_argument("--dtype", default="bf16")
    ap.add_argument("--concurrency", type=int, default=128)
    ap.add_argument("--requests", type=int, default=256)
    ap.add_argument("--num-predict", type=int, default=64)
    ap.add_argument("--prompt-words", type=int, default=24)
    ap.add_argument("--max-seq", type=int, default=1024)
    ap.add_argument("--workers", type=int, default=1,
                    help="DP workers; worker i pins cuda:i")
    ap.add_argument("--same-device", action="store_true",
                    help="all DP workers share --device (single-GPU DP "
                         "validation, BASELINE config 3 on one lease)")
    args = ap.parse_args()

    import torch
    if not torch.cuda.is_available():
        args.device, args.dtype, args.model = "cpu", "fp32", "tiny"
        if not os.environ.get("LT_KEEP_CONC"):
            args.concurrency, args.requests = 4, 16

    bus_port, http_port = _free_port(), _free_port()
    env = {"BUS_PORT": str(bus_port), "HTTP_PORT": str(http_port),
           "LT_CONC": str(args.concurrency), "LT_MODEL": args.model,
           "LT_DEVICE": args.device, "LT_DTYPE": args.dtype,
           "LT_MAXSEQ": str(args.max_seq)}
    print("spawning server...", file=sys.stderr, flush=True)
    server = _spawn(SERVER_SRC, env, "SERVER_READY")
    workers = []
    for i in range(args.workers):
        wenv = dict(env)
        wenv["LT_WID"] = f"lt-worker-{i}"
        if args.device.startswith("cuda") and not args.same_device:
            wenv["LT_DEVICE"] = f"cuda:{i}"
        print(f"spawning worker {i}...", file=sys.stderr, flush=True)
        workers.append(_spawn(WORKER_SRC, wenv, "WORKER_READY"))
    worker = workers[0]
    try:
        base = f"http://127.0.0.1:{http_port}"
        # registration wait
        import httpx
        with httpx.Client(base_url=base, timeout=30) as c:
            for _ in range(200):
                r = c.get("/health/workers")
                if r.status_code == 200 and \
                        r.json()["counts"]["total"] >= args.workers:
                    break
                time.sleep(0.1)
        print("workers registered; warmup...", file=sys.stderr, flush=True)
        # full-concurrency warmup: captures every hipGraph bucket the main
        # wave will hit, so one-time capture cost stays out of the numbers
        wa = argparse.Namespace(**vars(args))
        wa.requests = args.concurrency * 2
        asyncio.run(run_clients(base, wa))
        print("warmup done; main wave...", file=sys.stderr, flush=True)
        wall, ttfts, tokens, errs = asyncio.run(run_clients(base, args))
        out = {
            "metric": "output tokens/sec (node), /api/generate streaming",
            "value": round(sum(tokens) / wall, 1),
            "unit": "tokens/s",
            "requests": args.requests,
            "concurrency": args.concurrency,
            "num_predict": args.num_predict,
            "errors": errs,
            "wall_s": round(wall, 2),
            "ttft_p50_ms": round(statistics.median(ttfts), 2) if ttfts else None,
            "ttft_p95_ms": round(sorted(ttfts)[int(len(ttfts) * 0.95)], 2)
            if ttfts else None,
            "model": args.model,
            "device": args.device,
            "workers": args.workers,
        }
        print(json.dumps(out))
    finally:
        for p in workers + [server]:
            if p.poll() is None:
                p.send_signal(signal.SIGTERM)
        time.sleep(1.0 if os.environ.get("LT_PROFILE") else 0)
        for p in workers + [server]:
            if p.poll() is None:
                p.send_signal(signal.SIGKILL)
        for p in workers + [server]:
            p.wait(timeout=10)


if __name__ == "__main__":
    main()
