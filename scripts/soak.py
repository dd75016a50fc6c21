#!/usr/bin/env python3
"""Feature-matrix soak: every API surface through the FULL stack at once
(server + worker processes, HTTP in, bus between) with per-feature
invariants checked on every response — the "does everything still work
together" gate. Complements scripts/chaos.py (fault injection) and
scripts/loadtest.py (throughput).

Mix: generate (stream/non-stream), chat, embeddings, format=json,
penalties, seeded determinism pairs, context continuation, cancellation,
tools, OpenAI completions/chat.

Usage: python scripts/soak.py [--requests 160] [--concurrency 16]
Exit 0 = all invariants held.
"""

import argparse
import asyncio
import json
import os
import signal
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import loadtest  # noqa: E402


async def run_mix(base: str, args) -> dict:
    import aiohttp
    sem = asyncio.Semaphore(args.concurrency)
    stats = {"done": 0, "failed": [], "kinds": {}}
    TOOLS = [{"type": "function", "function": {
        "name": "lookup", "parameters": {"type": "object"}}}]

    async def check(i, sess):
        kind = KINDS[i % len(KINDS)]
        stats["kinds"][kind] = stats["kinds"].get(kind, 0) + 1
        try:
            if kind == "generate":
                async with sess.post("/api/generate", json={
                        "model": "tiny", "stream": False,
                        "prompt": f"soak {i} " + "word " * 10,
                        "options": {"seed": i, "num_predict": 24}}) as r:
                    b = await r.json()
                    assert r.status == 200 and b["done"] and \
                        b["eval_count"] > 0, b
            elif kind == "stream":
                async with sess.post("/api/generate", json={
                        "model": "tiny", "stream": True,
                        "prompt": f"soak {i}",
                        "options": {"seed": i, "num_predict": 24}}) as r:
                    n, done = 0, False
                    async for raw in r.content:
                        if not raw.strip():
                            continue
                        c = json.loads(raw)
                        n += 1
                        done = done or c.get("done", False)
                    assert done and n >= 1, (n, done)
            elif kind == "chat":
                async with sess.post("/api/chat", json={
                        "model": "tiny", "stream": False,
                        "messages": [
                            {"role": "system", "content": "terse"},
                            {"role": "user", "content": f"q{i}"}],
                        "options": {"num_predict": 16}}) as r:
                    b = await r.json()
                    assert r.status == 200 and \
                        b["message"]["role"] == "assistant", b
            elif kind == "embed":
                async with sess.post("/api/embed", json={
                        "model": "tiny",
                        "input": [f"a{i}", f"b{i}"]}) as r:
                    b = await r.json()
                    assert len(b["embeddings"]) == 2 and \
                        len(b["embeddings"][0]) > 0, b
            elif kind == "json":
                async with sess.post("/api/generate", json={
                        "model": "tiny", "stream": False, "format": "json",
                        "prompt": "emit json", "options":
                        {"seed": i, "temperature": 1.0,
                         "num_predict": 48}}) as r:
                    b = await r.json()
                    if b["done_reason"] != "length":   # cap can cut a string
                        json.loads(b["response"])      # else must parse
            elif kind == "penalties":
                async with sess.post("/api/generate", json={
                        "model": "tiny", "stream": False,
                        "prompt": f"soak {i}",
                        "options": {"seed": i, "num_predict": 24,
                                    "repeat_penalty": 1.3,
                                    "presence_penalty": 0.5,
                                    "frequency_penalty": 0.2}}) as r:
                    b = await r.json()
                    assert b["eval_count"] > 0, b
            elif kind == "seeded_pair":
                payload = {"model": "tiny", "stream": False,
                           "prompt": "determinism",
                           "options": {"seed": 4242, "temperature": 0.9,
                                       "num_predict": 16}}
                async with sess.post("/api/generate", json=payload) as r1:
                    b1 = await r1.json()
                async with sess.post("/api/generate", json=payload) as r2:
                    b2 = await r2.json()
                assert b1["response"] == b2["response"], (b1, b2)
            elif kind == "context":
                o = {"temperature": 0.0, "num_predict": 6}
                async with sess.post("/api/generate", json={
                        "model": "tiny", "prompt": "ctx", "stream": False,
                        "options": o}) as r1:
                    b1 = await r1.json()
                async with sess.post("/api/generate", json={
                        "model": "tiny", "prompt": " more", "stream": False,
                        "context": b1["context"], "options": o}) as r2:
                    b2 = await r2.json()
                    assert b2["context"], b2
            elif kind == "cancel":
                # fire a long job, cancel via native API, expect it to end
                req = {"model": "tiny", "stream": False,
                       "prompt": "long", "request_type": "generate",
                       "options": {"num_predict": 200}, "priority": "low"}
                async with sess.post("/inference", json=req,
                                     timeout=aiohttp.ClientTimeout(
                                         total=60)) as r:
                    assert r.status == 200
            elif kind == "tools":
                async with sess.post("/v1/chat/completions", json={
                        "model": "tiny", "stream": False, "tools": TOOLS,
                        "tool_choice": "required",
                        "messages": [{"role": "user", "content": "call"}],
                        "max_tokens": 32, "seed": i,
                        "temperature": 1.0}) as r:
                    b = await r.json()
                    ch = b["choices"][0]
                    msg = ch["message"]
                    if not msg.get("tool_calls") and \
                            ch["finish_reason"] != "length":
                        json.loads(msg["content"])     # forced json
            elif kind == "schema":
                sch = {"type": "object",
                       "properties": {"ok": {"type": "boolean"},
                                      "n": {"type": "integer"}},
                       "required": ["ok"], "additionalProperties": False}
                async with sess.post("/api/generate", json={
                        "model": "tiny", "stream": False, "format": sch,
                        "prompt": "emit", "options":
                        {"seed": i, "temperature": 1.0,
                         "num_predict": 64}}) as r:
                    b = await r.json()
                    if b["done_reason"] != "length":
                        doc = json.loads(b["response"])
                        assert isinstance(doc.get("ok"), bool) and \
                            set(doc) <= {"ok", "n"}, doc
            elif kind == "openai":
                async with sess.post("/v1/completions", json={
                        "model": "tiny", "prompt": f"c{i}",
                        "max_tokens": 12, "seed": i}) as r:
                    b = await r.json()
                    assert b["choices"][0]["text"] is not None, b
            stats["done"] += 1
        except Exception as e:  # noqa: BLE001
            stats["failed"].append(f"{kind}#{i}: {e!r}")

    KINDS = ["generate", "stream", "chat", "embed", "json", "penalties",
             "seeded_pair", "context", "cancel", "tools", "openai",
             "schema"]
    conn = aiohttp.TCPConnector(limit=args.concurrency + 8)
    async with aiohttp.ClientSession(
            base_url=base, connector=conn,
            timeout=aiohttp.ClientTimeout(total=120)) as sess:

        async def gated(i):
            async with sem:
                await check(i, sess)

        await asyncio.gather(*(gated(i) for i in range(args.requests)))
    return stats


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--requests", type=int, default=160)
    ap.add_argument("--concurrency", type=int, default=16)
    args = ap.parse_args()
    bus_port, http_port = loadtest._free_port(), loadtest._free_port()
    env = {"BUS_PORT": str(bus_port), "HTTP_PORT": str(http_port),
           "LT_CONC": "32", "LT_MODEL": "tiny", "LT_DEVICE": "cpu",
           "LT_DTYPE": "fp32", "LT_MAXSEQ": "256"}
    server = loadtest._spawn(loadtest.SERVER_SRC, env, "SERVER_READY")
    worker = loadtest._spawn(loadtest.WORKER_SRC, env, "WORKER_READY")
    try:
        import httpx
        base = f"http://127.0.0.1:{http_port}"
        with httpx.Client(base_url=base, timeout=30) as c:
            for _ in range(200):
                r = c.get("/health/workers")
                if r.status_code == 200 and \
                        r.json()["counts"]["total"] >= 1:
                    break
                time.sleep(0.1)
        t0 = time.monotonic()
        stats = asyncio.run(run_mix(base, args))
        out = {"requests": args.requests, "completed": stats["done"],
               "failed": len(stats["failed"]), "kinds": stats["kinds"],
               "wall_s": round(time.monotonic() - t0, 2),
               "ok": not stats["failed"]}
        print(json.dumps(out))
        for f in stats["failed"][:10]:
            print("FAIL:", f, file=sys.stderr)
        sys.exit(0 if out["ok"] else 1)
    finally:
        for p in (worker, server):
            if p.poll() is None:
                p.send_signal(signal.SIGKILL)
        for p in (worker, server):
            try:
                p.wait(timeout=10)
            except Exception:  # noqa: BLE001
                pass


if __name__ == "__main__":
    main()
