#!/usr/bin/env python3
"""Interactive-latency probe: SEQUENTIAL single streaming requests
through the full serving path — the TTFT a lone user sees (the
closed-loop loadtest's TTFT is cohort queueing at saturation).
Prints one JSON line with p50/p95 over --requests solo requests."""
import json, os, statistics, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
import loadtest

def main():
    import argparse
    ap = argparse.ArgumentParser()
    ap.add_argument("--requests", type=int, default=24)
    ap.add_argument("--num-predict", type=int, default=32)
    ap.add_argument("--prompt-words", type=int, default=24)
    args = ap.parse_args()
    import torch
    on_gpu = torch.cuda.is_available()
    bus_port, http_port = loadtest._free_port(), loadtest._free_port()
    env = {"BUS_PORT": str(bus_port), "HTTP_PORT": str(http_port),
           "LT_CONC": "8",
           "LT_MODEL": "llama3.2:3b" if on_gpu else "tiny",
           "LT_DEVICE": "cuda:0" if on_gpu else "cpu",
           "LT_DTYPE": "bf16" if on_gpu else "fp32",
           "LT_MAXSEQ": "1024"}
    server = loadtest._spawn(loadtest.SERVER_SRC, env, "SERVER_READY")
    worker = loadtest._spawn(loadtest.WORKER_SRC, {**env, "LT_WID": "probe"},
                             "WORKER_READY")
    base = f"http://127.0.0.1:{http_port}"
    import httpx
    try:
        with httpx.Client(base_url=base, timeout=60) as c:
            for _ in range(200):
                r = c.get("/health/workers")
                if r.status_code == 200 and r.json()["counts"]["total"] == 1:
                    break
                time.sleep(0.1)
        # Raw socket for the timed requests: httpx's stream iterator was
        # measured to deliver the whole short body at once, inflating solo
        # TTFT ~8x; recv() timestamps are the ground truth of when the
        # first token chunk hit the wire.
        import socket
        ttfts, totals = [], []
        for i in range(args.requests + 4):
            body = json.dumps({
                "model": env["LT_MODEL"],
                "prompt": f"probe {i}: " + "lorem " * args.prompt_words,
                "stream": True,
                # temperature 0.8: greedy decode on a random-init model
                # fixates on one token, often an empty-decoding special ->
                # no text chunks -> no observable TTFT (r02aa logs).
                "options": {"temperature": 0.8, "seed": i,
                            "num_predict": args.num_predict}})
            req = ("POST /api/generate HTTP/1.1\r\nHost: x\r\n"
                   "Content-Type: application/json\r\n"
                   f"Content-Length: {len(body)}\r\n"
                   "Connection: close\r\n\r\n" + body)
            s = socket.create_connection(("127.0.0.1", http_port))
            t0 = time.monotonic()
            s.sendall(req.encode())
            first, buf = None, b""
            while True:
                d = s.recv(65536)
                if not d:
                    break
                buf += d
                if first is None and b'"response"' in buf:
                    first = time.monotonic() - t0
            s.close()
            if i >= 4 and first is not None:  # skip warmup/captures
                ttfts.append(first * 1000)
                totals.append((time.monotonic() - t0) * 1000)
        if not ttfts:
            # Every chunk had empty text (possible with the tiny synthetic
            # model: greedy decode can settle on an empty-decoding token,
            # and empty pieces are not streamed) -> no TTFT observable.
            print(json.dumps({"metric": "solo-request TTFT",
                              "error": "no text chunks observed"}))
            return
        print(json.dumps({
            "metric": "solo-request TTFT, /api/generate streaming",
            "requests": args.requests,
            "ttft_p50_ms": round(statistics.median(ttfts), 2),
            "ttft_p95_ms": round(sorted(ttfts)[int(0.95 * len(ttfts))], 2),
            "total_p50_ms": round(statistics.median(totals), 2),
            "num_predict": args.num_predict,
            "model": env["LT_MODEL"]}))
    finally:
        server.kill(); worker.kill()

if __name__ == "__main__":
    main()
