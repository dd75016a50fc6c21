#!/usr/bin/env python3
"""Wire-level stream trace: dump per-recv arrival times/sizes for a few
solo /api/generate streaming requests through the full serving path.
Diagnostic for TTFT==total (no chunks before the end) seen on GPU."""
import json, os, socket, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
import loadtest


def main():
    import argparse
    ap = argparse.ArgumentParser()
    ap.add_argument("--requests", type=int, default=4)
    ap.add_argument("--num-predict", type=int, default=64)
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
    worker = loadtest._spawn(loadtest.WORKER_SRC, {**env, "LT_WID": "tr"},
                             "WORKER_READY")
    import httpx
    try:
        with httpx.Client(base_url=f"http://127.0.0.1:{http_port}",
                          timeout=60) as c:
            for _ in range(200):
                r = c.get("/health/workers")
                if r.status_code == 200 and \
                        r.json()["counts"]["total"] == 1:
                    break
                time.sleep(0.1)
        for i in range(args.requests):
            body = json.dumps({
                "model": env["LT_MODEL"],
                "prompt": f"trace {i}: the quick brown fox",
                "stream": True,
                "options": {"temperature": 0.8, "seed": i,
                            "num_predict": args.num_predict}})
            req = ("POST /api/generate HTTP/1.1\r\nHost: x\r\n"
                   "Content-Type: application/json\r\n"
                   f"Content-Length: {len(body)}\r\n"
                   "Connection: close\r\n\r\n" + body)
            s = socket.create_connection(("127.0.0.1", http_port))
            t0 = time.monotonic()
            s.sendall(req.encode())
            arrivals, buf = [], b""
            while True:
                d = s.recv(65536)
                if not d:
                    break
                arrivals.append((round((time.monotonic() - t0) * 1000, 1),
                                 len(d)))
                buf += d
            s.close()
            nlines = buf.count(b'"response"')
            print(f"req {i}: {len(arrivals)} recvs, {nlines} ndjson lines, "
                  f"arrivals={arrivals[:20]}", flush=True)
            if i == 0:
                # first 400 bytes of body for eyeball check
                print("  head:", repr(buf[:400]), flush=True)
    finally:
        server.kill(); worker.kill()


if __name__ == "__main__":
    main()
