#!/usr/bin/env python3
"""Full-stack serving benchmark: concurrent OpenAI-style requests
through the real HTTP server (uvicorn) + EngineLoop + continuous
batching, with live prefills interleaving decodes — the workload the
batch-held-constant bench.py cannot show (reference parity: the system
test POSTs /v1/completions, reference test/system.sh:70-77).

GPU box: python scripts/bench_http.py [--model llama2-7b] [--clients 32]
Prints one JSON line: completed requests, wall, generated tok/s, p50/p95
request latency.
"""
import argparse
import json
import os
import statistics
import sys
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default=None)
    p.add_argument("--clients", type=int, default=32)
    p.add_argument("--requests", type=int, default=128)
    p.add_argument("--max-tokens", type=int, default=48)
    p.add_argument("--port", type=int, default=18080)
    args = p.parse_args()

    import torch
    if args.model is None:
        args.model = "llama2-7b" if torch.cuda.is_available() else "tiny-llama"

    from runbooks_amd.serve import Engine
    from runbooks_amd.serve.http import build_app

    eng = Engine(args.model,
                 dtype=torch.bfloat16 if torch.cuda.is_available()
                 else torch.float32,
                 kv_blocks=None if torch.cuda.is_available() else 4096,
                 seed=7)
    app = build_app(eng, model_name=args.model)

    import uvicorn
    cfg = uvicorn.Config(app, host="127.0.0.1", port=args.port,
                         log_level="error")
    server = uvicorn.Server(cfg)
    t = threading.Thread(target=server.run, daemon=True)
    t.start()

    import httpx
    base = f"http://127.0.0.1:{args.port}"
    for _ in range(300):
        try:
            if httpx.get(base + "/", timeout=1).status_code == 200:
                break
        except Exception:
            time.sleep(0.2)

    lat = []
    done = [0]
    gen_tokens = [0]
    lock = threading.Lock()
    work = list(range(args.requests))

    errors = []

    def client(cid: int):
        with httpx.Client(timeout=300) as c:
            while True:
                with lock:
                    if not work:
                        return
                    i = work.pop()
                prompt = ("the quick brown fox %d jumps over the lazy dog "
                          % i) * (2 + i % 6)
                t0 = time.perf_counter()
                try:
                    r = c.post(base + "/v1/completions", json={
                        "model": args.model, "prompt": prompt,
                        "max_tokens": args.max_tokens, "temperature": 0.0})
                    dt = time.perf_counter() - t0
                    assert r.status_code == 200, r.text[:300]
                    usage = r.json()["usage"]
                except Exception as e:
                    with lock:
                        errors.append(repr(e)[:200])
                    return
                with lock:
                    lat.append(dt)
                    done[0] += 1
                    gen_tokens[0] += usage["completion_tokens"]

    t0 = time.perf_counter()
    threads = [threading.Thread(target=client, args=(i,))
               for i in range(args.clients)]
    for th in threads:
        th.start()
    for th in threads:
        th.join()
    wall = time.perf_counter() - t0
    if errors:
        print(json.dumps({"errors": errors[:5], "n_errors": len(errors)}),
              flush=True)
        sys.exit(1)
    lat.sort()
    print(json.dumps({
        "metric": "http_serve_generated_tokens_per_sec",
        "value": round(gen_tokens[0] / wall, 1),
        "requests": done[0], "clients": args.clients,
        "wall_s": round(wall, 2),
        "requests_per_sec": round(done[0] / wall, 2),
        "p50_latency_s": round(statistics.median(lat), 3),
        "p95_latency_s": round(lat[int(len(lat) * 0.95) - 1], 3),
        "model": args.model,
        "max_tokens": args.max_tokens,
    }), flush=True)
    server.should_exit = True


if __name__ == "__main__":
    main()
