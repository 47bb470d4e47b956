#!/usr/bin/env python3
"""Measure end-to-end serving latency of examples/serve.py's /generate
endpoint (in-process TestClient -- same handler code path as uvicorn,
minus kernel TCP; measures request marshalling + graphed decode).

    python tools/serve_latency.py [--hidden 1024] [--vocab 50000]
                                  [--steps 64] [--reps 20]

Prints one JSON line: p50/p95 ms per request and generated tok/s.
"""
import argparse
import json
import os
import statistics
import sys
import time


sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from examples.serve import build_app, load_model  # noqa: E402


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--hidden", type=int, default=1024)
    ap.add_argument("--vocab", type=int, default=50000)
    ap.add_argument("--steps", type=int, default=64)
    ap.add_argument("--reps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=3)
    args = ap.parse_args(argv)
    args.demo, args.ckpt = True, ""
    model, device = load_model(args)
    app = build_app(model, device)
    from fastapi.testclient import TestClient
    client = TestClient(app)
    body = {"tokens": [1, 2, 3, 4], "steps": args.steps, "temperature": 0.0}
    for _ in range(args.warmup):
        r = client.post("/generate", json=body)
        assert r.status_code == 200, r.text
    lat = []
    for _ in range(args.reps):
        t0 = time.perf_counter()
        r = client.post("/generate", json=body)
        lat.append((time.perf_counter() - t0) * 1e3)
        assert r.status_code == 200 and len(r.json()["generated"]) == args.steps
    lat.sort()
    p50 = statistics.median(lat)
    result = {
        "metric": "serve_latency", "device": str(device),
        "hidden": args.hidden, "vocab": args.vocab, "steps": args.steps,
        "reps": args.reps, "p50_ms": round(p50, 3),
        "p95_ms": round(lat[int(0.95 * len(lat)) - 1], 3),
        "tok_per_s": round(args.steps / (p50 / 1e3), 1),
    }
    print(json.dumps(result))
    return result


if __name__ == "__main__":
    main()
