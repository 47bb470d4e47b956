#!/usr/bin/env python3
"""Multi-tenant serving benchmark: drive the continuous batcher with a
closed-loop synthetic workload and report request throughput + latency
percentiles, vs the solo (one-at-a-time) baseline.

    python tools/batcher_bench.py --hidden 1024 --vocab 50000 \
        --clients 32 --steps 64 --requests 128

Closed loop: `--clients` concurrent clients each submit, wait, and
immediately submit again until `--requests` total requests finish.
"""
import argparse
import json
import os
import statistics
import sys
import threading
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from lstm_tensorspark_amd.config import ModelConfig
from lstm_tensorspark_amd.models.lstm import LSTMModel
from lstm_tensorspark_amd.serving import ContinuousBatcher


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--hidden", type=int, default=1024)
    ap.add_argument("--vocab", type=int, default=50000)
    ap.add_argument("--layers", type=int, default=2)
    ap.add_argument("--clients", type=int, default=32)
    ap.add_argument("--steps", type=int, default=64)
    ap.add_argument("--prompt", type=int, default=8)
    ap.add_argument("--requests", type=int, default=128)
    ap.add_argument("--max-batch", type=int, default=64)
    ap.add_argument("--solo-baseline", type=int, default=8,
                    help="also time N one-at-a-time requests (0 = skip)")
    args = ap.parse_args(argv)
    torch.manual_seed(0)
    dev = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    cfg = ModelConfig(input_size=args.hidden, hidden_size=args.hidden,
                      num_layers=args.layers, head="lm",
                      vocab_size=args.vocab)
    m = LSTMModel(cfg).to(dev)
    if dev.type == "cuda":
        m = m.bfloat16()
    m.eval()

    solo_ms = None
    if args.solo_baseline:
        prompt = torch.randint(0, args.vocab, (args.prompt, 1), device=dev)
        m.generate(prompt, steps=args.steps, use_graph=False)  # warm
        t0 = time.perf_counter()
        for _ in range(args.solo_baseline):
            m.generate(prompt, steps=args.steps, use_graph=False)
        solo_ms = (time.perf_counter() - t0) / args.solo_baseline * 1e3

    eng = ContinuousBatcher(m, max_batch=args.max_batch)
    eng.start()
    lat: list = []
    done = threading.Semaphore(0)
    remaining = [args.requests]
    lock = threading.Lock()

    def client(ci):
        g = torch.Generator().manual_seed(ci)
        while True:
            with lock:
                if remaining[0] <= 0:
                    break
                remaining[0] -= 1
            toks = torch.randint(0, args.vocab, (args.prompt,),
                                 generator=g).tolist()
            t0 = time.perf_counter()
            rid = eng.submit(toks, args.steps)
            out = eng.result(rid, timeout=600)
            dt = (time.perf_counter() - t0) * 1e3
            assert len(out) == args.prompt + args.steps
            with lock:
                lat.append(dt)
        done.release()

    t0 = time.perf_counter()
    threads = [threading.Thread(target=client, args=(i,))
               for i in range(args.clients)]
    for t in threads:
        t.start()
    for _ in threads:
        done.acquire()
    wall = time.perf_counter() - t0
    occupancy = eng.stats()["mean_occupancy"]
    eng.stop()
    lat.sort()
    n = len(lat)
    result = {
        "metric": "batcher_requests_per_s", "device": str(dev),
        "clients": args.clients, "steps": args.steps,
        "requests": n, "max_batch": args.max_batch,
        "req_per_s": round(n / wall, 2),
        "tok_per_s": round(n * args.steps / wall, 1),
        "p50_ms": round(statistics.median(lat), 2),
        "p95_ms": round(lat[max(0, int(0.95 * n) - 1)], 2),
        "mean_occupancy": occupancy,
        "solo_ms": round(solo_ms, 2) if solo_ms is not None else None,
        "speedup_vs_solo": (round(solo_ms * n / (wall * 1e3), 2)
                            if solo_ms is not None else None),
    }
    print(json.dumps(result))
    return result


if __name__ == "__main__":
    main()
