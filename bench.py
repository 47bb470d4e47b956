#!/usr/bin/env python3
"""Flagship training-step benchmark (driver contract).

Measures BASELINE.json's metric: seq/sec (whole node) training a 2-layer
LSTM h=1024 seq=128, bf16 compute + fp32 master, synthetic data,
random-init weights, data-parallel over RCCL/xGMI for --gpus N.

Launched either directly (N=1) or via torch.distributed.run with one rank
per GPU (reads RANK/LOCAL_RANK/WORLD_SIZE/MASTER_* from the env).

Rank 0 prints ONE JSON line with the whole-job aggregate metric.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from lstm_tensorspark_amd.config import PRESETS, get_preset
from lstm_tensorspark_amd.parallel.launch import cleanup, init_from_env
from lstm_tensorspark_amd.trainer import Trainer


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1,
                   help="world size (informational; actual world from env)")
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--preset", default="mi355x-1gpu", choices=sorted(PRESETS))
    p.add_argument("--batch", type=int, default=0,
                   help="override per-rank batch size")
    p.add_argument("--dtype", default="",
                   help="override compute dtype (bf16 | fp8)")
    p.add_argument("--cpu", action="store_true",
                   help="allow CPU run (small shapes; for plumbing checks)")
    return p.parse_args()


def main():
    args = parse_args()
    ctx = init_from_env(backend="auto")
    n_gpus = max(ctx.world_size, 1)

    cfg = get_preset(args.preset)
    cfg.train.log_interval = 0
    cfg.train.ckpt_dir = ""
    if args.batch:
        cfg.train.batch_size = args.batch
    if args.dtype:
        cfg.train.dtype = args.dtype
    on_gpu = ctx.device.type == "cuda"
    if not on_gpu:
        if not args.cpu:
            print("ERROR: no GPU visible; pass --cpu for a plumbing run",
                  file=sys.stderr)
            sys.exit(2)
        # plumbing-scale shapes so a CPU check finishes quickly
        cfg.model.input_size = min(cfg.model.input_size, 128)
        cfg.model.hidden_size = min(cfg.model.hidden_size, 128)
        cfg.train.batch_size = min(cfg.train.batch_size, 8)
        cfg.train.seq_len = min(cfg.train.seq_len, 32)
        cfg.train.dtype = "fp32"

    trainer = Trainer(cfg, dist_ctx=ctx)
    B, T = cfg.train.batch_size, cfg.train.seq_len
    dev, dt = trainer.device, trainer.dtype

    # Device-resident synthetic batch pool (data generation is not the
    # metric; every timed step still does full fwd+bwd+clip+optimizer).
    gen = torch.Generator(device=dev)
    gen.manual_seed(1234 + ctx.rank)
    pool = []
    for _ in range(4):
        # TIME-MAJOR (T, B, ·): the engine's native layout
        if cfg.model.head == "lm":
            stream = torch.randint(0, cfg.model.vocab_size, (T + 1, B),
                                   generator=gen, device=dev)
            x, y = stream[:-1], stream[1:]
        else:
            x = torch.randn(T, B, cfg.model.input_size, generator=gen,
                            device=dev, dtype=torch.float32).to(dt)
            y = torch.randint(0, cfg.model.num_classes, (T, B), generator=gen,
                              device=dev)
        pool.append((x, y))

    def sync():
        if on_gpu:
            torch.cuda.synchronize()
        if ctx.world_size > 1:
            torch.distributed.barrier()
        if on_gpu:
            torch.cuda.synchronize()

    for i in range(args.warmup):
        trainer.train_step(pool[i % len(pool)])

    sync()
    t0 = time.perf_counter()
    for i in range(args.steps):
        trainer.train_step(pool[i % len(pool)])
    if on_gpu:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    sync()

    # MAX elapsed over ranks = whole-job wall time
    if ctx.world_size > 1:
        et = torch.tensor([elapsed], dtype=torch.float64)
        if ctx.backend == "nccl":
            et = et.to(dev)
        torch.distributed.all_reduce(et, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(et.item())

    seqs_total = B * n_gpus * args.steps
    value = seqs_total / elapsed
    if ctx.rank == 0:
        result = {
            "metric": "seq/sec",
            "value": round(value, 2),
            "unit": "seq/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(1e3 * elapsed / args.steps, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # reference has no published numbers
            "dtype": cfg.train.dtype if on_gpu else "fp32",
            "data": "synthetic",
            "config": {
                "model": f"lstm-{cfg.model.num_layers}l-h{cfg.model.hidden_size}",
                "global_batch": B * n_gpus,
                "seq_len": T,
                "parallelism": f"dp{n_gpus}",
                "preset": args.preset,
            },
        }
        print(json.dumps(result))
    cleanup()


if __name__ == "__main__":
    main()
