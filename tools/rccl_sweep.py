#!/usr/bin/env python3
"""Bucket-size x RCCL-algorithm sweep for the DP gradient all-reduce
(SURVEY.md §2.4 / VERDICT r1 item 2b).

Measures, at the current world size:
  1. raw all-reduce bandwidth for message sizes bracketing the headline
     gradient volume (67 MB across ~4-16 MB buckets), and
  2. end-to-end train-step time of the headline model for each
     (bucket_mb, reduce_dtype) combination,
writing one JSON line each to profiles/rccl_sweep_w{N}.jsonl (rank 0).

RCCL algorithm/protocol are fixed at PROCESS START by env (NCCL_ALGO /
NCCL_PROTO are read at communicator init), so sweep those by re-invoking:

    for algo in '' Ring Tree; do
      NCCL_ALGO=$algo python -m torch.distributed.run --nnodes=1 \
        --nproc-per-node 8 --master-addr 127.0.0.1 tools/rccl_sweep.py
    done

At world_size=1 (no torchrun) the tool runs degenerate (no process group,
comm rows skipped, step rows measured locally) so the CPU tier can execute
it (tests/test_trainer_cpu.py::test_rccl_sweep_tool_degenerate).
"""
import argparse
import json
import os
import sys
import time

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from lstm_tensorspark_amd.config import Config, ModelConfig, TrainConfig, DataConfig
from lstm_tensorspark_amd.parallel.launch import init_from_env, cleanup
from lstm_tensorspark_amd.trainer import Trainer


def bench_allreduce(ctx, sizes_mb, iters=20, dtype=torch.bfloat16):
    rows = []
    for mb in sizes_mb:
        n = mb * (1 << 20) // dtype.itemsize
        t = torch.ones(n, dtype=dtype, device=ctx.device)
        for _ in range(3):
            dist.all_reduce(t)
        if ctx.device.type == "cuda":
            torch.cuda.synchronize()
        dist.barrier()
        t0 = time.perf_counter()
        for _ in range(iters):
            dist.all_reduce(t)
        if ctx.device.type == "cuda":
            torch.cuda.synchronize()
        el = (time.perf_counter() - t0) / iters
        # ring all-reduce algorithmic bandwidth: 2*(n-1)/n * bytes / time
        bytes_ = n * dtype.itemsize
        busbw = 2 * (ctx.world_size - 1) / ctx.world_size * bytes_ / el / 1e9
        rows.append({"kind": "allreduce", "mb": mb, "ms": round(el * 1e3, 3),
                     "busbw_gbps": round(busbw, 1),
                     "algo": os.environ.get("NCCL_ALGO", ""),
                     "proto": os.environ.get("NCCL_PROTO", ""),
                     "world": ctx.world_size,
                     "dtype": str(dtype).split(".")[-1]})
    return rows


def bench_step(ctx, bucket_mb, reduce_dtype, batch, steps=8, warmup=3,
               hidden=1024, seq_len=128):
    cfg = Config(
        model=ModelConfig(input_size=hidden, hidden_size=hidden, num_layers=2),
        train=TrainConfig(batch_size=batch, seq_len=seq_len, lr=0.01,
                          steps=steps, log_interval=0,
                          dtype="bf16" if ctx.device.type == "cuda" else "fp32"),
        data=DataConfig(task="random"),
    )
    cfg.dist.bucket_mb = bucket_mb
    cfg.dist.reduce_dtype = reduce_dtype
    t = Trainer(cfg, dist_ctx=ctx)
    for i in range(warmup):
        t.train_step(t.data.batch(i))
    if ctx.device.type == "cuda":
        torch.cuda.synchronize()
    if dist.is_initialized():
        dist.barrier()
    t0 = time.perf_counter()
    for i in range(steps):
        t.train_step(t.data.batch(warmup + i))
    if ctx.device.type == "cuda":
        torch.cuda.synchronize()
    el = (time.perf_counter() - t0) / steps
    row = {"kind": "step", "bucket_mb": bucket_mb,
           "reduce_dtype": reduce_dtype or "bf16",
           "ms_per_step": round(el * 1e3, 2),
           "seq_per_sec": round(batch * ctx.world_size / el, 1),
           "algo": os.environ.get("NCCL_ALGO", ""),
           "proto": os.environ.get("NCCL_PROTO", ""),
           "world": ctx.world_size, "batch_per_gpu": batch}
    if t.bucketer is not None:
        row["num_buckets"] = len(t.bucketer.buckets)
    del t
    return row


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--buckets", default="4,8,16,32,64")
    ap.add_argument("--batch", type=int, default=0,
                    help="per-GPU batch (default: 16384 GPU / 8 CPU)")
    ap.add_argument("--sizes-mb", default="1,4,8,16,32,64")
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--hidden", type=int, default=1024)
    ap.add_argument("--seq-len", type=int, default=128)
    ap.add_argument("--out", default="")
    args = ap.parse_args(argv)

    ctx = init_from_env()
    on_gpu = ctx.device.type == "cuda"
    batch = args.batch or (16384 if on_gpu else 8)
    rows = []
    if ctx.world_size > 1:
        rows += bench_allreduce(ctx, [int(s) for s in args.sizes_mb.split(",")])
    for bmb in [int(b) for b in args.buckets.split(",")]:
        for rd in ("", "fp32"):
            rows.append(bench_step(ctx, bmb, rd, batch, steps=args.steps,
                                   hidden=args.hidden, seq_len=args.seq_len))
            if ctx.is_main:
                print(json.dumps(rows[-1]), flush=True)
    if ctx.is_main:
        out = args.out or os.path.join(
            os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
            "profiles", f"rccl_sweep_w{ctx.world_size}.jsonl")
        with open(out, "a") as f:
            for r in rows:
                f.write(json.dumps(r) + "\n")
        best = min((r for r in rows if r["kind"] == "step"),
                   key=lambda r: r["ms_per_step"])
        print(json.dumps({"best": best, "written": out}), flush=True)
    cleanup()


if __name__ == "__main__":
    main()
