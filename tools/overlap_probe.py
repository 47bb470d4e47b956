#!/usr/bin/env python3
"""Backward/all-reduce overlap measurement (SURVEY.md §7 hard part 3,
VERDICT r1 item 2c): a recurrent backward produces all layers' dW near the
END of the time loop, so transformer DDP bucket intuition may not hold --
this probe MEASURES when each bucket's all-reduce launches relative to
backward, and how much communication actually hides.

Per rank it wraps GradBucketer._launch_reduce / _on_grad_ready to stamp
CUDA events, then reports per bucket:
  t_ready_ms   when the bucket's last grad landed (vs backward start)
  t_bwd_ms     total backward span
  overlap_pct  (t_bwd - t_ready)/t_bwd -- how much backward remained to
               hide that bucket's all-reduce under
plus the step-time delta between overlapped (hook-launched) and deferred
(all reduces after backward) modes -- the actual value of overlap.

Runs at world_size=1 degenerate (events recorded, reduce skipped) so the
CPU tier can exercise it; real numbers need >= 2 GPUs under torchrun.
Writes JSON to profiles/overlap_w{N}.jsonl on rank 0.
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


class TimingBucketer:
    """Instrumentation wrapper: stamps an event when each bucket becomes
    ready (last grad accumulated) without changing reduce behavior."""

    def __init__(self, bucketer, device):
        self.b = bucketer
        self.on_gpu = device.type == "cuda"
        self.ready_events = {}
        self.ready_wall = {}
        orig = bucketer._on_grad_ready
        # own countdown: at world_size=1 the bucketer's hook early-returns
        # without touching .pending, so readiness is tracked here
        self._left = {}

        def hook(p):
            bucket = bucketer._param_bucket[p]
            i = bucketer.buckets.index(bucket)
            self._left[i] = self._left.get(i, len(bucket.params)) - 1
            orig(p)
            if self._left[i] == 0:  # this grad completed the bucket
                self._left[i] = len(bucket.params)
                if self.on_gpu:
                    ev = torch.cuda.Event(enable_timing=True)
                    ev.record()
                    self.ready_events[i] = ev
                self.ready_wall[i] = time.perf_counter()

        # rebind the registered hooks' target
        bucketer._on_grad_ready = hook
        for h in bucketer._hooks:
            h.remove()
        bucketer._hooks = [
            p.register_post_accumulate_grad_hook(hook)
            for p in bucketer.params
        ]


def run(ctx, batch, seq_len, hidden, layers, steps):
    cfg = Config(
        model=ModelConfig(input_size=hidden, hidden_size=hidden,
                          num_layers=layers),
        train=TrainConfig(batch_size=batch, seq_len=seq_len, lr=0.01,
                          steps=steps, log_interval=0,
                          dtype="bf16" if ctx.device.type == "cuda" else "fp32"),
        data=DataConfig(task="random"),
    )
    cfg.dist.bucket_mb = 16
    t = Trainer(cfg, dist_ctx=ctx)
    if t.bucketer is None:
        from lstm_tensorspark_amd.parallel.ddp import GradBucketer
        t.bucketer = GradBucketer(t.model.parameters(), bucket_mb=16)
    tim = TimingBucketer(t.bucketer, ctx.device)
    on_gpu = ctx.device.type == "cuda"

    rows = []
    for it in range(steps):
        x, y = t.data.batch(it)
        t.bucketer.zero_grad()
        out, _ = t.model(x)
        loss = t.model.loss(out, y)
        if on_gpu:
            torch.cuda.synchronize()
        ev0 = torch.cuda.Event(enable_timing=True) if on_gpu else None
        if on_gpu:
            ev0.record()
        w0 = time.perf_counter()
        t.bucketer.start_backward(will_reduce=True)
        loss.backward()
        evb = torch.cuda.Event(enable_timing=True) if on_gpu else None
        if on_gpu:
            evb.record()
        wb = time.perf_counter()
        t.bucketer.finalize()
        if on_gpu:
            torch.cuda.synchronize()
        wf = time.perf_counter()
        if it < 2:
            continue  # warmup
        bwd_ms = (ev0.elapsed_time(evb) if on_gpu else (wb - w0) * 1e3)
        per_bucket = []
        for i in range(len(t.bucketer.buckets)):
            if on_gpu and i in tim.ready_events:
                rdy = ev0.elapsed_time(tim.ready_events[i])
            elif i in tim.ready_wall:
                rdy = (tim.ready_wall[i] - w0) * 1e3
            else:
                continue
            per_bucket.append({
                "bucket": i,
                "mb": round(t.bucketer.buckets[i].numel()
                            * t.bucketer.buckets[i].flat.element_size()
                            / (1 << 20), 1),
                "t_ready_ms": round(rdy, 3),
                "overlap_window_pct": round(100 * max(0.0, bwd_ms - rdy)
                                            / max(bwd_ms, 1e-9), 1),
            })
        rows.append({"step": it, "bwd_ms": round(bwd_ms, 3),
                     "finalize_ms": round((wf - wb) * 1e3, 3),
                     "buckets": per_bucket})
    return t, rows


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=0)
    ap.add_argument("--seq-len", type=int, default=128)
    ap.add_argument("--hidden", type=int, default=1024)
    ap.add_argument("--layers", type=int, default=2)
    ap.add_argument("--steps", type=int, default=6)
    ap.add_argument("--out", default="")
    args = ap.parse_args(argv)
    ctx = init_from_env()
    on_gpu = ctx.device.type == "cuda"
    batch = args.batch or (16384 if on_gpu else 4)
    if not on_gpu and args.hidden > 128:
        args.hidden, args.seq_len = 32, 16  # CPU degenerate stays fast
    t, rows = run(ctx, batch, args.seq_len, args.hidden, args.layers,
                  args.steps)

    # A/B: hook-overlapped vs deferred-all-reduce (reduces after backward)
    def timed_steps(overlapped, n=4):
        for i in range(2):
            t.train_step(t.data.batch(100 + i))
        if on_gpu:
            torch.cuda.synchronize()
        if dist.is_initialized():
            dist.barrier()
        t0 = time.perf_counter()
        for i in range(n):
            x, y = t.data.batch(200 + i)
            t.bucketer.zero_grad()
            out, _ = t.model(x)
            loss = t.model.loss(out, y)
            t.bucketer.start_backward(will_reduce=overlapped)
            loss.backward()
            t.bucketer.finalize()  # deferred mode launches everything here
            t.optimizer.step()
        if on_gpu:
            torch.cuda.synchronize()
        return (time.perf_counter() - t0) / n * 1e3

    ms_overlap = timed_steps(True)
    ms_defer = timed_steps(False)
    summary = {"world": ctx.world_size,
               "ms_overlapped": round(ms_overlap, 2),
               "ms_deferred": round(ms_defer, 2),
               "overlap_saving_pct": round(100 * (ms_defer - ms_overlap)
                                           / max(ms_defer, 1e-9), 1),
               "per_step": rows[-1] if rows else None}
    if ctx.is_main:
        print(json.dumps(summary), flush=True)
        out = args.out or os.path.join(
            os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
            "profiles", f"overlap_w{ctx.world_size}.jsonl")
        with open(out, "a") as f:
            f.write(json.dumps(summary) + "\n")
    cleanup()
    return summary


if __name__ == "__main__":
    main()
