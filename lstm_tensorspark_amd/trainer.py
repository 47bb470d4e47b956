"""Trainer: the train()/fit() API of the engine.

Replaces the reference's driver epoch loop: the Spark
distribute->train->aggregate round becomes, per step and per rank,

    synth batch (rank-seeded)                      [data.py]
    forward (fused HIP kernels on GPU)             [ops/lstm_layer.py]
    backward -> bucketed async all-reduce over RCCL [parallel/ddp.py]
    grad clip (K6) -> fused optimizer (K7)         [ops/optim.py]
    rank0: log / checkpoint at interval

Truncated BPTT (SURVEY.md §5 long-context row): when train.bptt > 0 and
< seq_len, each batch is streamed in windows of bptt steps with the (h, c)
carry detached between windows -- activation memory is O(bptt*B*H), not
O(T*B*H).

Composable knobs on top (docs/TRAINING.md has the full contracts):
gradient accumulation (accum_steps), LR warmup + cosine/linear decay,
periodic held-out eval (eval_interval), auto-resume, fault injection,
opt-in whole-step hipGraph capture.
"""
from __future__ import annotations

import json
import os
import time
from typing import Optional

import torch
import torch.distributed as dist

from .checkpoint import load_checkpoint, save_checkpoint
from .config import Config
from .data import SyntheticSeqData
from .models.lstm import LSTMModel
from .ops.lstm_layer import enable_side_dw, wait_side_stream
from .ops.optim import build_optimizer
from .parallel.ddp import GradBucketer, broadcast_parameters
from .parallel.launch import DistContext
from .utils.logging import get_logger


def compute_dtype(cfg: Config, device: torch.device) -> torch.dtype:
    # "fp8" = bf16 weights/activations with the recurrent GEMM quantized to
    # OCP e4m3 inside the C++ driver (experimental; 2x MFMA rate)
    if cfg.train.dtype in ("bf16", "fp8") and device.type == "cuda":
        return torch.bfloat16
    return torch.float32


class Trainer:
    def __init__(self, cfg: Config, model: Optional[LSTMModel] = None,
                 dist_ctx: Optional[DistContext] = None, device=None):
        self.cfg = cfg
        self.ctx = dist_ctx or DistContext()
        self.device = torch.device(device) if device is not None else self.ctx.device
        self.dtype = compute_dtype(cfg, self.device)
        self.log = get_logger(rank=self.ctx.rank)

        torch.manual_seed(cfg.train.seed)  # same init on every rank
        self.model = model if model is not None else LSTMModel(cfg.model)
        self.model = self.model.to(self.device)
        if self.dtype != torch.float32:
            self.model = self.model.to(self.dtype)
        broadcast_parameters(self.model)  # belt-and-braces weight sync at init
        if self.ctx.world_size > 1 and self.ctx.rank > 0:
            # init above must be seed-identical on every rank, but the
            # ONGOING rng stream (dropout masks) should not be: identical
            # masks across ranks correlate the regularization noise.
            # Diverge rank>0 only -- rank 0 and world-1 trajectories stay
            # bitwise identical to the single-process run.
            from .data import mix64
            torch.manual_seed(mix64(cfg.train.seed, 3, self.ctx.rank))

        self.optimizer = build_optimizer(cfg.train, self.model.parameters())
        self.bucketer = None
        if self.ctx.world_size > 1:
            rd = {"fp32": torch.float32, "": None}.get(
                getattr(cfg.dist, "reduce_dtype", ""), None)
            self.bucketer = GradBucketer(
                self.model.parameters(), bucket_mb=cfg.dist.bucket_mb,
                reduce_dtype=rd,
                bucket_order=getattr(cfg.dist, "bucket_order", "reverse"))
        self.data = SyntheticSeqData(cfg, rank=self.ctx.rank,
                                     world_size=self.ctx.world_size,
                                     device=self.device, dtype=self.dtype)
        # Side-stream dW overlap (ops/lstm_layer.py) is measured NEUTRAL at
        # HBM-sized batches (the step is bound by the combined HBM+MFMA
        # envelope, not by dependencies): 234.2 vs 235.4 ms/step at B=16384.
        # Keep it off by default for a simpler stream graph; opt in via
        # dist.side_dw for small-model regimes.  Only safe when autograd
        # STORES (not accumulates) weight grads.
        if self.device.type == "cuda":
            from .ops.extension import extension_available, get_extension
            if extension_available():
                get_extension().set_fp8_fwd(cfg.train.dtype == "fp8")
                from .models import lstm as _lstm_mod
                _lstm_mod.enable_fp8_head(
                    getattr(cfg.train, "fp8_head", False))
                if getattr(cfg.train, "autotune", False):
                    from .ops.autotune import tune_fwd
                    r = tune_fwd(cfg.train.batch_size, cfg.model.hidden_size)
                    if self.ctx.is_main:
                        self.log.info(f"autotune fwd path: {r}")
        single_window = cfg.train.bptt <= 0 or cfg.train.bptt >= cfg.train.seq_len
        no_accum = getattr(cfg.train, "accum_steps", 1) <= 1
        # side_dw is only safe when autograd STORES weight grads; both BPTT
        # windows and micro-batch accumulation make it ACCUMULATE them
        enable_side_dw(getattr(cfg.dist, "side_dw", False)
                       and self.bucketer is None and single_window
                       and no_accum and self.device.type == "cuda")
        self.step = 0
        self.history: list[dict] = []
        # hipGraph capture state (train.cuda_graph)
        self._graph = None
        self._graph_io = None
        self._graph_eligible = (
            cfg.train.cuda_graph and self.device.type == "cuda"
            and self.ctx.world_size == 1 and self.bucketer is None
            and cfg.train.optimizer == "sgd" and single_window
            and cfg.model.dropout == 0.0
            # lr is baked into the captured kernels; accumulation needs
            # multiple backwards per step -- both fall back to eager
            and no_accum and not cfg.train.lr_schedule
            and cfg.train.warmup_steps <= 0)

    # ------------------------------------------------------------------
    def _bptt_windows(self, x, y):
        """Split (x, y) along time (dim 0 -- time-major) into truncated-BPTT
        windows."""
        T = x.shape[0]
        k = self.cfg.train.bptt
        if k <= 0 or k >= T:
            return [(x, y)]
        per_step_y = y.dim() >= 2 and y.shape[0] == T
        wins = []
        for s in range(0, T, k):
            e = min(s + k, T)
            wins.append((x[s:e], y[s:e] if per_step_y else y))
        return wins

    def _lr_at(self, step: int) -> float:
        """LR schedule as a pure function of the optimizer step (so resume
        lands on the same value): linear warmup 0 -> lr over warmup_steps,
        then constant / cosine / linear decay to min_lr at train.steps."""
        t = self.cfg.train
        if t.warmup_steps > 0 and step < t.warmup_steps:
            return t.lr * (step + 1) / t.warmup_steps
        if not t.lr_schedule:
            return t.lr
        total = max(t.steps, t.warmup_steps + 1)
        frac = min(1.0, (step - t.warmup_steps)
                   / max(1, total - t.warmup_steps))
        if t.lr_schedule == "cosine":
            import math
            c = 0.5 * (1.0 + math.cos(math.pi * frac))
        elif t.lr_schedule == "linear":
            c = 1.0 - frac
        else:
            raise ValueError(f"unknown lr_schedule {t.lr_schedule!r}")
        return t.min_lr + (t.lr - t.min_lr) * c

    def train_step(self, batch) -> dict:
        """One optimizer step.  `batch` is one (x, y) pair, or a list of
        them for gradient accumulation (grads are summed across
        micro-batches with the loss scaled 1/n, so the update equals one
        step on the concatenated batch)."""
        cfg = self.cfg
        if cfg.train.fault_step == self.step and cfg.train.fault_rank == self.ctx.rank:
            raise RuntimeError(
                f"injected fault on rank {self.ctx.rank} at step {self.step}")
        batches = batch if isinstance(batch, list) else [batch]
        if self._graph_eligible:
            if len(batches) != 1:
                raise ValueError(
                    "micro-batch lists need accum_steps > 1 in the config "
                    "(graph capture is single-backward; it would silently "
                    "drop the extra micro-batches)")
            return self._graphed_step(*batches[0])
        if self.bucketer is not None:
            self.bucketer.zero_grad()
        else:
            self.optimizer.zero_grad()

        total_loss = 0.0
        n_micro = len(batches)
        for bi, (x, y) in enumerate(batches):
            wins = self._bptt_windows(x, y)
            state = None
            for wi, (wx, wy) in enumerate(wins):
                out, state = self.model(wx, state)
                state = self.model.detach_state(state)
                loss = self.model.loss(out, wy)
                scale = len(wins) * n_micro
                if scale > 1:
                    loss = loss / scale
                if self.bucketer is not None:
                    # grads accumulate across windows/micro-batches into the
                    # bucket buffers; only the LAST backward launches the
                    # all-reduces.
                    self.bucketer.start_backward(
                        will_reduce=(bi == n_micro - 1
                                     and wi == len(wins) - 1))
                loss.backward()
                total_loss += float(loss.detach())
        if self.bucketer is not None:
            self.bucketer.finalize()
        # grads may still be in flight on the dW side stream
        wait_side_stream()

        norm = None
        if cfg.train.grad_clip > 0:
            norm = self.optimizer.clip_grad_norm(cfg.train.grad_clip)
        if cfg.train.lr_schedule or cfg.train.warmup_steps > 0:
            self.optimizer.lr = self._lr_at(self.step)
        self.optimizer.step()
        # invalidate serving-weight caches (model.generate's gemv layout)
        self.model._weights_version = getattr(
            self.model, "_weights_version", 0) + 1
        return {"loss": total_loss,
                "grad_norm": float(norm) if norm is not None else None}

    # -- hipGraph-captured step (train.cuda_graph) ----------------------
    def _graph_state_tensors(self):
        """Every tensor the capture warmup mutates: params, fp32 masters and
        momentum buffers (graph eligibility pins the optimizer to SGD)."""
        ts = [p for p in self.model.parameters()]
        ts += self.optimizer.masters
        ts += getattr(self.optimizer, "bufs", [])
        return ts

    def _graph_body(self):
        """One full training step on the static input tensors.  Everything
        here must be capture-safe: no host syncs, no new host-side state,
        stable tensor storage (grads zeroed in place, loss/norm read back
        from the tensors captured on the first call)."""
        gx, gy = self._graph_io
        self.optimizer.zero_grad(set_to_none=False)
        out, _ = self.model(gx)
        loss = self.model.loss(out, gy)
        loss.backward()
        norm = None
        if self.cfg.train.grad_clip > 0:
            norm = self.optimizer.clip_grad_norm(self.cfg.train.grad_clip)
        self.optimizer.step()
        self._g_loss = loss.detach()
        self._g_norm = norm

    def _graphed_step(self, x, y) -> dict:
        if self._graph is None:
            self._graph_io = (x.clone(), y.clone())
            # Snapshot every tensor the warmup/capture mutates (params,
            # fp32 masters, momentum) so the graphed trajectory is
            # STEP-EXACT vs eager: warmup twice eagerly (creates grads /
            # hipBLASLt workspaces / momentum buffers at stable addresses),
            # capture (records, does not execute), restore the snapshot
            # in place (the graph holds raw addresses -- copy_, never
            # rebind), then replay once = exactly one optimizer step on
            # this batch (tests/test_gpu_train.py asserts per-step loss
            # equality with the eager path).
            snap = [(t, t.detach().clone()) for t in self._graph_state_tensors()]
            sc0 = self.optimizer.step_count
            for _ in range(2):
                self._graph_body()
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self._graph_body()
            self._graph = g
            for t, s in snap:
                t.detach().copy_(s)
            self.optimizer.step_count = sc0 + 1  # the one replayed step
            self._graph.replay()
            self.model._weights_version = getattr(
                self.model, "_weights_version", 0) + 1
            self.log.info("train step captured as hipGraph (step-exact)")
        else:
            self._graph_io[0].copy_(x, non_blocking=True)
            self._graph_io[1].copy_(y, non_blocking=True)
            self._graph.replay()
            self.optimizer.step_count += 1
            self.model._weights_version = getattr(
                self.model, "_weights_version", 0) + 1
        return {"loss": float(self._g_loss),
                "grad_norm": float(self._g_norm)
                if self._g_norm is not None else None}

    def resume_latest(self) -> bool:
        """Auto-resume: load the newest checkpoint in ckpt_dir, if any.
        Part of the failure-recovery story: this engine is
        single-node/fixed-world -- a failed rank fail-fasts the whole job
        (ProcessGroup timeout is the watchdog) and the relaunched job calls
        resume_latest() to continue from the last saved step."""
        d = self.cfg.train.ckpt_dir
        if not d or not os.path.isdir(d):
            return False
        cands = [os.path.join(d, f) for f in os.listdir(d) if f.endswith(".pt")]
        if not cands:
            return False
        latest = max(cands, key=os.path.getmtime)
        self.load(latest)
        return True

    def fit(self, steps: Optional[int] = None) -> list:
        """Train for `steps` optimizer steps (default: config.train.steps).

        Fail-fast: any exception tears down the process group before
        re-raising so surviving ranks see a clean abort instead of a hang.
        """
        try:
            return self._fit(steps)
        except Exception:
            if dist.is_initialized():
                try:
                    dist.destroy_process_group()
                except Exception:
                    pass
            raise

    def _fit(self, steps: Optional[int] = None) -> list:
        cfg = self.cfg
        steps = steps if steps is not None else cfg.train.steps
        t_last = time.perf_counter()
        seqs_since = 0
        start_step = self.step
        accum = getattr(cfg.train, "accum_steps", 1)
        while self.step < start_step + steps:
            if accum > 1:
                # micro-batch data streams: step s consumes data steps
                # [s*accum, (s+1)*accum) so every micro-batch is distinct
                # and the mapping is resume-deterministic
                batch = [self.data.batch(self.step * accum + i)
                         for i in range(accum)]
            else:
                batch = self.data.batch(self.step)
            info = self.train_step(batch)
            self.step += 1
            seqs_since += cfg.train.batch_size * self.ctx.world_size * accum
            if cfg.train.log_interval and self.step % cfg.train.log_interval == 0:
                if self.device.type == "cuda":
                    torch.cuda.synchronize()
                now = time.perf_counter()
                rec = {
                    "step": self.step,
                    "loss": round(info["loss"], 6),
                    "grad_norm": info["grad_norm"],
                    "seq_per_sec": round(seqs_since / (now - t_last), 2),
                    "ms_per_step": round(1e3 * (now - t_last)
                                         / max(1, cfg.train.log_interval), 3),
                }
                self.history.append(rec)
                if self.ctx.is_main:
                    self.log.info(json.dumps(rec))
                t_last, seqs_since = now, 0
            ei = getattr(cfg.train, "eval_interval", 0)
            if ei and self.step % ei == 0:
                # all ranks reach this at the same step (same counters), so
                # evaluate()'s all-reduce stays collective-aligned
                r = self.evaluate(steps=cfg.train.eval_steps)
                rec = {"step": self.step,
                       "eval_loss": round(r["eval_loss"], 6)}
                self.history.append(rec)
                if self.ctx.is_main:
                    self.log.info(json.dumps(rec))
                t_last = time.perf_counter()  # don't bill eval to seq/s
            if (cfg.train.ckpt_interval and cfg.train.ckpt_dir
                    and self.step % cfg.train.ckpt_interval == 0):
                self.save(os.path.join(cfg.train.ckpt_dir,
                                       f"ckpt_step{self.step}.pt"))
        if cfg.train.ckpt_dir:
            self.save(os.path.join(cfg.train.ckpt_dir, "ckpt_final.pt"))
        return self.history

    def export_history(self, path: str):
        """Write the step/eval history as CSV (pandas/spreadsheet-ready;
        the TensorBoard package is not in this image, and the JSONL log +
        this CSV cover the same observability need -- SURVEY.md §5
        metrics row)."""
        import csv
        cols: list = []
        for r in self.history:
            for k in r:
                if k not in cols:
                    cols.append(k)
        with open(path, "w", newline="") as f:
            w = csv.DictWriter(f, fieldnames=cols)
            w.writeheader()
            w.writerows(self.history)

    # alias matching the reference's API surface
    train = fit

    @torch.no_grad()
    def evaluate(self, steps: int = 10, data=None) -> dict:
        """Average loss over `steps` held-out synthetic batches (the eval
        counterpart of the reference's Session.run(logits) path, SURVEY.md
        §3.3).  Uses the structurally disjoint eval RNG stream (data.mix64
        stream=1) so eval batches never coincide with training batches.
        All-reduces the mean loss (and accuracy, for logit heads)
        across ranks."""
        self.model.eval()
        src = data if data is not None else self.data
        logit_head = self.cfg.model.head in ("classifier", "seq_classifier",
                                             "lm")
        total, acc, n = 0.0, 0.0, 0
        with torch.no_grad():
            for i in range(steps):
                x, y = src.batch(i, stream=1)
                out, _ = self.model(x)
                total += float(self.model.loss(out, y))
                if logit_head:
                    acc += float((out.argmax(-1) == y).float().mean())
                n += 1
        self.model.train()
        mean = total / max(n, 1)
        mean_acc = acc / max(n, 1)
        if dist.is_initialized() and self.ctx.world_size > 1:
            t = torch.tensor([mean, mean_acc], dtype=torch.float64)
            if self.ctx.backend == "nccl":
                t = t.to(self.device)
            dist.all_reduce(t, op=dist.ReduceOp.AVG)
            mean, mean_acc = float(t[0].item()), float(t[1].item())
        r = {"eval_loss": mean, "steps": n}
        if logit_head:
            r["eval_acc"] = mean_acc
        return r

    # ------------------------------------------------------------------
    def save(self, path: str):
        if not self.ctx.is_main:
            if dist.is_initialized():
                dist.barrier()
            return
        save_checkpoint(path, self.model, self.optimizer, step=self.step,
                        config=self.cfg.to_dict())
        self.log.info(f"checkpoint saved: {path} (step {self.step})")
        if dist.is_initialized():
            dist.barrier()

    def load_weights(self, path: str) -> dict:
        """Weights-only init from a safetensors artifact (fine-tune /
        continued-pretraining entry: fresh optimizer and step counter,
        pretrained weights).  Returns the file's metadata."""
        from .checkpoint import load_weights_safetensors
        meta = load_weights_safetensors(path, self.model)
        broadcast_parameters(self.model)   # keep ranks bitwise identical
        self.optimizer.sync_masters()
        self.model._weights_version = getattr(
            self.model, "_weights_version", 0) + 1
        self.log.info(f"weights loaded: {path}")
        return meta

    def load(self, path: str):
        payload = load_checkpoint(path, self.model, self.optimizer)
        self.step = payload["step"]
        if self.ctx.world_size > 1 and self.ctx.rank > 0:
            # the checkpoint carries rank 0's rng state; re-diverge this
            # rank's mask stream (mixing the step keeps the post-resume
            # stream from replaying the pre-save one)
            from .data import mix64
            torch.manual_seed(mix64(self.cfg.train.seed, 3, self.ctx.rank,
                                    self.step))
        self.model._weights_version = getattr(
            self.model, "_weights_version", 0) + 1
        self.log.info(f"checkpoint loaded: {path} (step {self.step})")
        return payload
