"""Optimizers with fp32 master weights + fused multi-tensor GPU kernels.

Replaces the reference's per-executor optimizer apply (SURVEY.md §2.3 rows
B3/O1): K6 = fused multi-tensor global-norm clip, K7 = fused multi-tensor
SGD/Adam apply. On GPU the updates run as ONE kernel launch over all
parameters (grouped pointer lists); on CPU (and as the numerics oracle)
torch ops implement identical math.

Master-weight policy: parameters may be bf16 (compute copy); each has an
fp32 master. step() updates the master in fp32 from the (possibly bf16)
gradient, then writes the rounded bf16 copy back into the parameter.
"""
from __future__ import annotations

import math
from typing import Iterable, List

import torch

from . import extension


class MasterOptimizer:
    """Base: holds params + fp32 masters + state; fused path on GPU."""

    def __init__(self, params: Iterable[torch.nn.Parameter]):
        self.params: List[torch.nn.Parameter] = [p for p in params if p.requires_grad]
        if not self.params:
            raise ValueError("no parameters to optimize")
        self.masters = [p.detach().float().clone() for p in self.params]
        self.step_count = 0

    def sync_masters(self):
        """Re-derive the fp32 masters from the CURRENT parameter values
        (required after loading weights from outside the optimizer, e.g.
        a safetensors fine-tune init -- otherwise the first step snaps
        the params back to the stale masters)."""
        with torch.no_grad():
            for m, p in zip(self.masters, self.params):
                m.copy_(p.detach().float())

    # -- grad clip (K6) ----------------------------------------------------
    def global_grad_norm(self) -> torch.Tensor:
        grads = [p.grad for p in self.params if p.grad is not None]
        dev = self.params[0].device
        if not grads:
            return torch.zeros((), device=dev)
        if dev.type == "cuda" and not extension.force_ref():
            ext = extension.get_extension()
            return ext.multi_tensor_l2norm(grads)
        sq = torch.zeros((), dtype=torch.float32, device=dev)
        for g in grads:
            sq = sq + g.float().pow(2).sum()
        return sq.sqrt()

    def clip_grad_norm(self, max_norm: float) -> torch.Tensor:
        norm = self.global_grad_norm()
        grads = [p.grad for p in self.params if p.grad is not None]
        dev = self.params[0].device
        if dev.type == "cuda" and not extension.force_ref():
            ext = extension.get_extension()
            ext.multi_tensor_clip_scale(grads, norm, float(max_norm))
        else:
            coef = max_norm / (norm + 1e-6)
            if float(coef) < 1.0:
                for g in grads:
                    g.mul_(coef.to(g.dtype))
        return norm

    def zero_grad(self, set_to_none: bool = True):
        # set_to_none=False keeps grad storage stable (required inside a
        # captured hipGraph, where tensors must not be re-allocated)
        for p in self.params:
            if set_to_none:
                p.grad = None
            elif p.grad is not None:
                p.grad.zero_()

    # -- checkpoint --------------------------------------------------------
    def state_dict(self) -> dict:
        return {"masters": self.masters, "step": self.step_count,
                **self._extra_state()}

    def load_state_dict(self, sd: dict):
        assert len(sd["masters"]) == len(self.masters)
        for m, src in zip(self.masters, sd["masters"]):
            m.copy_(src)
        self.step_count = sd["step"]
        self._load_extra_state(sd)
        # re-sync compute copies from masters (bitwise resume)
        with torch.no_grad():
            for p, m in zip(self.params, self.masters):
                p.copy_(m.to(p.dtype))

    def _extra_state(self) -> dict:
        return {}

    def _load_extra_state(self, sd: dict):
        pass


class MasterSGD(MasterOptimizer):
    def __init__(self, params, lr: float, momentum: float = 0.0,
                 weight_decay: float = 0.0):
        super().__init__(params)
        self.lr, self.momentum, self.weight_decay = lr, momentum, weight_decay
        self.bufs = [torch.zeros_like(m) for m in self.masters] if momentum else []

    @torch.no_grad()
    def step(self):
        self.step_count += 1
        dev = self.params[0].device
        use_hip = dev.type == "cuda" and not extension.force_ref()
        if use_hip:
            ext = extension.get_extension()
            grads = [p.grad if p.grad is not None else torch.zeros_like(p)
                     for p in self.params]
            ext.fused_sgd(
                [p.data for p in self.params], self.masters, grads,
                self.bufs if self.momentum else [],
                self.lr, self.momentum, self.weight_decay,
            )
            return
        for idx, (p, m) in enumerate(zip(self.params, self.masters)):
            if p.grad is None:
                continue
            g = p.grad.float()
            if self.weight_decay:
                g = g.add(m, alpha=self.weight_decay)
            if self.momentum:
                buf = self.bufs[idx]
                buf.mul_(self.momentum).add_(g)
                g = buf
            m.add_(g, alpha=-self.lr)
            p.copy_(m.to(p.dtype))

    def _extra_state(self):
        return {"momentum_bufs": self.bufs}

    def _load_extra_state(self, sd):
        if self.momentum:
            for b, src in zip(self.bufs, sd["momentum_bufs"]):
                b.copy_(src)


class MasterAdam(MasterOptimizer):
    def __init__(self, params, lr: float, betas=(0.9, 0.999), eps: float = 1e-8,
                 weight_decay: float = 0.0):
        super().__init__(params)
        self.lr, self.betas, self.eps = lr, betas, eps
        self.weight_decay = weight_decay
        self.exp_avg = [torch.zeros_like(m) for m in self.masters]
        self.exp_avg_sq = [torch.zeros_like(m) for m in self.masters]

    @torch.no_grad()
    def step(self):
        self.step_count += 1
        b1, b2 = self.betas
        bc1 = 1.0 - b1 ** self.step_count
        bc2 = 1.0 - b2 ** self.step_count
        dev = self.params[0].device
        use_hip = dev.type == "cuda" and not extension.force_ref()
        if use_hip:
            ext = extension.get_extension()
            grads = [p.grad if p.grad is not None else torch.zeros_like(p)
                     for p in self.params]
            ext.fused_adam(
                [p.data for p in self.params], self.masters, grads,
                self.exp_avg, self.exp_avg_sq,
                self.lr, b1, b2, self.eps, self.weight_decay, bc1, bc2,
            )
            return
        for idx, (p, m) in enumerate(zip(self.params, self.masters)):
            if p.grad is None:
                continue
            g = p.grad.float()
            if self.weight_decay:
                g = g.add(m, alpha=self.weight_decay)
            ea, eas = self.exp_avg[idx], self.exp_avg_sq[idx]
            ea.mul_(b1).add_(g, alpha=1 - b1)
            eas.mul_(b2).addcmul_(g, g, value=1 - b2)
            denom = (eas / bc2).sqrt_().add_(self.eps)
            m.addcdiv_(ea / bc1, denom, value=-self.lr)
            p.copy_(m.to(p.dtype))

    def _extra_state(self):
        return {"exp_avg": self.exp_avg, "exp_avg_sq": self.exp_avg_sq}

    def _load_extra_state(self, sd):
        for b, src in zip(self.exp_avg, sd["exp_avg"]):
            b.copy_(src)
        for b, src in zip(self.exp_avg_sq, sd["exp_avg_sq"]):
            b.copy_(src)


def build_optimizer(train_cfg, params) -> MasterOptimizer:
    if train_cfg.optimizer == "sgd":
        return MasterSGD(params, lr=train_cfg.lr, momentum=train_cfg.momentum,
                         weight_decay=train_cfg.weight_decay)
    if train_cfg.optimizer == "adam":
        return MasterAdam(params, lr=train_cfg.lr, betas=train_cfg.adam_betas,
                          eps=train_cfg.adam_eps,
                          weight_decay=train_cfg.weight_decay)
    raise ValueError(f"unknown optimizer {train_cfg.optimizer!r}")
