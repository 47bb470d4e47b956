"""Fused log-softmax cross-entropy [K3] -- hand-written for the V=50k LM
head: one online-softmax pass per row in the
forward, fused (softmax - onehot) in the backward, no (N,V) fp32
intermediate.  CPU / non-bf16 paths use torch.nn.functional.cross_entropy.
"""
from __future__ import annotations

import torch
import torch.nn.functional as F

from . import extension


class FusedCrossEntropyFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, targets):
        ext = extension.get_extension()
        loss, lse = ext.xent_fwd(logits, targets)
        ctx.save_for_backward(logits, targets, lse)
        return loss.mean()

    @staticmethod
    def backward(ctx, grad_out):
        logits, targets, lse = ctx.saved_tensors
        ext = extension.get_extension()
        # grad_out stays on device (no host sync): the kernel reads it
        dlogits = ext.xent_bwd(logits, targets, lse, grad_out)
        return dlogits, None


def cross_entropy(logits: torch.Tensor, targets: torch.Tensor) -> torch.Tensor:
    """Mean cross-entropy over flattened (N, V) logits."""
    if (extension.use_hip(logits.device, logits.dtype)
            and logits.dim() == 2 and targets.dtype == torch.long):
        return FusedCrossEntropyFn.apply(logits, targets)
    return F.cross_entropy(logits.float(), targets)
