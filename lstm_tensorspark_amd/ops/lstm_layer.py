"""One LSTM layer (one direction) over a full sequence as an autograd op.

This is the MI355X replacement for the reference's per-timestep TF graph ops
(SURVEY.md §2.3, rows F1-F5 / B1-B2): instead of ~10 graph nodes per
timestep replayed by Session.run, the whole-sequence forward is

  1. ONE library GEMM (hipBLASLt via torch.matmul) for the input projection
     of ALL timesteps:  pre = x.view(T*B, X) @ W_x + b          [K1a]
  2. a C++-driven time loop of ONE fused hand-written CDNA4 kernel per step:
     MFMA recurrent GEMM h_{t-1} @ W_h + gate activations + cell-state
     update + h/c/act stores, all fused                          [K1b+K2]

and the backward is
  3. a C++ time loop of a fused pointwise-backward kernel + recurrent
     dgates @ W_h^T MFMA kernel                                  [K4+K5a]
  4. THREE library GEMMs for the weight/input gradients:
     dW_h = hprev^T @ dgates, dW_x = x^T @ dgates, dx = dgates @ W_x^T
     (batched over all timesteps -- no per-step dW round trips)  [K5b]

Everything is TIME-MAJOR (T, B, ·): the whole-sequence hidden state lives
in ONE (T+1, B, H) buffer holding h0 at slot 0 (forward) or slot T
(reverse), so h_{t-1} for the dW_h GEMM is a contiguous leading-dim slice
-- no torch.cat, no padding, zero copies (functional.py documents the slot
convention).  The layer's (T,B,H) output is itself a contiguous view of
that buffer, so the next layer's input GEMM reshapes it for free.

On CPU the reference implementations in functional.py run instead (same
signatures, same precision policy), so the identical Trainer code path is
exercised by the CPU test tiers.
"""
from __future__ import annotations

from typing import Optional

import torch

from . import extension, functional
from .functional import h_prev_view, h_seq_view


def _seq_fwd(pre, w_h, h0, c0, reverse, use_hip):
    if use_hip:
        ext = extension.get_extension()
        return ext.lstm_seq_fwd(pre, w_h, h0, c0, reverse)
    return functional.ref_seq_fwd(pre, w_h, h0, c0, reverse)


def _seq_bwd(d_hseq, d_hT, d_cT, w_h, acts, c_seq, c0, reverse, use_hip):
    if use_hip:
        ext = extension.get_extension()
        return ext.lstm_seq_bwd(d_hseq, d_hT, d_cT, w_h, acts, c_seq, c0, reverse)
    return functional.ref_seq_bwd(d_hseq, d_hT, d_cT, w_h, acts, c_seq, c0, reverse)


class LSTMLayerFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, h0, c0, w_x, w_h, b, reverse: bool):
        # x: (T,B,X) compute dtype; h0: (B,H); c0: (B,H) fp32
        T, B, X = x.shape
        H = w_h.shape[0]
        ctx.set_materialize_grads(False)
        use_hip = extension.use_hip(x.device, x.dtype)
        # [K1a] input projection for all timesteps in one library GEMM
        pre = torch.addmm(b.to(x.dtype), x.reshape(T * B, X), w_x).view(T, B, 4 * H)
        h_all, c_seq, acts, hT, cT = _seq_fwd(pre, w_h, h0, c0, reverse, use_hip)
        ctx.save_for_backward(x, c0, w_x, w_h, h_all, c_seq, acts)
        ctx.reverse = reverse
        ctx.use_hip = use_hip
        return h_seq_view(h_all, reverse), hT, cT

    @staticmethod
    def backward(ctx, d_hseq, d_hT, d_cT):
        x, c0, w_x, w_h, h_all, c_seq, acts = ctx.saved_tensors
        T, B, X = x.shape
        H = w_h.shape[0]
        # set_materialize_grads(False): unused outputs arrive as None and the
        # backward kernels treat null pointers as zero.
        dgates, dh0, dc0 = _seq_bwd(
            d_hseq.contiguous() if d_hseq is not None else None,
            d_hT.contiguous() if d_hT is not None else None,
            d_cT.contiguous() if d_cT is not None else None,
            w_h, acts, c_seq, c0, ctx.reverse, ctx.use_hip,
        )
        dg_flat = dgates.reshape(T * B, 4 * H)
        # [K5b] weight/input gradients: flat contiguous library GEMMs, each
        # gated on needs_input_grad (dx for the FIRST layer is a whole
        # (T*B,4H)x(4H,X) GEMM that nothing consumes -- skip it).
        ng = ctx.needs_input_grad
        dw_h = dw_x = db = dx = None
        if ng[4]:
            hprev = h_prev_view(h_all, ctx.reverse).reshape(T * B, H)
            dw_h = (hprev.t() @ dg_flat).to(w_h.dtype)
        if ng[3]:
            dw_x = (x.reshape(T * B, X).t() @ dg_flat).to(w_x.dtype)
        if ng[5]:
            # sum with fp32 accumulation WITHOUT materializing an fp32 copy
            db = dg_flat.sum(dim=0, dtype=functional._acc_dtype(dg_flat.dtype)).to(w_x.dtype)
        if ng[0]:
            dx = (dg_flat @ w_x.t()).view(T, B, X)
        return dx, dh0, dc0, dw_x, dw_h, db, None


def lstm_layer(
    x: torch.Tensor,
    h0: torch.Tensor,
    c0: torch.Tensor,
    w_x: torch.Tensor,
    w_h: torch.Tensor,
    b: torch.Tensor,
    reverse: bool = False,
):
    """Run one LSTM layer (one direction) over (T,B,X) -> (T,B,H).

    Returns (h_seq, hT, cT); cT is fp32.
    """
    return LSTMLayerFn.apply(x, h0, c0, w_x, w_h, b, reverse)
