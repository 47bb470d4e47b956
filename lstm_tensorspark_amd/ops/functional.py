"""Reference (pure-PyTorch) implementations of the LSTM sequence kernels.

These mirror the HIP extension's signatures exactly (csrc/): every HIP
kernel's numerics test compares against the function here in fp32
(SURVEY.md §4.2 "Unit: kernel math" tier). They are also the compute path
on CPU, where the whole stack must run at world_size=1 (BASELINE configs[0]).

LAYOUT: the engine is TIME-MAJOR — sequences are (T, B, ·), the layout the
recurrent hot path wants on MI355X: every per-timestep tile is one dense
contiguous block, the (T+1, B, H) hidden buffer makes h_{t-1} for BOTH
directions a contiguous leading-dim slice (no torch.cat, no padding), and
all four weight/input-grad GEMMs in ops/lstm_layer.py run on flat
contiguous views.

Hidden-state buffer convention (shared with csrc/bindings.cpp):
    forward:  h_all[0] = h0,  h_all[t+1] = h_t   -> h_prev(t) = h_all[t]
    reverse:  h_all[T] = h0,  h_all[t]   = h_t   -> h_prev(t) = h_all[t+1]
so  h_seq = h_all[1:] (fwd) / h_all[:T] (rev)   [contiguous view]
and hprev_seq = h_all[:T] (fwd) / h_all[1:] (rev) [contiguous view].

Math (standard LSTM, gate order i, f, g, o along the 4H axis, gate-BLOCKED
layout [i|f|g|o] -- the HIP kernel assigns one output-column range per gate
to the same wavefront so the cell update is lane-local):

    raw_t   = pre_t + h_{t-1} @ W_h            # pre_t = x_t @ W_x + b
    i,f,o   = sigmoid(raw[..., :H]), ...       # computed in fp32
    g       = tanh(raw[..., 2H:3H])
    c_t     = f * c_{t-1} + i * g              # cell state kept in fp32
    h_t     = o * tanh(c_t)                    # stored in compute dtype

Precision policy (matches the HIP kernels): MFMA accumulates in fp32,
pointwise math in fp32, h/activations stored in the compute dtype (bf16 on
GPU), cell state always fp32, master weights fp32 (ops/optim.py).
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch


def _acc_dtype(dt: torch.dtype) -> torch.dtype:
    """Accumulation dtype: fp32 for bf16/fp16/fp32, fp64 stays fp64 (so the
    fp64 gradcheck/numerics tests are exact)."""
    return torch.float64 if dt == torch.float64 else torch.float32


def time_order(T: int, reverse: bool):
    return range(T - 1, -1, -1) if reverse else range(T)


def h_slot(t: int, reverse: bool) -> int:
    """Slot of h_t's output in the (T+1, B, H) hidden buffer."""
    return t if reverse else t + 1


def h_seq_view(h_all: torch.Tensor, reverse: bool) -> torch.Tensor:
    """(T, B, H) contiguous view of the per-step outputs."""
    T = h_all.shape[0] - 1
    return h_all[:T] if reverse else h_all[1:]


def h_prev_view(h_all: torch.Tensor, reverse: bool) -> torch.Tensor:
    """(T, B, H) contiguous view of h_{t-1} per step (h0 at its slot)."""
    T = h_all.shape[0] - 1
    return h_all[1:] if reverse else h_all[:T]


def ref_seq_fwd(
    pre: torch.Tensor,      # (T, B, 4H) compute dtype: x @ W_x + b
    w_h: torch.Tensor,      # (H, 4H)
    h0: torch.Tensor,       # (B, H) compute dtype
    c0: torch.Tensor,       # (B, H) fp32
    reverse: bool = False,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, torch.Tensor, torch.Tensor]:
    """Returns (h_all, c_seq, acts, hT, cT).

    h_all: (T+1,B,H) compute dtype, slot convention above.
    c_seq: (T,B,H) fp32 post-update cell states (natural time order).
    acts:  (T,B,4H) compute dtype, post-activation gates [i|f|g|o].
    """
    T, B, H4 = pre.shape
    H = H4 // 4
    dt = pre.dtype
    dev = pre.device
    at = _acc_dtype(dt)
    h_all = torch.empty(T + 1, B, H, dtype=dt, device=dev)
    c_seq = torch.empty(T, B, H, dtype=at, device=dev)
    acts = torch.empty(T, B, H4, dtype=dt, device=dev)
    h_all[T if reverse else 0] = h0
    h, c = h0, c0.to(at)
    for t in time_order(T, reverse):
        raw = (pre[t] + h @ w_h).to(at)
        i = torch.sigmoid(raw[:, :H])
        f = torch.sigmoid(raw[:, H : 2 * H])
        g = torch.tanh(raw[:, 2 * H : 3 * H])
        o = torch.sigmoid(raw[:, 3 * H :])
        c = f * c + i * g
        h32 = o * torch.tanh(c)
        h = h32.to(dt)
        acts[t, :, :H] = i.to(dt)
        acts[t, :, H : 2 * H] = f.to(dt)
        acts[t, :, 2 * H : 3 * H] = g.to(dt)
        acts[t, :, 3 * H :] = o.to(dt)
        c_seq[t] = c
        h_all[h_slot(t, reverse)] = h
    return h_all, c_seq, acts, h, c


def ref_seq_bwd(
    d_hseq: Optional[torch.Tensor],  # (T,B,H) upstream grad per step, or None
    d_hT: Optional[torch.Tensor],    # (B,H) grad of final hidden, or None
    d_cT: Optional[torch.Tensor],    # (B,H) fp32 grad of final cell, or None
    w_h: torch.Tensor,               # (H,4H)
    acts: torch.Tensor,              # (T,B,4H) saved post-activation gates
    c_seq: torch.Tensor,             # (T,B,H) fp32 saved cell states
    c0: torch.Tensor,                # (B,H) fp32 initial cell state
    reverse: bool = False,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Returns (dgates, dh0, dc0).

    dgates: (T,B,4H) compute dtype -- gradients w.r.t. RAW gates (pre-
            activation). The caller turns these into dW_x/dW_h/db/dx with
            flat contiguous GEMMs (ops/lstm_layer.py).
    dh0:    (B,H) compute dtype.  dc0: (B,H) fp32.
    """
    T, B, H4 = acts.shape
    H = H4 // 4
    dt = acts.dtype
    dev = acts.device
    at = _acc_dtype(dt)
    dgates = torch.empty(T, B, H4, dtype=dt, device=dev)
    w_h_t = w_h.t().contiguous()
    dh = torch.zeros(B, H, dtype=at, device=dev) if d_hT is None else d_hT.to(at)
    dc = torch.zeros(B, H, dtype=at, device=dev) if d_cT is None else d_cT.to(at)
    order = list(time_order(T, reverse))
    for k in range(T - 1, -1, -1):
        t = order[k]
        if d_hseq is not None:
            dh = dh + d_hseq[t].to(at)
        i = acts[t, :, :H].to(at)
        f = acts[t, :, H : 2 * H].to(at)
        g = acts[t, :, 2 * H : 3 * H].to(at)
        o = acts[t, :, 3 * H :].to(at)
        c_t = c_seq[t]
        c_prev = c_seq[order[k - 1]] if k > 0 else c0.to(at)
        tc = torch.tanh(c_t)
        do = dh * tc
        dc = dc + dh * o * (1.0 - tc * tc)
        di = dc * g
        dg = dc * i
        df = dc * c_prev
        dc = dc * f  # carry to previous step
        draw = torch.cat(
            [
                di * i * (1.0 - i),
                df * f * (1.0 - f),
                dg * (1.0 - g * g),
                do * o * (1.0 - o),
            ],
            dim=1,
        ).to(dt)
        dgates[t] = draw
        dh = (draw @ w_h_t).to(at)
    return dgates, dh.to(dt), dc
