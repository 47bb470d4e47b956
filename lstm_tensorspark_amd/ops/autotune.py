"""Opt-in runtime autotune for the recurrent forward path.

The default split-vs-fused choice is a measured heuristic (B*H >= 2M,
csrc/bindings.cpp); `tune_fwd` replaces it with an on-device measurement
for THIS trainer's exact (B, H): it times a short synthetic sequence
through both paths and pins the winner via set_split_fwd.  Enabled by
`train.autotune = true` (off by default: the heuristic is already right
at the preset shapes, and first-step timing on a busy GPU adds noise).
"""
from __future__ import annotations

import torch

from . import extension


def tune_fwd(B: int, H: int, T: int = 8, iters: int = 5) -> dict:
    """Measure split (hipBLASLt + pointwise) vs fused (hand MFMA kernel)
    for the (B, H) recurrent step shape and pin the faster path.

    Returns {"split_us", "fused_us", "choice"}.  No-op (heuristic kept)
    without a GPU or when H % 64 != 0."""
    if not torch.cuda.is_available() or H % 64 != 0:
        return {"choice": "heuristic"}
    ext = extension.get_extension()
    dev = "cuda"
    g = torch.Generator(device=dev).manual_seed(0)
    pre = torch.randn(T, B, 4 * H, generator=g, device=dev).bfloat16()
    w_h = (torch.randn(H, 4 * H, generator=g, device=dev) / H ** 0.5).bfloat16()
    h0 = torch.randn(B, H, generator=g, device=dev).bfloat16().tanh()
    c0 = torch.randn(B, H, generator=g, device=dev)

    def time_mode(mode: int) -> float:
        ext.set_split_fwd(mode)
        for _ in range(2):
            ext.lstm_seq_fwd(pre, w_h, h0, c0, False)
        torch.cuda.synchronize()
        s = torch.cuda.Event(enable_timing=True)
        e = torch.cuda.Event(enable_timing=True)
        s.record()
        for _ in range(iters):
            ext.lstm_seq_fwd(pre, w_h, h0, c0, False)
        e.record()
        torch.cuda.synchronize()
        return 1e3 * s.elapsed_time(e) / (iters * T)  # us / step

    try:
        split_us = time_mode(1)
        fused_us = time_mode(0)
    except Exception:
        ext.set_split_fwd(-1)
        raise
    choice = 1 if split_us <= fused_us else 0
    ext.set_split_fwd(choice)
    return {"split_us": round(split_us, 1), "fused_us": round(fused_us, 1),
            "choice": "split" if choice else "fused"}
