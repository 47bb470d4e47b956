#!/usr/bin/env python3
"""Numerics probe for the MT256 rec_fwd variant: find WHERE the output
mismatches the reference (which rows/cols), to diagnose fragment-mapping
bugs.  Runs RAW_DEBUG so output = pre + h @ W_h exactly."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def probe(B, H, variant):
    from lstm_tensorspark_amd.ops import extension
    ext = extension.get_extension()
    g = torch.Generator(device="cuda").manual_seed(0)
    h_prev = torch.randn(B, H, generator=g, device="cuda").bfloat16()
    w_h = (torch.randn(H, 4 * H, generator=g, device="cuda") / H ** 0.5).bfloat16()
    pre = torch.randn(B, 4 * H, generator=g, device="cuda").bfloat16()
    c_prev = torch.randn(B, H, generator=g, device="cuda")
    ext.set_fwd_variant(variant)
    _, _, acts = ext.lstm_rec_fwd_step(h_prev, w_h, pre, c_prev, True)
    ref = pre.float() + h_prev.float() @ w_h.float()
    err = (acts.float() - ref).abs()
    bad = err > 0.05 * ref.abs().max()
    print(f"B={B} H={H} v={variant}: max_err={err.max().item():.4f} "
          f"bad={int(bad.sum())}/{bad.numel()}")
    if bad.any():
        rows = bad.any(dim=1)
        cols = bad.any(dim=0)
        print("  bad rows:", torch.nonzero(rows).flatten()[:12].tolist(),
              "... count", int(rows.sum()))
        print("  bad cols:", torch.nonzero(cols).flatten()[:12].tolist(),
              "... count", int(cols.sum()))
        r = int(torch.nonzero(rows).flatten()[0])
        c = int(torch.nonzero(bad[r]).flatten()[0])
        print(f"  sample ({r},{c}): got {acts[r,c].item():.4f} "
              f"want {ref[r,c].item():.4f}")
        # is got(r,c) == ref at some other (r2,c2)? search row/col swaps
        v = acts[r, c].float()
        hits = (ref - v).abs() < 2e-2
        idx = torch.nonzero(hits)
        print(f"  value found at ref positions: {idx[:6].tolist()}")


if __name__ == "__main__":
    variants = [int(v) for v in sys.argv[1].split(",")] if len(sys.argv) > 1 \
        else [2, 32, 512, 520]
    reps = int(sys.argv[2]) if len(sys.argv) > 2 else 1
    # multi-run race screen across shapes (sync-structure edits are NEW
    # templates per the guide's two-lane discipline): nsteps = 1, 2, 4, 16
    for rep in range(reps):
        for v in variants:
            probe(256, 64, v)
            probe(256, 128, v)
            probe(256, 256, v)
            probe(512, 64, v)
            probe(4096, 1024, v)


def dump(B, H, variant, path):
    from lstm_tensorspark_amd.ops import extension
    ext = extension.get_extension()
    g = torch.Generator(device="cuda").manual_seed(0)
    h_prev = torch.randn(B, H, generator=g, device="cuda").bfloat16()
    w_h = (torch.randn(H, 4 * H, generator=g, device="cuda") / H ** 0.5).bfloat16()
    pre = torch.randn(B, 4 * H, generator=g, device="cuda").bfloat16()
    c_prev = torch.randn(B, H, generator=g, device="cuda")
    ext.set_fwd_variant(variant)
    _, _, acts = ext.lstm_rec_fwd_step(h_prev, w_h, pre, c_prev, True)
    ref = pre.float() + h_prev.float() @ w_h.float()
    torch.save({"acts": acts.float().cpu(), "ref": ref.cpu()}, path)
    print("dumped", path)
