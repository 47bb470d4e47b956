#!/usr/bin/env python3
"""Focused runner for PMC capture: N iterations of ONE rec_fwd variant
(so rocprofv3 --pmc attribution is a single kernel symbol).

Usage: rocprofv3 --pmc <counters> -- python tools/pmc_rec_fwd.py --variant 32
"""
import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--variant", type=int, default=32)
    ap.add_argument("--B", type=int, default=4096)
    ap.add_argument("--H", type=int, default=1024)
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--raw", action="store_true", help="RAW_DEBUG (no epilogue)")
    args = ap.parse_args()
    B, H = args.B, args.H

    from lstm_tensorspark_amd.ops import extension
    ext = extension.get_extension()
    g = torch.Generator(device="cuda").manual_seed(0)
    h_prev = torch.randn(B, H, generator=g, device="cuda").bfloat16()
    w_h = (torch.randn(H, 4 * H, generator=g, device="cuda") / H ** 0.5).bfloat16()
    pre = torch.randn(B, 4 * H, generator=g, device="cuda").bfloat16()
    c_prev = torch.randn(B, H, generator=g, device="cuda")
    ext.set_fwd_variant(args.variant)
    for _ in range(10):
        ext.lstm_rec_fwd_step(h_prev, w_h, pre, c_prev, args.raw)
    torch.cuda.synchronize()
    s, e = torch.cuda.Event(enable_timing=True), torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(args.iters):
        ext.lstm_rec_fwd_step(h_prev, w_h, pre, c_prev, args.raw)
    e.record()
    torch.cuda.synchronize()
    print(f"variant={args.variant} raw={args.raw} "
          f"us={1e3 * s.elapsed_time(e) / args.iters:.1f}")


if __name__ == "__main__":
    main()
