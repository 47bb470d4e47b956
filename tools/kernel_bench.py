#!/usr/bin/env python3
"""Microbenchmark of the hot kernels in isolation (GPU box).

Prints one JSON line per timing so the results can be committed under
profiles/ and compared across kernel revisions.
"""
import argparse
import json
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def time_op(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    start = torch.cuda.Event(enable_timing=True)
    end = torch.cuda.Event(enable_timing=True)
    start.record()
    for _ in range(iters):
        fn()
    end.record()
    torch.cuda.synchronize()
    return start.elapsed_time(end) / iters  # ms


def report(name, ms, flops=None, bytes_=None, **extra):
    rec = {"name": name, "ms": round(ms, 4)}
    if flops:
        rec["tflops"] = round(flops / ms / 1e9, 1)
    if bytes_:
        rec["gbps"] = round(bytes_ / ms / 1e6, 1)
    rec.update(extra)
    print(json.dumps(rec), flush=True)
    return rec


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--B", type=int, default=1024)
    ap.add_argument("--H", type=int, default=1024)
    ap.add_argument("--T", type=int, default=128)
    ap.add_argument("--iters", type=int, default=50)
    args = ap.parse_args()
    B, H, T = args.B, args.H, args.T

    from lstm_tensorspark_amd.ops import extension
    ext = extension.get_extension()
    dev = "cuda"
    g = torch.Generator(device=dev).manual_seed(0)

    h_prev = torch.randn(B, H, generator=g, device=dev).bfloat16()
    w_h = (torch.randn(H, 4 * H, generator=g, device=dev) / H ** 0.5).bfloat16()
    w_t = w_h.t().contiguous()
    pre1 = torch.randn(B, 4 * H, generator=g, device=dev).bfloat16()
    c_prev = torch.randn(B, H, generator=g, device=dev)

    rec_flops = 2 * B * 4 * H * H

    # ---- rec_fwd variant sweep (within-probe A/B): correctness then time
    ref_raw = pre1.float() + h_prev.float() @ w_h.float()
    i_, f_, g_, o_ = ref_raw.chunk(4, dim=1)
    c_ref = f_.sigmoid() * c_prev + i_.sigmoid() * g_.tanh()
    h_ref = o_.sigmoid() * torch.tanh(c_ref)
    default_variant = ext.get_fwd_variant()
    best = (None, 1e9)
    for v in [0, 1, 2, 3, 4, 5, 6, 7, 8, 9, 10, 11, 12, 13, 14, 15,
              18, 19, 26, 27,   # 16+x = BM128
              32, 40,           # MT256 3buf-BK32 (8=XCD swizzle)
              33, 41,           # MT256 2buf-BK64
              512, 520,         # 8-phase BK64 counted-vmcnt (8=XCD swizzle)
              1024, 1032,       # 2-phase merged BK64 (8=XCD swizzle)
              3072, 3075,       # 2-phase BM=128 row tile (3=TUNE3)
              4096, 4104]:      # overlap 128x256 BK32, 2 blocks/CU
        ext.set_fwd_variant(v)
        h_out, c_out, acts_out = ext.lstm_rec_fwd_step(h_prev, w_h, pre1,
                                                       c_prev, False)
        err_h = (h_out.float() - h_ref).abs().max().item()
        err_c = (c_out - c_ref).abs().max().item()
        ok = err_h < 0.05 and err_c < 0.08
        ms = time_op(lambda: ext.lstm_rec_fwd_step(h_prev, w_h, pre1,
                                                   c_prev, False), args.iters)
        report(f"rec_fwd_step variant={v}", ms, rec_flops, B=B, H=H,
               correct=ok, err_h=round(err_h, 4))
        if ok and ms < best[1]:
            best = (v, ms)
    print(json.dumps({"best_variant": best[0], "ms": round(best[1], 4)}),
          flush=True)
    ext.set_fwd_variant(best[0] if best[0] is not None else default_variant)
    # ablation: RAW_DEBUG = GEMM + pre-add + acts stores, NO activations/
    # cell update/h/c stores -- isolates the epilogue cost
    report("rec_fwd_step RAW(no-epilogue)", time_op(
        lambda: ext.lstm_rec_fwd_step(h_prev, w_h, pre1, c_prev, True),
        args.iters), rec_flops, B=B, H=H)

    # sequence forward (transpose amortized over T steps)
    pre = torch.randn(T, B, 4 * H, generator=g, device=dev).bfloat16()  # time-major
    h0 = torch.randn(B, H, generator=g, device=dev).bfloat16()
    c0 = torch.randn(B, H, generator=g, device=dev)
    # correctness: persistent vs per-step path on identical inputs
    ext.set_persistent_fwd(True)
    outs_p = ext.lstm_seq_fwd(pre, w_h, h0, c0, False)
    ext.set_persistent_fwd(False)
    outs_s = ext.lstm_seq_fwd(pre, w_h, h0, c0, False)
    perr = max((a.float() - b.float()).abs().max().item()
               for a, b in zip(outs_p, outs_s))
    for on in (False, True):
        ext.set_persistent_fwd(on)
        ms = time_op(lambda: ext.lstm_seq_fwd(pre, w_h, h0, c0, False),
                     max(args.iters // 10, 3))
        report(f"seq_fwd persistent={on}", ms, rec_flops * T, B=B, H=H, T=T,
               us_per_step=round(1e3 * ms / T, 2),
               vs_per_step_err=round(perr, 4) if on else None)
    ext.set_persistent_fwd(True)
    ms = ms

    # sequence backward: split vs fused (512-thr) vs fused-ov A/B
    h_all, c_seq, acts, hT, cT = ext.lstm_seq_fwd(pre, w_h, h0, c0, False)
    d_hseq = torch.randn(T, B, H, generator=g, device=dev).bfloat16()
    bwd_default = ext.get_fused_bwd()
    for mode, tag in ((0, "split"), (-1, "auto"), (1, "fused512"),
                      (3, "fused512-swz"), (2, "fused-ov")):
        ext.set_fused_bwd(mode)
        try:
            ms = time_op(lambda: ext.lstm_seq_bwd(d_hseq, None, None, w_h,
                                                  acts, c_seq, c0, False),
                         max(args.iters // 10, 3))
        except Exception as e:
            report(f"seq_bwd {tag}", 0.0, note=f"skipped: {e}")
            continue
        report(f"seq_bwd {tag}", ms, 2 * rec_flops * T, B=B, H=H, T=T,
               us_per_step=round(1e3 * ms / T, 2))
    ext.set_fused_bwd(bwd_default)

    # components: library GEMMs
    dg1 = torch.randn(B, 4 * H, generator=g, device=dev).bfloat16()
    report("mm dh=dgates@WhT (per step)", time_op(lambda: dg1 @ w_h.t(), args.iters),
           rec_flops)
    # the fwd recurrent shape on hipBLASLt (split-path candidate):
    raw_buf = torch.empty(B, 4 * H, device=dev).bfloat16()
    report("mm fwd h@Wh (per step)", time_op(
        lambda: torch.mm(h_prev, w_h, out=raw_buf), args.iters), rec_flops)
    report("addmm fwd pre+h@Wh (per step)", time_op(
        lambda: torch.addmm(pre1, h_prev, w_h, out=raw_buf), args.iters), rec_flops)
    x_flat = torch.randn(B * T, H, generator=g, device=dev).bfloat16()
    w_x = (torch.randn(H, 4 * H, generator=g, device=dev) / H ** 0.5).bfloat16()
    report("pre-GEMM (BT,X)@(X,4H)", time_op(lambda: x_flat @ w_x, 10),
           2 * B * T * H * 4 * H)
    dg_flat = torch.randn(B * T, 4 * H, generator=g, device=dev).bfloat16()
    report("dW GEMM (H,BT)@(BT,4H)", time_op(lambda: x_flat.t() @ dg_flat, 10),
           2 * B * T * H * 4 * H)

    # full layer autograd fwd+bwd
    from lstm_tensorspark_amd.ops.lstm_layer import lstm_layer
    x = torch.randn(T, B, H, generator=g, device=dev).bfloat16()
    wx_p = w_x.clone().requires_grad_(True)
    wh_p = w_h.clone().requires_grad_(True)
    b_p = torch.zeros(4 * H, device=dev).bfloat16().requires_grad_(True)
    h0z = torch.zeros(B, H, device=dev).bfloat16()
    c0z = torch.zeros(B, H, device=dev)
    gout = torch.randn(T, B, H, generator=g, device=dev).bfloat16()

    def layer_fb():
        hs, _, _ = lstm_layer(x, h0z, c0z, wx_p, wh_p, b_p)
        hs.backward(gout)
        wx_p.grad = wh_p.grad = b_p.grad = None

    total_flops = (2 * B * T * 4 * H * H) * 2 + rec_flops * T * 3 + 2 * B * T * H * 4 * H * 2
    report("layer fwd+bwd", time_op(layer_fb, 5, warmup=3), total_flops,
           B=B, H=H, T=T)

    # pointwise bwd kernel isolated
    dc = torch.zeros(B, H, device=dev)
    dg_out = torch.empty(B, 4 * H, device=dev).bfloat16()
    report("pw_bwd isolated (via seq T=1)", time_op(
        lambda: ext.lstm_seq_bwd(None, d_hseq[0].contiguous(), None, w_h,
                                 acts[:1].contiguous(), c_seq[:1].contiguous(),
                                 c0, False), args.iters))


if __name__ == "__main__":
    main()
