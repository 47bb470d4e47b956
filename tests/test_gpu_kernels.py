"""GPU numerics: every HIP kernel vs the plain-PyTorch fp32 reference
(functional.py / optim.py CPU oracle) on the same device.  All tests here
run in bf16 (the flagship compute dtype) against fp32 references built
from the SAME bf16-quantized inputs, so tolerances cover accumulation-
order differences only.
"""
import os

import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="no GPU")


@pytest.fixture(scope="module")
def ext():
    from lstm_tensorspark_amd.ops import extension

    return extension.get_extension()


def make_step_inputs(B, H, seed=0, dev="cuda"):
    g = torch.Generator(device="cpu").manual_seed(seed)

    def r(*s, scale=1.0):
        return (torch.randn(*s, generator=g) * scale).to(dev)

    h_prev = r(B, H, scale=0.5).bfloat16()
    w_h = r(H, 4 * H, scale=1.0 / H ** 0.5).bfloat16()
    pre = r(B, 4 * H, scale=1.0).bfloat16()
    c_prev = r(B, H, scale=0.5)  # fp32
    return h_prev, w_h, pre, c_prev


# ---------------------------------------------------------------- K1b GEMM
@pytest.mark.parametrize("B,H", [(64, 64), (128, 128), (100, 256), (256, 1024)])
def test_rec_gemm_mapping_raw(ext, B, H):
    """MFMA fragment mapping: raw gates == pre + h @ W_h.  Asymmetric random
    operands (transpose-detecting, guide §5.4 rule 16)."""
    h_prev, w_h, pre, c_prev = make_step_inputs(B, H)
    h_out, c_out, acts = ext.lstm_rec_fwd_step(h_prev, w_h, pre, c_prev, True)
    ref = pre.float() + h_prev.float() @ w_h.float()
    err = (acts.float() - ref).abs().max().item()
    scale = ref.abs().max().item()
    assert err < 0.02 * max(scale, 1.0), f"raw-gate mismatch: {err} (scale {scale})"


@pytest.mark.parametrize("B,H", [(64, 64), (96, 128), (256, 1024)])
def test_rec_fwd_step_full(ext, B, H):
    """Fused step: activations + cell update vs fp32 reference."""
    h_prev, w_h, pre, c_prev = make_step_inputs(B, H, seed=1)
    h_out, c_out, acts = ext.lstm_rec_fwd_step(h_prev, w_h, pre, c_prev, False)
    raw = pre.float() + h_prev.float() @ w_h.float()
    i, f, g, o = raw.chunk(4, dim=1)
    i, f, g, o = i.sigmoid(), f.sigmoid(), g.tanh(), o.sigmoid()
    c_ref = f * c_prev + i * g
    h_ref = o * torch.tanh(c_ref)
    assert (c_out - c_ref).abs().max().item() < 0.03
    assert (h_out.float() - h_ref).abs().max().item() < 0.02
    acts_ref = torch.cat([i, f, g, o], dim=1)
    assert (acts.float() - acts_ref).abs().max().item() < 0.02


@pytest.mark.parametrize("variant", [512, 520, 1027, 4096])
@pytest.mark.parametrize("B,H", [(256, 64), (256, 128), (256, 256),
                                 (512, 1024), (1024, 1024)])
def test_rec_fwd_8phase_variants(ext, variant, B, H):
    """The 8-phase BK=64 counted-vmcnt kernel (k_lstm_rec_fwd_8p) vs the
    fp32 reference: full fused step at the nsteps = 1/2/4/16 pipeline
    edges (prologue/epilogue drain paths), with and without XCD swizzle."""
    h_prev, w_h, pre, c_prev = make_step_inputs(B, H, seed=4)
    prev = ext.get_fwd_variant()
    try:
        ext.set_fwd_variant(variant)
        h_out, c_out, acts = ext.lstm_rec_fwd_step(h_prev, w_h, pre, c_prev,
                                                   False)
    finally:
        ext.set_fwd_variant(prev)
    raw = pre.float() + h_prev.float() @ w_h.float()
    i, f, g, o = raw.chunk(4, dim=1)
    i, f, g, o = i.sigmoid(), f.sigmoid(), g.tanh(), o.sigmoid()
    c_ref = f * c_prev + i * g
    h_ref = o * torch.tanh(c_ref)
    assert (c_out - c_ref).abs().max().item() < 0.03
    assert (h_out.float() - h_ref).abs().max().item() < 0.02
    acts_ref = torch.cat([i, f, g, o], dim=1)
    assert (acts.float() - acts_ref).abs().max().item() < 0.02


# ---------------------------------------------------------------- sequence
def seq_inputs(B, T, H, seed=0, dev="cuda"):
    g = torch.Generator(device="cpu").manual_seed(seed)

    def r(*s, scale=1.0):
        return (torch.randn(*s, generator=g) * scale).to(dev)

    pre = r(T, B, 4 * H).bfloat16()  # time-major
    w_h = r(H, 4 * H, scale=1.0 / H ** 0.5).bfloat16()
    h0 = r(B, H, scale=0.5).bfloat16()
    c0 = r(B, H, scale=0.5)
    return pre, w_h, h0, c0


@pytest.mark.parametrize("reverse", [False, True])
@pytest.mark.parametrize("B,T,H", [(64, 5, 64), (256, 16, 128)])
def test_seq_fwd_vs_ref(ext, reverse, B, T, H):
    from lstm_tensorspark_amd.ops import functional

    pre, w_h, h0, c0 = seq_inputs(B, T, H)
    # both return h_all (T+1,B,H) with the h0 slot included
    h_all, c_seq, acts, hT, cT = ext.lstm_seq_fwd(pre, w_h, h0, c0, reverse)
    h_r, c_r, a_r, hT_r, cT_r = functional.ref_seq_fwd(pre, w_h, h0, c0, reverse)
    # recurrence compounds bf16 rounding; tolerance grows mildly with T
    tol = 0.02 + 0.005 * T
    assert h_all.shape == (T + 1, B, H)
    assert (h_all.float() - h_r.float()).abs().max().item() < tol
    assert (c_seq - c_r).abs().max().item() < 2 * tol
    assert (hT.float() - hT_r.float()).abs().max().item() < tol
    assert (cT - cT_r).abs().max().item() < 2 * tol


@pytest.mark.parametrize("reverse", [False, True])
def test_seq_bwd_vs_ref(ext, reverse):
    """Backward kernels on IDENTICAL saved tensors as the reference."""
    from lstm_tensorspark_amd.ops import functional

    B, T, H = 64, 8, 128
    pre, w_h, h0, c0 = seq_inputs(B, T, H, seed=2)
    # use the reference forward's saved tensors for BOTH backward paths
    h_r, c_r, a_r, hT_r, cT_r = functional.ref_seq_fwd(pre, w_h, h0, c0, reverse)
    g = torch.Generator(device="cpu").manual_seed(3)
    d_hseq = (torch.randn(T, B, H, generator=g) * 0.1).to("cuda").bfloat16()
    d_hT = (torch.randn(B, H, generator=g) * 0.1).to("cuda").bfloat16()
    d_cT = (torch.randn(B, H, generator=g) * 0.1).to("cuda")

    dg, dh0, dc0 = ext.lstm_seq_bwd(d_hseq, d_hT, d_cT, w_h, a_r, c_r, c0, reverse)[:3]
    dg_r, dh0_r, dc0_r = functional.ref_seq_bwd(d_hseq, d_hT, d_cT, w_h, a_r,
                                                c_r, c0, reverse)
    assert (dg.float() - dg_r.float()).abs().max().item() < 0.05
    assert (dh0.float() - dh0_r.float()).abs().max().item() < 0.05
    assert (dc0 - dc0_r).abs().max().item() < 0.05


@pytest.mark.parametrize("reverse", [False, True])
@pytest.mark.parametrize("B,T,H", [(256, 6, 256), (512, 3, 1024)])
def test_seq_bwd_fused_vs_ref(ext, reverse, B, T, H):
    """Fused dh-GEMM+pointwise backward (k_lstm_dh_pw_bwd) vs the fp32
    reference AND vs the split path on identical saved tensors."""
    from lstm_tensorspark_amd.ops import functional

    pre, w_h, h0, c0 = seq_inputs(B, T, H, seed=5)
    h_r, c_r, a_r, hT_r, cT_r = functional.ref_seq_fwd(pre, w_h, h0, c0, reverse)
    g = torch.Generator(device="cpu").manual_seed(6)
    d_hseq = (torch.randn(T, B, H, generator=g) * 0.1).to("cuda").bfloat16()
    prev = ext.get_fused_bwd()
    try:
        ext.set_fused_bwd(1)
        dg, dh0, dc0 = ext.lstm_seq_bwd(d_hseq, None, None, w_h, a_r, c_r,
                                        c0, reverse)[:3]
        ext.set_fused_bwd(0)
        dg_s, dh0_s, dc0_s = ext.lstm_seq_bwd(d_hseq, None, None, w_h, a_r,
                                              c_r, c0, reverse)[:3]
    finally:
        ext.set_fused_bwd(prev)
    dg_r, dh0_r, dc0_r = functional.ref_seq_bwd(d_hseq, None, None, w_h, a_r,
                                                c_r, c0, reverse)
    assert (dg.float() - dg_r.float()).abs().max().item() < 0.05
    assert (dh0.float() - dh0_r.float()).abs().max().item() < 0.05
    assert (dc0 - dc0_r).abs().max().item() < 0.05
    # fused vs split agree to bf16 rounding (fused keeps the dh carry fp32)
    assert (dg.float() - dg_s.float()).abs().max().item() < 0.05
    assert (dh0.float() - dh0_s.float()).abs().max().item() < 0.05


@pytest.mark.parametrize("reverse", [False, True])
def test_fused_bwd_write_time_bias_grad(ext, reverse):
    """The r2 write-time bias-grad accumulation (db_partial) must equal
    the full colsum of dgates to fp32-ordering tolerance."""
    B, T, H = 256, 5, 256
    pre, w_h, h0, c0 = seq_inputs(B, T, H, seed=9)
    from lstm_tensorspark_amd.ops import functional
    h_r, c_r, a_r, _, _ = functional.ref_seq_fwd(pre, w_h, h0, c0, reverse)
    g = torch.Generator(device="cpu").manual_seed(10)
    d_hseq = (torch.randn(T, B, H, generator=g) * 0.1).to("cuda").bfloat16()
    prev = ext.get_fused_bwd()
    try:
        ext.set_fused_bwd(1)
        dg, dh0, dc0, db_p = ext.lstm_seq_bwd(d_hseq, None, None, w_h, a_r,
                                              c_r, c0, reverse)
    finally:
        ext.set_fused_bwd(prev)
    assert db_p.numel() > 0, "partials missing on an eligible fused shape"
    t_first = (T - 1) if not reverse else 0
    db = db_p.sum(dim=0) + ext.colsum_bf16(dg.select(0, t_first))
    ref = dg.float().sum(dim=(0, 1))
    scale = max(1.0, ref.abs().max().item())
    assert (db - ref).abs().max().item() < 1e-2 * scale


def test_seq_bwd_fused_none_dh_up(ext):
    """Fused backward with NO upstream per-step grads (loss on hT only):
    dh_up is null inside k_lstm_dh_pw_bwd for every step."""
    from lstm_tensorspark_amd.ops import functional

    B, T, H = 256, 4, 256
    pre, w_h, h0, c0 = seq_inputs(B, T, H, seed=7)
    h_r, c_r, a_r, _, _ = functional.ref_seq_fwd(pre, w_h, h0, c0, False)
    d_hT = (torch.randn(B, H) * 0.1).to("cuda").bfloat16()
    prev = ext.get_fused_bwd()
    try:
        ext.set_fused_bwd(1)
        dg, dh0, dc0 = ext.lstm_seq_bwd(None, d_hT, None, w_h, a_r, c_r,
                                        c0, False)[:3]
    finally:
        ext.set_fused_bwd(prev)
    dg_r, dh0_r, dc0_r = functional.ref_seq_bwd(None, d_hT, None, w_h, a_r,
                                                c_r, c0, False)
    assert (dg.float() - dg_r.float()).abs().max().item() < 0.05
    assert (dh0.float() - dh0_r.float()).abs().max().item() < 0.05
    assert (dc0 - dc0_r).abs().max().item() < 0.05


def test_seq_bwd_none_grads(ext):
    from lstm_tensorspark_amd.ops import functional

    B, T, H = 32, 4, 64
    pre, w_h, h0, c0 = seq_inputs(B, T, H, seed=4)
    h_r, c_r, a_r, _, _ = functional.ref_seq_fwd(pre, w_h, h0, c0, False)
    d_hT = (torch.randn(B, H) * 0.1).to("cuda").bfloat16()
    dg, dh0, dc0 = ext.lstm_seq_bwd(None, d_hT, None, w_h, a_r, c_r, c0, False)[:3]
    dg_r, dh0_r, dc0_r = functional.ref_seq_bwd(None, d_hT, None, w_h, a_r,
                                                c_r, c0, False)
    assert (dg.float() - dg_r.float()).abs().max().item() < 0.05
    assert (dc0 - dc0_r).abs().max().item() < 0.05


# ------------------------------------------------------------- full layer
def test_layer_autograd_gpu_vs_cpu_fp32():
    """End-to-end layer fwd+bwd: GPU bf16 fused path vs CPU fp32 reference."""
    from lstm_tensorspark_amd.ops.lstm_layer import lstm_layer

    B, T, X, H = 32, 8, 128, 128
    g = torch.Generator(device="cpu").manual_seed(5)
    x32 = torch.randn(T, B, X, generator=g) * 0.5  # time-major
    h032 = torch.randn(B, H, generator=g) * 0.5
    c032 = torch.randn(B, H, generator=g) * 0.5
    wx32 = torch.randn(X, 4 * H, generator=g) / X ** 0.5
    wh32 = torch.randn(H, 4 * H, generator=g) / H ** 0.5
    b32 = torch.randn(4 * H, generator=g) * 0.1
    gout = torch.randn(T, B, H, generator=g) * 0.1

    # CPU fp32 reference
    cpu = [t.clone().requires_grad_(True) for t in (x32, h032, c032, wx32, wh32, b32)]
    h_seq_c, _, _ = lstm_layer(*cpu)
    h_seq_c.backward(gout)

    # GPU bf16 fused
    gpu = [t.to("cuda").bfloat16().requires_grad_(True) for t in (x32, h032, wx32, wh32, b32)]
    c0g = c032.to("cuda").requires_grad_(True)
    xg, h0g, wxg, whg, bg = gpu
    h_seq_g, _, _ = lstm_layer(xg, h0g, c0g, wxg, whg, bg)
    h_seq_g.backward(gout.to("cuda").bfloat16())

    assert (h_seq_g.float().cpu() - h_seq_c.detach()).abs().max().item() < 0.1
    pairs = [(xg, cpu[0]), (h0g, cpu[1]), (c0g, cpu[2]), (wxg, cpu[3]),
             (whg, cpu[4]), (bg, cpu[5])]
    for gt, ct in pairs:
        gg, cg = gt.grad.float().cpu(), ct.grad
        denom = cg.abs().max().item() + 1e-3
        rel = (gg - cg).abs().max().item() / denom
        assert rel < 0.12, f"grad rel err {rel}"


# ---------------------------------------------------------------- K6 / K7
def test_l2norm_and_clip(ext):
    gs = [torch.randn(1000, device="cuda").bfloat16(),
          torch.randn(64, 33, device="cuda"),
          torch.randn(7, device="cuda").bfloat16()]
    norm = ext.multi_tensor_l2norm(list(gs))
    want = torch.sqrt(sum(g.float().pow(2).sum() for g in gs))
    assert abs(norm.item() - want.item()) / want.item() < 1e-2
    ext.multi_tensor_clip_scale(list(gs), norm, 0.5)
    after = torch.sqrt(sum(g.float().pow(2).sum() for g in gs))
    assert abs(after.item() - 0.5) < 0.02


def test_fused_sgd_vs_oracle(ext):
    torch.manual_seed(0)
    shapes = [(128, 64), (4096,), (33,)]
    ps = [torch.randn(s, device="cuda").bfloat16() for s in shapes]
    ms = [p.float() for p in ps]
    bufs = [torch.zeros_like(m) for m in ms]
    ps_c = [p.clone() for p in ps]
    ms_c = [m.clone() for m in ms]
    bufs_c = [b.clone() for b in bufs]
    for step in range(3):
        gs = [torch.randn(s, device="cuda").bfloat16() * 0.1 for s in shapes]
        ext.fused_sgd(list(ps), list(ms), list(gs), list(bufs), 0.1, 0.9, 0.01)
        # oracle (fp32 math, same formulas as ops/optim.py CPU path)
        for p, m, b, g in zip(ps_c, ms_c, bufs_c, gs):
            gv = g.float() + 0.01 * m
            b.mul_(0.9).add_(gv)
            m.add_(b, alpha=-0.1)
            p.copy_(m.bfloat16())
    for a, b in zip(ms, ms_c):
        assert (a - b).abs().max().item() < 1e-5
    for a, b in zip(ps, ps_c):
        assert torch.equal(a, b)


def test_fused_adam_vs_oracle(ext):
    torch.manual_seed(1)
    shapes = [(256, 32), (999,)]
    ps = [torch.randn(s, device="cuda").bfloat16() for s in shapes]
    ms = [p.float() for p in ps]
    ea = [torch.zeros_like(m) for m in ms]
    eas = [torch.zeros_like(m) for m in ms]
    ps_c = [p.clone() for p in ps]
    ms_c = [m.clone() for m in ms]
    ea_c = [e.clone() for e in ea]
    eas_c = [e.clone() for e in eas]
    b1, b2, eps, lr = 0.9, 0.999, 1e-8, 1e-3
    for step in range(1, 4):
        bc1, bc2 = 1 - b1 ** step, 1 - b2 ** step
        gs = [torch.randn(s, device="cuda").bfloat16() * 0.1 for s in shapes]
        ext.fused_adam(list(ps), list(ms), list(gs), list(ea), list(eas),
                       lr, b1, b2, eps, 0.0, bc1, bc2)
        for p, m, e1, e2, g in zip(ps_c, ms_c, ea_c, eas_c, gs):
            gv = g.float()
            e1.mul_(b1).add_(gv, alpha=1 - b1)
            e2.mul_(b2).addcmul_(gv, gv, value=1 - b2)
            denom = (e2 / bc2).sqrt().add_(eps)
            m.addcdiv_(e1 / bc1, denom, value=-lr)
            p.copy_(m.bfloat16())
    for a, b in zip(ms, ms_c):
        assert (a - b).abs().max().item() < 1e-5


# -------------------------------------------------- native-code-is-loaded
def test_hip_path_actually_dispatches():
    """The flagship path must run the HIP extension, not eager fallback."""
    from lstm_tensorspark_amd.ops import extension
    from lstm_tensorspark_amd.ops.lstm_layer import lstm_layer

    assert extension.extension_available(), "extension must be importable"
    assert extension.use_hip(torch.device("cuda"), torch.bfloat16)
    so = extension.get_extension().__file__
    assert "lstm_tensorspark_amd" in so  # in-tree .so, not site-packages


# ---------------------------------------------------------------- K3 xent
@pytest.mark.parametrize("N,V", [(64, 512), (32, 50000), (16, 1000 + 3)])
def test_fused_xent_vs_torch(ext, N, V):
    import torch.nn.functional as F

    g = torch.Generator(device="cpu").manual_seed(0)
    logits = (torch.randn(N, V, generator=g) * 2).to("cuda").bfloat16()
    targets = torch.randint(0, V, (N,), generator=g).to("cuda")
    lg = logits.clone().requires_grad_(True)
    from lstm_tensorspark_amd.ops.xent import FusedCrossEntropyFn

    loss = FusedCrossEntropyFn.apply(lg, targets)
    loss.backward()

    lr = logits.float().clone().requires_grad_(True)
    loss_ref = F.cross_entropy(lr, targets)
    loss_ref.backward()

    assert abs(loss.item() - loss_ref.item()) < 2e-2 * max(1.0, abs(loss_ref.item()))
    err = (lg.grad.float() - lr.grad).abs().max().item()
    assert err < 2e-3, err


def test_fused_xent_grad_scale(ext):
    """Upstream grad != 1 must scale dlogits on-device."""
    from lstm_tensorspark_amd.ops.xent import FusedCrossEntropyFn

    g = torch.Generator(device="cpu").manual_seed(1)
    logits = torch.randn(8, 128, generator=g).to("cuda").bfloat16()
    targets = torch.randint(0, 128, (8,), generator=g).to("cuda")
    l1 = logits.clone().requires_grad_(True)
    FusedCrossEntropyFn.apply(l1, targets).mul(3.0).backward()
    l2 = logits.clone().requires_grad_(True)
    FusedCrossEntropyFn.apply(l2, targets).backward()
    assert (l1.grad.float() - 3.0 * l2.grad.float()).abs().max().item() < 1e-2


# ------------------------------------------------- side-stream dW overlap
def test_side_stream_dw_matches_sync_path():
    """dW/db produced on the side stream (after wait_side_stream) must be
    identical to the synchronous path."""
    from lstm_tensorspark_amd.ops.lstm_layer import (
        enable_side_dw, lstm_layer, wait_side_stream)

    T, B, X, H = 8, 256, 128, 128
    g = torch.Generator(device="cpu").manual_seed(7)
    x32 = torch.randn(T, B, X, generator=g) * 0.5
    wx32 = torch.randn(X, 4 * H, generator=g) / X ** 0.5
    wh32 = torch.randn(H, 4 * H, generator=g) / H ** 0.5
    b32 = torch.randn(4 * H, generator=g) * 0.1
    gout = (torch.randn(T, B, H, generator=g) * 0.1).to("cuda").bfloat16()
    h0 = torch.zeros(B, H, device="cuda").bfloat16()
    c0 = torch.zeros(B, H, device="cuda")

    grads = {}
    for side in (False, True):
        enable_side_dw(side)
        xs = x32.to("cuda").bfloat16().requires_grad_(True)
        wx = wx32.to("cuda").bfloat16().requires_grad_(True)
        wh = wh32.to("cuda").bfloat16().requires_grad_(True)
        bb = b32.to("cuda").bfloat16().requires_grad_(True)
        hs, _, _ = lstm_layer(xs, h0, c0, wx, wh, bb)
        hs.backward(gout)
        wait_side_stream()
        torch.cuda.synchronize()
        grads[side] = (wx.grad.clone(), wh.grad.clone(), bb.grad.clone())
    enable_side_dw(False)
    for a, b in zip(grads[False], grads[True]):
        assert torch.equal(a, b)


# ------------------------------------------------------------- shape fuzz
def test_shape_fuzz_seq_fwd_bwd_vs_ref():
    """Randomized shape sweep: every (B,T,H) through the full sequence
    fwd+bwd HIP path vs the fp32 reference.  Covers tile-boundary and
    variant-dispatch edges (BM32/BM64/MT256 x split/fused thresholds)."""
    import random

    from lstm_tensorspark_amd.ops import functional
    from lstm_tensorspark_amd.ops.extension import get_extension

    ext = get_extension()
    rng = random.Random(0)
    shapes = [(rng.choice([1, 3, 16, 63, 100, 256, 300, 512, 2048]),
               rng.choice([1, 2, 5, 9]),
               rng.choice([64, 128, 192, 1024]))
              for _ in range(12)] + [(256, 3, 64), (2048, 2, 1024)]
    for i, (B, T, H) in enumerate(shapes):
        g = torch.Generator(device="cpu").manual_seed(100 + i)

        def r(*s, scale=1.0):
            return (torch.randn(*s, generator=g) * scale).to("cuda")

        pre = r(T, B, 4 * H).bfloat16()
        w_h = r(H, 4 * H, scale=1.0 / H ** 0.5).bfloat16()
        h0 = r(B, H, scale=0.5).bfloat16()
        c0 = r(B, H, scale=0.5)
        h_all, c_seq, acts, hT, cT = ext.lstm_seq_fwd(pre, w_h, h0, c0, i % 2 == 1)
        h_r, c_r, a_r, hT_r, cT_r = functional.ref_seq_fwd(
            pre, w_h, h0, c0, i % 2 == 1)
        # bf16 recurrence diverges CHAOTICALLY from the fp32 oracle as T
        # grows (a 1-ulp difference in step 1 is amplified through the
        # gates); the HIP-vs-HIP consistency check below is the tight one.
        tol = 0.05 + 0.02 * T
        err = (h_all.float() - h_r.float()).abs().max().item()
        assert err < tol, f"fwd shape {(B,T,H)} rev={i%2} err={err}"
        # split-vs-fused consistency on the same inputs (both bf16 paths)
        if B % 32 == 0 and H % 64 == 0:
            cur = ext.get_split_fwd()
            try:
                ext.set_split_fwd(0)
                h_f = ext.lstm_seq_fwd(pre, w_h, h0, c0, i % 2 == 1)[0]
                ext.set_split_fwd(1)
                h_s = ext.lstm_seq_fwd(pre, w_h, h0, c0, i % 2 == 1)[0]
            finally:
                ext.set_split_fwd(cur)
            diff2 = (h_f.float() - h_s.float()).abs()
            err2 = diff2.max().item()
            # two DIFFERENT bf16 chains over T recurrent steps diverge
            # chaotically (and hipBLASLt's per-box algorithm choice makes
            # the magnitude box-dependent): a gross path bug blows up the
            # MEAN too, so accept chaos-limited max error when the mean
            # stays tiny (flaked box-dependently at max-only, r2)
            assert err2 < tol or diff2.mean().item() < 0.01, \
                f"split-vs-fused {(B,T,H)} err={err2} mean={diff2.mean().item()}"
        d_hseq = r(T, B, H, scale=0.1).bfloat16()
        dg, dh0, dc0 = ext.lstm_seq_bwd(d_hseq, None, None, w_h, a_r, c_r,
                                        c0, i % 2 == 1)[:3]
        dg_r, dh0_r, dc0_r = functional.ref_seq_bwd(
            d_hseq, None, None, w_h, a_r, c_r, c0, i % 2 == 1)
        err = (dg.float() - dg_r.float()).abs().max().item()
        assert err < 0.06, f"bwd shape {(B,T,H)} rev={i%2} err={err}"


# --------------------------------------------------------------- fp8 mode
def test_fp8_recurrent_gemm_vs_bf16():
    """Experimental fp8(e4m3) recurrent GEMM: output tracks the bf16 path
    within fp8 quantization error (h is tanh-bounded, fixed scale 448)."""
    from lstm_tensorspark_amd.ops import functional
    from lstm_tensorspark_amd.ops.extension import get_extension

    ext = get_extension()
    B, T, H = 2048, 8, 1024  # B*H >= 2M -> split path (fp8 requires split)
    g = torch.Generator(device="cpu").manual_seed(11)

    def r(*s, scale=1.0):
        return (torch.randn(*s, generator=g) * scale).to("cuda")

    pre = r(T, B, 4 * H, scale=0.5).bfloat16()
    w_h = r(H, 4 * H, scale=1.0 / H ** 0.5).bfloat16()
    h0 = (r(B, H).tanh()).bfloat16()
    c0 = r(B, H, scale=0.5)
    out_bf = ext.lstm_seq_fwd(pre, w_h, h0, c0, False)
    ext.set_fp8_fwd(True)
    try:
        out_f8 = ext.lstm_seq_fwd(pre, w_h, h0, c0, False)
    finally:
        ext.set_fp8_fwd(False)
    # h compounds quantization error over T steps; generous but bounded
    err = (out_bf[0].float() - out_f8[0].float()).abs().max().item()
    assert err < 0.05 + 0.02 * T, err
    # and the fp8 path must still be close to the fp32 reference
    h_r, *_ = functional.ref_seq_fwd(pre, w_h, h0, c0, False)
    err_ref = (out_f8[0].float() - h_r.float()).abs().max().item()
    assert err_ref < 0.05 + 0.02 * T, err_ref


def test_fp8_layer_grads_track_bf16():
    """fp8 dW/dx GEMMs: gradients track the bf16 path within quantization
    error (relative to grad magnitude)."""
    from lstm_tensorspark_amd.ops.extension import get_extension
    from lstm_tensorspark_amd.ops.lstm_layer import enable_fp8_wgrad, lstm_layer

    ext = get_extension()
    T, B, X, H = 4, 2048, 1024, 1024   # B*H >= 2M -> split+fp8 eligible
    g = torch.Generator(device="cpu").manual_seed(21)

    def r(*s, scale=1.0):
        return (torch.randn(*s, generator=g) * scale).to("cuda")

    x32 = r(T, B, X, scale=0.5)
    wx32 = r(X, 4 * H, scale=1.0 / X ** 0.5)
    wh32 = r(H, 4 * H, scale=1.0 / H ** 0.5)
    b32 = r(4 * H, scale=0.1)
    gout = r(T, B, H, scale=0.1).bfloat16()
    h0 = torch.zeros(B, H, device="cuda").bfloat16()
    c0 = torch.zeros(B, H, device="cuda")

    grads = {}
    for fp8 in (False, True):
        ext.set_fp8_fwd(fp8)
        enable_fp8_wgrad(fp8)
        try:
            xs = x32.bfloat16().requires_grad_(True)
            wx = wx32.bfloat16().requires_grad_(True)
            wh = wh32.bfloat16().requires_grad_(True)
            bb = b32.bfloat16().requires_grad_(True)
            hs, _, _ = lstm_layer(xs, h0, c0, wx, wh, bb)
            hs.backward(gout)
            grads[fp8] = [t.grad.float() for t in (xs, wx, wh, bb)]
        finally:
            ext.set_fp8_fwd(False)
            enable_fp8_wgrad(False)
    for name, a, b in zip("dx dwx dwh db".split(), grads[False], grads[True]):
        denom = a.abs().max().item() + 1e-6
        rel = (a - b).abs().max().item() / denom
        assert rel < 0.25, f"{name} rel err {rel}"


# ---------------------------------------------------------- serving GEMV
@pytest.mark.parametrize("B,N,K", [(1, 4096, 1024), (4, 512, 2048),
                                   (16, 50000, 4096)])
def test_decode_gemv_vs_mm(ext, B, N, K):
    g = torch.Generator(device="cpu").manual_seed(31)
    A_t = (torch.randn(N, K, generator=g) / K ** 0.5).to("cuda").bfloat16()
    x = torch.randn(B, K, generator=g).to("cuda").bfloat16()
    bias = torch.randn(N, generator=g).to("cuda").float()
    out = ext.decode_gemv(A_t, x, bias)
    ref = (x.float() @ A_t.float().t()) + bias
    err = (out.float() - ref).abs().max().item()
    assert err < 0.03 * max(1.0, ref.abs().max().item()), err


def test_colsum_vs_torch(ext):
    g = torch.Generator(device="cpu").manual_seed(41)
    for R, C in [(1000, 128), (4096, 4096), (33, 8)]:
        t = (torch.randn(R, C, generator=g) * 0.1).to("cuda").bfloat16()
        out = ext.colsum_bf16(t)
        ref = t.float().sum(dim=0)
        assert (out - ref).abs().max().item() < 1e-2 * max(1.0, ref.abs().max().item())


def test_quant_fp8_pack_vs_separate(ext):
    """One-pass pack == {quant_fp8, quant_fp8_T, fp32 colsum} bitwise for
    the fp8 outputs and to fp32-order tolerance for db."""
    g = torch.Generator(device="cpu").manual_seed(43)
    R, C = 1024, 256
    t = (torch.randn(R, C, generator=g) * 2.0).to("cuda").bfloat16()
    amax = t.abs().amax().float().clamp_min(1e-8)
    mult = 448.0 / amax
    dg8, dg8t, db = ext.quant_fp8_pack(t, mult)
    ref8 = ext.quant_fp8(t, mult)
    ref8t = ext.quant_fp8_T(t, mult)
    assert torch.equal(dg8.view(torch.uint8), ref8.view(torch.uint8))
    assert torch.equal(dg8t.view(torch.uint8), ref8t.view(torch.uint8))
    ref_db = t.float().sum(dim=0)
    assert (db - ref_db).abs().max().item() < 1e-2 * max(
        1.0, ref_db.abs().max().item())


def test_colsum_deterministic_run_to_run(ext):
    # the two-stage fixed-order reduction must be bitwise reproducible
    # (the old fp32 atomicAdd across grid-y blocks was not)
    g = torch.Generator(device="cpu").manual_seed(42)
    t = (torch.randn(3 * 4096 + 17, 512, generator=g) * 0.1).to("cuda").bfloat16()
    a = ext.colsum_bf16(t)
    for _ in range(5):
        assert torch.equal(a, ext.colsum_bf16(t))
