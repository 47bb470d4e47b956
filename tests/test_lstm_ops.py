"""Manual-backward correctness of the LSTM layer op (CPU, fp64 oracle).

This is the backward-math tier: the same dgates/dh/dc recurrences the HIP
kernels implement are validated here against torch autograd exactly
(fp64), so GPU kernel tests only need to match functional.py.
"""
import pytest
import torch

from lstm_tensorspark_amd.ops.lstm_layer import lstm_layer


def autograd_ref(x, h0, c0, w_x, w_h, b, reverse):
    # time-major: x (T,B,X)
    T, B, X = x.shape
    H = w_h.shape[0]
    pre = x @ w_x + b
    h, c = h0, c0
    hs = [None] * T
    ts = range(T - 1, -1, -1) if reverse else range(T)
    for t in ts:
        raw = pre[t] + h @ w_h
        i = raw[:, :H].sigmoid()
        f = raw[:, H:2 * H].sigmoid()
        g = raw[:, 2 * H:3 * H].tanh()
        o = raw[:, 3 * H:].sigmoid()
        c = f * c + i * g
        h = o * torch.tanh(c)
        hs[t] = h
    return torch.stack(hs, 0), h, c


def make_inputs(B, T, X, H, dtype=torch.float64, seed=0):
    g = torch.Generator().manual_seed(seed)

    def r(*s, scale=1.0):
        return (torch.randn(*s, generator=g, dtype=torch.float32) * scale).to(dtype)

    return (r(T, B, X), r(B, H), r(B, H), r(X, 4 * H, scale=0.3),
            r(H, 4 * H, scale=0.3), r(4 * H, scale=0.1))


@pytest.mark.parametrize("reverse", [False, True])
@pytest.mark.parametrize("shape", [(4, 7, 5, 6), (2, 1, 3, 4), (3, 9, 8, 8)])
def test_manual_backward_matches_autograd_fp64(reverse, shape):
    B, T, X, H = shape
    tensors = [t.requires_grad_(True) for t in make_inputs(B, T, X, H)]
    x, h0, c0, w_x, w_h, b = tensors
    g = torch.Generator().manual_seed(1)
    gouts = [torch.randn(T, B, H, generator=g, dtype=torch.float32).double(),
             torch.randn(B, H, generator=g, dtype=torch.float32).double(),
             torch.randn(B, H, generator=g, dtype=torch.float32).double()]

    hseq_r, hT_r, cT_r = autograd_ref(x, h0, c0, w_x, w_h, b, reverse)
    loss_r = sum((o * go).sum() for o, go in zip((hseq_r, hT_r, cT_r), gouts))
    grads_r = torch.autograd.grad(loss_r, tensors)

    t2 = [t.detach().clone().requires_grad_(True) for t in tensors]
    hseq, hT, cT = lstm_layer(*t2[:6], reverse=reverse)
    assert torch.equal(hseq, hseq_r.detach()) or (hseq - hseq_r).abs().max() < 1e-14
    loss = sum((o * go).sum() for o, go in zip((hseq, hT, cT), gouts))
    grads = torch.autograd.grad(loss, t2)
    for name, a, r in zip("x h0 c0 w_x w_h b".split(), grads, grads_r):
        assert (a - r).abs().max() < 1e-11, f"d{name} mismatch"


def test_gradcheck_small():
    B, T, X, H = 2, 3, 4, 3
    tensors = [t.requires_grad_(True) for t in make_inputs(B, T, X, H, seed=3)]

    def fn(*args):
        hseq, hT, cT = lstm_layer(*args, reverse=False)
        return hseq, hT, cT

    assert torch.autograd.gradcheck(fn, tuple(tensors), eps=1e-6, atol=1e-8)


@pytest.mark.parametrize("bidi", [False, True])
def test_parity_vs_torch_nn_lstm(bidi):
    """Full-stack parity against torch.nn.LSTM (fp64): layer math, stacking,
    bidirectional concat, h0/c0 handling."""
    from lstm_tensorspark_amd.config import ModelConfig
    from lstm_tensorspark_amd.models.lstm import LSTMModel

    torch.manual_seed(0)
    B, T, X, H, L = 3, 6, 5, 4, 2
    cfg = ModelConfig(input_size=X, hidden_size=H, num_layers=L,
                      bidirectional=bidi, head="none", dropout=0.0)
    model = LSTMModel(cfg).double()
    ref = torch.nn.LSTM(X, H, num_layers=L, bidirectional=bidi,
                        batch_first=False).double()  # time-major, like us
    # copy weights: torch LSTM w_ih (4H, in) -> our W_x (in, 4H); same gate
    # order (i, f, g, o); our b = b_ih + b_hh
    with torch.no_grad():
        for l in range(L):
            for d in range(2 if bidi else 1):
                sfx = f"_reverse" if d == 1 else ""
                name = f"layer{l}{'_rev' if d == 1 else ''}"
                model.weight(f"{name}/W_x").copy_(
                    getattr(ref, f"weight_ih_l{l}{sfx}").t())
                model.weight(f"{name}/W_h").copy_(
                    getattr(ref, f"weight_hh_l{l}{sfx}").t())
                model.weight(f"{name}/b").copy_(
                    getattr(ref, f"bias_ih_l{l}{sfx}")
                    + getattr(ref, f"bias_hh_l{l}{sfx}"))
    x = torch.randn(T, B, X, dtype=torch.float64)
    out_ref, (hn, cn) = ref(x)
    out, state = model.backbone(x)
    assert (out - out_ref).abs().max() < 1e-12
    # final states: our state list is [(l0 fwd), (l0 rev), (l1 fwd), ...]
    D = 2 if bidi else 1
    for l in range(L):
        for d in range(D):
            hT, cT = state[l * D + d]
            assert (hT - hn[l * D + d]).abs().max() < 1e-12
            assert (cT - cn[l * D + d]).abs().max() < 1e-12


def test_grad_parity_vs_torch_nn_lstm():
    from lstm_tensorspark_amd.config import ModelConfig
    from lstm_tensorspark_amd.models.lstm import LSTMModel

    torch.manual_seed(0)
    B, T, X, H, L = 2, 5, 4, 3, 2
    cfg = ModelConfig(input_size=X, hidden_size=H, num_layers=L,
                      bidirectional=True, head="none")
    model = LSTMModel(cfg).double()
    ref = torch.nn.LSTM(X, H, num_layers=L, bidirectional=True,
                        batch_first=False).double()  # time-major, like us
    with torch.no_grad():
        for l in range(L):
            for d in range(2):
                sfx = "_reverse" if d == 1 else ""
                name = f"layer{l}{'_rev' if d == 1 else ''}"
                model.weight(f"{name}/W_x").copy_(getattr(ref, f"weight_ih_l{l}{sfx}").t())
                model.weight(f"{name}/W_h").copy_(getattr(ref, f"weight_hh_l{l}{sfx}").t())
                model.weight(f"{name}/b").copy_(
                    getattr(ref, f"bias_ih_l{l}{sfx}") + getattr(ref, f"bias_hh_l{l}{sfx}"))
    x = torch.randn(T, B, X, dtype=torch.float64)
    gout = torch.randn(T, B, 2 * H, dtype=torch.float64)

    out_ref, _ = ref(x)
    out_ref.backward(gout)
    out, _ = model.backbone(x)
    out.backward(gout)
    for l in range(L):
        for d in range(2):
            sfx = "_reverse" if d == 1 else ""
            name = f"layer{l}{'_rev' if d == 1 else ''}"
            assert (model.weight(f"{name}/W_x").grad
                    - getattr(ref, f"weight_ih_l{l}{sfx}").grad.t()).abs().max() < 1e-11
            assert (model.weight(f"{name}/W_h").grad
                    - getattr(ref, f"weight_hh_l{l}{sfx}").grad.t()).abs().max() < 1e-11
            assert (model.weight(f"{name}/b").grad
                    - getattr(ref, f"bias_ih_l{l}{sfx}").grad).abs().max() < 1e-11


def test_reverse_equals_flipped_forward():
    """reverse=True over x must equal reverse=False over time-flipped x,
    time-flipped back (same weights, fp64 exact)."""
    tensors = make_inputs(3, 7, 5, 6, seed=11)
    x, h0, c0, w_x, w_h, b = tensors
    h_rev, hT_r, cT_r = lstm_layer(x, h0, c0, w_x, w_h, b, reverse=True)
    h_fwd, hT_f, cT_f = lstm_layer(x.flip(0), h0, c0, w_x, w_h, b,
                                   reverse=False)
    assert torch.equal(h_rev, h_fwd.flip(0))
    assert torch.equal(hT_r, hT_f) and torch.equal(cT_r, cT_f)


def test_bptt_equals_full_when_window_covers_seq():
    """bptt == seq_len must be bitwise-identical to no truncation."""
    from lstm_tensorspark_amd.config import Config, ModelConfig, TrainConfig

    from lstm_tensorspark_amd.trainer import Trainer

    def run(bptt):
        cfg = Config(
            model=ModelConfig(input_size=8, hidden_size=8, num_classes=4),
            train=TrainConfig(batch_size=3, seq_len=6, bptt=bptt,
                              log_interval=0, seed=5))
        t = Trainer(cfg, device="cpu")
        t.train_step(t.data.batch(0))
        return {k: v.detach().clone() for k, v in t.model.weight_dict().items()}

    w_full = run(0)
    w_win = run(6)
    for k in w_full:
        assert torch.equal(w_full[k], w_win[k]), k


def test_carry_state_continuity():
    """Splitting a sequence into two halves with carried (h, c) state must
    reproduce the single-pass forward exactly (fp64)."""
    tensors = make_inputs(2, 8, 4, 5, seed=13)
    x, h0, c0, w_x, w_h, b = tensors
    h_all, hT, cT = lstm_layer(x, h0, c0, w_x, w_h, b)
    h1, hT1, cT1 = lstm_layer(x[:4], h0, c0, w_x, w_h, b)
    h2, hT2, cT2 = lstm_layer(x[4:], hT1, cT1, w_x, w_h, b)
    assert torch.equal(torch.cat([h1, h2], dim=0), h_all)
    assert torch.equal(hT2, hT) and torch.equal(cT2, cT)


def test_cpu_shape_fuzz_vs_autograd():
    """Randomized CPU fp64 fuzz: manual backward vs torch autograd on
    arbitrary (B,T,X,H), both directions."""
    import random

    rng = random.Random(7)
    for i in range(6):
        B = rng.choice([1, 2, 5])
        T = rng.choice([1, 3, 8])
        X = rng.choice([1, 4, 9])
        H = rng.choice([2, 6, 11])
        reverse = bool(i % 2)
        tensors = [t.requires_grad_(True)
                   for t in make_inputs(B, T, X, H, seed=200 + i)]
        g = torch.Generator().manual_seed(300 + i)
        gout = torch.randn(T, B, H, generator=g, dtype=torch.float32).double()

        hseq_r, _, _ = autograd_ref(*tensors, reverse)
        loss_r = (hseq_r * gout).sum()
        grads_r = torch.autograd.grad(loss_r, tensors)

        t2 = [t.detach().clone().requires_grad_(True) for t in tensors]
        hseq, _, _ = lstm_layer(*t2, reverse=reverse)
        loss = (hseq * gout).sum()
        grads = torch.autograd.grad(loss, t2)
        for name, a, r in zip("x h0 c0 w_x w_h b".split(), grads, grads_r):
            err = (a - r).abs().max().item()
            assert err < 1e-11, f"{(B,T,X,H,reverse)} d{name}: {err}"
