"""Property-based tests (hypothesis) for invariant-heavy CPU paths:
the RNG stream mixer, the sampling filters, and config serialization.
These complement the example-based suites with randomized coverage of
the full input envelopes.
"""
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from lstm_tensorspark_amd.data import mix64

_SMALL = dict(max_examples=50, deadline=None)


@given(seed=st.integers(0, 2**31 - 1), stream=st.integers(0, 3),
       rank=st.integers(0, 63), step=st.integers(0, 10**7))
@settings(**_SMALL)
def test_mix64_in_u64_range(seed, stream, rank, step):
    v = mix64(seed, stream, rank, step)
    assert 0 <= v < 2**64


@given(seed=st.integers(0, 2**20), rank=st.integers(0, 7),
       step=st.integers(0, 10**6), d_rank=st.integers(1, 7),
       d_step=st.integers(1, 10**6))
@settings(**_SMALL)
def test_mix64_rank_step_disjoint(seed, rank, step, d_rank, d_step):
    """No (rank, step) pair may alias another's stream -- the exact
    failure mode of the r1 linear mix (rank 0 step 7919 == rank 1 step 0)."""
    base = mix64(seed, 0, rank, step)
    assert base != mix64(seed, 0, rank + d_rank, step)
    assert base != mix64(seed, 0, rank, step + d_step)
    assert base != mix64(seed, 1, rank, step)  # train vs eval stream


@given(order=st.permutations([11, 22, 33]))
@settings(**_SMALL)
def test_mix64_order_sensitive(order):
    """The mix must absorb position, not just values (a commutative mix
    would alias (rank=a, step=b) with (rank=b, step=a))."""
    if tuple(order) != (11, 22, 33):
        assert mix64(*order) != mix64(11, 22, 33)


@given(vocab=st.integers(4, 200), k=st.integers(1, 200),
       seed=st.integers(0, 1000))
@settings(**_SMALL)
def test_sample_top_k_support(vocab, k, seed):
    """Sampled tokens always lie inside the top-k logit set; top_k >= V
    degrades to plain sampling without error."""
    from lstm_tensorspark_amd.models.lstm import LSTMModel
    g = torch.Generator().manual_seed(seed)
    logits = torch.randn(3, vocab, generator=g)
    tok = LSTMModel._sample(logits, temperature=1.0, generator=g, top_k=k)
    assert tok.shape == (3,)
    kk = min(k, vocab)
    topk = logits.topk(kk, dim=-1).indices
    assert (tok.unsqueeze(-1) == topk).any(-1).all()


@given(vocab=st.integers(4, 200), p=st.floats(0.05, 0.999),
       seed=st.integers(0, 1000))
@settings(**_SMALL)
def test_sample_top_p_support(vocab, p, seed):
    """Nucleus sampling only emits tokens from the smallest prefix of the
    sorted distribution whose mass reaches p (plus the boundary token)."""
    from lstm_tensorspark_amd.models.lstm import LSTMModel
    g = torch.Generator().manual_seed(seed)
    logits = torch.randn(2, vocab, generator=g)
    tok = LSTMModel._sample(logits, temperature=1.0, generator=g, top_p=p)
    probs = torch.softmax(logits, dim=-1)
    srt, idx = probs.sort(dim=-1, descending=True)
    cum = srt.cumsum(-1)
    for r in range(2):
        nucleus_n = int((cum[r] - srt[r] < p).sum())  # tokens kept
        assert (tok[r] == idx[r, :nucleus_n]).any()


@given(seed=st.integers(0, 2**31 - 1))
@settings(max_examples=20, deadline=None)
def test_lm_assoc_multiplier_always_bijective(seed):
    """For any config seed the lm_assoc affine map must be a bijection on
    Z_V (gcd(a, V) == 1) for awkward V too."""
    import math
    for V in (50000, 65536, 99, 30030):
        a = 2 * (seed % 1000) + 1
        while math.gcd(a, V) != 1:
            a += 2
        assert math.gcd(a, V) == 1
        assert len({(a * t) % V for t in range(min(V, 512))}) == min(V, 512)


@given(hidden=st.integers(8, 64), layers=st.integers(1, 3),
       bptt=st.integers(0, 16), clip=st.floats(0.0, 10.0))
@settings(max_examples=15, deadline=None)
def test_config_dict_roundtrip(hidden, layers, bptt, clip):
    from lstm_tensorspark_amd.config import Config, ModelConfig, TrainConfig
    cfg = Config(model=ModelConfig(input_size=8, hidden_size=hidden,
                                   num_layers=layers),
                 train=TrainConfig(bptt=bptt, grad_clip=clip))
    back = Config.from_dict(cfg.to_dict())
    assert back.to_dict() == cfg.to_dict()


@given(n=st.integers(10, 5000), t=st.integers(1, 64), b=st.integers(1, 8),
       step=st.integers(0, 1000), rank=st.integers(0, 7))
@settings(max_examples=25, deadline=None)
def test_lm_file_windows_always_in_bounds(tmp_path_factory, n, t, b, step,
                                          rank):
    """Any corpus length >= T+1 yields valid next-token windows for any
    (rank, step); shorter corpora raise cleanly."""
    import numpy as np
    from lstm_tensorspark_amd.config import (Config, ModelConfig,
                                             TrainConfig, DataConfig)
    from lstm_tensorspark_amd.data import SyntheticSeqData
    d = tmp_path_factory.mktemp("lmf")
    path = str(d / "c.bin")
    np.random.default_rng(n).integers(0, 250, n).astype(
        np.uint16).tofile(path)
    cfg = Config(
        model=ModelConfig(input_size=8, hidden_size=8, num_layers=1,
                          head="lm", vocab_size=256),
        train=TrainConfig(batch_size=b, seq_len=t, seed=7),
        data=DataConfig(task="lm_file", path=path),
    )
    src = SyntheticSeqData(cfg, rank=rank)
    if n < t + 1:
        import pytest
        with pytest.raises(ValueError):
            src.batch(step)
        return
    x, y = src.batch(step)
    assert x.shape == (t, b) and y.shape == (t, b)
    assert 0 <= int(x.min()) and int(x.max()) < 256
    assert torch.equal(x[1:], y[:-1])


@given(lr=st.floats(1e-4, 10.0), min_lr_frac=st.floats(0.0, 0.9),
       warmup=st.integers(0, 50), total=st.integers(1, 500),
       kind=st.sampled_from(["cosine", "linear"]))
@settings(max_examples=40, deadline=None)
def test_lr_schedule_invariants(lr, min_lr_frac, warmup, total, kind):
    """Any schedule config: values stay in [min(min_lr, warmup ramp), lr],
    warmup is non-decreasing, decay is non-increasing, and the budget
    boundary lands exactly on min_lr."""
    from lstm_tensorspark_amd.config import Config, ModelConfig, TrainConfig
    from lstm_tensorspark_amd.trainer import Trainer
    min_lr = lr * min_lr_frac
    cfg = Config(model=ModelConfig(input_size=8, hidden_size=8,
                                   num_classes=4),
                 train=TrainConfig(batch_size=2, seq_len=4, lr=lr,
                                   min_lr=min_lr, warmup_steps=warmup,
                                   lr_schedule=kind, steps=total,
                                   log_interval=0))
    t = Trainer(cfg)
    vals = [t._lr_at(s) for s in range(0, max(total, warmup) + 2)]
    assert all(0.0 <= v <= lr + 1e-12 for v in vals)
    for a, b in zip(vals[:warmup], vals[1:warmup]):
        assert a <= b + 1e-12                      # warmup non-decreasing
    decay = vals[warmup:]
    for a, b in zip(decay, decay[1:]):
        assert b <= a + 1e-9 * lr                  # decay non-increasing
    if total > warmup:
        assert abs(t._lr_at(total) - min_lr) <= 1e-9 * max(lr, 1.0)


@given(accum=st.integers(1, 4), bptt=st.sampled_from([0, 4, 8]),
       seed=st.integers(0, 100))
@settings(max_examples=12, deadline=None)
def test_accum_bptt_grad_scale_invariant(accum, bptt, seed):
    """For any (accum_steps, bptt) combination the accumulated grad must
    equal the mean of the per-micro-batch full-window grads (the 1/n and
    1/windows scalings must compose correctly)."""
    from lstm_tensorspark_amd.config import (Config, ModelConfig,
                                             TrainConfig, DataConfig)
    from lstm_tensorspark_amd.trainer import Trainer
    def cfg():
        return Config(
            model=ModelConfig(input_size=6, hidden_size=8, num_classes=5),
            train=TrainConfig(batch_size=3, seq_len=8, bptt=bptt,
                              momentum=0.0, lr=0.1, seed=seed,
                              log_interval=0),
            data=DataConfig(task="copy"),
        )
    tA = Trainer(cfg())
    batches = [tA.data.batch(i) for i in range(accum)]
    tA.train_step(batches if accum > 1 else batches[0])
    gA = {n: p.grad.detach().clone()
          for n, p in zip(tA.model.weight_dict(), tA.model.parameters())}

    tB = Trainer(cfg())
    acc = {}
    for b in batches:
        tB.optimizer.zero_grad()
        tB.train_step(b)                      # full-window single batch
        for n, p in zip(tB.model.weight_dict(), tB.model.parameters()):
            acc[n] = acc.get(n, 0) + p.grad.detach().clone() / accum
        # undo the weight update so every micro-batch sees the same weights
        tB2 = Trainer(cfg())
        for p, q in zip(tB.model.parameters(), tB2.model.parameters()):
            with torch.no_grad():
                p.copy_(q)
        tB.optimizer.sync_masters()
    for n in gA:
        assert torch.allclose(gA[n], acc[n], atol=2e-6, rtol=1e-4), n
