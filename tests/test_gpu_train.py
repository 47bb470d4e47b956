"""GPU integration: single-GPU bf16 training through the fused kernels
(SURVEY.md §4.2 "Integration: single-GPU train" tier)."""
import os

import pytest
import torch

pytestmark = pytest.mark.gpu

from lstm_tensorspark_amd.config import Config, ModelConfig, TrainConfig, DataConfig
from lstm_tensorspark_amd.trainer import Trainer


def copy_cfg(**train_kw):
    kw = dict(batch_size=64, seq_len=32, lr=0.5, momentum=0.9, steps=40,
              log_interval=0, seed=7, dtype="bf16")
    kw.update(train_kw)
    return Config(
        model=ModelConfig(input_size=64, hidden_size=128, num_layers=1,
                          num_classes=8),
        train=TrainConfig(**kw),
        data=DataConfig(task="copy", copy_delay=1),
    )


def test_gpu_loss_decreases_on_copy_task():
    t = Trainer(copy_cfg(), device="cuda:0")
    first = t.train_step(t.data.batch(0))["loss"]
    t.fit(40)
    last = t.train_step(t.data.batch(41))["loss"]
    assert last < first * 0.6, (first, last)


def test_gpu_bidirectional_multilayer_step():
    cfg = copy_cfg(steps=2)
    cfg.model.bidirectional = True
    cfg.model.num_layers = 2
    cfg.model.dropout = 0.1
    t = Trainer(cfg, device="cuda:0")
    info = t.train_step(t.data.batch(0))
    assert info["loss"] == info["loss"]


def test_gpu_lm_head_adam_clip_bptt():
    cfg = Config(
        model=ModelConfig(input_size=128, hidden_size=128, num_layers=2,
                          head="lm", vocab_size=512),
        train=TrainConfig(batch_size=16, seq_len=64, bptt=32, lr=1e-3,
                          optimizer="adam", grad_clip=1.0, steps=2,
                          log_interval=0, dtype="bf16"),
        data=DataConfig(task="lm"),
    )
    t = Trainer(cfg, device="cuda:0")
    t.fit(2)
    assert t.step == 2


def test_gpu_checkpoint_resume_bitwise(tmp_path):
    cfg = copy_cfg(steps=3)
    cfg.train.ckpt_dir = str(tmp_path)
    tA = Trainer(cfg, device="cuda:0")
    tA.fit(2)
    mid = os.path.join(str(tmp_path), "mid.pt")
    tA.save(mid)
    tA.fit(2)
    tB = Trainer(copy_cfg(steps=3), device="cuda:0")
    tB.load(mid)
    tB.fit(2)
    for k, vA in tA.model.weight_dict().items():
        vB = tB.model.weight_dict()[k]
        assert torch.equal(vA, vB), f"resume not bitwise for {k}"


def test_gpu_fused_path_convergence():
    """Training convergence THROUGH the r2 fused kernels (2ph fwd +
    dh_pw_bwd bwd engage only when the grid fills the chip, so the small
    copy-task tests never exercise them end-to-end): B=4096, h=1024 on
    the copy task must learn."""
    cfg = Config(
        model=ModelConfig(input_size=64, hidden_size=1024, num_layers=2,
                          num_classes=8),
        train=TrainConfig(batch_size=4096, seq_len=16, lr=0.3, momentum=0.9,
                          steps=30, log_interval=0, seed=9, dtype="bf16"),
        data=DataConfig(task="copy", copy_delay=1),
    )
    t = Trainer(cfg, device="cuda:0")
    first = t.train_step(t.data.batch(0))["loss"]
    t.fit(30)
    last = t.train_step(t.data.batch(31))["loss"]
    # calibrated against the measured trajectory (2.08 -> 1.93 @30 steps;
    # the fused and split paths produce IDENTICAL losses to 4 decimals --
    # diagnosis run in git history): assert a solid monotone decrease,
    # not full convergence (big-batch copy learns slowly at these steps)
    assert last < first - 0.05, (first, last)


def test_graft_smoke_entry():
    import __graft_entry__

    __graft_entry__.smoke()


def test_gpu_cuda_graph_step_matches_eager():
    """Graphed steps (after the capture step) must track eager training:
    same config/data, losses finite and decreasing comparably."""
    cfg = copy_cfg(steps=8)
    cfg.train.cuda_graph = True
    tg = Trainer(cfg, device="cuda:0")
    assert tg._graph_eligible
    losses_g = [tg.train_step(tg.data.batch(i))["loss"] for i in range(8)]
    assert tg._graph is not None
    te = Trainer(copy_cfg(steps=8), device="cuda:0")
    losses_e = [te.train_step(te.data.batch(i))["loss"] for i in range(8)]
    # same trend: both fall; graphed path numerically sane
    assert all(l == l for l in losses_g)
    assert losses_g[-1] < losses_g[0]
    assert losses_e[-1] < losses_e[0]


def test_gpu_cuda_graph_step_exact_parity():
    """STEP-EXACT graph capture: the capture-step
    snapshot/restore must make the graphed weight trajectory equal eager
    per step, not just 'both decrease'.  Same kernels in the same order
    on the same data -> per-step losses must agree to bf16 noise."""
    cfg = copy_cfg(steps=6)
    cfg.train.cuda_graph = True
    tg = Trainer(cfg, device="cuda:0")
    assert tg._graph_eligible
    losses_g = [tg.train_step(tg.data.batch(i))["loss"] for i in range(6)]
    te = Trainer(copy_cfg(steps=6), device="cuda:0")
    losses_e = [te.train_step(te.data.batch(i))["loss"] for i in range(6)]
    for i, (lg, le) in enumerate(zip(losses_g, losses_e)):
        assert abs(lg - le) <= 1e-4 + 2e-3 * abs(le), (i, losses_g, losses_e)
    # weights after N steps must match eagerly-trained weights closely
    for k, vg in tg.model.weight_dict().items():
        ve = te.model.weight_dict()[k]
        assert torch.allclose(vg.float(), ve.float(), rtol=1e-2, atol=1e-3), k


def test_gpu_cuda_graph_ineligible_falls_back():
    cfg = copy_cfg(steps=2, bptt=8)   # multi-window -> ineligible
    cfg.train.cuda_graph = True
    t = Trainer(cfg, device="cuda:0")
    assert not t._graph_eligible
    info = t.train_step(t.data.batch(0))
    assert info["loss"] == info["loss"]


def test_gpu_lm_generate():
    from lstm_tensorspark_amd.config import ModelConfig
    from lstm_tensorspark_amd.models.lstm import LSTMModel

    torch.manual_seed(5)
    cfg = ModelConfig(input_size=128, hidden_size=128, num_layers=2,
                      head="lm", vocab_size=512)
    m = LSTMModel(cfg).to("cuda").bfloat16()
    prompt = torch.randint(0, 512, (8, 4), device="cuda")
    out = m.generate(prompt, steps=16, use_gemv=False)  # graph, library mm
    assert out.shape == (24, 4)
    assert torch.equal(out[:8], prompt)
    assert (out < 512).all() and (out >= 0).all()
    # graphed decode must match the eager loop token for token (greedy)
    out_eager = m.generate(prompt, steps=16, use_graph=False)
    assert torch.equal(out, out_eager)
    # gemv serving path: same API, valid tokens, deterministic
    g1 = m.generate(prompt, steps=16, use_gemv=True)
    g2 = m.generate(prompt, steps=16, use_gemv=True)
    assert torch.equal(g1, g2) and g1.shape == (24, 4)
    assert (g1 < 512).all() and (g1 >= 0).all()


def test_gpu_generate_beam_and_filters():
    """Beam search + top-k sampling on the GPU eager decode path:
    beam_size=1 reduces to greedy (same logits tensor, same tie-break),
    beams stay in-vocab and deterministic, seeded top-k is reproducible."""
    from lstm_tensorspark_amd.config import ModelConfig
    from lstm_tensorspark_amd.models.lstm import LSTMModel

    torch.manual_seed(9)
    cfg = ModelConfig(input_size=128, hidden_size=128, num_layers=2,
                      head="lm", vocab_size=512)
    m = LSTMModel(cfg).to("cuda").bfloat16()
    prompt = torch.randint(0, 512, (6, 1), device="cuda")
    seq1, s1 = m.generate_beam(prompt, steps=8, beam_size=1)
    greedy = m.generate(prompt, steps=8, use_graph=False)
    assert torch.equal(seq1, greedy)
    seq4, s4 = m.generate_beam(prompt, steps=8, beam_size=4)
    # NOTE: no s4 >= s1 assertion -- beam search may prune the greedy
    # path, so that inequality is not guaranteed in general
    assert seq4.shape == (14, 1) and s4 == s4
    assert (seq4 >= 0).all() and (seq4 < 512).all()
    seq4b, s4b = m.generate_beam(prompt, steps=8, beam_size=4)
    assert torch.equal(seq4, seq4b) and s4 == s4b
    g = torch.Generator(device="cuda").manual_seed(3)
    a = m.generate(prompt, steps=8, temperature=1.0, top_k=8, generator=g)
    g2 = torch.Generator(device="cuda").manual_seed(3)
    b = m.generate(prompt, steps=8, temperature=1.0, top_k=8, generator=g2)
    assert torch.equal(a, b)


def test_gpu_fp8_train_step():
    """fp8 mode end-to-end trainer step: finite loss, weights update."""
    cfg = copy_cfg(steps=2)
    cfg.train.dtype = "fp8"
    cfg.model.input_size = 1024
    cfg.model.hidden_size = 1024
    cfg.train.batch_size = 2048      # B*H >= 2M -> split+fp8 path
    t = Trainer(cfg, device="cuda:0")
    w0 = t.model.weight("layer0/W_h").detach().clone()
    info = t.train_step(t.data.batch(0))
    assert info["loss"] == info["loss"]
    assert not torch.equal(w0, t.model.weight("layer0/W_h"))


def test_gpu_autotune_picks_and_pins():
    from lstm_tensorspark_amd.ops.autotune import tune_fwd
    from lstm_tensorspark_amd.ops.extension import get_extension

    ext = get_extension()
    try:
        r = tune_fwd(4096, 1024, T=4, iters=3)
        assert r["choice"] in ("split", "fused")
        assert ext.get_split_fwd() in (0, 1)
    finally:
        ext.set_split_fwd(-1)


def test_gpu_continuous_batcher():
    """Serving engine on real kernels: a lone request matches solo decode
    exactly (identical path); a mixed batch produces valid, complete,
    deterministic results.  (Exact multi-request parity is asserted on
    CPU fp32 -- bf16 batched-vs-solo GEMM tie-breaks preclude it here.)"""
    from lstm_tensorspark_amd.config import ModelConfig
    from lstm_tensorspark_amd.models.lstm import LSTMModel
    from lstm_tensorspark_amd.serving import ContinuousBatcher

    torch.manual_seed(21)
    cfg = ModelConfig(input_size=128, hidden_size=128, num_layers=2,
                      head="lm", vocab_size=256)
    m = LSTMModel(cfg).to("cuda").bfloat16()

    # lone request == solo generate (same (1,1) forwards)
    eng = ContinuousBatcher(m)
    rid = eng.submit([5, 6, 7], 8)
    eng.drain()
    got = eng.result(rid, timeout=0)
    prompt = torch.tensor([5, 6, 7], device="cuda").unsqueeze(1)
    solo = m.generate(prompt, steps=8, use_graph=False)[:, 0].tolist()
    assert got == solo

    # mixed batch: valid tokens, right lengths, engine fully drains
    reqs = [([1, 2], 5), ([9], 7), ([3, 4, 5, 6], 3)]
    rids = [eng.submit(t, s) for t, s in reqs]
    eng.drain()
    for rid2, (toks, steps) in zip(rids, reqs):
        out = eng.result(rid2, timeout=0)
        assert len(out) == len(toks) + steps
        assert out[:len(toks)] == toks
        assert all(0 <= t < 256 for t in out)
    st = eng.stats()
    assert (st["active"], st["queued"], st["done_unclaimed"]) == (0, 0, 0)
