"""Checkpoint layout + bitwise save->load->resume (SURVEY.md §4.2 plumbing)."""
import os

import torch

from lstm_tensorspark_amd.config import Config, ModelConfig, TrainConfig, DataConfig
from lstm_tensorspark_amd.checkpoint import load_checkpoint, save_checkpoint
from lstm_tensorspark_amd.trainer import Trainer


def small_cfg(tmp, **train_kw):
    kw = dict(batch_size=4, seq_len=8, lr=0.05, momentum=0.9, steps=4,
              log_interval=0, ckpt_dir=str(tmp))
    kw.update(train_kw)
    return Config(
        model=ModelConfig(input_size=12, hidden_size=10, num_layers=2,
                          num_classes=7),
        train=TrainConfig(**kw),
        data=DataConfig(task="copy"),
    )


def test_layout_flat_named_weight_dict(tmp_path):
    cfg = small_cfg(tmp_path)
    t = Trainer(cfg)
    path = os.path.join(str(tmp_path), "c.pt")
    save_checkpoint(path, t.model, t.optimizer, step=3, config=cfg.to_dict())
    payload = torch.load(path, map_location="cpu", weights_only=False)
    keys = set(payload["weights"])
    assert {"layer0/W_x", "layer0/W_h", "layer0/b",
            "layer1/W_x", "layer1/W_h", "layer1/b",
            "head/W", "head/b"} == keys
    assert payload["step"] == 3
    assert payload["optimizer"] is not None
    assert "rng" in payload and "torch" in payload["rng"]
    assert payload["config"]["model"]["hidden_size"] == 10


def test_bitwise_resume(tmp_path):
    # Run A: 3 steps, save, 3 more steps.
    cfg = small_cfg(tmp_path)
    tA = Trainer(cfg)
    tA.fit(3)
    mid = os.path.join(str(tmp_path), "mid.pt")
    tA.save(mid)
    tA.fit(3)
    wA = {k: v.detach().clone() for k, v in tA.model.weight_dict().items()}

    # Run B: fresh trainer, load the mid checkpoint, 3 steps.
    tB = Trainer(small_cfg(tmp_path))
    tB.load(mid)
    assert tB.step == 3
    tB.fit(3)
    for k, vA in wA.items():
        vB = tB.model.weight_dict()[k]
        assert torch.equal(vA, vB), f"resume not bitwise for {k}"
    # optimizer masters identical too
    for mA, mB in zip(tA.optimizer.masters, tB.optimizer.masters):
        assert torch.equal(mA, mB)


def test_bitwise_resume_adam_bptt(tmp_path):
    cfg = small_cfg(tmp_path, optimizer="adam", lr=1e-3, bptt=3, seq_len=8,
                    grad_clip=1.0)
    tA = Trainer(cfg)
    tA.fit(2)
    mid = os.path.join(str(tmp_path), "mid.pt")
    tA.save(mid)
    tA.fit(2)
    tB = Trainer(small_cfg(tmp_path, optimizer="adam", lr=1e-3, bptt=3,
                           seq_len=8, grad_clip=1.0))
    tB.load(mid)
    tB.fit(2)
    for k, vA in tA.model.weight_dict().items():
        assert torch.equal(vA, tB.model.weight_dict()[k]), k


def test_mismatched_names_rejected(tmp_path):
    cfg = small_cfg(tmp_path)
    t = Trainer(cfg)
    p = os.path.join(str(tmp_path), "c.pt")
    save_checkpoint(p, t.model)
    cfg2 = small_cfg(tmp_path)
    cfg2.model.num_layers = 1
    t2 = Trainer(cfg2)
    import pytest
    with pytest.raises(ValueError):
        load_checkpoint(p, t2.model)


def test_fault_injection_and_auto_resume(tmp_path):
    """Failure-detection tier: a rank fault aborts the run; a relaunched
    trainer auto-resumes from the newest checkpoint and reproduces the
    uninterrupted run bitwise."""
    import pytest

    # uninterrupted oracle
    cfgA = small_cfg(tmp_path / "a", ckpt_interval=2, steps=6)
    tA = Trainer(cfgA)
    tA.fit(6)

    # faulting run: dies at step 5 (after the step-4 checkpoint)
    cfgB = small_cfg(tmp_path / "b", ckpt_interval=2, steps=6)
    cfgB.train.fault_step = 5
    tB = Trainer(cfgB)
    with pytest.raises(RuntimeError, match="injected fault"):
        tB.fit(6)
    assert tB.step == 5  # 5 steps completed (fault raised entering step 6)

    # relaunch + auto-resume from latest checkpoint, finish the job
    cfgC = small_cfg(tmp_path / "b", ckpt_interval=2, steps=6)
    tC = Trainer(cfgC)
    assert tC.resume_latest()
    tC.fit(6 - tC.step)
    for k, vA in tA.model.weight_dict().items():
        assert torch.equal(vA, tC.model.weight_dict()[k]), k


def test_safetensors_export_roundtrip(tmp_path):
    from lstm_tensorspark_amd.checkpoint import (
        export_weights_safetensors, load_weights_safetensors)
    cfg = small_cfg(tmp_path)
    tA = Trainer(cfg)
    tA.fit(2)
    path = os.path.join(str(tmp_path), "w.safetensors")
    export_weights_safetensors(path, tA.model, metadata={"step": 2})
    tB = Trainer(small_cfg(tmp_path))
    meta = load_weights_safetensors(path, tB.model)
    assert meta["step"] == "2" and meta["format"] == "lstm_tensorspark_amd"
    for k, vA in tA.model.weight_dict().items():
        assert torch.equal(vA, tB.model.weight_dict()[k]), k
    # strict name check: wrong architecture is rejected
    bad = Trainer(Config(
        model=ModelConfig(input_size=12, hidden_size=10, num_layers=3,
                          num_classes=7),
        train=TrainConfig(batch_size=4, seq_len=8, steps=1, log_interval=0,
                          ckpt_dir=str(tmp_path)),
        data=DataConfig(task="copy"),
    ))
    try:
        load_weights_safetensors(path, bad.model)
        assert False, "expected mismatch error"
    except ValueError as e:
        assert "mismatch" in str(e)


def test_ckpt_to_safetensors_tool(tmp_path):
    """End-to-end deployment pipeline: train -> .pt checkpoint ->
    converter tool -> .safetensors -> serve.load_model."""
    import argparse
    from tools.ckpt_to_safetensors import convert
    import examples.serve as serve

    cfg = Config(
        model=ModelConfig(input_size=16, hidden_size=16, num_layers=1,
                          head="lm", vocab_size=32),
        train=TrainConfig(batch_size=4, seq_len=8, steps=2, log_interval=0,
                          ckpt_dir=str(tmp_path)),
        data=DataConfig(task="lm"),
    )
    t = Trainer(cfg)
    t.fit(2)
    src = os.path.join(str(tmp_path), "c.pt")
    save_checkpoint(src, t.model, t.optimizer, step=2, config=cfg.to_dict())
    dst = os.path.join(str(tmp_path), "m.safetensors")
    meta = convert(src, dst)
    assert meta["step"] == "2"
    args = argparse.Namespace(demo=False, ckpt=dst, hidden=1, vocab=1)
    model, _ = serve.load_model(args)
    assert model.cfg.vocab_size == 32
    for k, v in t.model.weight_dict().items():
        assert torch.equal(v, model.weight_dict()[k]), k


def test_load_checkpoint_from_older_config_schema(tmp_path):
    """A checkpoint saved before accum/schedule/eval/data-file fields
    existed must still load (dataclass defaults fill absent keys)."""
    cfg = small_cfg(tmp_path)
    t = Trainer(cfg)
    t.fit(2)
    path = os.path.join(str(tmp_path), "old.pt")
    save_checkpoint(path, t.model, t.optimizer, step=2, config=cfg.to_dict())
    payload = torch.load(path, map_location="cpu", weights_only=False)
    for k in ("accum_steps", "lr_schedule", "warmup_steps", "min_lr",
              "eval_interval", "eval_steps"):
        payload["config"]["train"].pop(k, None)
    for k in ("path", "file_dtype", "eval_frac"):
        payload["config"]["data"].pop(k, None)
    torch.save(payload, path)
    t2 = Trainer(small_cfg(tmp_path))
    t2.load(path)
    assert t2.step == 2
    from lstm_tensorspark_amd.config import Config
    back = Config.from_dict(torch.load(path, map_location="cpu",
                                       weights_only=False)["config"])
    assert back.train.accum_steps == 1 and back.data.eval_frac == 0.0


def test_trainer_load_weights_finetune(tmp_path):
    """Fine-tune entry: load a safetensors artifact into a fresh Trainer --
    weights land, fp32 masters re-sync (first step must move FROM the
    loaded weights, not snap back to stale masters), step stays 0."""
    from lstm_tensorspark_amd.checkpoint import export_weights_safetensors

    cfgA = small_cfg(tmp_path)
    tA = Trainer(cfgA)
    tA.fit(3)
    path = os.path.join(str(tmp_path), "w.safetensors")
    export_weights_safetensors(path, tA.model, metadata={"step": 3})

    tB = Trainer(small_cfg(tmp_path))
    meta = tB.load_weights(path)
    assert meta["step"] == "3" and tB.step == 0
    for k, v in tA.model.weight_dict().items():
        assert torch.equal(v, tB.model.weight_dict()[k]), k
    # masters were re-synced: one step from identical grads moves both
    # trainers identically
    wd0 = {k: v.detach().clone() for k, v in tB.model.weight_dict().items()}
    tB.train_step(tB.data.batch(0))
    moved = any(not torch.equal(wd0[k], v)
                for k, v in tB.model.weight_dict().items())
    assert moved
    for m, p in zip(tB.optimizer.masters, tB.optimizer.params):
        assert torch.allclose(m, p.detach().float(), atol=1e-6)
