import json

import pytest

from lstm_tensorspark_amd.config import Config, PRESETS, get_preset


def test_default_roundtrip():
    c = Config()
    d = c.to_dict()
    c2 = Config.from_dict(d)
    assert c2.to_dict() == d


def test_presets_cover_baseline_configs():
    # one preset per BASELINE.json config
    assert len(PRESETS) == 5
    p1 = get_preset("mi355x-1gpu")
    assert p1.model.hidden_size == 1024
    assert p1.model.num_layers == 2
    assert p1.train.seq_len == 128
    assert p1.train.dtype == "bf16"
    p3 = get_preset("bidir-4l")
    assert p3.model.bidirectional and p3.model.num_layers == 4
    assert p3.model.hidden_size == 2048 and p3.train.seq_len == 512
    p4 = get_preset("lm-50k")
    assert p4.model.head == "lm" and p4.model.vocab_size == 50000
    assert p4.train.bptt == 256 and p4.train.grad_clip > 0
    assert p4.model.hidden_size == 4096


def test_overrides():
    c = Config().apply_overrides(
        ["train.lr=0.5", "model.hidden_size=64", "model.bidirectional=true",
         "dist.backend=gloo"])
    assert c.train.lr == 0.5
    assert c.model.hidden_size == 64
    assert c.model.bidirectional is True
    assert c.dist.backend == "gloo"


def test_bad_override_rejected():
    with pytest.raises(ValueError):
        Config().apply_overrides(["train.nope=1"])


def test_unknown_key_rejected():
    with pytest.raises(ValueError):
        Config.from_dict({"model": {"hidden_sizee": 8}})


def test_yaml_load(tmp_path):
    p = tmp_path / "c.yaml"
    p.write_text("model:\n  hidden_size: 96\ntrain:\n  seq_len: 17\n")
    c = Config.from_yaml(str(p))
    assert c.model.hidden_size == 96 and c.train.seq_len == 17


def test_json_serializable():
    json.loads(Config().to_json())


def test_example_yamls_load():
    import glob
    import os

    from lstm_tensorspark_amd.config import Config

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    paths = sorted(glob.glob(os.path.join(root, "examples", "*.yaml")))
    assert len(paths) >= 2
    for p in paths:
        cfg = Config.from_yaml(p)
        assert cfg.train.batch_size > 0 and cfg.model.hidden_size > 0


def test_fp8_dtype_plumbing_cpu_fallback():
    import torch

    from lstm_tensorspark_amd.config import Config, ModelConfig, TrainConfig
    from lstm_tensorspark_amd.trainer import Trainer, compute_dtype

    cfg = Config(model=ModelConfig(input_size=16, hidden_size=16),
                 train=TrainConfig(batch_size=2, seq_len=4, dtype="fp8",
                                   log_interval=0))
    # on CPU, fp8 mode degrades to the fp32 reference path cleanly
    assert compute_dtype(cfg, torch.device("cpu")) == torch.float32
    t = Trainer(cfg, device="cpu")
    info = t.train_step(t.data.batch(0))
    assert info["loss"] == info["loss"]


def test_fp8_head_field_roundtrip(tmp_path):
    """The r2 train.fp8_head knob survives dict/YAML round trips and the
    CLI --set override path."""
    import yaml

    from lstm_tensorspark_amd.config import Config, ModelConfig, TrainConfig, DataConfig

    c = Config(model=ModelConfig(), train=TrainConfig(fp8_head=True),
               data=DataConfig())
    d = c.to_dict()
    assert d["train"]["fp8_head"] is True
    c2 = Config.from_dict(d)
    assert c2.train.fp8_head is True
    p = tmp_path / "c.yaml"
    p.write_text(yaml.safe_dump(d))
    c3 = Config.from_dict(yaml.safe_load(p.read_text()))
    assert c3.train.fp8_head is True


def test_example_yamls_load():
    """Every checked-in example YAML must parse into a valid Config."""
    import glob
    import os
    from lstm_tensorspark_amd.config import Config
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    paths = sorted(glob.glob(os.path.join(repo, "examples", "*.yaml")))
    assert len(paths) >= 3
    for p in paths:
        cfg = Config.from_yaml(p)
        assert cfg.model.hidden_size > 0, p
