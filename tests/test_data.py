import torch

from lstm_tensorspark_amd.config import Config, ModelConfig, TrainConfig, DataConfig
from lstm_tensorspark_amd.data import SyntheticSeqData


def _cfg(task="random", **tk):
    return Config(
        model=ModelConfig(input_size=16, hidden_size=8, num_classes=10,
                          vocab_size=50 if task == "lm" else 0,
                          head="lm" if task == "lm" else "seq_classifier"),
        train=TrainConfig(batch_size=4, seq_len=12, **tk),
        data=DataConfig(task=task),
    )


def test_deterministic_per_step():
    d = SyntheticSeqData(_cfg(), rank=0)
    x1, y1 = d.batch(7)
    x2, y2 = d.batch(7)
    assert torch.equal(x1, x2) and torch.equal(y1, y2)


def test_steps_differ():
    d = SyntheticSeqData(_cfg(), rank=0)
    x1, _ = d.batch(1)
    x2, _ = d.batch(2)
    assert not torch.equal(x1, x2)


def test_ranks_disjoint():
    d0 = SyntheticSeqData(_cfg(), rank=0, world_size=2)
    d1 = SyntheticSeqData(_cfg(), rank=1, world_size=2)
    x0, _ = d0.batch(3)
    x1, _ = d1.batch(3)
    assert not torch.equal(x0, x1)


def test_copy_task_learnable_structure():
    cfg = _cfg("copy")
    cfg.data.copy_delay = 2
    d = SyntheticSeqData(cfg, rank=0)
    x, y = d.batch(0)
    assert x.shape == (12, 4, 16) and y.shape == (12, 4)  # time-major
    # target at t equals input class at t-delay
    toks = x.argmax(-1)
    assert torch.equal(y[2:], toks[:-2])
    assert (y[:2] == 0).all()


def test_lm_task_shapes():
    d = SyntheticSeqData(_cfg("lm"))
    x, y = d.batch(0)
    assert x.dtype == torch.long and x.shape == (12, 4)  # time-major
    # next-token prediction: y is x shifted by one
    assert torch.equal(x[1:], y[:-1])


def test_lm_assoc_learnable_structure():
    cfg = _cfg("lm")
    cfg.data.task = "lm_assoc"
    d = SyntheticSeqData(cfg, rank=0)
    x, y = d.batch(0)
    assert x.shape == (12, 4) and y.shape == (12, 4)
    # y follows the affine map of x (deterministic given the seed)
    a = 2 * (cfg.train.seed % 1000) + 1
    c = (cfg.train.seed // 7) % cfg.model.vocab_size
    assert torch.equal(y, (x * a + c) % cfg.model.vocab_size)
