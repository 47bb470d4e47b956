import math

import torch

from lstm_tensorspark_amd.config import Config, ModelConfig, TrainConfig, DataConfig
from lstm_tensorspark_amd.data import SyntheticSeqData, mix64


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
    V = cfg.model.vocab_size
    a = 2 * (cfg.train.seed % 1000) + 1
    while math.gcd(a, V) != 1:
        a += 2
    c = (cfg.train.seed // 7) % V
    assert torch.equal(y, (x * a + c) % V)


def test_lm_assoc_map_bijective_nonpow2_vocab():
    # regression: an odd multiplier is only bijective mod power-of-two V;
    # the generator must pick a coprime to V (e.g. V=50000, seed giving
    # a % 5 == 0 previously produced a non-bijective map)
    cfg = _cfg("lm")
    cfg.data.task = "lm_assoc"
    cfg.model.vocab_size = 50
    cfg.train.seed = 2  # 2*(2%1000)+1 = 5 -> gcd(5, 50) = 5, must be fixed up
    d = SyntheticSeqData(cfg, rank=0)
    x, y = d.batch(0)
    a = 5
    while math.gcd(a, 50) != 1:
        a += 2
    assert math.gcd(a, 50) == 1
    c = (2 // 7) % 50
    assert torch.equal(y, (x * a + c) % 50)


def test_rng_streams_collision_free():
    """The (seed, stream, rank, step) -> generator-seed map must be
    injective over the operating envelope (ranks <= 8, long trainings,
    train + eval streams).  The old linear mix collided at step spans
    > 7919."""
    seed = 1234
    seen = set()
    n = 0
    steps = list(range(3000)) + [7919, 100_000, 1_000_000, 10_000_000]
    for rank in range(8):
        for stream in (0, 1):
            for step in steps:
                seen.add(mix64(seed, stream, rank, step))
                n += 1
    assert len(seen) == n
    # the exact collision from VERDICT: rank 0 step 7919 vs rank 1 step 0
    assert mix64(seed, 0, 0, 7919) != mix64(seed, 0, 1, 0)


def test_eval_stream_disjoint_from_train():
    d = SyntheticSeqData(_cfg(), rank=0)
    xt, _ = d.batch(5, stream=0)
    xe, _ = d.batch(5, stream=1)
    assert not torch.equal(xt, xe)


# ---------------------------------------------------------------- lm_file


def _file_cfg(path, vocab=300, file_dtype="uint16", B=4, T=8):
    from lstm_tensorspark_amd.config import (Config, ModelConfig,
                                             TrainConfig, DataConfig)
    return Config(
        model=ModelConfig(input_size=16, hidden_size=16, num_layers=1,
                          head="lm", vocab_size=vocab),
        train=TrainConfig(batch_size=B, seq_len=T, seed=11, steps=2,
                          log_interval=0),
        data=DataConfig(task="lm_file", path=path, file_dtype=file_dtype),
    )


def test_lm_file_bin_deterministic_and_shifted(tmp_path):
    import numpy as np
    from lstm_tensorspark_amd.data import SyntheticSeqData
    toks = np.random.default_rng(0).integers(0, 300, 4096).astype(np.uint16)
    p = str(tmp_path / "corpus.bin")
    toks.tofile(p)
    d = SyntheticSeqData(_file_cfg(p))
    x, y = d.batch(3)
    assert x.shape == (8, 4) and y.shape == (8, 4) and x.dtype == torch.long
    assert torch.equal(x[1:], y[:-1])          # next-token shift
    x2, y2 = d.batch(3)
    assert torch.equal(x, x2) and torch.equal(y, y2)   # resumable
    # windows really come from the file
    flat = torch.from_numpy(toks.astype("int64"))
    col = torch.cat([x[:, 0], y[-1:, 0]])
    starts = (flat.unfold(0, 9, 1) == col).all(-1)
    assert bool(starts.any())


def test_lm_file_ranks_disjoint_streams(tmp_path):
    import numpy as np
    from lstm_tensorspark_amd.data import SyntheticSeqData
    np.random.default_rng(1).integers(0, 300, 4096).astype(
        np.uint16).tofile(str(tmp_path / "c.bin"))
    cfg = _file_cfg(str(tmp_path / "c.bin"))
    d0 = SyntheticSeqData(cfg, rank=0, world_size=2)
    d1 = SyntheticSeqData(cfg, rank=1, world_size=2)
    x0, _ = d0.batch(0)
    x1, _ = d1.batch(0)
    assert not torch.equal(x0, x1)


def test_lm_file_text_byte_level(tmp_path):
    from lstm_tensorspark_amd.data import SyntheticSeqData
    p = str(tmp_path / "corpus.txt")
    with open(p, "w") as f:
        f.write("the quick brown fox jumps over the lazy dog " * 40)
    d = SyntheticSeqData(_file_cfg(p, vocab=256))
    x, y = d.batch(0)
    assert x.max() < 256 and torch.equal(x[1:], y[:-1])


def test_lm_file_trains(tmp_path):
    import numpy as np
    from lstm_tensorspark_amd.trainer import Trainer
    np.random.default_rng(2).integers(0, 300, 2048).astype(
        np.uint16).tofile(str(tmp_path / "c.bin"))
    t = Trainer(_file_cfg(str(tmp_path / "c.bin")))
    t.fit(2)
    assert t.step == 2


def test_lm_file_errors(tmp_path):
    import numpy as np
    import pytest
    from lstm_tensorspark_amd.data import SyntheticSeqData
    # out-of-range token vs vocab
    np.array([5, 999], dtype=np.uint16).tofile(str(tmp_path / "bad.bin"))
    with pytest.raises(ValueError, match="out of range"):
        SyntheticSeqData(_file_cfg(str(tmp_path / "bad.bin"),
                                   vocab=100)).batch(0)
    # corpus shorter than a window
    np.arange(4, dtype=np.uint16).tofile(str(tmp_path / "short.bin"))
    with pytest.raises(ValueError, match="tokens"):
        SyntheticSeqData(_file_cfg(str(tmp_path / "short.bin"))).batch(0)
    with pytest.raises(ValueError, match="requires data.path"):
        SyntheticSeqData(_file_cfg("")).batch(0)


def test_lm_file_eval_stream_disjoint(tmp_path):
    """evaluate()'s stream=1 draws different windows than training
    stream=0 at the same step for file-backed data too."""
    import numpy as np
    from lstm_tensorspark_amd.data import SyntheticSeqData
    np.random.default_rng(3).integers(0, 300, 8192).astype(
        np.uint16).tofile(str(tmp_path / "c.bin"))
    d = SyntheticSeqData(_file_cfg(str(tmp_path / "c.bin")))
    x_tr, _ = d.batch(5, stream=0)
    x_ev, _ = d.batch(5, stream=1)
    assert not torch.equal(x_tr, x_ev)


def test_lm_file_eval_frac_held_out(tmp_path):
    """With eval_frac, train windows never touch the corpus tail and eval
    windows never leave it."""
    import numpy as np
    from lstm_tensorspark_amd.data import SyntheticSeqData
    N, T = 1000, 8
    toks = np.arange(N, dtype=np.uint16)  # position-unique tokens
    p = str(tmp_path / "c.bin")
    toks.tofile(p)
    cfg = _file_cfg(p, vocab=1200, T=T)
    cfg.data.eval_frac = 0.2
    d = SyntheticSeqData(cfg)
    split = int(N * 0.8)
    flat = torch.from_numpy(toks.astype("int64"))
    for step in range(20):
        x, y = d.batch(step, stream=0)
        xe, ye = d.batch(step, stream=1)
        # recover window start positions from the identifiable corpus
        for b in range(x.shape[1]):
            col = torch.cat([x[:, b], y[-1:, b]])
            starts = (flat.unfold(0, T + 1, 1) == col).all(-1).nonzero()
            assert all(int(s) + T < split for s in starts), "train hit tail"
            cole = torch.cat([xe[:, b], ye[-1:, b]])
            se = (flat.unfold(0, T + 1, 1) == cole).all(-1).nonzero()
            assert any(int(s) >= split for s in se), "eval not in tail"
