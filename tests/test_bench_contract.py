"""The driver depends on bench.py's exact output contract: one JSON line
from rank 0 with the required keys.  Run on CPU at plumbing scale."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_cpu_contract():
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"),
         "--cpu", "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=300, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    rec = json.loads(lines[0])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in rec, key
    assert rec["data"] == "synthetic"
    assert rec["scaling"] == "weak"
    assert rec["value"] > 0
    for key in ("model", "global_batch", "seq_len", "parallelism"):
        assert key in rec["config"], key


def test_train_cli_smoke(tmp_path):
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "train.py"),
         "--preset", "plumbing-cpu", "--set", "train.steps=2",
         "--set", "train.batch_size=4", "--set", "train.seq_len=8",
         "--set", f"train.ckpt_dir={tmp_path}", "--set", "train.log_interval=1"],
        capture_output=True, text=True, timeout=300, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    assert os.path.exists(os.path.join(str(tmp_path), "ckpt_final.pt"))


def test_train_cli_new_knobs(tmp_path):
    """accum_steps / lr_schedule / eval_interval all plumb through the
    generic --set CLI overrides into a real run."""
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "train.py"),
         "--preset", "plumbing-cpu", "--set", "train.steps=4",
         "--set", "train.batch_size=4", "--set", "train.seq_len=8",
         "--set", "train.accum_steps=2", "--set", "train.lr_schedule=cosine",
         "--set", "train.warmup_steps=2", "--set", "train.eval_interval=2",
         "--set", "train.eval_steps=1", "--set", "train.log_interval=1",
         "--set", f"train.ckpt_dir={tmp_path}"],
        capture_output=True, text=True, timeout=300, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    assert "eval_loss" in out.stdout + out.stderr
    assert os.path.exists(os.path.join(str(tmp_path), "ckpt_final.pt"))


def test_bench_dp2_contract():
    """The driver's exact multi-rank invocation shape (torchrun world 2,
    gloo on CPU): exactly ONE JSON line, from rank 0, with dp2 config and
    the whole-job aggregate value."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29617", os.path.join(REPO, "bench.py"),
         "--gpus", "2", "--steps", "2", "--warmup", "1", "--cpu",
         "--preset", "plumbing-cpu"],
        capture_output=True, text=True, timeout=600, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines()
             if l.startswith("{\"metric\"")]
    assert len(lines) == 1, out.stdout
    rec = json.loads(lines[0])
    assert rec["n_gpus"] == 2
    assert rec["config"]["parallelism"] == "dp2"
    assert rec["value"] > 0 and rec["ms_per_step"] > 0


def test_train_cli_auto_resume(tmp_path):
    """--auto-resume: a relaunched job picks up the newest checkpoint in
    ckpt_dir and continues (the failure-recovery entry point)."""
    base = ["--preset", "plumbing-cpu", "--set", "train.batch_size=4",
            "--set", "train.seq_len=8", "--set", "train.log_interval=1",
            "--set", f"train.ckpt_dir={tmp_path}"]
    r1 = subprocess.run(
        [sys.executable, os.path.join(REPO, "train.py"),
         "--set", "train.steps=2"] + base,
        capture_output=True, text=True, timeout=300, cwd=REPO)
    assert r1.returncode == 0, r1.stderr[-2000:]
    r2 = subprocess.run(
        [sys.executable, os.path.join(REPO, "train.py"),
         "--set", "train.steps=4", "--auto-resume"] + base,
        capture_output=True, text=True, timeout=300, cwd=REPO)
    assert r2.returncode == 0, r2.stderr[-2000:]
    logs = r2.stdout + r2.stderr
    assert "checkpoint loaded" in logs          # it resumed
    assert '"step": 4' in logs                  # ran 3..4 (total budget 4)
    assert '"step": 5' not in logs              # ...and STOPPED at 4
    assert '"step": 1}' not in logs and '"step": 1,' not in logs


def test_train_cli_crash_and_auto_resume_recovery(tmp_path):
    """Full recovery story: a run crashes mid-training (injected fault),
    the relaunch with --auto-resume picks up the last periodic checkpoint
    and finishes the budget."""
    base = ["--preset", "plumbing-cpu", "--set", "train.batch_size=4",
            "--set", "train.seq_len=8", "--set", "train.log_interval=1",
            "--set", "train.steps=6", "--set", "train.ckpt_interval=2",
            "--set", f"train.ckpt_dir={tmp_path}"]
    crash = subprocess.run(
        [sys.executable, os.path.join(REPO, "train.py"),
         "--set", "train.fault_step=4"] + base,
        capture_output=True, text=True, timeout=300, cwd=REPO)
    assert crash.returncode != 0                    # it really died
    assert "injected fault" in crash.stderr
    assert os.path.exists(os.path.join(str(tmp_path), "ckpt_step4.pt"))
    relaunch = subprocess.run(
        [sys.executable, os.path.join(REPO, "train.py"),
         "--auto-resume"] + base,
        capture_output=True, text=True, timeout=300, cwd=REPO)
    assert relaunch.returncode == 0, relaunch.stderr[-2000:]
    logs = relaunch.stdout + relaunch.stderr
    assert "checkpoint loaded" in logs
    assert '"step": 6' in logs                      # budget completed
    assert os.path.exists(os.path.join(str(tmp_path), "ckpt_final.pt"))
