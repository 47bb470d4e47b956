"""Typed configuration for the MI355X-native LSTM training engine.

Replaces the reference's spark-submit/argparse CLI flag set (SURVEY.md §5,
"Config / flag system" row: dataset path, hidden size, layers, epochs,
learning rate, #partitions -> here: dataclass config + YAML + CLI overrides).
The five BASELINE.json configs are checked-in presets (PRESETS below).
"""
from __future__ import annotations

import dataclasses
import json
from dataclasses import dataclass, field


@dataclass
class ModelConfig:
    input_size: int = 128
    hidden_size: int = 128
    num_layers: int = 1
    bidirectional: bool = False
    dropout: float = 0.0          # inter-layer dropout (not on recurrent path)
    head: str = "seq_classifier"  # "seq_classifier" | "classifier" | "lm" | "none"
    num_classes: int = 128        # for classifier heads
    vocab_size: int = 0           # for head == "lm" (embedding + vocab projection)

    @property
    def num_directions(self) -> int:
        return 2 if self.bidirectional else 1


@dataclass
class TrainConfig:
    batch_size: int = 32          # per-rank batch size (weak scaling)
    seq_len: int = 32
    bptt: int = 0                 # truncated-BPTT window; 0 => full backprop through seq_len
    lr: float = 0.1
    momentum: float = 0.9
    weight_decay: float = 0.0
    optimizer: str = "sgd"        # "sgd" | "adam"
    adam_betas: tuple = (0.9, 0.999)
    adam_eps: float = 1e-8
    grad_clip: float = 0.0        # global-norm clip; 0 => off
    dtype: str = "fp32"           # "fp32" | "bf16" | "fp8" (bf16 + e4m3 recurrent GEMM)
    fp8_head: bool = False        # e4m3 LM-head forward GEMM (rowwise scales,
                                  # bf16 backward); quality-verified with bf16
                                  # recurrence -- the lm-50k speed knob
    steps: int = 100
    accum_steps: int = 1          # micro-batches accumulated per optimizer
                                  # step (loss scaled 1/accum; effective
                                  # batch = batch_size*accum_steps*world)
    lr_schedule: str = ""         # "" constant | "cosine" | "linear" decay
                                  # from lr to min_lr over train.steps
    warmup_steps: int = 0         # linear LR warmup 0 -> lr first
    min_lr: float = 0.0           # decay floor for cosine/linear
    log_interval: int = 10
    eval_interval: int = 0        # run evaluate() every N steps in fit()
    eval_steps: int = 10          # batches per periodic eval
    ckpt_interval: int = 0        # 0 => only at end if ckpt_dir set
    ckpt_dir: str = ""
    seed: int = 1234
    # capture the whole train step in a hipGraph (torch.cuda.CUDAGraph) and
    # replay it: removes ALL per-step launch/dispatch overhead.  Eligible
    # when: GPU, world_size==1, optimizer=="sgd" (no host-side step-count
    # scalars), single BPTT window, dropout==0.  Ineligible configs fall
    # back to eager silently.
    cuda_graph: bool = False
    # measure split-vs-fused recurrent path for THIS shape at init instead
    # of the built-in heuristic (ops/autotune.py)
    autotune: bool = False
    # fault injection (failure-detection tests): raise on this rank at this
    # step; -1 = disabled
    fault_step: int = -1
    fault_rank: int = 0


@dataclass
class DistConfig:
    backend: str = "auto"         # "auto" | "gloo" | "nccl" (nccl IS RCCL on ROCm)
    bucket_mb: int = 16           # gradient all-reduce bucket size, sized for xGMI links
    side_dw: bool = False         # dW GEMMs on a side stream (neutral at HBM-sized B)
    reduce_dtype: str = ""        # "" native (bf16) | "fp32" wire precision for all-reduce
    bucket_order: str = "reverse" # "reverse" registration heuristic |
                                  # "measured": rebuild buckets to the
                                  # grad-arrival order observed in step 1


@dataclass
class DataConfig:
    task: str = "random"          # "random" | "copy" | "lm" | "lm_assoc" | "lm_file"
    copy_delay: int = 4           # for the learnable copy task
    path: str = ""                # lm_file: token file (.bin uint16/uint32
                                  # memmap, or raw text read byte-level)
    file_dtype: str = "uint16"    # lm_file .bin element type
    eval_frac: float = 0.0        # lm_file: reserve the corpus tail for
                                  # eval (stream=1) -- a true held-out
                                  # split; 0 = eval samples everywhere


@dataclass
class Config:
    model: ModelConfig = field(default_factory=ModelConfig)
    train: TrainConfig = field(default_factory=TrainConfig)
    dist: DistConfig = field(default_factory=DistConfig)
    data: DataConfig = field(default_factory=DataConfig)

    # ------------------------------------------------------------------ io
    def to_dict(self) -> dict:
        return dataclasses.asdict(self)

    @classmethod
    def from_dict(cls, d: dict) -> "Config":
        def sub(klass, key):
            kw = dict(d.get(key, {}) or {})
            fields = {f.name for f in dataclasses.fields(klass)}
            unknown = set(kw) - fields
            if unknown:
                raise ValueError(f"unknown {key} config keys: {sorted(unknown)}")
            if klass is TrainConfig and "adam_betas" in kw:
                kw["adam_betas"] = tuple(kw["adam_betas"])
            return klass(**kw)

        unknown = set(d) - {"model", "train", "dist", "data"}
        if unknown:
            raise ValueError(f"unknown config sections: {sorted(unknown)}")
        return cls(
            model=sub(ModelConfig, "model"),
            train=sub(TrainConfig, "train"),
            dist=sub(DistConfig, "dist"),
            data=sub(DataConfig, "data"),
        )

    def to_json(self) -> str:
        return json.dumps(self.to_dict(), indent=2)

    @classmethod
    def from_yaml(cls, path: str) -> "Config":
        import yaml

        with open(path) as f:
            return cls.from_dict(yaml.safe_load(f) or {})

    def apply_overrides(self, overrides: list) -> "Config":
        """Apply 'section.key=value' CLI overrides, returning a new Config."""
        d = self.to_dict()
        for ov in overrides:
            key, _, val = ov.partition("=")
            sec, _, name = key.partition(".")
            if sec not in d or name not in d[sec]:
                raise ValueError(f"unknown config override: {ov}")
            cur = d[sec][name]
            if isinstance(cur, bool):
                d[sec][name] = val.lower() in ("1", "true", "yes")
            elif isinstance(cur, int):
                d[sec][name] = int(val)
            elif isinstance(cur, float):
                d[sec][name] = float(val)
            elif isinstance(cur, (list, tuple)):
                d[sec][name] = json.loads(val)
            else:
                d[sec][name] = val
        return Config.from_dict(d)


# ---------------------------------------------------------------------------
# The five BASELINE.json configs as presets (BASELINE.md measurement ladder).
# ---------------------------------------------------------------------------
def _preset0() -> Config:
    """configs[0]: 1-layer LSTM h=128 seq=32 on CPU, world_size=1 (plumbing)."""
    return Config(
        model=ModelConfig(input_size=128, hidden_size=128, num_layers=1),
        train=TrainConfig(batch_size=32, seq_len=32, dtype="fp32", lr=0.05),
    )


def _preset1() -> Config:
    """configs[1]: 2-layer LSTM h=1024 seq=128 bf16 on one MI355X.

    batch_size 16384/GPU: sized for 288 GB HBM3E (~140 GB live) -- bigger
    per-step shards raise GEMM efficiency and amortize the recurrent loop
    (measured on one MI355X: 71.8k seq/s at 16384, 68.1k at 8192, 64.9k at
    4096, 41.7k at 1024) and enlarge the all-reduce buckets.
    """
    return Config(
        model=ModelConfig(input_size=1024, hidden_size=1024, num_layers=2,
                          num_classes=128),
        train=TrainConfig(batch_size=16384, seq_len=128, dtype="bf16", lr=0.05),
    )


def _preset2() -> Config:
    """configs[2]: 2-layer LSTM h=1024 seq=128, DP over RCCL/xGMI."""
    c = _preset1()
    c.dist.backend = "nccl"
    c.dist.bucket_mb = 16
    return c


def _preset3() -> Config:
    """configs[3]: 4-layer bidirectional LSTM h=2048 seq=512 DP=8."""
    # batch 1024/GPU: measured 814 seq/s vs 407 at 128 on one MI355X
    return Config(
        model=ModelConfig(input_size=2048, hidden_size=2048, num_layers=4,
                          bidirectional=True, dropout=0.1, num_classes=128),
        train=TrainConfig(batch_size=1024, seq_len=512, dtype="bf16", lr=0.05),
        dist=DistConfig(backend="nccl", bucket_mb=32),
    )


def _preset4() -> Config:
    """configs[4]: LSTM-LM h=4096 vocab=50k, truncated BPTT=256, grad-clip, DP=8."""
    return Config(
        model=ModelConfig(input_size=4096, hidden_size=4096, num_layers=2,
                          head="lm", vocab_size=50000, dropout=0.1),
        # batch 512/GPU: measured 698 seq/s vs 381 at 64 on one MI355X
        train=TrainConfig(batch_size=512, seq_len=512, bptt=256, dtype="bf16",
                          lr=1e-3, optimizer="adam", grad_clip=1.0),
        dist=DistConfig(backend="nccl", bucket_mb=32),
        data=DataConfig(task="lm"),
    )


PRESETS = {
    "plumbing-cpu": _preset0,
    "mi355x-1gpu": _preset1,
    "mi355x-dp8": _preset2,
    "bidir-4l": _preset3,
    "lm-50k": _preset4,
}


def get_preset(name: str) -> Config:
    if name not in PRESETS:
        raise KeyError(f"unknown preset {name!r}; available: {sorted(PRESETS)}")
    return PRESETS[name]()
