"""Synthetic sequence data, generated per-rank from a rank-seeded RNG.

Replaces the reference's Spark RDD dataset handling (SURVEY.md §2.1 "Dataset
handling": file reader + sc.parallelize into partitions). There is no
network in this environment, so data is synthetic by contract
(BASELINE.json: "synthetic sequence data / random-init weights"); the
per-rank generator plays the role of one RDD partition: deterministic,
disjoint-by-seed, and resumable (the generator state is (seed, step)).

`task="lm_file"` additionally covers the reference's actual file-reader
path for user-supplied corpora: a memmapped token file sampled by
rank-seeded random windows (see `batch`), so the same determinism /
disjointness / resume contract holds for real data too.
"""
from __future__ import annotations

import torch

from .config import Config

_M64 = (1 << 64) - 1


def mix64(*vals: int) -> int:
    """Collision-resistant 64-bit mix of an integer tuple (splitmix64
    absorb/finalize chain).  Used to derive RNG stream seeds from
    (seed, stream, rank, step) so no two streams ever coincide -- the old
    linear mix ``seed*1000003 + rank*7919 + step`` collided whenever step
    spans exceeded 7919 (rank 0 step 7919 == rank 1 step 0)."""
    x = 0
    for v in vals:
        x = (x + (v & _M64) + 0x9E3779B97F4A7C15) & _M64
        z = x
        z = ((z ^ (z >> 30)) * 0xBF58476D1CE4E5B9) & _M64
        z = ((z ^ (z >> 27)) * 0x94D049BB133111EB) & _M64
        x = z ^ (z >> 31)
    return x


class SyntheticSeqData:
    """Deterministic per-rank synthetic batches.

    Each (rank, step) pair maps to a unique RNG stream so that
    - two ranks never see the same batch (data parallelism is real work),
    - resume from checkpoint regenerates the identical batch sequence,
    - tests can reproduce any batch in isolation.
    """

    def __init__(self, cfg: Config, rank: int = 0, world_size: int = 1,
                 device: str = "cpu", dtype: torch.dtype = torch.float32):
        self.cfg = cfg
        self.rank = rank
        self.world_size = world_size
        self.device = torch.device(device)
        self.dtype = dtype
        self.task = cfg.data.task
        m, t = cfg.model, cfg.train
        self.batch_size = t.batch_size
        self.seq_len = t.seq_len
        self.input_size = m.input_size
        self.base_seed = t.seed

    def _gen(self, step: int, stream: int = 0) -> torch.Generator:
        # On GPU, generate device-side: an HBM-sized batch (e.g. 16384x128
        # sequences) takes ~10 s to randn on the host but <10 ms on-device.
        # Determinism contract holds per (seed, stream, rank, step,
        # device-kind).  stream 0 = training, 1 = eval -- structurally
        # disjoint, and mix64 is collision-free over the whole tuple (the
        # old linear mix collided across (rank, step) pairs;
        # tests/test_data.py asserts disjointness).
        dev = self.device if self.device.type == "cuda" else torch.device("cpu")
        g = torch.Generator(device=dev)
        g.manual_seed(mix64(self.base_seed, stream, self.rank, step))
        return g

    def batch(self, step: int, stream: int = 0):
        """Return (x, y) for a training step (TIME-MAJOR).

        x: (T, B, input_size) float for feature tasks, (T, B) long for "lm".
        y: per-head target -- (T, B) for per-step heads, (B,) for
           "classifier" -- see models/heads.
        """
        g = self._gen(step, stream)
        dev = g.device
        B, T = self.batch_size, self.seq_len
        task = self.task
        if task == "random":
            x = torch.randn(T, B, self.input_size, generator=g, device=dev)
            head = self.cfg.model.head
            if head == "classifier":
                y = torch.randint(0, self.cfg.model.num_classes, (B,),
                                  generator=g, device=dev)
            elif head == "seq_classifier":
                y = torch.randint(0, self.cfg.model.num_classes, (T, B),
                                  generator=g, device=dev)
            else:
                y = torch.randn(T, B, self.cfg.model.hidden_size, generator=g,
                                device=dev)
            return (x.to(self.device, self.dtype), y.to(self.device))
        if task == "copy":
            # Learnable task: predict the input class seen `delay` steps ago.
            # Input: one-hot of a random class sequence; target: the shifted
            # sequence (first `delay` targets are class 0 = "blank").
            C = self.cfg.model.num_classes
            d = self.cfg.data.copy_delay
            toks = torch.randint(1, min(C, self.input_size), (T, B),
                                 generator=g, device=dev)
            x = torch.zeros(T, B, self.input_size, device=dev)
            x.scatter_(2, toks.unsqueeze(-1), 1.0)
            y = torch.zeros(T, B, dtype=torch.long, device=dev)
            if d < T:
                y[d:] = toks[: T - d]
            return (x.to(self.device, self.dtype), y.to(self.device))
        if task == "lm":
            V = self.cfg.model.vocab_size
            assert V > 1, "lm task requires model.vocab_size"
            stream = torch.randint(0, V, (T + 1, B), generator=g, device=dev)
            x, y = stream[:-1].contiguous(), stream[1:].contiguous()
            return (x.to(self.device), y.to(self.device))
        if task == "lm_assoc":
            # LEARNABLE next-token task: the sequence follows a fixed affine
            # map  tok_{t+1} = (a*tok_t + c) mod V  (a,c from the config
            # seed), so a capable model drives the loss from ln(V) toward 0.
            # The plain "lm" task is uniform random (loss pinned at ln V) --
            # right for throughput, useless for convergence checks.
            V = self.cfg.model.vocab_size
            assert V > 1, "lm_assoc requires model.vocab_size"
            # a must be coprime to V for tok -> a*tok+c (mod V) to be a
            # bijection (odd alone only suffices for power-of-two V; for
            # V=50000 an a divisible by 5 is non-bijective)
            import math
            a = 2 * (self.base_seed % 1000) + 1
            while math.gcd(a, V) != 1:
                a += 2
            c = (self.base_seed // 7) % V
            t0 = torch.randint(0, V, (1, B), generator=g, device=dev)
            toks = [t0]
            for _ in range(T):
                toks.append((toks[-1] * a + c) % V)
            stream = torch.cat(toks, dim=0)
            x, y = stream[:-1].contiguous(), stream[1:].contiguous()
            return (x.to(self.device), y.to(self.device))
        if task == "lm_file":
            # File-backed LM corpus (the reference's "file reader +
            # sc.parallelize" dataset row): a token file is
            # memmapped once and each (rank, step) draws B random (T+1)
            # windows from it through the same mix64 stream machinery --
            # deterministic, resumable, and disjoint across ranks without
            # any physical sharding (random-offset sampling IS the
            # partitioning; no driver round-trip, no data shipping).
            arr = self._file_tokens()
            N = arr.shape[0]
            # held-out split: train windows start in [0, split-T), eval
            # (stream=1) windows in [split, N-T) -- windows cannot cross
            # the boundary, so the eval tail is never trained on
            f = getattr(self.cfg.data, "eval_frac", 0.0)
            lo, hi = 0, N - T
            if 0.0 < f < 1.0:
                split = int(N * (1.0 - f))
                lo, hi = (split, N - T) if stream == 1 else (0, split - T)
            if hi <= lo:
                raise ValueError(
                    f"lm_file corpus has {N} tokens: stream {stream} split "
                    f"[{lo}, {hi}) has no room for seq_len+1={T + 1} "
                    f"windows (eval_frac={f})")
            # offsets on CPU (memmap gather is a host op), batch to device
            gc = torch.Generator()
            gc.manual_seed(mix64(self.base_seed, stream, self.rank, step))
            off = torch.randint(lo, hi, (B,), generator=gc)
            idx = (off.unsqueeze(0)
                   + torch.arange(T + 1).unsqueeze(1)).numpy()  # (T+1, B)
            import numpy as np
            win = torch.from_numpy(
                arr[idx.reshape(-1)].astype(np.int64)).reshape(T + 1, B)
            x, y = win[:-1].contiguous(), win[1:].contiguous()
            return (x.to(self.device), y.to(self.device))
        raise ValueError(f"unknown data task {task!r}")

    def _file_tokens(self):
        """Memmap (or byte-read) the lm_file corpus once; cached.  Returns
        a numpy array -- a .bin stays an mmap so batch gathers touch only
        the sampled pages (a multi-GB corpus never fully loads)."""
        cached = getattr(self, "_file_arr", None)
        if cached is not None:
            return cached
        import numpy as np
        path = self.cfg.data.path
        if not path:
            raise ValueError("data.task='lm_file' requires data.path")
        V = self.cfg.model.vocab_size
        if path.endswith(".bin"):
            dt = np.dtype(self.cfg.data.file_dtype)
            arr = np.memmap(path, dtype=dt, mode="r")
            # range check on a bounded sample (full scan would page in the
            # whole corpus); out-of-range tokens beyond it surface as an
            # embedding index error with a clear message at first use
            sample = arr[: 1 << 20]
            mx = int(sample.max()) if sample.shape[0] else 0
        else:
            # raw text: byte-level tokens (V must be >= 256)
            with open(path, "rb") as f:
                arr = np.frombuffer(f.read(), dtype=np.uint8)
            mx = int(arr.max()) if arr.shape[0] else 0
        if mx >= V:
            raise ValueError(
                f"lm_file token {mx} out of range for vocab_size={V}")
        self._file_arr = arr
        return self._file_arr

    def __iter__(self):
        step = 0
        while True:
            yield self.batch(step)
            step += 1
