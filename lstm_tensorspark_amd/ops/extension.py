"""Loader for the in-tree HIP extension (_lstm_hip.so, built from csrc/).

Policy (per the build contract): on a GPU box the hand-written CDNA4 kernels
ARE the compute path -- if a CUDA/HIP device is present and the extension
cannot be imported, ops fail loudly instead of silently falling back to
eager PyTorch. On CPU the reference implementations (functional.py) run.

Set LSTM_TS_FORCE_REF=1 to force the reference path (used by numerics tests
that compare HIP kernels against the fp32 reference on the same GPU).
"""
from __future__ import annotations

import importlib
import os

import torch

_EXT = None
_EXT_ERR: Exception | None = None
_TRIED = False


def _try_load():
    global _EXT, _EXT_ERR, _TRIED
    if _TRIED:
        return
    _TRIED = True
    try:
        # The .so is built in-tree into this package directory by
        # csrc/build.py (driven by __graft_entry__.build()).
        _EXT = importlib.import_module("lstm_tensorspark_amd._lstm_hip")
    except Exception as e:  # pragma: no cover - exercised only on GPU boxes
        _EXT, _EXT_ERR = None, e


def extension_available() -> bool:
    _try_load()
    return _EXT is not None


def get_extension():
    """Return the HIP extension module, failing loudly on a GPU box."""
    _try_load()
    if _EXT is None:
        raise RuntimeError(
            "lstm_tensorspark_amd HIP extension (_lstm_hip) is not built but a "
            "GPU is present; run `python csrc/build.py` (or __graft_entry__."
            "build()) first. Silent eager fallback on GPU is disabled by "
            f"policy. Import error: {_EXT_ERR!r}"
        )
    return _EXT


def force_ref() -> bool:
    return os.environ.get("LSTM_TS_FORCE_REF", "0") == "1"


def use_hip(device: torch.device | str,
            dtype: "torch.dtype | None" = None) -> bool:
    """Decide the backend: HIP kernels on GPU for bf16 (the flagship compute
    dtype), reference ops on CPU or for fp32-on-GPU (numerics/debug runs)."""
    dev = torch.device(device)
    if dev.type != "cuda":
        return False
    if force_ref():
        return False
    if dtype is not None and dtype != torch.bfloat16:
        return False
    # GPU path: the extension is mandatory (fail loudly if missing).
    get_extension()
    return True
