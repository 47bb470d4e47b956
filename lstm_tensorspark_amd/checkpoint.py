"""Checkpoint save/resume with the reference's logical layout.

SURVEY.md §5 Checkpoint row: a flat named-weight dict
{layer_i[_rev]/{W_x,W_h,b}, embed/W, head/{W,b}} + optimizer state + RNG
state + step counter, written by rank 0. Resume restores bitwise (tested
by tests/test_checkpoint.py): model weights, fp32 masters, optimizer
buffers, data-stream position, and torch RNG states.
"""
from __future__ import annotations

import os
import tempfile
from typing import Optional

import torch

FORMAT_VERSION = 1


def _rng_state() -> dict:
    st = {"torch": torch.get_rng_state()}
    if torch.cuda.is_available():
        st["cuda"] = torch.cuda.get_rng_state_all()
    return st


def _load_rng_state(st: dict):
    torch.set_rng_state(st["torch"])
    if torch.cuda.is_available() and "cuda" in st:
        torch.cuda.set_rng_state_all(st["cuda"])


def save_checkpoint(path: str, model, optimizer=None, step: int = 0,
                    config: Optional[dict] = None, extra: Optional[dict] = None):
    """Atomic checkpoint write (tmp + rename)."""
    weights = {name: p.detach().cpu() for name, p in model.weight_dict().items()}
    payload = {
        "format_version": FORMAT_VERSION,
        "weights": weights,
        "optimizer": optimizer.state_dict() if optimizer is not None else None,
        "step": step,
        "rng": _rng_state(),
        "config": config,
        "extra": extra or {},
    }
    d = os.path.dirname(os.path.abspath(path))
    os.makedirs(d, exist_ok=True)
    fd, tmp = tempfile.mkstemp(dir=d, suffix=".tmp")
    os.close(fd)
    try:
        torch.save(payload, tmp)
        os.replace(tmp, path)
    finally:
        if os.path.exists(tmp):
            os.unlink(tmp)


def export_weights_safetensors(path: str, model, metadata: Optional[dict] = None):
    """Weights-only export in safetensors format (zero-copy mmap load,
    no pickle) -- the interchange format for serving/deployment.  The full
    training checkpoint stays torch.save (optimizer + RNG state are
    arbitrary python objects); this exports just the flat named-weight
    dict {layer_i[_rev]/{W_x,W_h,b}, embed/W, head/{W,b}}.
    """
    from safetensors.torch import save_file
    weights = {name: p.detach().cpu().contiguous()
               for name, p in model.weight_dict().items()}
    meta = {"format": "lstm_tensorspark_amd", "version": str(FORMAT_VERSION)}
    for k, v in (metadata or {}).items():
        meta[str(k)] = str(v)  # safetensors metadata must be str->str
    d = os.path.dirname(os.path.abspath(path))
    os.makedirs(d, exist_ok=True)
    fd, tmp = tempfile.mkstemp(dir=d, suffix=".tmp")
    os.close(fd)
    try:
        save_file(weights, tmp, metadata=meta)
        os.replace(tmp, path)
    finally:
        if os.path.exists(tmp):
            os.unlink(tmp)


def load_weights_safetensors(path: str, model) -> dict:
    """Load a safetensors weight export into `model` (strict name match).
    Returns the file's metadata dict."""
    from safetensors import safe_open
    wd = model.weight_dict()
    with safe_open(path, framework="pt", device="cpu") as f:
        names = set(f.keys())
        missing = names ^ set(wd)
        if missing:
            raise ValueError(
                f"safetensors/model weight-name mismatch: {sorted(missing)}")
        with torch.no_grad():
            for name, p in wd.items():
                p.copy_(f.get_tensor(name).to(p.device, p.dtype))
        return dict(f.metadata() or {})


def load_checkpoint(path: str, model=None, optimizer=None,
                    restore_rng: bool = True, map_location="cpu") -> dict:
    payload = torch.load(path, map_location=map_location, weights_only=False)
    if payload.get("format_version") != FORMAT_VERSION:
        raise ValueError(f"unsupported checkpoint version in {path}")
    if model is not None:
        wd = model.weight_dict()
        missing = set(payload["weights"]) ^ set(wd)
        if missing:
            raise ValueError(f"checkpoint/model weight-name mismatch: {sorted(missing)}")
        with torch.no_grad():
            for name, p in wd.items():
                p.copy_(payload["weights"][name].to(p.device, p.dtype))
    if optimizer is not None and payload["optimizer"] is not None:
        sd = payload["optimizer"]
        # move saved tensors to the optimizer's device
        dev = optimizer.params[0].device

        def to_dev(x):
            if isinstance(x, torch.Tensor):
                return x.to(dev)
            if isinstance(x, list):
                return [to_dev(v) for v in x]
            return x

        optimizer.load_state_dict({k: to_dev(v) for k, v in sd.items()})
    if restore_rng:
        _load_rng_state(payload["rng"])
    return payload
