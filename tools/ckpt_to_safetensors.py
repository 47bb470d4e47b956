#!/usr/bin/env python3
"""Convert a training checkpoint (.pt) to a safetensors deployment
artifact: weights only, model config embedded as JSON in the file
metadata so `examples/serve.py --ckpt out.safetensors` needs no
side-channel config.

    python tools/ckpt_to_safetensors.py ckpt_final.pt model.safetensors
"""
import argparse
import json
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from lstm_tensorspark_amd.checkpoint import export_weights_safetensors  # noqa: E402
from lstm_tensorspark_amd.config import Config  # noqa: E402
from lstm_tensorspark_amd.models.lstm import LSTMModel  # noqa: E402


def convert(src: str, dst: str) -> dict:
    payload = torch.load(src, map_location="cpu", weights_only=False)
    cfg = Config.from_dict(payload["config"])
    model = LSTMModel(cfg.model)
    wd = model.weight_dict()
    with torch.no_grad():
        for name, p in wd.items():
            p.copy_(payload["weights"][name].to(p.dtype))
    meta = {"config": json.dumps(payload["config"]["model"]),
            "step": str(payload.get("step", 0))}
    export_weights_safetensors(dst, model, metadata=meta)
    return meta


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("src", help="training checkpoint (.pt)")
    ap.add_argument("dst", help="output .safetensors path")
    args = ap.parse_args(argv)
    meta = convert(args.src, args.dst)
    print(json.dumps({"out": args.dst, "step": meta["step"],
                      "bytes": os.path.getsize(args.dst)}))


if __name__ == "__main__":
    main()
