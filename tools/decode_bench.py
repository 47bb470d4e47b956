#!/usr/bin/env python3
"""LM decode (serving) throughput: tokens/s of model.generate()."""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from lstm_tensorspark_amd.config import ModelConfig
from lstm_tensorspark_amd.models.lstm import LSTMModel


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--B", type=int, default=256)
    ap.add_argument("--H", type=int, default=4096)
    ap.add_argument("--vocab", type=int, default=50000)
    ap.add_argument("--layers", type=int, default=2)
    ap.add_argument("--prompt", type=int, default=32)
    ap.add_argument("--steps", type=int, default=128)
    args = ap.parse_args()
    torch.manual_seed(0)
    cfg = ModelConfig(input_size=args.H, hidden_size=args.H,
                      num_layers=args.layers, head="lm", vocab_size=args.vocab)
    m = LSTMModel(cfg).to("cuda").bfloat16()
    prompt = torch.randint(0, args.vocab, (args.prompt, args.B), device="cuda")
    m.generate(prompt, steps=32)  # warmup (captures the decode graph)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    out = m.generate(prompt, steps=args.steps)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    toks = args.steps * args.B
    print({"decode_tokens_per_s": round(toks / dt, 1),
           "ms_per_token_step": round(1e3 * dt / args.steps, 3),
           "B": args.B, "H": args.H, "vocab": args.vocab,
           "layers": args.layers})
    assert out.shape == (args.prompt + args.steps, args.B)


if __name__ == "__main__":
    main()
