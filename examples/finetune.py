#!/usr/bin/env python3
"""Fine-tune a pretrained LM on a token corpus.

Ties the deployment pipeline together: safetensors weights in
(`Trainer.load_weights` -- fresh optimizer, fp32 masters re-synced),
file-backed corpus with a held-out tail, LR warmup+cosine, gradient
accumulation, periodic held-out eval, checkpoints out.

    python examples/finetune.py --weights model.safetensors \
        --corpus corpus.bin --vocab 50000 --steps 500
"""
import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from lstm_tensorspark_amd.config import (Config, DataConfig, ModelConfig,
                                         TrainConfig)
from lstm_tensorspark_amd.trainer import Trainer


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--weights", required=True,
                    help="pretrained .safetensors (config in metadata)")
    ap.add_argument("--corpus", required=True, help="token corpus (.bin/.txt)")
    ap.add_argument("--vocab", type=int, default=50000)
    ap.add_argument("--hidden", type=int, default=1024)
    ap.add_argument("--layers", type=int, default=2)
    ap.add_argument("--steps", type=int, default=500)
    ap.add_argument("--batch", type=int, default=256)
    ap.add_argument("--seq-len", type=int, default=256)
    ap.add_argument("--accum", type=int, default=1)
    ap.add_argument("--lr", type=float, default=3e-4)
    ap.add_argument("--out", default="finetune_ckpts")
    args = ap.parse_args(argv)

    import json as _json
    from safetensors import safe_open
    with safe_open(args.weights, framework="pt", device="cpu") as f:
        meta = dict(f.metadata() or {})
    if "config" in meta:
        mcfg = ModelConfig(**_json.loads(meta["config"]))
    else:
        mcfg = ModelConfig(input_size=args.hidden, hidden_size=args.hidden,
                           num_layers=args.layers, head="lm",
                           vocab_size=args.vocab)
    cfg = Config(
        model=mcfg,
        train=TrainConfig(batch_size=args.batch, seq_len=args.seq_len,
                          accum_steps=args.accum, lr=args.lr,
                          optimizer="adam", grad_clip=1.0, dtype="bf16",
                          steps=args.steps, lr_schedule="cosine",
                          warmup_steps=max(1, args.steps // 20),
                          min_lr=args.lr / 10, eval_interval=50,
                          eval_steps=5, ckpt_dir=args.out,
                          log_interval=10),
        data=DataConfig(task="lm_file", path=args.corpus, eval_frac=0.02),
    )
    t = Trainer(cfg)
    t.load_weights(args.weights)
    t.fit()
    r = t.evaluate(steps=10)
    print(json.dumps({"final_eval": r, "steps": t.step,
                      "ckpt_dir": args.out}))
    return r


if __name__ == "__main__":
    main()
