#!/usr/bin/env python3
"""CLI train entry: replaces the reference's spark-submit driver script.

  python train.py --preset mi355x-1gpu --set train.steps=100
  torchrun --nproc-per-node 8 train.py --preset mi355x-dp8
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from lstm_tensorspark_amd.config import PRESETS, Config, get_preset
from lstm_tensorspark_amd.parallel.launch import cleanup, init_from_env
from lstm_tensorspark_amd.trainer import Trainer


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--preset", default=None, choices=sorted(PRESETS),
                   help="named BASELINE config")
    p.add_argument("--config", default=None, help="YAML config file")
    p.add_argument("--set", action="append", default=[], metavar="SEC.KEY=V",
                   help="config override, repeatable")
    p.add_argument("--resume", default=None, help="checkpoint to resume from")
    p.add_argument("--auto-resume", action="store_true",
                   help="resume from the newest checkpoint in "
                        "train.ckpt_dir if one exists (the relaunch-after-"
                        "failure entry point -- SURVEY.md §5 recovery row)")
    args = p.parse_args()

    if args.preset:
        cfg = get_preset(args.preset)
    elif args.config:
        cfg = Config.from_yaml(args.config)
    else:
        cfg = Config()
    if args.set:
        cfg = cfg.apply_overrides(args.set)

    ctx = init_from_env(backend=cfg.dist.backend)
    trainer = Trainer(cfg, dist_ctx=ctx)
    if args.resume:
        trainer.load(args.resume)
    elif args.auto_resume:
        trainer.resume_latest()
    # train.steps is the TOTAL budget (the LR schedule decays over it),
    # so a resumed run finishes the remaining steps rather than adding
    # a whole new budget on top
    trainer.fit(max(0, cfg.train.steps - trainer.step))
    cleanup()


if __name__ == "__main__":
    main()
