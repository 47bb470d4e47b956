#!/usr/bin/env python3
"""torch.profiler attribution of one training step (op-level, GPU time)."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from lstm_tensorspark_amd.config import get_preset
from lstm_tensorspark_amd.trainer import Trainer


def main():
    cfg = get_preset(os.environ.get("PRESET", "mi355x-1gpu"))
    cfg.train.log_interval = 0
    cfg.train.ckpt_dir = ""
    t = Trainer(cfg, device="cuda:0")
    batch = t.data.batch(0)
    for _ in range(3):
        t.train_step(batch)
    torch.cuda.synchronize()
    from torch.profiler import ProfilerActivity, profile
    with profile(activities=[ProfilerActivity.CUDA, ProfilerActivity.CPU]) as prof:
        for _ in range(3):
            t.train_step(batch)
        torch.cuda.synchronize()
    print(prof.key_averages().table(sort_by="cuda_time_total", row_limit=28,
                                    max_name_column_width=60))


if __name__ == "__main__":
    main()
