"""Structured per-step logging (SURVEY.md §5 Metrics row): rank-0 JSONL on
stdout, consumed by the bench harness; replaces the reference driver's
per-epoch print + Spark UI."""
from __future__ import annotations

import logging
import sys

_LOGGERS = {}


def get_logger(rank: int = 0, name: str = "lstm_ts") -> logging.Logger:
    key = (name, rank)
    if key in _LOGGERS:
        return _LOGGERS[key]
    log = logging.getLogger(f"{name}.r{rank}")
    log.setLevel(logging.INFO)
    log.propagate = False
    if not log.handlers:
        h = logging.StreamHandler(sys.stderr)
        h.setFormatter(logging.Formatter(
            f"[%(asctime)s r{rank}] %(message)s", datefmt="%H:%M:%S"))
        log.addHandler(h)
    _LOGGERS[key] = log
    return log
