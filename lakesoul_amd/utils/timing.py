"""Opt-in phase timing (LAKESOUL_TIMING=1): accumulates named phase
durations; GPU phases synchronize so numbers are honest. Zero overhead
when disabled."""

from __future__ import annotations

import os
import time
from collections import defaultdict
from contextlib import contextmanager

ENABLED = os.environ.get("LAKESOUL_TIMING", "0") == "1"
_acc = defaultdict(float)
_cnt = defaultdict(int)


@contextmanager
def phase(name: str, sync_gpu: bool = False):
    if not ENABLED:
        yield
        return
    import torch

    if sync_gpu and torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    try:
        yield
    finally:
        if sync_gpu and torch.cuda.is_available():
            torch.cuda.synchronize()
        _acc[name] += time.perf_counter() - t0
        _cnt[name] += 1


def report() -> str:
    lines = [f"{k:24s} {_acc[k]*1000:9.1f} ms  ({_cnt[k]} calls)" for k in sorted(_acc, key=lambda k: -_acc[k])]
    return "\n".join(lines)


def reset():
    _acc.clear()
    _cnt.clear()
