"""Memory accounting / profiling hooks — the jemalloc-profiling analog
(reference: lakesoul-io/src/mem.rs jemalloc prof_gdump + pprof dumps,
LoggedMemoryPool for tests).

``snapshot()`` gives one consistent view across the three pools that
matter on an MI355X node:

- host RSS / peak RSS (the cgroup-capped CPU side),
- the torch CUDA caching allocator (allocated/reserved/peak — HBM),
- pinned-host bytes in flight (the staging pool the scan pipeline uses).

``track()`` context manager logs deltas around a block (LoggedMemoryPool
analog); the gateway exposes snapshots at /metrics/memory.
"""

from __future__ import annotations

import contextlib
import os
from typing import Dict, Optional


def _rss_bytes() -> Dict[str, int]:
    out = {"rss": 0, "peak_rss": 0}
    try:
        with open("/proc/self/status") as f:
            for line in f:
                if line.startswith("VmRSS:"):
                    out["rss"] = int(line.split()[1]) * 1024
                elif line.startswith("VmHWM:"):
                    out["peak_rss"] = int(line.split()[1]) * 1024
    except OSError:  # pragma: no cover
        pass
    return out


def snapshot() -> Dict[str, int]:
    snap: Dict[str, int] = dict(_rss_bytes())
    try:
        import torch

        if torch.cuda.is_available():
            snap["hbm_allocated"] = torch.cuda.memory_allocated()
            snap["hbm_reserved"] = torch.cuda.memory_reserved()
            snap["hbm_peak_allocated"] = torch.cuda.max_memory_allocated()
            st = torch.cuda.memory_stats()
            snap["hbm_alloc_retries"] = int(st.get("num_alloc_retries", 0))
        else:
            snap["hbm_allocated"] = 0
            snap["hbm_reserved"] = 0
            snap["hbm_peak_allocated"] = 0
    except Exception:  # pragma: no cover
        pass
    return snap


@contextlib.contextmanager
def track(label: str, log=None):
    """Log host/HBM deltas around a block (LoggedMemoryPool analog)."""
    before = snapshot()
    try:
        yield before
    finally:
        after = snapshot()
        delta = {k: after.get(k, 0) - before.get(k, 0) for k in after}
        msg = (f"[memprof] {label}: rss {delta.get('rss', 0) / 1e6:+.1f} MB, "
               f"hbm {delta.get('hbm_allocated', 0) / 1e6:+.1f} MB "
               f"(peak rss {after.get('peak_rss', 0) / 1e9:.2f} GB)")
        (log or (lambda s: print(s, flush=True)))(msg)


def reset_peaks() -> None:
    try:
        import torch

        if torch.cuda.is_available():
            torch.cuda.reset_peak_memory_stats()
    except Exception:  # pragma: no cover
        pass
