"""HBM-resident scan cache.

The MI355X-native counterpart of the reference's disk page cache
(``rust/lakesoul-io/src/cache/``): with 288 GB HBM3E per GPU, decoded
merge-on-read results stay resident and repeated scans of an unchanged
snapshot skip IO/decode/merge entirely. Keys are the unit's exact file
list (immutable files + snapshot-versioned metadata make this safe: any
commit changes the file list and so the key). LRU by bytes.

Enable per-scan with ``options={"scan_cache": "1"}`` in IOConfig or env
``LAKESOUL_SCAN_CACHE=1``; capacity via ``LAKESOUL_SCAN_CACHE_BYTES``
(default 32 GiB). bench.py does NOT enable it (every timed step performs
the full decode+merge).
"""

from __future__ import annotations

import os
import threading
from collections import OrderedDict
from typing import Optional, Tuple

from .batch import Batch


def _batch_bytes(b: Batch) -> int:
    total = 0
    for c in b.columns.values():
        for t in (c.data, c.offsets, c.bytes_, c.validity):
            if t is not None:
                total += t.numel() * t.element_size()
    return total


class HbmScanCache:
    def __init__(self, capacity_bytes: Optional[int] = None):
        self.capacity = capacity_bytes or int(
            os.environ.get("LAKESOUL_SCAN_CACHE_BYTES", str(32 * 1024**3))
        )
        self._lock = threading.Lock()
        self._map: "OrderedDict[Tuple, Tuple[Batch, int]]" = OrderedDict()
        self._bytes = 0
        self.hits = 0
        self.misses = 0

    def key(self, unit, read_cols) -> Tuple:
        return (tuple(unit.files), tuple(read_cols))

    def get(self, key) -> Optional[Batch]:
        with self._lock:
            item = self._map.get(key)
            if item is None:
                self.misses += 1
                return None
            self._map.move_to_end(key)
            self.hits += 1
            return item[0]

    def put(self, key, batch: Batch) -> None:
        nb = _batch_bytes(batch)
        if nb > self.capacity:
            return
        with self._lock:
            if key in self._map:
                return
            self._map[key] = (batch, nb)
            self._bytes += nb
            while self._bytes > self.capacity and self._map:
                _, (old, ob) = self._map.popitem(last=False)
                self._bytes -= ob

    def stats(self) -> dict:
        with self._lock:
            return {
                "entries": len(self._map),
                "bytes": self._bytes,
                "capacity": self.capacity,
                "hits": self.hits,
                "misses": self.misses,
            }

    def clear(self) -> None:
        with self._lock:
            self._map.clear()
            self._bytes = 0


_cache: Optional[HbmScanCache] = None


def scan_cache() -> HbmScanCache:
    global _cache
    if _cache is None:
        _cache = HbmScanCache()
    return _cache
