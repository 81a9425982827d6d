"""Streaming (incremental) table source.

Analog of the reference's Flink ``LakeSoulSource`` + dynamic split
enumerator (``source/LakeSoulAllPartitionDynamicSplitEnumerator.java``):
polls the metadata for new partition versions and yields the newly
committed rows. The consumed position is (partition_desc -> version), so
a reader resumed from a saved position never re-reads or skips commits.
"""

from __future__ import annotations

import time
from typing import Dict, Iterator, Optional


class TableStream:
    def __init__(self, table, start_versions: Optional[Dict[str, int]] = None,
                 device: Optional[str] = None, columns=None):
        self.table = table
        self.positions: Dict[str, int] = dict(start_versions or {})
        self.device = device
        self.columns = columns

    def poll(self):
        """Return (batches, advanced): new rows committed since the last
        poll, as a list of Batch, and whether the position advanced."""
        from ..io.reader import LakeSoulScan

        out = []
        advanced = False
        for desc in self.table.partition_descs():
            cur = self.table.client.get_latest_version(self.table.table_id, desc)
            if cur is None:
                continue
            start = self.positions.get(desc, -1)
            if cur <= start:
                continue
            scan = LakeSoulScan(
                self.table,
                columns=self.columns,
                partitions=[desc],
                device=self.device,
                incremental=(start, cur),
            )
            for b in scan.iter_batches():
                if b is not None and b.num_rows:
                    out.append(b)
            self.positions[desc] = cur
            advanced = True
        return out, advanced

    def __iter__(self) -> Iterator:
        while True:
            batches, advanced = self.poll()
            for b in batches:
                yield b
            if not advanced:
                return  # caller re-enters (or use run() for a daemon loop)

    def run(self, interval_s: float = 1.0, max_polls: Optional[int] = None):
        polls = 0
        while max_polls is None or polls < max_polls:
            batches, _ = self.poll()
            for b in batches:
                yield b
            polls += 1
            if max_polls is None or polls < max_polls:
                time.sleep(interval_s)
