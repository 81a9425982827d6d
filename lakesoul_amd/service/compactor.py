"""Auto-compaction service.

Analog of the reference's PG-trigger + Flink listener pipeline
(``script/meta_init.sql:102-150`` pg_notify('lakesoul_compaction_notify')
-> ``entry/clean/NewCleanJob.java`` / CompactionBroadcastProcessFunction):
polls the metadata for partitions with >= 10 delta versions since the
last compaction (the trigger's rule, via MetaClient.compaction_needed)
and compacts them; optionally removes files replaced by old compactions
(CleanExpiredData analog).
"""

from __future__ import annotations

import time
from typing import List, Optional, Tuple


class CompactionService:
    def __init__(self, catalog, namespaces: Optional[List[str]] = None,
                 cleanup: bool = True, device: Optional[str] = None):
        self.catalog = catalog
        self.namespaces = namespaces
        self.cleanup = cleanup
        self.device = device

    def scan_once(self) -> List[Tuple[str, str]]:
        """One poll cycle: compact every partition that needs it.
        Returns the (table_name, partition_desc) pairs compacted."""
        compacted = []
        namespaces = self.namespaces or self.catalog.list_namespaces()
        for ns in namespaces:
            for name in self.catalog.list_tables(ns):
                t = self.catalog.table(name, ns)
                for desc in t.partition_descs():
                    if t.client.compaction_needed(t.table_id, desc):
                        t.compaction(desc, device=self.device)
                        compacted.append((name, desc))
                if self.cleanup:
                    from ..io.compaction import cleanup_discarded_files

                    cleanup_discarded_files(t)
        return compacted

    def run(self, interval_s: float = 30.0, max_cycles: Optional[int] = None):
        """Poll loop (daemon mode)."""
        cycles = 0
        while max_cycles is None or cycles < max_cycles:
            self.scan_once()
            cycles += 1
            if max_cycles is None or cycles < max_cycles:
                time.sleep(interval_s)
