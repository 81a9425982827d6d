"""IO configuration.

MI355X-native analog of the reference's ``LakeSoulIOConfig``
(``rust/lakesoul-io/src/config/mod.rs:40-116``): typed fields with the same
defaults, plus an untyped option map with ``LAKESOUL_<KEY>`` environment
fallback (reference: ``config/mod.rs:160-165``).
"""

from __future__ import annotations

import os
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from . import constants


@dataclass
class IOConfig:
    """Configuration for a single reader/writer session."""

    prefix: str = ""
    files: List[str] = field(default_factory=list)
    primary_keys: List[str] = field(default_factory=list)
    range_partitions: List[str] = field(default_factory=list)
    hash_bucket_num: int = constants.DEFAULT_HASH_BUCKET_NUM
    aux_sort_cols: List[str] = field(default_factory=list)
    batch_size: int = constants.DEFAULT_BATCH_SIZE
    max_row_group_size: int = constants.DEFAULT_MAX_ROW_GROUP_SIZE
    max_row_group_num_values: int = constants.DEFAULT_MAX_ROW_GROUP_NUM_VALUES
    prefetch_size: int = constants.DEFAULT_PREFETCH_SIZE
    compression: str = constants.DEFAULT_COMPRESSION
    compression_level: int = constants.DEFAULT_COMPRESSION_LEVEL
    merge_operators: Dict[str, str] = field(default_factory=dict)
    default_column_value: Dict[str, str] = field(default_factory=dict)
    max_file_size: Optional[int] = None
    # "cuda" | "cpu" | None (auto)
    device: Optional[str] = None
    options: Dict[str, str] = field(default_factory=dict)

    # ------------------------------------------------------------------ #

    def option(self, key: str, default: Optional[str] = None) -> Optional[str]:
        """Look up an untyped option, falling back to ``LAKESOUL_<KEY>`` env
        (reference behavior: config/mod.rs:160-165)."""
        if key in self.options:
            return self.options[key]
        env_key = "LAKESOUL_" + key.upper()
        if env_key in os.environ:
            return os.environ[env_key]
        return default

    def effective_hash_bucket_num(self) -> int:
        # reference clamps non-positive values to 1 (config/mod.rs:220-224)
        return max(1, int(self.hash_bucket_num))

    def resolve_device(self) -> str:
        if self.device is not None:
            return self.device
        try:
            import torch

            if torch.cuda.is_available():
                return "cuda"
        except Exception:
            pass
        return "cpu"
