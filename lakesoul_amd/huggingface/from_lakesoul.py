"""HuggingFace datasets loader (reference:
``python/src/lakesoul/huggingface/from_lakesoul.py``)."""

from __future__ import annotations

from typing import Optional, Sequence


def from_lakesoul(
    table,
    columns: Optional[Sequence[str]] = None,
    partitions: Optional[Sequence[str]] = None,
    filters=None,
    split: str = "train",
):
    """Load a LakeSoul table (merge-on-read) as a huggingface
    ``datasets.Dataset``."""
    try:
        import datasets
    except ImportError as e:  # pragma: no cover
        raise ImportError("huggingface 'datasets' package is required") from e

    tbl = table.scan(columns=columns, partitions=partitions, filters=filters).to_arrow()
    return datasets.Dataset(datasets.table.InMemoryTable(tbl), split=split)
