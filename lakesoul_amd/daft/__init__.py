"""Daft source/sink (reference: ``python/src/lakesoul/daft/``).

Daft is not installed in this build image; the adapters are importable
and raise a clear error on use without daft. With daft present:

    from lakesoul_amd.daft import read_lakesoul, write_lakesoul
    df = read_lakesoul(table)
    write_lakesoul(df, table)
"""

from __future__ import annotations

from typing import Optional, Sequence


def _require_daft():
    try:
        import daft

        return daft
    except ImportError as e:
        raise ImportError(
            "daft is not installed in this environment; "
            "lakesoul_amd.daft needs the 'daft' package"
        ) from e


def read_lakesoul(table, columns: Optional[Sequence[str]] = None, filters=None):
    daft = _require_daft()
    tbl = table.scan(columns=columns, filters=filters, device="cpu").to_arrow()
    return daft.from_arrow(tbl)


def write_lakesoul(df, table) -> None:
    _require_daft()
    table.write(df.to_arrow())
