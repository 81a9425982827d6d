from .from_lakesoul import from_lakesoul  # noqa: F401
