from .cdc import CdcIngestor  # noqa: F401
