from .dataset import LakeSoulIterableDataset  # noqa: F401
