from .dataset import LakeSoulArrowDataset  # noqa: F401
