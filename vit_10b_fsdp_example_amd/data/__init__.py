from .datasets import FakeImageNetDataset, build_datasets
from .loader import DeviceLoader

__all__ = ["FakeImageNetDataset", "build_datasets", "DeviceLoader"]
