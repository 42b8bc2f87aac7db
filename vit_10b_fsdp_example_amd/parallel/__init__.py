from .fsdp import FullyShardedDataParallel
from .checkpoint import CheckpointWrapper, checkpoint_module
from .comm import CommContext

__all__ = [
    "FullyShardedDataParallel",
    "CheckpointWrapper",
    "checkpoint_module",
    "CommContext",
]
