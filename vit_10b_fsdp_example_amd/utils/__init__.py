from .meters import SmoothedValue
from .scheduler import get_warmup_cosine_scheduler
from .ckpt import save_ckpt, load_ckpt

__all__ = [
    "SmoothedValue",
    "get_warmup_cosine_scheduler",
    "save_ckpt",
    "load_ckpt",
]
