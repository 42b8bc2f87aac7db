"""Loader for the in-tree HIP extension (csrc/ -> vit_10b_fsdp_example_amd/_C.so).

The extension is built ahead of time for gfx950 with
``python setup.py build_ext --inplace`` (driven by __graft_entry__.build),
so the .so lives inside the package and travels with the repo snapshot
to GPU boxes.  There is deliberately NO JIT fallback and NO silent eager
fallback on GPU: if a GPU is present and the extension is missing, ops
raise, so a benchmark can never silently run on un-optimized PyTorch
kernels.  Set VITFSDP_ALLOW_EAGER=1 to override for debugging only.
"""

import importlib
import os

import torch

_EXT = None
_TRIED = False


def ext():
    """Return the compiled _C module, or None if unavailable."""
    global _EXT, _TRIED
    if not _TRIED:
        _TRIED = True
        try:
            _EXT = importlib.import_module("vit_10b_fsdp_example_amd._C")
        except ImportError:
            _EXT = None
    return _EXT


def has_ext():
    return ext() is not None


def allow_eager_on_gpu():
    return os.environ.get("VITFSDP_ALLOW_EAGER", "0") == "1"


def use_hip(*tensors):
    """Decide whether the HIP kernel path should run for these tensors.

    True  -> tensors are on GPU and the extension is loaded.
    False -> CPU tensors (eager torch path, used by CPU tests).
    Raises -> GPU tensors but no extension: that would silently benchmark
    eager PyTorch instead of our CDNA4 kernels.
    """
    on_gpu = any(t.is_cuda for t in tensors if torch.is_tensor(t))
    if not on_gpu:
        return False
    if has_ext():
        return True
    if allow_eager_on_gpu():
        return False
    raise RuntimeError(
        "vit_10b_fsdp_example_amd._C HIP extension is not built but tensors "
        "are on GPU. Build it with `python setup.py build_ext --inplace` "
        "(or run __graft_entry__.build()). Set VITFSDP_ALLOW_EAGER=1 to "
        "debug with eager PyTorch kernels."
    )
