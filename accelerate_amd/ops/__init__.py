"""HIP/CDNA4 kernel pack (gfx950) and its Python wrappers.

The native extension ``accelerate_amd/_C*.so`` is built in-tree by
``setup.py build_ext --inplace`` (driven by ``__graft_entry__.build()``),
compiling ``ops/csrc/*.hip`` with ``hipcc --offload-arch=gfx950``.

Policy: on a GPU box the HIP kernels are THE path — ops raise if the
extension is missing rather than silently falling back to eager torch.
On CPU-only boxes (unit tests) a reference torch implementation runs.
"""

import importlib
import os

import torch

_EXT = None
_EXT_CHECKED = False


def _load_extension(required: bool = None):
    """Import accelerate_amd._C. required=None -> required iff a GPU is present."""
    global _EXT, _EXT_CHECKED
    if _EXT is None and not _EXT_CHECKED:
        _EXT_CHECKED = True
        try:
            _EXT = importlib.import_module("accelerate_amd._C")
        except ImportError as e:
            _EXT = None
            _import_error = e
    if _EXT is None:
        if required is None:
            required = torch.cuda.is_available()
        if required:
            raise RuntimeError(
                "accelerate_amd's HIP extension (accelerate_amd._C) is not built. "
                "Run `python setup.py build_ext --inplace` (or `python -c 'import __graft_entry__; "
                "__graft_entry__.build()'`) to compile the gfx950 kernels. "
                "The framework does not silently fall back to eager torch on a GPU."
            )
    return _EXT


def has_extension() -> bool:
    return _load_extension(required=False) is not None


from .grad_scaler import GradScaler  # noqa: E402
from .clip_grad import clip_grad_norm_, get_grad_norm  # noqa: E402
from .optim import FusedAdamW  # noqa: E402

__all__ = ["GradScaler", "FusedAdamW", "clip_grad_norm_", "get_grad_norm", "has_extension"]
