"""Capability probes (reference: utils/imports.py).

Every probe is lru_cached and must never initialize the HIP runtime —
`is_hip_available` goes through ``torch.cuda.is_available`` only when asked
to (it is safe on ROCm: device_count uses the driver, not a context), and
`is_hip_available(check_device=False)` stays driver-free for fork-launched
notebooks (reference: imports.py:119-127 uses NVML for the same reason; the
MI355X analogue is amdsmi).
"""

import importlib
import importlib.metadata
from functools import lru_cache


def _is_package_available(pkg_name: str, metadata_name: str = None) -> bool:
    package_exists = importlib.util.find_spec(pkg_name) is not None
    if package_exists:
        try:
            _ = importlib.metadata.metadata(metadata_name or pkg_name)
            return True
        except importlib.metadata.PackageNotFoundError:
            # namespace packages (e.g. our own in-tree modules) have no metadata
            return package_exists
    return False


@lru_cache
def is_torch_distributed_available() -> bool:
    import torch.distributed

    return torch.distributed.is_available()


@lru_cache
def is_hip_available(check_device: bool = True) -> bool:
    """True when torch was built for ROCm and (optionally) a GPU is visible."""
    import torch

    if getattr(torch.version, "hip", None) is None:
        return False
    if not check_device:
        return True
    return torch.cuda.is_available()


# torch on ROCm exposes HIP devices through the ``cuda`` namespace; this
# framework is MI355X-only so "cuda available" == "HIP available".
def is_cuda_available() -> bool:
    import torch

    return torch.cuda.is_available()


@lru_cache
def is_bf16_available(ignore_hip: bool = False) -> bool:
    import torch

    if torch.cuda.is_available():
        return torch.cuda.is_bf16_supported()
    return not ignore_hip


@lru_cache
def is_fp8_available() -> bool:
    """CDNA4 (gfx950) has OCP fp8 (e4m3fn/e5m2) MFMA support."""
    import torch

    if not torch.cuda.is_available():
        return False
    arch = torch.cuda.get_device_properties(0).gcnArchName
    return "gfx95" in arch or "gfx94" in arch


@lru_cache
def is_mi355x() -> bool:
    import torch

    if not torch.cuda.is_available():
        return False
    return "gfx950" in torch.cuda.get_device_properties(0).gcnArchName


@lru_cache
def is_safetensors_available() -> bool:
    return _is_package_available("safetensors")


@lru_cache
def is_transformers_available() -> bool:
    return _is_package_available("transformers")


@lru_cache
def is_datasets_available() -> bool:
    return _is_package_available("datasets")


@lru_cache
def is_tensorboard_available() -> bool:
    return _is_package_available("tensorboard") or _is_package_available("tensorboardX")


@lru_cache
def is_wandb_available() -> bool:
    return _is_package_available("wandb")


@lru_cache
def is_mlflow_available() -> bool:
    return _is_package_available("mlflow")


@lru_cache
def is_rich_available() -> bool:
    return _is_package_available("rich")


@lru_cache
def is_numpy_available() -> bool:
    return _is_package_available("numpy")


@lru_cache
def is_pandas_available() -> bool:
    return _is_package_available("pandas")


@lru_cache
def is_torchdata_stateful_dataloader_available() -> bool:
    if not _is_package_available("torchdata"):
        return False
    try:
        from torchdata.stateful_dataloader import StatefulDataLoader  # noqa: F401

        return True
    except ImportError:
        return False


@lru_cache
def is_amd_kernels_available() -> bool:
    """True when the in-tree HIP extension (accelerate_amd._C) is importable."""
    try:
        from accelerate_amd.ops import _load_extension

        return _load_extension(required=False) is not None
    except Exception:
        return False


@lru_cache
def is_rocm_available() -> bool:
    import torch

    return bool(getattr(torch.version, "hip", None))


@lru_cache
def is_peft_available() -> bool:
    return _is_package_available("peft")


@lru_cache
def is_torchvision_available() -> bool:
    return _is_package_available("torchvision")


@lru_cache
def is_timm_available() -> bool:
    return _is_package_available("timm")


@lru_cache
def is_pandas_available() -> bool:
    return _is_package_available("pandas")


@lru_cache
def is_matplotlib_available() -> bool:
    return _is_package_available("matplotlib")


@lru_cache
def is_pytest_available() -> bool:
    return _is_package_available("pytest")


@lru_cache
def is_triton_available() -> bool:
    return _is_package_available("triton")


@lru_cache
def is_torchdata_available() -> bool:
    return _is_package_available("torchdata")


def is_weights_only_available() -> bool:
    return True  # torch >= 2.4 in this stack
