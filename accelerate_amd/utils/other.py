"""Misc utilities (reference: utils/other.py)."""

import os
import socket
from contextlib import closing

import torch

from ..state import PartialState


def wait_for_everyone():
    """Barrier across all processes (module-level convenience)."""
    PartialState().wait_for_everyone()


def extract_model_from_parallel(model, keep_fp32_wrapper: bool = True, recursive: bool = False):
    """Unwrap DistributedDataParallelEngine / ShardedModel / compiled wrappers
    (reference: utils/other.py extract_model_from_parallel)."""
    from ..parallel.ddp import DistributedDataParallelEngine

    options = (DistributedDataParallelEngine, torch.nn.parallel.DistributedDataParallel, torch.nn.DataParallel)
    try:
        from ..parallel.fsdp import ShardedModel

        options = options + (ShardedModel,)
    except ImportError:
        pass

    is_compiled = getattr(model, "_orig_mod", None) is not None
    if is_compiled:
        compiled_model = model
        model = model._orig_mod

    while isinstance(model, options):
        model = model.module

    if recursive:
        for name, module in model.named_children():
            setattr(model, name, extract_model_from_parallel(module, keep_fp32_wrapper, recursive))

    if not keep_fp32_wrapper:
        forward = model.forward
        original_forward = model.__dict__.pop("_original_forward", None)
        if original_forward is not None:
            while hasattr(forward, "__wrapped__"):
                forward = forward.__wrapped__
                if forward == original_forward:
                    break
            model.forward = forward

    if is_compiled:
        compiled_model._orig_mod = model
        model = compiled_model
    return model


def is_port_in_use(port: int = None) -> bool:
    if port is None:
        port = 29500
    with closing(socket.socket(socket.AF_INET, socket.SOCK_STREAM)) as s:
        return s.connect_ex(("localhost", port)) == 0


def get_free_port() -> int:
    with closing(socket.socket(socket.AF_INET, socket.SOCK_STREAM)) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def check_os_kernel():
    """Warn on old Linux kernels (reference: utils/other.py:531)."""
    import platform

    info = platform.uname()
    if info.system != "Linux":
        return
    try:
        from packaging.version import parse

        if parse(info.release.split("-")[0]) < parse("5.5"):
            import warnings

            warnings.warn(
                f"Detected kernel version {info.release.split('-')[0]}, which is below the recommended minimum of 5.5.0; "
                "this can cause the process to hang.",
                UserWarning,
            )
    except Exception:
        pass


def convert_bytes(size: int) -> str:
    """Human-readable bytes."""
    for unit in ["bytes", "KB", "MB", "GB", "TB"]:
        if size < 1024.0:
            return f"{round(size, 2)} {unit}"
        size /= 1024.0
    return f"{round(size, 2)} PB"
