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


def get_module_children_bottom_up(model) -> list:
    """All submodules, deepest first (reference: utils/other.py export)."""
    children = list(model.modules())[1:]
    return list(reversed(children))


def is_compiled_module(module) -> bool:
    import torch

    return isinstance(module, torch._dynamo.eval_frame.OptimizedModule)


def compile_regions(module, compile_rest: bool = True, **compile_kwargs):
    """Regional compilation (reference: utils/other.py:106; BASELINE.md
    regional-compilation rows): compile each REPEATED block of a model
    individually so dynamo's cache is hit once per unique structure —
    compile time drops ~5x on LLMs vs whole-model compile. Repeated blocks
    are the same-class children of any ``nn.ModuleList`` (decoder layers);
    with ``compile_rest`` the remaining modules (embeddings, head) are
    compiled as one region each at the root.

    Mutates ``module`` in place (children are swapped for their
    ``OptimizedModule`` wrappers) and returns it.
    """
    import torch
    import torch.nn as nn

    compiled_any = False
    for sub in module.modules():
        if isinstance(sub, nn.ModuleList) and len(sub) >= 2:
            classes = {type(c) for c in sub}
            if len(classes) == 1:
                for i, child in enumerate(sub):
                    sub[i] = torch.compile(child, **compile_kwargs)
                compiled_any = True
    if compile_rest and compiled_any:
        for name, child in module.named_children():
            if isinstance(child, nn.ModuleList) or is_compiled_module(child):
                continue
            if any(True for _ in child.parameters()):
                setattr(module, name, torch.compile(child, **compile_kwargs))
    module._regions_compiled = compiled_any
    return module


def has_compiled_regions(module) -> bool:
    return bool(getattr(module, "_regions_compiled", False))


def merge_dicts(source: dict, destination: dict) -> dict:
    """Recursively merge ``source`` into ``destination`` (reference export)."""
    for key, value in source.items():
        if isinstance(value, dict):
            node = destination.setdefault(key, {})
            merge_dicts(value, node)
        else:
            destination[key] = value
    return destination


def convert_dict_to_env_variables(current_env: dict) -> list:
    """Render an env dict as KEY=value strings, skipping unprintables
    (reference export, used by the launch CLI)."""
    forbidden = set("\n\r")
    valid = []
    for key, value in current_env.items():
        value = str(value)
        if forbidden & set(value):
            import logging

            logging.getLogger(__name__).warning(f"skipping env var {key} (newline in value)")
            continue
        valid.append(f"{key}={value}")
    return valid


def write_basic_config(mixed_precision: str = "no", save_location: str = None):
    """Write a default single-node config YAML without the interactive Q&A
    (reference: utils/other.py write_basic_config — notebook convenience)."""
    import os

    import torch
    import yaml

    if save_location is None:
        save_location = os.path.join(
            os.path.expanduser("~"), ".cache", "accelerate_amd", "default_config.yaml"
        )
    os.makedirs(os.path.dirname(save_location), exist_ok=True)
    n_gpus = torch.cuda.device_count()
    config = {
        "compute_environment": "LOCAL_MACHINE",
        "distributed_type": "MULTI_GPU" if n_gpus > 1 else "NO",
        "num_processes": max(n_gpus, 1),
        "mixed_precision": mixed_precision,
        "use_cpu": n_gpus == 0,
    }
    with open(save_location, "w") as f:
        yaml.safe_dump(config, f)
    return save_location


def save(obj, f, save_on_each_node: bool = False, safe_serialization: bool = False):
    """Save ``obj`` on the main process — or every node's main process
    (reference: utils/other.py save; re-exported by checkpointing)."""
    import torch

    from ..state import PartialState

    state = PartialState()
    if safe_serialization:
        import safetensors.torch

        save_func = lambda obj, f: safetensors.torch.save_file(obj, f, metadata={"format": "pt"})
    else:
        save_func = torch.save
    if state.is_main_process and not save_on_each_node:
        save_func(obj, f)
    elif state.is_local_main_process and save_on_each_node:
        save_func(obj, f)
