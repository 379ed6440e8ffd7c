"""Big-model init & dispatch (reference: big_modeling.py).

`init_empty_weights` materializes models on the meta device; `dispatch_model`
splits them across MI355X GPUs / CPU / disk per a device map sized for
288 GB HBM3E per GPU, attaching AlignDevicesHooks for offloaded blocks.
"""

import os
from contextlib import contextmanager
from functools import wraps
from typing import Dict, List, Optional, Union

import torch
import torch.nn as nn

from .hooks import (
    AlignDevicesHook,
    CpuOffload,
    LayerwiseCastingHook,
    UserCpuOffloadHook,
    add_hook_to_module,
    attach_align_device_hook,
    attach_align_device_hook_on_blocks,
)
from .logging import get_logger
from .utils.imports import is_hip_available
from .utils.modeling import (
    check_device_map,
    find_tied_parameters,
    get_balanced_memory,
    infer_auto_device_map,
    load_checkpoint_in_model,
    retie_parameters,
)
from .utils.offload import OffloadedWeightsLoader, extract_submodules_state_dict, offload_state_dict
from .utils.other import check_os_kernel

logger = get_logger(__name__)


@contextmanager
def init_empty_weights(include_buffers: bool = False):
    """Initialize a model with meta-device parameters — zero host RAM
    (reference: big_modeling.py:62)."""
    with init_on_device(torch.device("meta"), include_buffers=include_buffers) as f:
        yield f


def _rehome_parameter(registered: nn.Parameter, device: torch.device) -> nn.Parameter:
    """Rebuild a just-registered Parameter on ``device``, preserving its
    concrete class and any attributes a Parameter subclass stashed on the
    instance (the contract transformers' custom params rely on)."""
    attrs = dict(registered.__dict__)
    attrs["requires_grad"] = registered.requires_grad
    return type(registered)(registered.to(device), **attrs)


@contextmanager
def init_on_device(device: torch.device, include_buffers: bool = False):
    """Run model construction with every parameter landing on ``device``.

    Behavior parity with reference big_modeling.py:98-176; mechanism is
    ours. With ``include_buffers`` the torch device-mode context already
    routes every tensor constructor (params, buffers, intermediates) to the
    target device, so no patching is needed at all. Without it, only
    ``nn.Module.register_parameter`` is intercepted: the module registers
    its parameter normally, then the slot is swapped for a re-homed copy —
    buffers keep their computed (host) values, which is what lets meta-init
    models keep usable RoPE/position tables.
    """
    if include_buffers:
        with device:
            yield
        return

    unpatched = nn.Module.register_parameter

    def register_and_rehome(module, name, param):
        unpatched(module, name, param)
        slot = module._parameters.get(name)
        if slot is not None:
            module._parameters[name] = _rehome_parameter(slot, device)

    nn.Module.register_parameter = register_and_rehome
    try:
        yield
    finally:
        nn.Module.register_parameter = unpatched


def cpu_offload(
    model: nn.Module,
    execution_device: Optional[torch.device] = None,
    offload_buffers: bool = False,
    state_dict: Optional[Dict[str, torch.Tensor]] = None,
    preload_module_classes: Optional[List[str]] = None,
):
    """Full CPU offload with per-leaf onload hooks (reference: big_modeling.py:179)."""
    if execution_device is None:
        execution_device = next(iter(model.parameters())).device
    if state_dict is None:
        state_dict = {n: p.to("cpu") for n, p in model.state_dict().items()}

    add_hook_to_module(model, AlignDevicesHook(io_same_device=True), append=True)
    attach_align_device_hook(
        model,
        execution_device=execution_device,
        offload=True,
        offload_buffers=offload_buffers,
        weights_map=state_dict,
        preload_module_classes=preload_module_classes,
    )
    return model


def cpu_offload_with_hook(
    model: nn.Module,
    execution_device: Optional[Union[int, str, torch.device]] = None,
    prev_module_hook: Optional[UserCpuOffloadHook] = None,
):
    """Offload model to CPU, onloading the WHOLE model on forward; returns a
    handle to chain pipelines (reference: big_modeling.py:225)."""
    hook = CpuOffload(execution_device=execution_device, prev_module_hook=prev_module_hook)
    add_hook_to_module(model, hook, append=True)
    user_hook = UserCpuOffloadHook(model, hook)
    return model, user_hook


def disk_offload(
    model: nn.Module,
    offload_dir: Union[str, os.PathLike],
    execution_device: Optional[torch.device] = None,
    offload_buffers: bool = False,
    preload_module_classes: Optional[List[str]] = None,
):
    """Offload all weights to memmaps on disk (reference: big_modeling.py:269)."""
    if not os.path.isdir(offload_dir) or not os.path.isfile(os.path.join(offload_dir, "index.json")):
        offload_state_dict(offload_dir, model.state_dict())
    if execution_device is None:
        execution_device = next(iter(model.parameters())).device
    weights_map = OffloadedWeightsLoader(save_folder=offload_dir)

    add_hook_to_module(model, AlignDevicesHook(io_same_device=True), append=True)
    attach_align_device_hook(
        model,
        execution_device=execution_device,
        offload=True,
        offload_buffers=offload_buffers,
        weights_map=weights_map,
        preload_module_classes=preload_module_classes,
    )
    return model


def dispatch_model(
    model: nn.Module,
    device_map: Dict[str, Union[int, str, torch.device]],
    main_device: Optional[torch.device] = None,
    state_dict: Optional[Dict[str, torch.Tensor]] = None,
    offload_dir: Optional[Union[str, os.PathLike]] = None,
    offload_index: Optional[Dict[str, str]] = None,
    offload_buffers: bool = False,
    skip_keys: Optional[Union[str, List[str]]] = None,
    preload_module_classes: Optional[List[str]] = None,
    force_hooks: bool = False,
):
    """Dispatch a model across devices per the device map
    (reference: big_modeling.py:315-517)."""
    check_os_kernel()

    # Error early if the device map is incomplete.
    check_device_map(model, device_map)

    # If we only have one device, we can simply .to() the model
    if len(set(device_map.values())) == 1 and not force_hooks:
        device = list(device_map.values())[0]
        if device != "disk":
            return model.to(device)
        raise ValueError("You can't offload the whole model to disk without hooks; pass force_hooks=True.")

    if main_device is None:
        if set(device_map.values()) == {"cpu"} or set(device_map.values()) == {"cpu", "disk"}:
            main_device = "cpu"
        else:
            main_device = [d for d in device_map.values() if d not in ["cpu", "disk"]][0]

    if main_device != "cpu":
        cpu_modules = [name for name, device in device_map.items() if device == "cpu"]
        if state_dict is None and len(cpu_modules) > 0:
            state_dict = extract_submodules_state_dict(model.state_dict(), cpu_modules)

    disk_modules = [name for name, device in device_map.items() if device == "disk"]
    if offload_dir is None and offload_index is None and len(disk_modules) > 0:
        raise ValueError(
            "We need an `offload_dir` to dispatch this model according to this `device_map`, the following submodules "
            f"need to be offloaded: {', '.join(disk_modules)}."
        )
    if len(disk_modules) > 0 and offload_index is None:
        if not os.path.isdir(offload_dir) or not os.path.isfile(os.path.join(offload_dir, "index.json")):
            disk_state_dict = extract_submodules_state_dict(model.state_dict(), disk_modules)
            offload_state_dict(offload_dir, disk_state_dict)

    execution_device = {
        name: main_device if device in ["cpu", "disk"] else device for name, device in device_map.items()
    }
    execution_device[""] = main_device
    offloaded_devices = ["disk"] if main_device == "cpu" else ["cpu", "disk"]
    offload = {name: device in offloaded_devices for name, device in device_map.items()}
    save_folder = offload_dir if len(disk_modules) > 0 else None
    if state_dict is not None or save_folder is not None or offload_index is not None:
        device = main_device if offload_index is not None else None
        weights_map = OffloadedWeightsLoader(
            state_dict=state_dict, save_folder=save_folder, index=offload_index, device=device
        )
    else:
        weights_map = None

    # When dispatching the model's parameters to the devices specified in device_map, we want to avoid allocating memory several times for the
    # tied parameters. The dictionary tied_params_map keeps track of the already allocated data for a given tied parameter (represented by its
    # original pointer) on each device.
    tied_params = find_tied_parameters(model)
    tied_params_map = {}
    for group in tied_params:
        for param_name in group:
            # data_ptr() is enough here, as `find_tied_parameters` finds tied params simply by comparing `param1 is param2`
            obj = model
            for part in param_name.split("."):
                obj = getattr(obj, part)
            tied_params_map[obj.data_ptr()] = {}

    attach_align_device_hook_on_blocks(
        model,
        execution_device=execution_device,
        offload=offload,
        offload_buffers=offload_buffers,
        weights_map=weights_map,
        skip_keys=skip_keys,
        preload_module_classes=preload_module_classes,
        tied_params_map=tied_params_map,
    )

    # warn if any params are on meta without offload
    offloaded_devices_str = " and ".join([device for device in set(device_map.values()) if device in ("cpu", "disk")])
    if len(offloaded_devices_str) > 0:
        logger.warning(f"Some parameters are on the meta device because they were offloaded to the {offloaded_devices_str}.")

    # Attach a .to() poison and hf_device_map for downstream awareness
    model.hf_device_map = dict(device_map)

    def add_warning(fn, model):
        @wraps(fn)
        def wrapper(*args, **kwargs):
            warning_msg = "You shouldn't move a model that is dispatched using accelerate hooks."
            if str(fn.__name__) == "to":
                to_device = torch._C._nn._parse_to(*args, **kwargs)[0]
                if to_device is not None:
                    logger.warning(warning_msg)
            else:
                logger.warning(warning_msg)
            for param in model.parameters():
                if param.device == torch.device("meta"):
                    raise RuntimeError("You can't move a model that has some modules offloaded to cpu or disk.")
            return fn(*args, **kwargs)

        return wrapper

    # Make sure to update _accelerate_added_attributes in hooks.py if you add any hook
    model.to = add_warning(model.to, model)
    if is_hip_available():
        model.cuda = add_warning(model.cuda, model)

    retie_parameters(model, tied_params)
    return model


def load_checkpoint_and_dispatch(
    model: nn.Module,
    checkpoint: Union[str, os.PathLike],
    device_map: Optional[Union[str, Dict[str, Union[int, str, torch.device]]]] = None,
    max_memory: Optional[Dict[Union[int, str], Union[int, str]]] = None,
    no_split_module_classes: Optional[List[str]] = None,
    offload_folder: Optional[Union[str, os.PathLike]] = None,
    offload_buffers: bool = False,
    dtype: Optional[Union[str, torch.dtype]] = None,
    offload_state_dict: Optional[bool] = None,
    skip_keys: Optional[Union[str, List[str]]] = None,
    preload_module_classes: Optional[List[str]] = None,
    force_hooks: bool = False,
    strict: bool = False,
):
    """Three phases (reference behavior: big_modeling.py:520): resolve a
    device map from its policy name, stream checkpoint shards into place
    (offload writes included), then attach dispatch hooks."""
    _POLICIES = ("auto", "balanced", "balanced_low_0", "sequential")
    if isinstance(device_map, str):
        if device_map not in _POLICIES:
            raise ValueError(
                f"unknown device_map policy {device_map!r}; expected one of "
                f"{', '.join(_POLICIES)} (or an explicit dict)"
            )
        budget = max_memory
        if device_map != "sequential":
            # balanced policies pre-split the memory budget evenly so the
            # greedy allocator fills devices at matching rates
            budget = get_balanced_memory(
                model,
                max_memory=max_memory,
                no_split_module_classes=no_split_module_classes,
                dtype=dtype,
                low_zero=(device_map == "balanced_low_0"),
            )
        device_map = infer_auto_device_map(
            model,
            max_memory=budget,
            no_split_module_classes=no_split_module_classes,
            dtype=dtype,
            offload_buffers=offload_buffers,
        )

    spills_to_disk = device_map is not None and "disk" in device_map.values()
    load_checkpoint_in_model(
        model,
        checkpoint,
        device_map=device_map,
        offload_folder=offload_folder,
        dtype=dtype,
        offload_state_dict=spills_to_disk if offload_state_dict is None else offload_state_dict,
        offload_buffers=offload_buffers,
        strict=strict,
    )
    if device_map is None:
        return model
    return dispatch_model(
        model,
        device_map=device_map,
        offload_dir=offload_folder,
        offload_buffers=offload_buffers,
        skip_keys=skip_keys,
        preload_module_classes=preload_module_classes,
        force_hooks=force_hooks,
    )


def attach_layerwise_casting_hooks(
    module: nn.Module,
    storage_dtype: torch.dtype,
    compute_dtype: torch.dtype,
    skip_modules_pattern=None,
    skip_modules_classes=None,
    non_blocking: bool = False,
    _prefix: str = "",
):
    """(reference: big_modeling.py:661)"""
    import re

    should_skip = (skip_modules_classes is not None and isinstance(module, skip_modules_classes)) or (
        skip_modules_pattern is not None and any(re.search(p, _prefix) for p in skip_modules_pattern)
    )
    if should_skip:
        return
    if isinstance(module, (nn.Linear, nn.Conv1d, nn.Conv2d, nn.Conv3d, nn.Embedding, nn.LayerNorm)) or (
        len(list(module.children())) == 0 and len(list(module.parameters(recurse=False))) > 0
    ):
        add_hook_to_module(
            module, LayerwiseCastingHook(storage_dtype=storage_dtype, compute_dtype=compute_dtype, non_blocking=non_blocking),
            append=True,
        )
        return
    for name, child in module.named_children():
        child_prefix = f"{_prefix}.{name}" if _prefix else name
        attach_layerwise_casting_hooks(
            child, storage_dtype, compute_dtype, skip_modules_pattern, skip_modules_classes, non_blocking, child_prefix
        )
