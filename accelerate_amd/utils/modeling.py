"""Model introspection, device-map solving and checkpoint IO
(reference: utils/modeling.py — same behavior contract, our own
implementation, with memory constants sized for MI355X's 288 GB HBM3E).
"""

import json
import os
import re
import shutil
import tempfile
from collections import defaultdict
from contextlib import contextmanager
from typing import Dict, List, Optional, Set, Tuple, Union

import torch
import torch.nn as nn

from .constants import SAFE_WEIGHTS_INDEX_NAME, SAFE_WEIGHTS_NAME, WEIGHTS_INDEX_NAME, WEIGHTS_NAME
from .imports import is_safetensors_available
from ..logging import get_logger

logger = get_logger(__name__)

WEIGHTS_INDEX_PATTERN = r"pytorch_model\.bin\.index\.json"


def dtype_byte_size(dtype: torch.dtype) -> float:
    """Bytes per element of ``dtype`` (reference: modeling.py)."""
    if dtype == torch.bool:
        return 1 / 8
    if dtype in (torch.float8_e4m3fn, torch.float8_e5m2, torch.int8, torch.uint8):
        return 1
    bit_search = re.search(r"[^\d](\d+)(_|$)", str(dtype))
    if bit_search is None:
        raise ValueError(f"`dtype` is not a valid dtype: {dtype}.")
    return int(bit_search.groups()[0]) / 8


def id_tensor_storage(tensor: torch.Tensor):
    """Unique id of the storage backing a tensor."""
    return tensor.device, tensor.untyped_storage().data_ptr(), tensor.untyped_storage().nbytes()


def named_module_tensors(module: nn.Module, include_buffers: bool = True, recurse: bool = False,
                         remove_non_persistent: bool = False):
    """Yield (name, tensor) over parameters (and buffers) of a module."""
    yield from module.named_parameters(recurse=recurse)
    if include_buffers:
        non_persistent = get_non_persistent_buffers(module, recurse=recurse) if remove_non_persistent else set()
        for name, buf in module.named_buffers(recurse=recurse):
            if name not in non_persistent:
                yield name, buf


def get_non_persistent_buffers(module: nn.Module, recurse: bool = False) -> Set[str]:
    non_persistent = set(module._non_persistent_buffers_set)
    if recurse:
        for name, mod in module.named_modules():
            if name == "":
                continue
            non_persistent |= {f"{name}.{b}" for b in mod._non_persistent_buffers_set}
    return non_persistent


def find_tied_parameters(model: nn.Module) -> List[List[str]]:
    """Groups of parameter names sharing the same storage
    (reference: modeling.py:567)."""
    all_named_parameters = dict(model.named_parameters(remove_duplicate=False))
    no_duplicate_named_parameters = dict(model.named_parameters(remove_duplicate=True))
    tied_param_groups = defaultdict(list)
    for name, param in all_named_parameters.items():
        if name not in no_duplicate_named_parameters:
            # find the canonical name
            for main_name, main_param in no_duplicate_named_parameters.items():
                if main_param is param:
                    tied_param_groups[main_name].append(name)
                    break
    return [sorted([name] + tied) for name, tied in tied_param_groups.items()]


def retie_parameters(model: nn.Module, tied_params: List[List[str]]):
    """Re-tie parameter groups after dispatch (reference: modeling.py:622)."""
    for tied_group in tied_params:
        param_to_tie = None
        # first find a non-meta param to act as source
        for param_name in tied_group:
            module = model
            splits = param_name.split(".")
            for split in splits[:-1]:
                module = getattr(module, split)
            param = getattr(module, splits[-1])
            if param_to_tie is None and param.device != torch.device("meta"):
                param_to_tie = param
                break
        if param_to_tie is not None:
            for param_name in tied_group:
                module = model
                splits = param_name.split(".")
                for split in splits[:-1]:
                    module = getattr(module, split)
                setattr(module, splits[-1], param_to_tie)


def _get_proper_dtype(dtype: Union[str, torch.dtype]) -> torch.dtype:
    if isinstance(dtype, str):
        dtype = dtype.replace("torch.", "")
        dtype = getattr(torch, dtype)
    return dtype


def compute_module_sizes(model: nn.Module, dtype: Optional[Union[str, torch.dtype]] = None,
                         special_dtypes: Optional[Dict[str, torch.dtype]] = None,
                         buffers_only: bool = False) -> Dict[str, int]:
    """Size in bytes of each submodule (key "" = whole model)
    (reference: modeling.py:664)."""
    if dtype is not None:
        dtype = _get_proper_dtype(dtype)
        dtype_size = dtype_byte_size(dtype)
    if special_dtypes is not None:
        special_dtypes = {k: _get_proper_dtype(v) for k, v in special_dtypes.items()}
        special_dtypes_size = {k: dtype_byte_size(v) for k, v in special_dtypes.items()}
    module_sizes = defaultdict(int)
    if buffers_only:
        module_list = model.named_buffers(recurse=True)
    else:
        module_list = named_module_tensors(model, recurse=True)
    for name, tensor in module_list:
        if special_dtypes is not None and name in special_dtypes:
            size = tensor.numel() * special_dtypes_size[name]
        elif dtype is None:
            size = tensor.numel() * dtype_byte_size(tensor.dtype)
        elif str(tensor.dtype).startswith(("torch.uint", "torch.int", "torch.bool")):
            size = tensor.numel() * dtype_byte_size(tensor.dtype)
        else:
            size = tensor.numel() * min(dtype_size, dtype_byte_size(tensor.dtype))
        name_parts = name.split(".")
        for idx in range(len(name_parts) + 1):
            module_sizes[".".join(name_parts[:idx])] += size
    return module_sizes


def compute_module_total_buffer_size(model, dtype=None, special_dtypes=None) -> int:
    module_sizes = compute_module_sizes(model, dtype=dtype, special_dtypes=special_dtypes, buffers_only=True)
    return module_sizes.get("", 0)


def get_max_layer_size(modules: List[Tuple[str, nn.Module]], module_sizes: Dict[str, int],
                       no_split_module_classes: List[str]) -> Tuple[int, List[str]]:
    """Largest 'layer' (no-split block or leaf) and the names attaining it
    (reference: modeling.py get_max_layer_size)."""
    max_size = 0
    layer_names = []
    modules_to_treat = modules.copy()
    while len(modules_to_treat) > 0:
        module_name, module = modules_to_treat.pop(0)
        modules_children = list(module.named_children()) if isinstance(module, nn.Module) else []
        if len(modules_children) == 0 or module.__class__.__name__ in no_split_module_classes:
            size = module_sizes[module_name]
            if size > max_size:
                max_size = size
                layer_names = [module_name]
            elif size == max_size:
                layer_names.append(module_name)
        else:
            modules_to_treat = [(f"{module_name}.{n}", v) for n, v in modules_children] + modules_to_treat
    return max_size, layer_names


def convert_file_size_to_int(size: Union[int, str]) -> int:
    if isinstance(size, int):
        return size
    mult = {"GB": 10**9, "MB": 10**6, "KB": 10**3, "GIB": 2**30, "MIB": 2**20, "KIB": 2**10}
    size = size.upper().strip()
    for suffix, m in sorted(mult.items(), key=lambda kv: -len(kv[0])):
        if size.endswith(suffix):
            return int(float(size[: -len(suffix)]) * m)
    if size.endswith("B"):
        return int(size[:-1])
    return int(size)


def get_max_memory(max_memory: Optional[Dict[Union[int, str], Union[int, str]]] = None):
    """Per-device memory budget. Defaults: ~95% of each MI355X's 288 GB
    HBM3E (the allocator reserve is small on CDNA) and all-but-2 GiB of host
    RAM (reference: modeling.py:757 uses 90%/CPU-virtual)."""
    import psutil

    if max_memory is None:
        max_memory = {}
        if torch.cuda.is_available():
            for i in range(torch.cuda.device_count()):
                try:
                    _ = torch.tensor([0], device=i)
                    free, total = torch.cuda.mem_get_info(i)
                    max_memory[i] = int(0.95 * free)
                except Exception:
                    continue
        max_memory["cpu"] = max(0, psutil.virtual_memory().available - 2 * 2**30)
        return max_memory

    for key in list(max_memory.keys()):
        if isinstance(max_memory[key], str):
            max_memory[key] = convert_file_size_to_int(max_memory[key])
    # sort gpu keys numerically, cpu/disk last
    gpu_keys = sorted(k for k in max_memory if isinstance(k, int))
    other_keys = [k for k in max_memory if not isinstance(k, int)]
    return {**{k: max_memory[k] for k in gpu_keys}, **{k: max_memory[k] for k in other_keys}}


def get_balanced_memory(model: nn.Module, max_memory=None, no_split_module_classes=None,
                        dtype=None, special_dtypes=None, low_zero: bool = False):
    """Even split across GPUs with leaf-size buffer (reference: modeling.py:931)."""
    user_not_set_max_memory = max_memory is None
    max_memory = get_max_memory(max_memory)
    if no_split_module_classes is None:
        no_split_module_classes = []
    elif not isinstance(no_split_module_classes, (list, tuple)):
        no_split_module_classes = [no_split_module_classes]

    num_devices = len([d for d in max_memory if max_memory[d] > 0 and isinstance(d, int)])
    if num_devices == 0:
        return max_memory
    if num_devices == 1:
        low_zero = False
        if user_not_set_max_memory:
            for key in max_memory.keys():
                if isinstance(key, int):
                    max_memory[key] *= 0.9  # 90% to leave room for activations

    module_sizes = compute_module_sizes(model, dtype=dtype, special_dtypes=special_dtypes)
    per_gpu = module_sizes[""] // (num_devices - 1 if low_zero else num_devices)

    # buffer: biggest × 1.25 of the leaf-module sizes so the greedy allocator
    # never gets stuck on a large tail block (reference: modeling.py:1010-1035)
    leaves = [n for n in module_sizes if len(n) > 0 and len(list(_get_module(model, n).named_children())) == 0]
    mean_leaves = int(sum(module_sizes[n] for n in leaves) / max(len(leaves), 1))
    buffer = int(1.25 * max([module_sizes[n] for n in leaves], default=0))
    per_gpu += buffer

    max_memory = get_max_memory(max_memory) if user_not_set_max_memory else max_memory
    gpus_idx_list = sorted(k for k in max_memory if isinstance(k, int) and max_memory[k] > 0)
    for idx in gpus_idx_list[:-1]:
        max_memory[idx] = min(max_memory[0] if low_zero and idx == 0 else per_gpu, max_memory[idx])
    if low_zero:
        min_zero = max(0, module_sizes[""] - sum(max_memory[i] for i in range(1, num_devices)))
        max_memory[0] = min(min_zero, max_memory[0])
    return max_memory


def _get_module(model, name):
    module = model
    if name == "":
        return model
    for part in name.split("."):
        module = getattr(module, part)
    return module


def check_device_map(model: nn.Module, device_map: Dict[str, Union[int, str, torch.device]]):
    """Every tensor of the model must be covered by the map
    (reference: modeling.py check_device_map)."""
    all_model_tensors = [name for name, _ in model.state_dict().items()]
    for module_name in device_map.keys():
        if module_name == "":
            all_model_tensors.clear()
            break
        all_model_tensors = [
            name for name in all_model_tensors if not name.startswith(module_name + ".") and name != module_name
        ]
    if len(all_model_tensors) > 0:
        non_covered = ", ".join(all_model_tensors[:10])
        raise ValueError(
            f"The device_map provided does not give any device for the following parameters: {non_covered}"
        )


def infer_auto_device_map(
    model: nn.Module,
    max_memory: Optional[Dict] = None,
    no_split_module_classes: Optional[List[str]] = None,
    dtype: Optional[Union[str, torch.dtype]] = None,
    special_dtypes: Optional[Dict[str, torch.dtype]] = None,
    verbose: bool = False,
    clean_result: bool = True,
    offload_buffers: bool = False,
) -> Dict[str, Union[int, str]]:
    """Greedy device-map allocator (reference: modeling.py:1295-1601).

    Walks top-level modules in order, filling GPUs first (in index order),
    then CPU, then disk. On every GPU except the last we reserve headroom for
    the largest layer still to come, so during offloaded inference the
    largest offloaded block can be onloaded for execution. Tied parameters
    are placed together with their first-seen owner.
    """
    max_memory = get_max_memory(max_memory)
    if no_split_module_classes is None:
        no_split_module_classes = []
    elif not isinstance(no_split_module_classes, (list, tuple)):
        no_split_module_classes = [no_split_module_classes]

    devices = list(max_memory.keys())
    if "disk" not in devices:
        devices.append("disk")
    gpus = [d for d in devices if d not in ("cpu", "disk")]

    # Devices that need to keep space for a potential offloaded layer:
    # every gpu except the last one if cpu/disk offload could happen
    if "mps" in gpus:
        main_devices = ["mps"]
    elif len(gpus) > 0:
        main_devices = [gpus[0], "cpu"]
    else:
        main_devices = ["cpu"]

    module_sizes = compute_module_sizes(model, dtype=dtype, special_dtypes=special_dtypes)
    tied_parameters = find_tied_parameters(model)

    device_map: Dict[str, Union[int, str]] = {}
    current_device = 0
    current_memory_used = 0
    device_memory_used: Dict = {}
    device_buffer_sizes: Dict = {}

    modules_to_treat = (
        [(n, p) for n, p in model.named_parameters(recurse=False)]
        + [(n, m) for n, m in model.named_children()]
        + [(n, b) for n, b in model.named_buffers(recurse=False)]
    )

    # largest layer reservation
    max_layer_size, max_layer_names = get_max_layer_size(
        [(n, m) for n, m in model.named_children() if isinstance(m, nn.Module)], module_sizes, no_split_module_classes
    )

    while len(modules_to_treat) > 0:
        name, module = modules_to_treat.pop(0)
        if verbose:
            print(f"\nTreating module {name}.")
        # max size in the remaining layers may have changed since we took one, so we reset it
        max_layer_names = [n for n in max_layer_names if n != name and not n.startswith(name + ".")]
        if len(max_layer_names) == 0:
            max_layer_size, max_layer_names = get_max_layer_size(
                [(n, m) for n, m in modules_to_treat if isinstance(m, nn.Module)],
                module_sizes,
                no_split_module_classes,
            )

        module_size = module_sizes[name]

        # tied params handling: find tied groups with one member inside this module
        tied_param_groups = [
            group for group in tied_parameters
            if any(p.startswith(name + ".") or p == name for p in group)
            and not all(p.startswith(name + ".") or p == name for p in group)
        ]
        tied_params_outside = sum(
            [[p for p in group if not (p.startswith(name + ".") or p == name)] for group in tied_param_groups], []
        )

        device = devices[current_device]
        current_max_size = max_memory[device] if device != "disk" else None
        current_memory_reserved = 0
        # Reserve headroom for the largest remaining layer on the main device
        if devices[current_device] in main_devices:
            current_max_size = current_max_size - max_layer_size if current_max_size is not None else None
            current_memory_reserved = max_layer_size

        module_size_with_ties = module_size + sum(module_sizes.get(p, 0) for p in tied_params_outside)

        if current_max_size is not None and current_memory_used + module_size_with_ties > current_max_size:
            # try to split
            modules_children = (
                [] if isinstance(module, (nn.Parameter, torch.Tensor)) else list(module.named_children())
            )
            if len(modules_children) == 0 or module.__class__.__name__ in no_split_module_classes:
                # can't split: next device
                if verbose:
                    print(
                        f"Not enough space on {devices[current_device]} to put {name} "
                        f"(space available {current_max_size - current_memory_used}, module size {module_size_with_ties})."
                    )
                device_memory_used[device] = current_memory_used + current_memory_reserved
                current_device += 1
                modules_to_treat = [(name, module)] + modules_to_treat
                current_memory_used = 0
            else:
                if verbose:
                    print(f"Splitting {name}.")
                modules_children = list(module.named_parameters(recurse=False)) + modules_children
                modules_to_treat = [(f"{name}.{n}", v) for n, v in modules_children] + modules_to_treat
                # Update the max layer size.
                max_layer_size, max_layer_names = get_max_layer_size(
                    [(n, m) for n, m in modules_to_treat if isinstance(m, nn.Module)],
                    module_sizes,
                    no_split_module_classes,
                )
        else:
            # placement succeeds; place tied params here too
            if verbose:
                print(f"Putting {name} (size={module_size_with_ties}) on {devices[current_device]}.")
            current_memory_used += module_size_with_ties
            device_memory_used[device] = current_memory_used + current_memory_reserved
            device_map[name] = devices[current_device]
            for p in tied_params_outside:
                device_map[p] = devices[current_device]
                # remove tied modules from the queue (they're placed)
                modules_to_treat = [
                    (n, m) for n, m in modules_to_treat if n != p and not p.startswith(n + ".")
                ]
            if not offload_buffers and isinstance(module, nn.Module):
                current_buffer_size = compute_module_total_buffer_size(module, dtype=dtype, special_dtypes=special_dtypes)
                device_buffer_sizes[device] = device_buffer_sizes.get(device, 0) + current_buffer_size

    if clean_result:
        device_map = clean_device_map(device_map)
    return device_map


def clean_device_map(device_map: Dict[str, Union[int, str]], module_name: str = ""):
    """Collapse children all on the same device into their parent entry
    (reference: modeling.py clean_device_map)."""
    prefix = "" if module_name == "" else f"{module_name}."
    values = [v for k, v in device_map.items() if k.startswith(prefix)]
    if len(set(values)) == 1 and len(values) > 1:
        for k in [k for k in device_map if k.startswith(prefix)]:
            del device_map[k]
        device_map[module_name] = values[0]
    # Recurse over the children
    children_modules = [k for k in device_map.keys() if k.startswith(prefix) and len(k) > len(module_name)]
    idx = len(module_name.split(".")) + 1 if len(module_name) > 0 else 1
    children_modules = set(".".join(k.split(".")[:idx]) for k in children_modules)
    for child in children_modules:
        clean_device_map(device_map, module_name=child)
    return device_map


def set_module_tensor_to_device(
    module: nn.Module,
    tensor_name: str,
    device: Union[int, str, torch.device],
    value: Optional[torch.Tensor] = None,
    dtype: Optional[Union[str, torch.dtype]] = None,
    fp16_statistics=None,
    tied_params_map: Optional[Dict[int, Dict[torch.device, torch.Tensor]]] = None,
):
    """Move/assign one tensor of a module to a device, handling meta init
    (reference: modeling.py:227-438)."""
    if "." in tensor_name:
        splits = tensor_name.split(".")
        for split in splits[:-1]:
            new_module = getattr(module, split)
            if new_module is None:
                raise ValueError(f"{module} has no attribute {split}.")
            module = new_module
        tensor_name = splits[-1]

    if tensor_name not in module._parameters and tensor_name not in module._buffers:
        raise ValueError(f"{module} does not have a parameter or a buffer named {tensor_name}.")
    is_buffer = tensor_name in module._buffers
    old_value = getattr(module, tensor_name)

    # tied-pointer dedup: reuse an already-placed tensor for the same storage
    if (
        value is not None
        and tied_params_map is not None
        and value.data_ptr() in tied_params_map
        and device in tied_params_map[value.data_ptr()]
    ):
        module._parameters[tensor_name] = tied_params_map[value.data_ptr()][device]
        return
    elif (
        tied_params_map is not None
        and old_value.data_ptr() in tied_params_map
        and device in tied_params_map[old_value.data_ptr()]
    ):
        if is_buffer:
            module._buffers[tensor_name] = tied_params_map[old_value.data_ptr()][device]
        else:
            module._parameters[tensor_name] = tied_params_map[old_value.data_ptr()][device]
        return

    if old_value.device == torch.device("meta") and device not in ["meta", torch.device("meta")] and value is None:
        raise ValueError(f"{tensor_name} is on the meta device, we need a `value` to put in on {device}.")

    param = module._parameters[tensor_name] if tensor_name in module._parameters else None
    if value is not None:
        if dtype is None:
            # For compatibility with PyTorch load_state_dict which converts state dict dtype to existing dtype in model
            value = value.to(old_value.dtype)
        elif not str(value.dtype).startswith(("torch.uint", "torch.int", "torch.bool")):
            value = value.to(dtype)

    device_quantization = None
    with torch.no_grad():
        if value is None:
            new_value = old_value.to(device)
            if dtype is not None and device in ["meta", torch.device("meta")]:
                if not str(old_value.dtype).startswith(("torch.uint", "torch.int", "torch.bool")):
                    new_value = new_value.to(dtype)
                if not is_buffer:
                    module._parameters[tensor_name] = type(param)(new_value, requires_grad=old_value.requires_grad)
        elif isinstance(value, torch.Tensor):
            new_value = value.to(device)
        else:
            new_value = torch.tensor(value, device=device)
        if is_buffer:
            module._buffers[tensor_name] = new_value
        elif value is not None or torch.device(device) != module._parameters[tensor_name].device:
            param_cls = type(module._parameters[tensor_name])
            kwargs = module._parameters[tensor_name].__dict__
            new_value = param_cls(new_value, requires_grad=old_value.requires_grad)
            module._parameters[tensor_name] = new_value

        # register in the tied-params map so later placements reuse this tensor
        if (
            tied_params_map is not None
            and old_value.data_ptr() in tied_params_map
        ):
            tied_params_map[old_value.data_ptr()][device] = new_value
        if (
            value is not None
            and tied_params_map is not None
            and value.data_ptr() in tied_params_map
        ):
            tied_params_map[value.data_ptr()][device] = new_value


def get_state_dict_offloaded_model(model: nn.Module) -> Dict[str, torch.Tensor]:
    """Full state dict of a dispatched model with offloaded parts pulled in
    (reference: modeling.py:1732)."""
    from ..hooks import AlignDevicesHook

    state_dict = {}
    placeholders = set()
    for name, module in model.named_modules():
        if name == "":
            continue
        if hasattr(module, "_hf_hook") and isinstance(module._hf_hook, AlignDevicesHook) and module._hf_hook.offload:
            original_device = module._hf_hook.execution_device
            # assign the device to which the offloaded parameters will be sent
            module._hf_hook.execution_device = "cpu"
            module._hf_hook.pre_forward(module)
            for key, param in module.state_dict().items():
                full_key = f"{name}.{key}"
                state_dict[full_key] = param.clone().cpu()
            module._hf_hook.post_forward(module, None)
            module._hf_hook.execution_device = original_device
    # non-offloaded tensors
    for key, param in model.state_dict().items():
        if key not in state_dict:
            if param.device == torch.device("meta"):
                placeholders.add(key)
                continue
            state_dict[key] = param.cpu() if param.device.type != "cpu" else param
    for key in list(placeholders):
        if key in state_dict:
            placeholders.remove(key)
    if placeholders:
        logger.warning(f"The following tensors were not saved because they were still on meta device: {placeholders}")
    return state_dict


def load_state_dict(checkpoint_file, device_map=None):
    """Load a (safetensors or torch) checkpoint file, optionally only the
    weights needed per device (reference: modeling.py:1637)."""
    if str(checkpoint_file).endswith(".safetensors"):
        import safetensors.torch

        return safetensors.torch.load_file(checkpoint_file, device="cpu")
    return torch.load(checkpoint_file, map_location="cpu", weights_only=True)


def load_checkpoint_in_model(
    model: nn.Module,
    checkpoint: Union[str, os.PathLike],
    device_map: Optional[Dict[str, Union[int, str]]] = None,
    offload_folder: Optional[Union[str, os.PathLike]] = None,
    dtype: Optional[Union[str, torch.dtype]] = None,
    offload_state_dict: bool = False,
    offload_buffers: bool = False,
    keep_in_fp32_modules: Optional[List[str]] = None,
    strict: bool = False,
    full_state_dict: bool = True,
):
    """Load a (possibly sharded) checkpoint into a (possibly meta) model,
    honoring a device map with disk offload (reference: modeling.py:1805)."""
    if offload_folder is None and device_map is not None and "disk" in device_map.values():
        raise ValueError("At least one of the model submodule will be offloaded to disk, please pass along an `offload_folder`.")
    elif offload_folder is not None and device_map is not None and "disk" in device_map.values():
        os.makedirs(offload_folder, exist_ok=True)
    if dtype is not None:
        dtype = _get_proper_dtype(dtype)

    checkpoint_files = None
    index_filename = None
    checkpoint = str(checkpoint)
    if os.path.isfile(checkpoint):
        if checkpoint.endswith(".index.json"):
            index_filename = checkpoint
        else:
            checkpoint_files = [checkpoint]
    elif os.path.isdir(checkpoint):
        # check for index first
        potential_index = [f for f in os.listdir(checkpoint) if f.endswith(".index.json")]
        potential_safetensors = [f for f in os.listdir(checkpoint) if f.endswith(".safetensors")]
        potential_bin = [f for f in os.listdir(checkpoint) if f.endswith(".bin")]
        if len(potential_index) == 1:
            index_filename = os.path.join(checkpoint, potential_index[0])
        elif len(potential_safetensors) > 0:
            checkpoint_files = [os.path.join(checkpoint, f) for f in sorted(potential_safetensors)]
        elif len(potential_bin) > 0:
            checkpoint_files = [os.path.join(checkpoint, f) for f in sorted(potential_bin)]
        else:
            raise ValueError(f"{checkpoint} containing no .index.json, .safetensors or .bin files can't be loaded.")
    else:
        raise ValueError(f"`checkpoint` should be the path to a file or directory, got {checkpoint}.")

    if index_filename is not None:
        checkpoint_folder = os.path.split(index_filename)[0]
        with open(index_filename) as f:
            index = json.load(f)
        if "weight_map" in index:
            index = index["weight_map"]
        checkpoint_files = sorted(set(index.values()))
        checkpoint_files = [os.path.join(checkpoint_folder, f) for f in checkpoint_files]

    from .offload import offload_weight, save_offload_index

    offload_index = {}
    buffer_names = [name for name, _ in model.named_buffers()]
    tied_params = find_tied_parameters(model)
    for checkpoint_file in checkpoint_files:
        state_dict = load_state_dict(checkpoint_file)
        for param_name, param in state_dict.items():
            if param_name not in dict(model.named_parameters()) and param_name not in dict(model.named_buffers()):
                if strict:
                    raise RuntimeError(f"Unexpected key {param_name} in checkpoint")
                continue
            module_name = param_name
            while device_map is not None and module_name not in device_map:
                if "." not in module_name:
                    module_name = ""
                    break
                module_name = module_name.rsplit(".", 1)[0]
            param_device = device_map[module_name] if device_map is not None else 0 if torch.cuda.is_available() else "cpu"
            target_dtype = dtype
            if (
                keep_in_fp32_modules is not None
                and dtype == torch.float16
                and any(m in param_name for m in keep_in_fp32_modules)
            ):
                target_dtype = torch.float32

            if param_device == "disk":
                if offload_buffers or param_name not in buffer_names:
                    set_module_tensor_to_device(model, param_name, "meta", dtype=target_dtype)
                    offload_weight(param, param_name, offload_folder, index=offload_index)
                else:
                    set_module_tensor_to_device(model, param_name, "cpu", value=param, dtype=target_dtype)
            elif param_device == "cpu" and offload_state_dict:
                set_module_tensor_to_device(model, param_name, "meta", dtype=target_dtype)
                offload_weight(param, param_name, offload_folder, index=offload_index)
            else:
                set_module_tensor_to_device(model, param_name, param_device, value=param, dtype=target_dtype)
        del state_dict

    if len(offload_index) > 0 and offload_folder is not None:
        save_offload_index(offload_index, offload_folder)
    retie_parameters(model, tied_params)


@contextmanager
def align_module_device(module: nn.Module, execution_device: Optional[torch.device] = None):
    """Context manager moving a (hooked or plain) module's params to its
    execution device (reference: modeling.py:2167)."""
    if hasattr(module, "_hf_hook") and getattr(module._hf_hook, "offload", False):
        if execution_device is not None:
            original_device = module._hf_hook.execution_device
            module._hf_hook.execution_device = execution_device
        try:
            module._hf_hook.pre_forward(module)
            yield
        finally:
            module._hf_hook.post_forward(module, None)
            if execution_device is not None:
                module._hf_hook.execution_device = original_device
    elif execution_device is not None:
        devices = {name: param.device for name, param in module.named_parameters(recurse=False)}
        try:
            for name in devices:
                set_module_tensor_to_device(module, name, execution_device)
            yield
        finally:
            for name, device in devices.items():
                set_module_tensor_to_device(module, name, device)
    else:
        yield


def shard_checkpoint(state_dict: Dict[str, torch.Tensor], max_shard_size: Union[int, str] = "10GB",
                     weights_name: str = SAFE_WEIGHTS_NAME):
    """Split a state dict into shards under max_shard_size; returns
    (shards dict, index or None)."""
    max_shard_size = convert_file_size_to_int(max_shard_size)
    sharded_state_dicts = [{}]
    last_block_size = 0
    storage_ids = {}
    for key, weight in state_dict.items():
        # tied weights go into the shard of their storage owner
        storage_key = id_tensor_storage(weight) if isinstance(weight, torch.Tensor) else None
        if storage_key is not None and storage_key in storage_ids:
            sharded_state_dicts[storage_ids[storage_key]][key] = weight
            continue
        weight_size = weight.numel() * dtype_byte_size(weight.dtype)
        if last_block_size + weight_size > max_shard_size and len(sharded_state_dicts[-1]) > 0:
            sharded_state_dicts.append({})
            last_block_size = 0
        sharded_state_dicts[-1][key] = weight
        last_block_size += weight_size
        if storage_key is not None:
            storage_ids[storage_key] = len(sharded_state_dicts) - 1

    if len(sharded_state_dicts) == 1:
        return {weights_name: sharded_state_dicts[0]}, None
    weight_map = {}
    shards = {}
    for idx, shard in enumerate(sharded_state_dicts):
        shard_file = weights_name.replace(".safetensors", f"-{idx + 1:05d}-of-{len(sharded_state_dicts):05d}.safetensors")
        shard_file = shard_file.replace(".bin", f"-{idx + 1:05d}-of-{len(sharded_state_dicts):05d}.bin")
        shards[shard_file] = shard
        for key in shard.keys():
            weight_map[key] = shard_file
    metadata = {"total_size": sum(
        w.numel() * dtype_byte_size(w.dtype) for sd in sharded_state_dicts for w in sd.values()
    )}
    index = {"metadata": metadata, "weight_map": weight_map}
    return shards, index


def save_model_weights(state_dict, save_directory, max_shard_size="10GB", safe_serialization=True,
                       is_main_process=True):
    """Write sharded weights + index json (reference: accelerator.py:3439)."""
    weights_name = SAFE_WEIGHTS_NAME if safe_serialization else WEIGHTS_NAME
    if safe_serialization:
        # dedup shared tensors: keep one name per storage, drop aliases
        ptrs = defaultdict(list)
        for name, tensor in state_dict.items():
            if isinstance(tensor, torch.Tensor) and tensor.device.type != "meta":
                ptrs[id_tensor_storage(tensor)].append(name)
        shared_ptrs = {ptr: names for ptr, names in ptrs.items() if len(names) > 1}
        for names in shared_ptrs.values():
            for name in names[1:]:
                del state_dict[name]
        state_dict = {k: v.contiguous() if isinstance(v, torch.Tensor) else v for k, v in state_dict.items()}

    shards, index = shard_checkpoint(state_dict, max_shard_size=max_shard_size, weights_name=weights_name)
    if not is_main_process:
        return
    os.makedirs(save_directory, exist_ok=True)
    for shard_file, shard in shards.items():
        path = os.path.join(save_directory, shard_file)
        if safe_serialization:
            import safetensors.torch

            safetensors.torch.save_file(shard, path, metadata={"format": "pt"})
        else:
            torch.save(shard, path)
    if index is not None:
        index_name = SAFE_WEIGHTS_INDEX_NAME if safe_serialization else WEIGHTS_INDEX_NAME
        with open(os.path.join(save_directory, index_name), "w") as f:
            f.write(json.dumps(index, indent=2, sort_keys=True) + "\n")
        logger.info(f"Model weight shards saved in {save_directory} with index {index_name}")
    else:
        logger.info(f"Model weights saved in {os.path.join(save_directory, weights_name)}")


def get_mixed_precision_context_manager(native_amp, mixed_precision, device, autocast_kwargs=None):
    from ..accelerator import get_mixed_precision_context_manager as _impl

    return _impl(native_amp, mixed_precision, device, autocast_kwargs)
