"""Module-hook runtime for big-model dispatch.

Behavior parity with the reference's hooks runtime (reference hooks.py —
ModelHook protocol :58, add_hook_to_module :147, AlignDevicesHook :242,
block attachment :586-717, CPU offload :720, layerwise casting :784),
organized around two small helpers here: ``_TensorWalk`` bundles the
repeated "which tensors does this hook manage" iteration, and the
device-map attachment normalizes its inputs once instead of branching on
scalar-vs-dict shapes throughout.

On MI355X the onload path is the hot loop of offloaded inference: weights
stream H2D from a PINNED host cache with non_blocking copies, and an
execution-order-learning prefetcher (`_OnloadPrefetcher`) onloads the NEXT
block's weights on a side stream while the current block computes; a
tied-pointer map guarantees shared weights cross PCIe once per step.
"""

import functools
import os
from typing import Dict, List, Mapping, Optional, Union

import torch
import torch.nn as nn

from .utils.modeling import (
    get_non_persistent_buffers,
    named_module_tensors,
    set_module_tensor_to_device,
)
from .utils.offload import PrefixedDataset
from .utils.operations import find_device, send_to_device


class _OnloadPrefetcher:
    """Execution-order-learning async H2D weight prefetcher.

    One instance per dispatched model (keyed on the shared weights source).
    On the first forward pass it records the order offloaded hooks fire;
    from the second pass on, when hook i runs it kicks hook i+1's weights
    H2D on a dedicated side stream from PINNED host copies — the next
    block's onload overlaps the current block's compute, and the ring wraps
    so autoregressive decode prefetches block 0 during the last block
    (north-star item; reference contrast: hooks.py:359-399 synchronous
    pageable onload).
    """

    def __init__(self):
        self.stream: Optional[torch.cuda.Stream] = None
        self.order: List["AlignDevicesHook"] = []
        self.position: Dict[int, int] = {}
        self.learning = True

    def note(self, hook: "AlignDevicesHook"):
        """Called at the end of every offloaded pre_forward."""
        if self.learning:
            if id(hook) in self.position:
                self.learning = False  # ring closed: order learned
            else:
                self.position[id(hook)] = len(self.order)
                self.order.append(hook)
        if not self.learning and len(self.order) > 1:
            nxt = self.order[(self.position[id(hook)] + 1) % len(self.order)]
            if nxt is not hook and nxt._prefetched is None:
                nxt._start_prefetch(self._get_stream())

    def _get_stream(self) -> torch.cuda.Stream:
        if self.stream is None:
            self.stream = torch.cuda.Stream()
        return self.stream


_prefetchers: Dict[int, _OnloadPrefetcher] = {}
_pinned_bytes = 0


def _prefetcher_for(weights_map) -> _OnloadPrefetcher:
    source = getattr(weights_map, "dataset", weights_map)  # unwrap PrefixedDataset
    key = id(source)
    if key not in _prefetchers:
        _prefetchers[key] = _OnloadPrefetcher()
    return _prefetchers[key]


def _async_onload_enabled() -> bool:
    return os.environ.get("ACCELERATE_AMD_ASYNC_ONLOAD", "1") == "1" and torch.cuda.is_available()


class ModelHook:
    """The hook protocol: init/pre_forward/post_forward/detach."""

    no_grad = False

    def init_hook(self, module):
        return module

    def pre_forward(self, module, *args, **kwargs):
        return args, kwargs

    def post_forward(self, module, output):
        return output

    def detach_hook(self, module):
        return module


class SequentialHook(ModelHook):
    """Run several hooks as one (composition order = argument order)."""

    def __init__(self, *hooks):
        self.hooks = hooks

    def init_hook(self, module):
        for h in self.hooks:
            module = h.init_hook(module)
        return module

    def pre_forward(self, module, *args, **kwargs):
        for h in self.hooks:
            args, kwargs = h.pre_forward(module, *args, **kwargs)
        return args, kwargs

    def post_forward(self, module, output):
        for h in self.hooks:
            output = h.post_forward(module, output)
        return output

    def detach_hook(self, module):
        for h in self.hooks:
            module = h.detach_hook(module)
        return module


def _install_forward(module, fn, original):
    wrapped = functools.update_wrapper(functools.partial(fn, module), original)
    # fx GraphModule freezes per-instance forward overrides; patch the class
    if "GraphModuleImpl" in str(type(module)):
        module.__class__.forward = wrapped
    else:
        module.forward = wrapped


def add_hook_to_module(module: nn.Module, hook: ModelHook, append: bool = False):
    """Wrap ``module.forward`` so ``hook`` runs around it.

    ``append=True`` composes with an existing hook instead of replacing it.
    The pristine forward is stashed once in ``_old_forward`` and survives
    any number of replacements.
    """
    existing = getattr(module, "_hf_hook", None)
    if append and existing is not None:
        remove_hook_from_module(module)
        hook = SequentialHook(existing, hook)

    if hasattr(module, "_old_forward"):
        pristine = module._old_forward  # re-hooking: keep the true original
    else:
        pristine = module.forward
        module._old_forward = pristine

    module = hook.init_hook(module)
    module._hf_hook = hook

    def hooked_forward(mod, *args, **kwargs):
        args, kwargs = mod._hf_hook.pre_forward(mod, *args, **kwargs)
        if mod._hf_hook.no_grad:
            with torch.no_grad():
                out = mod._old_forward(*args, **kwargs)
        else:
            out = mod._old_forward(*args, **kwargs)
        return mod._hf_hook.post_forward(mod, out)

    _install_forward(module, hooked_forward, pristine)
    return module


def remove_hook_from_module(module: nn.Module, recurse: bool = False):
    if hasattr(module, "_hf_hook"):
        module._hf_hook.detach_hook(module)
        delattr(module, "_hf_hook")
    if hasattr(module, "_old_forward"):
        if "GraphModuleImpl" in str(type(module)):
            module.__class__.forward = module._old_forward
        else:
            module.forward = module._old_forward
        delattr(module, "_old_forward")
    for name in getattr(module, "_accelerate_added_attributes", []):
        module.__dict__.pop(name, None)
    if hasattr(module, "_accelerate_added_attributes"):
        delattr(module, "_accelerate_added_attributes")
    if recurse:
        for child in module.children():
            remove_hook_from_module(child, recurse)
    return module


def remove_hook_from_submodules(module: nn.Module):
    remove_hook_from_module(module)
    for child in module.children():
        remove_hook_from_submodules(child)


def _attr_data_ptr(module, dotted: str) -> int:
    obj = module
    for piece in dotted.split("."):
        obj = getattr(obj, piece)
    return obj.data_ptr()


class AlignDevicesHook(ModelHook):
    """Onload managed tensors before forward, offload to meta after.

    ``place_submodules`` widens the managed set to the whole subtree;
    ``weights_map`` is the offloaded source of truth (memmap/safetensors
    loader or a plain dict); ``tied_params_map`` (data_ptr -> device ->
    tensor) deduplicates H2D copies of shared weights across hooks.
    """

    def __init__(
        self,
        execution_device: Optional[Union[int, str, torch.device]] = None,
        offload: bool = False,
        io_same_device: bool = False,
        weights_map: Optional[Mapping] = None,
        offload_buffers: bool = False,
        place_submodules: bool = False,
        skip_keys: Optional[Union[str, List[str]]] = None,
        tied_params_map: Optional[Dict[int, Dict[torch.device, torch.Tensor]]] = None,
    ):
        self.execution_device = execution_device
        self.offload = offload
        self.io_same_device = io_same_device
        self.weights_map = weights_map
        self.offload_buffers = offload_buffers
        self.place_submodules = place_submodules
        self.skip_keys = skip_keys
        self.tied_params_map = tied_params_map
        self.input_device = None
        self.tied_params_names = set()
        self.tied_pointers_to_remove = set()
        # async-onload machinery (wired in init_hook for CUDA offload hooks)
        self._module_ref = None
        self._prefetcher = None
        self._prefetched = None
        self._prefetch_event = None
        self._pinned_cache: Dict[str, torch.Tensor] = {}

    def __repr__(self):
        return (
            f"AlignDevicesHook(execution_device={self.execution_device}, offload={self.offload}, "
            f"io_same_device={self.io_same_device}, offload_buffers={self.offload_buffers}, "
            f"place_submodules={self.place_submodules}, skip_keys={repr(self.skip_keys)})"
        )

    # -- managed-tensor iteration -----------------------------------------

    def _managed(self, module):
        """Tensors this hook onloads/offloads each step (persistent only)."""
        return named_module_tensors(
            module,
            include_buffers=self.offload_buffers,
            recurse=self.place_submodules,
            remove_non_persistent=True,
        )

    def init_hook(self, module):
        if self.execution_device in ("meta", torch.device("meta")):
            self.tied_params_map = None  # meta data_ptrs all alias zero

        if not self.offload:
            if self.execution_device is not None:
                for name, _ in named_module_tensors(module, recurse=self.place_submodules):
                    set_module_tensor_to_device(
                        module, name, self.execution_device, tied_params_map=self.tied_params_map
                    )
            return module

        # offloaded block: remember where tensors lived, snapshot values if
        # no external weights_map was given, then vacate to meta
        self.original_devices = {
            name: t.device for name, t in named_module_tensors(module, recurse=self.place_submodules)
        }
        if self.weights_map is None:
            self.weights_map = {
                name: t.to("cpu")
                for name, t in named_module_tensors(
                    module, include_buffers=self.offload_buffers, recurse=self.place_submodules
                )
            }
        for name, _ in self._managed(module):
            # disk-backed maps reload to fresh pointers every access, so tie
            # membership is decided by the MODULE tensor's pointer up front
            if self.tied_params_map is not None and _attr_data_ptr(module, name) in self.tied_params_map:
                self.tied_params_names.add(name)
            set_module_tensor_to_device(module, name, "meta")

        if (
            _async_onload_enabled()
            and self.execution_device is not None
            and torch.device(self.execution_device).type == "cuda"
        ):
            self._module_ref = module
            self._prefetcher = _prefetcher_for(self.weights_map)

        if self.execution_device is not None:
            if not self.offload_buffers:
                # buffers stay resident on the execution device
                for name, _ in module.named_buffers(recurse=self.place_submodules):
                    set_module_tensor_to_device(
                        module, name, self.execution_device, tied_params_map=self.tied_params_map
                    )
            else:
                for name in get_non_persistent_buffers(module, recurse=self.place_submodules):
                    set_module_tensor_to_device(module, name, self.execution_device)
        return module

    def _pinned_value(self, name: str, value: torch.Tensor) -> torch.Tensor:
        """Serve H2D copies from a per-hook pinned-host cache: the copy then
        runs on the HIP copy engine without a staging bounce, and can be
        issued asynchronously on the prefetch stream."""
        if value.device.type != "cpu" or value.is_pinned():
            return value
        cached = self._pinned_cache.get(name)
        if cached is None:
            global _pinned_bytes
            budget = int(os.environ.get("ACCELERATE_AMD_PINNED_CACHE_MB", "65536")) << 20
            nbytes = value.numel() * value.element_size()
            if _pinned_bytes + nbytes > budget:
                return value  # cache full: fall back to a pageable copy
            try:
                cached = value.pin_memory()
            except RuntimeError:
                cached = value  # host pinning exhausted
            else:
                _pinned_bytes += nbytes
            self._pinned_cache[name] = cached
        return cached

    def _start_prefetch(self, stream: torch.cuda.Stream):
        """Issue this hook's weight onload on the side stream (called by the
        prefetcher when the PREVIOUS block's pre_forward runs)."""
        module = self._module_ref
        if module is None:
            return
        staged = {}
        with torch.cuda.stream(stream):
            for name, _ in self._managed(module):
                if name in self.tied_params_names:
                    continue  # tied weights go through the dedup slow path
                value = self.weights_map[name]
                if value is None:
                    continue
                staged[name] = self._pinned_value(name, value).to(self.execution_device, non_blocking=True)
        ev = torch.cuda.Event()
        ev.record(stream)
        self._prefetched, self._prefetch_event = staged, ev

    def pre_forward(self, module, *args, **kwargs):
        if self.io_same_device:
            self.input_device = find_device([args, kwargs])
        if self.offload:
            self.tied_pointers_to_remove = set()
            staged = self._prefetched
            self._prefetched = None
            if staged is not None:
                # compute stream consumes the side-stream copies
                torch.cuda.current_stream().wait_event(self._prefetch_event)
            for name, _ in self._managed(module):
                got = staged.pop(name, None) if staged else None
                if got is not None:
                    got.record_stream(torch.cuda.current_stream())
                    set_module_tensor_to_device(module, name, self.execution_device, value=got)
                    continue
                value = self.weights_map[name]
                if name in self.tied_params_names and value.data_ptr() not in self.tied_params_map:
                    self.tied_params_map[value.data_ptr()] = {}
                if (
                    value is not None
                    and self.tied_params_map is not None
                    and value.data_ptr() in self.tied_params_map
                ):
                    self.tied_pointers_to_remove.add((value.data_ptr(), self.execution_device))
                elif value is not None and self._prefetcher is not None:
                    value = self._pinned_value(name, value)
                set_module_tensor_to_device(
                    module, name, self.execution_device, value=value, tied_params_map=self.tied_params_map
                )
            if self._prefetcher is not None:
                self._prefetcher.note(self)
        return (
            send_to_device(args, self.execution_device),
            send_to_device(kwargs, self.execution_device, skip_keys=self.skip_keys),
        )

    def post_forward(self, module, output):
        if self.offload:
            for name, _ in self._managed(module):
                set_module_tensor_to_device(module, name, "meta")
            for ptr, device in self.tied_pointers_to_remove:
                if ptr in self.tied_params_map and device in self.tied_params_map[ptr]:
                    del self.tied_params_map[ptr][device]
            self.tied_pointers_to_remove = set()
        if self.io_same_device and self.input_device is not None:
            output = send_to_device(output, self.input_device, skip_keys=self.skip_keys)
        return output

    def detach_hook(self, module):
        if self.offload:
            for name, device in self.original_devices.items():
                if device != torch.device("meta"):
                    set_module_tensor_to_device(module, name, device, value=self.weights_map.get(name))
        return module


# ---------------------------------------------------------------------------
# attachment helpers (used by dispatch_model)
# ---------------------------------------------------------------------------


def attach_execution_device_hook(
    module: nn.Module,
    execution_device,
    skip_keys=None,
    preload_module_classes: Optional[List[str]] = None,
    tied_params_map=None,
):
    """Make sure every stateful submodule receives inputs on its device."""
    if not hasattr(module, "_hf_hook") and len(module.state_dict()) > 0:
        add_hook_to_module(
            module,
            AlignDevicesHook(execution_device, skip_keys=skip_keys, tied_params_map=tied_params_map),
        )
    if preload_module_classes is not None and module.__class__.__name__ in preload_module_classes:
        return  # subtree is managed as one block
    for child in module.children():
        attach_execution_device_hook(
            child,
            execution_device,
            skip_keys=skip_keys,
            preload_module_classes=preload_module_classes,
            tied_params_map=tied_params_map,
        )


def attach_align_device_hook(
    module: nn.Module,
    execution_device=None,
    offload: bool = False,
    weights_map: Optional[Mapping] = None,
    offload_buffers: bool = False,
    module_name: str = "",
    skip_keys=None,
    preload_module_classes: Optional[List[str]] = None,
    tied_params_map=None,
):
    """Per-leaf onload/offload hooks over a subtree (offloaded blocks)."""
    owns_tensors = any(True for _ in named_module_tensors(module))
    whole_subtree = (
        offload and preload_module_classes is not None and module.__class__.__name__ in preload_module_classes
    )
    if owns_tensors or whole_subtree:
        scoped_map = None
        if weights_map is not None:
            scoped_map = PrefixedDataset(weights_map, f"{module_name}." if module_name else "")
        add_hook_to_module(
            module,
            AlignDevicesHook(
                execution_device=execution_device,
                offload=offload,
                weights_map=scoped_map,
                offload_buffers=offload_buffers,
                place_submodules=whole_subtree,
                skip_keys=skip_keys,
                tied_params_map=tied_params_map,
            ),
            append=True,
        )
    if whole_subtree:
        return  # one hook covers everything below
    for child_name, child in module.named_children():
        attach_align_device_hook(
            child,
            execution_device=execution_device,
            offload=offload,
            weights_map=weights_map,
            offload_buffers=offload_buffers,
            module_name=f"{module_name}.{child_name}" if module_name else child_name,
            skip_keys=skip_keys,
            preload_module_classes=preload_module_classes,
            tied_params_map=tied_params_map,
        )


def attach_align_device_hook_on_blocks(
    module: nn.Module,
    execution_device=None,
    offload=False,
    weights_map: Mapping = None,
    offload_buffers: bool = False,
    module_name: str = "",
    skip_keys=None,
    preload_module_classes: Optional[List[str]] = None,
    tied_params_map=None,
):
    """Walk the device map: resident blocks get ONE whole-subtree hook (the
    root additionally pins outputs to the input device); offloaded blocks
    get per-leaf onload hooks plus input-device hooks for their interior."""
    # degenerate map: a single device / a single offload decision
    if not isinstance(execution_device, Mapping) and not isinstance(offload, dict):
        if offload:
            attach_align_device_hook(
                module,
                execution_device=execution_device,
                offload=True,
                weights_map=weights_map,
                offload_buffers=offload_buffers,
                module_name=module_name,
                skip_keys=skip_keys,
                tied_params_map=tied_params_map,
            )
        else:
            add_hook_to_module(
                module,
                AlignDevicesHook(
                    execution_device=execution_device,
                    io_same_device=True,
                    skip_keys=skip_keys,
                    place_submodules=True,
                    tied_params_map=tied_params_map,
                ),
            )
        return

    # normalize both maps to dicts over the same keys
    if not isinstance(execution_device, Mapping):
        execution_device = {key: execution_device for key in offload}
    if not isinstance(offload, Mapping):
        offload = {key: offload for key in execution_device}

    mapped = module_name in execution_device and module_name in offload
    if mapped and not offload[module_name]:
        add_hook_to_module(
            module,
            AlignDevicesHook(
                execution_device=execution_device[module_name],
                offload_buffers=offload_buffers,
                io_same_device=(module_name == ""),
                place_submodules=True,
                skip_keys=skip_keys,
                tied_params_map=tied_params_map,
            ),
        )
        attach_execution_device_hook(
            module, execution_device[module_name], skip_keys=skip_keys, tied_params_map=tied_params_map
        )
    elif mapped:
        attach_align_device_hook(
            module,
            execution_device=execution_device[module_name],
            offload=True,
            weights_map=weights_map,
            offload_buffers=offload_buffers,
            module_name=module_name,
            skip_keys=skip_keys,
            preload_module_classes=preload_module_classes,
            tied_params_map=tied_params_map,
        )
        if not hasattr(module, "_hf_hook"):
            add_hook_to_module(
                module,
                AlignDevicesHook(
                    execution_device=execution_device[module_name],
                    io_same_device=(module_name == ""),
                    skip_keys=skip_keys,
                    tied_params_map=tied_params_map,
                ),
            )
        attach_execution_device_hook(
            module,
            execution_device[module_name],
            preload_module_classes=preload_module_classes,
            skip_keys=skip_keys,
            tied_params_map=tied_params_map,
        )
    elif module_name == "":
        add_hook_to_module(
            module,
            AlignDevicesHook(
                execution_device=execution_device.get(""),
                io_same_device=True,
                skip_keys=skip_keys,
                tied_params_map=tied_params_map,
            ),
        )

    for child_name, child in module.named_children():
        attach_align_device_hook_on_blocks(
            child,
            execution_device=execution_device,
            offload=offload,
            weights_map=weights_map,
            offload_buffers=offload_buffers,
            module_name=f"{module_name}.{child_name}" if module_name else child_name,
            preload_module_classes=preload_module_classes,
            skip_keys=skip_keys,
            tied_params_map=tied_params_map,
        )


# ---------------------------------------------------------------------------
# sequential CPU offload + layerwise casting
# ---------------------------------------------------------------------------


class CpuOffload(ModelHook):
    """Whole-model CPU residency; onload to the execution device on forward.
    ``prev_module_hook`` chains pipelines: running block N offloads N-1."""

    def __init__(self, execution_device=None, prev_module_hook=None):
        self.prev_module_hook = prev_module_hook
        if execution_device is not None:
            self.execution_device = torch.device(execution_device)
        elif torch.cuda.is_available():
            self.execution_device = torch.device("cuda", torch.cuda.current_device())
        else:
            self.execution_device = torch.device("cpu")

    def init_hook(self, module):
        return module.to("cpu")

    def pre_forward(self, module, *args, **kwargs):
        if self.prev_module_hook is not None:
            self.prev_module_hook.offload()
            from .utils.memory import clear_device_cache

            clear_device_cache()
        module.to(self.execution_device)
        return (
            send_to_device(args, self.execution_device),
            send_to_device(kwargs, self.execution_device),
        )


class UserCpuOffloadHook:
    """Handle returned by ``cpu_offload_with_hook``: manual offload/remove."""

    def __init__(self, model, hook):
        self.model = model
        self.hook = hook

    def offload(self):
        self.hook.init_hook(self.model)

    def remove(self):
        remove_hook_from_module(self.model)


class LayerwiseCastingHook(ModelHook):
    """Store in ``storage_dtype``, compute in ``compute_dtype`` — per-layer
    upcast around forward (fp8/bf16 storage with fp32 compute)."""

    def __init__(self, storage_dtype: torch.dtype, compute_dtype: torch.dtype, non_blocking: bool = False):
        self.storage_dtype = storage_dtype
        self.compute_dtype = compute_dtype
        self.non_blocking = non_blocking

    def init_hook(self, module):
        module.to(dtype=self.storage_dtype, non_blocking=self.non_blocking)
        return module

    def pre_forward(self, module, *args, **kwargs):
        module.to(dtype=self.compute_dtype, non_blocking=self.non_blocking)
        return args, kwargs

    def post_forward(self, module, output):
        module.to(dtype=self.storage_dtype, non_blocking=self.non_blocking)
        return output
