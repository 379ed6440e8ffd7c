"""Collective ops & tensor utilities over RCCL/gloo (reference: utils/operations.py).

All cross-process communication the framework itself issues lives here
(reference inventory: SURVEY.md §2.8). On MI355X these run over RCCL on
xGMI; on CPU test worlds over gloo. Debug mode (``ACCELERATE_DEBUG_MODE=1``)
verifies operand shapes across ranks before every collective and raises a
per-rank shape table on mismatch (reference: operations.py:361-421).
"""

import pickle
from functools import update_wrapper, wraps
from typing import Any, Mapping

import torch

from ..state import PartialState
from .dataclasses import DistributedType


class DistributedOperationException(Exception):
    """Raised when ranks disagree on the operands of a collective."""


def is_torch_tensor(tensor):
    return isinstance(tensor, torch.Tensor)


def is_namedtuple(data):
    return isinstance(data, tuple) and hasattr(data, "_asdict") and hasattr(data, "_fields")


def honor_type(obj, generator):
    """Cast a generator to the same (possibly namedtuple) type as obj."""
    if is_namedtuple(obj):
        return type(obj)(*list(generator))
    return type(obj)(generator)


def recursively_apply(func, data, *args, test_type=is_torch_tensor, error_on_other_type=False, **kwargs):
    """Apply ``func`` to every leaf of a nested list/tuple/dict structure
    (reference: operations.py:82-133)."""
    if isinstance(data, (tuple, list)):
        return honor_type(
            data,
            (
                recursively_apply(
                    func, o, *args, test_type=test_type, error_on_other_type=error_on_other_type, **kwargs
                )
                for o in data
            ),
        )
    elif isinstance(data, Mapping):
        return type(data)(
            {
                k: recursively_apply(
                    func, v, *args, test_type=test_type, error_on_other_type=error_on_other_type, **kwargs
                )
                for k, v in data.items()
            }
        )
    elif test_type(data):
        return func(data, *args, **kwargs)
    elif error_on_other_type:
        raise TypeError(
            f"Unsupported types ({type(data)}) passed to `{func.__name__}`. Only nested "
            "list/tuple/dicts of objects that are valid for `is_torch_tensor` should be passed."
        )
    return data


def send_to_device(tensor, device, non_blocking=False, skip_keys=None):
    """Recursively move tensors to device. On MI355X ``non_blocking=True``
    pairs with pinned host staging for async H2D over a copy stream
    (reference: operations.py:136-185)."""
    if is_torch_tensor(tensor) or hasattr(tensor, "to"):
        if isinstance(device, str) and device not in ("cpu", "meta") and not device.startswith(("cuda", "hip")):
            device = torch.device(device)
        try:
            return tensor.to(device, non_blocking=non_blocking)
        except TypeError:  # .to() not accepting non_blocking
            return tensor.to(device)
        except AssertionError:
            return tensor
    elif isinstance(tensor, (tuple, list)):
        return honor_type(tensor, (send_to_device(t, device, non_blocking=non_blocking, skip_keys=skip_keys) for t in tensor))
    elif isinstance(tensor, Mapping):
        if isinstance(skip_keys, str):
            skip_keys = [skip_keys]
        elif skip_keys is None:
            skip_keys = []
        return type(tensor)(
            {
                k: t if k in skip_keys else send_to_device(t, device, non_blocking=non_blocking, skip_keys=skip_keys)
                for k, t in tensor.items()
            }
        )
    return tensor


def get_data_structure(data):
    """Nested structure of TensorInformation describing data (for dispatch headers)."""

    def _get_data_structure(tensor):
        return TensorInformation(shape=tensor.shape, dtype=tensor.dtype)

    return recursively_apply(_get_data_structure, data)


def get_shape(data):
    def _get_shape(tensor):
        return list(tensor.shape)

    return recursively_apply(_get_shape, data)


class TensorInformation:
    def __init__(self, shape, dtype):
        self.shape = shape
        self.dtype = dtype

    def __repr__(self):
        return f"TensorInformation(shape={self.shape}, dtype={self.dtype})"


def initialize_tensors(data_structure):
    """Materialize empty tensors matching a TensorInformation structure."""

    def _initialize_tensor(tensor_info):
        return torch.empty(*tensor_info.shape, dtype=tensor_info.dtype)

    return recursively_apply(_initialize_tensor, data_structure, test_type=lambda t: isinstance(t, TensorInformation))


def find_batch_size(data):
    """Find the batch size (dim 0) of the first tensor leaf."""
    if isinstance(data, (tuple, list)):
        for d in data:
            result = find_batch_size(d)
            if result is not None:
                return result
        return None
    elif isinstance(data, Mapping):
        for k in data.keys():
            result = find_batch_size(data[k])
            if result is not None:
                return result
        return None
    elif isinstance(data, torch.Tensor):
        return data.shape[0] if len(data.shape) >= 1 else None
    return None


def ignorant_find_batch_size(data):
    try:
        return find_batch_size(data)
    except (ValueError, TypeError):
        return None


def listify(data):
    """Tensor → nested lists of plain Python numbers."""

    def _convert_to_list(tensor):
        tensor = tensor.detach().cpu()
        if tensor.dtype == torch.bfloat16:
            tensor = tensor.to(torch.float32)
        return tensor.tolist()

    return recursively_apply(_convert_to_list, data)


def convert_to_fp32(tensor):
    """Recursively convert fp16/bf16 leaves to fp32 (reference: operations.py:889)."""

    def _convert_to_fp32(tensor):
        return tensor.float()

    def _is_fp16_bf16_tensor(tensor):
        return hasattr(tensor, "dtype") and tensor.dtype in (torch.float16, torch.bfloat16)

    return recursively_apply(_convert_to_fp32, tensor, test_type=_is_fp16_bf16_tensor)


class ConvertOutputsToFp32:
    """Wraps a (typically autocast) forward so outputs come back fp32
    (reference: operations.py:911-948). A class so the model stays picklable."""

    def __init__(self, model_forward):
        self.model_forward = model_forward
        update_wrapper(self, model_forward)

    def __call__(self, *args, **kwargs):
        return convert_to_fp32(self.model_forward(*args, **kwargs))

    def __getstate__(self):
        raise pickle.PicklingError(
            "Cannot pickle a prepared model with automatic mixed precision, please unwrap the model first."
        )


def convert_outputs_to_fp32(model_forward):
    model_forward = ConvertOutputsToFp32(model_forward)

    def forward(*args, **kwargs):
        return model_forward(*args, **kwargs)

    # To act like a decorator so that it can be popped when doing `extract_model_from_parallel`
    forward.__wrapped__ = model_forward
    return forward


# ---------------------------------------------------------------------------
# debug-mode operation verification (reference: operations.py:361-421)
# ---------------------------------------------------------------------------


def verify_operation(function):
    """In debug mode, gather operand shapes from all ranks first and raise a
    readable table if they disagree (desync detection)."""

    @wraps(function)
    def wrapper(*args, **kwargs):
        if PartialState().debug is False or not PartialState().use_distributed:
            return function(*args, **kwargs)
        operation = f"{function.__module__}.{function.__name__}"
        if "tensor" in kwargs:
            tensor = kwargs["tensor"]
        else:
            tensor = args[0]
        if PartialState().device.type != tensor.device.type and tensor.device.type != "cpu":
            raise DistributedOperationException(
                f"One or more of the tensors passed to {operation} were not on the {tensor.device.type} "
                "while the `Accelerator` is configured for a different device."
            )
        shapes = get_shape(tensor)
        output = gather_object([shapes])
        if output[0] is not None:
            are_same = output.count(output[0]) == len(output)
            if not are_same:
                process_shape_str = "\n  - ".join([f"Process {i}: {shape}" for i, shape in enumerate(output)])
                raise DistributedOperationException(
                    f"Cannot apply desired operation due to shape mismatches. "
                    f"All shapes across devices must be valid.\n\nOperation: `{operation}`\nInput shapes:\n  - {process_shape_str}"
                )
        return function(*args, **kwargs)

    return wrapper


def chained_operation(function):
    """Re-raise collective errors with the op name attached."""

    @wraps(function)
    def wrapper(*args, **kwargs):
        try:
            return function(*args, **kwargs)
        except DistributedOperationException as e:
            operation = f"{function.__module__}.{function.__name__}"
            raise DistributedOperationException(
                f"Error found while calling `{operation}`. Please see the earlier error for more details."
            ) from e

    return wrapper


# ---------------------------------------------------------------------------
# collectives
# ---------------------------------------------------------------------------


def _gpu_gather_one(tensor):
    state = PartialState()
    if tensor.ndim == 0:
        tensor = tensor.clone()[None]
    # RCCL all_gather_into_tensor over xGMI: one flat numel×n buffer, no
    # per-rank tensor list allocation (reference: operations.py:322-358).
    tensor = tensor.contiguous()
    if state.backend is not None and state.backend != "gloo":
        output_tensors = torch.empty(
            state.num_processes * tensor.numel(), dtype=tensor.dtype, device=tensor.device
        )
        torch.distributed.all_gather_into_tensor(output_tensors, tensor)
        return output_tensors.view(-1, *tensor.size()[1:])
    else:
        output_tensors = [torch.empty_like(tensor) for _ in range(state.num_processes)]
        torch.distributed.all_gather(output_tensors, tensor)
        return torch.cat(output_tensors, dim=0)


def _gpu_gather(tensor):
    return recursively_apply(_gpu_gather_one, tensor, error_on_other_type=True)


@verify_operation
def gather(tensor):
    """All-gather a (nested) tensor across ranks, concatenated on dim 0
    (reference: operations.py:425)."""
    state = PartialState()
    if not state.use_distributed:
        return tensor
    return _gpu_gather(tensor)


def gather_object(object: Any):
    """All-gather picklable objects, flattening ONE level: each rank
    passes a LIST and the result concatenates every rank's items — the
    reference semantics (operations.py:498-523 _gpu_gather_object) that
    gather_for_metrics' remainder slicing depends on (slicing must drop
    tail SAMPLES, not tail ranks). Non-distributed worlds return the
    object unchanged, as the reference does."""
    state = PartialState()
    if not state.use_distributed:
        return object
    output_objects = [None for _ in range(state.num_processes)]
    torch.distributed.all_gather_object(output_objects, object)
    return [x for y in output_objects for x in y]


def _gpu_broadcast(data, src=0):
    def _gpu_broadcast_one(tensor, src=0):
        torch.distributed.broadcast(tensor, src=src)
        return tensor

    return recursively_apply(_gpu_broadcast_one, data, error_on_other_type=True, src=src)


@verify_operation
def broadcast(tensor, from_process: int = 0):
    """Broadcast a (nested) tensor from one rank to all
    (reference: operations.py:601)."""
    state = PartialState()
    if not state.use_distributed:
        return tensor
    return _gpu_broadcast(tensor, src=from_process)


def broadcast_object_list(object_list, from_process: int = 0):
    """Broadcast a list of picklable objects (reference: operations.py:675-697)."""
    state = PartialState()
    if not state.use_distributed:
        return object_list
    torch.distributed.broadcast_object_list(object_list, src=from_process)
    return object_list


def slice_tensors(data, tensor_slice, process_index=None, num_processes=None):
    def _slice_tensor(tensor, tensor_slice):
        return tensor[tensor_slice]

    return recursively_apply(_slice_tensor, data, tensor_slice)


def concatenate(data, dim=0):
    """Concatenate a list of nested structures leaf-wise (reference: operations.py:622)."""
    if isinstance(data[0], (tuple, list)):
        return honor_type(data[0], (concatenate([d[i] for d in data], dim=dim) for i in range(len(data[0]))))
    elif isinstance(data[0], Mapping):
        return type(data[0])({k: concatenate([d[k] for d in data], dim=dim) for k in data[0].keys()})
    elif not isinstance(data[0], torch.Tensor):
        raise TypeError(f"Can only concatenate tensors but got {type(data[0])}")
    return torch.cat(data, dim=dim)


@chained_operation
def pad_across_processes(tensor, dim=0, pad_index=0, pad_first=False):
    """Pad ragged tensors to the max size across ranks so they can be gathered
    (reference: operations.py:750-803)."""

    def _pad_across_processes(tensor, dim=0, pad_index=0, pad_first=False):
        if dim >= len(tensor.shape) or dim < -len(tensor.shape):
            return tensor
        if dim < 0:
            dim += len(tensor.shape)

        # Gather all sizes
        size = torch.tensor(tensor.shape, device=tensor.device)[None]
        sizes = gather(size).cpu()
        # Then pad to the maximum size
        max_size = max(s[dim] for s in sizes)
        if max_size == tensor.shape[dim]:
            return tensor

        old_size = tensor.shape
        new_size = list(old_size)
        new_size[dim] = max_size
        new_tensor = tensor.new_zeros(tuple(new_size)) + pad_index
        if pad_first:
            indices = tuple(
                slice(max_size - old_size[dim], max_size) if i == dim else slice(None) for i in range(len(new_size))
            )
        else:
            indices = tuple(slice(0, old_size[dim]) if i == dim else slice(None) for i in range(len(new_size)))
        new_tensor[indices] = tensor
        return new_tensor

    return recursively_apply(
        _pad_across_processes, tensor, error_on_other_type=True, dim=dim, pad_index=pad_index, pad_first=pad_first
    )


def pad_input_tensors(tensor, batch_size, num_processes, dim=0):
    """Pad dim-0 so batch_size divides num_processes (used by dispatcher)."""

    def _pad_input_tensors(tensor, batch_size, num_processes, dim=0):
        remainder = batch_size // num_processes
        last_inputs = batch_size - (remainder * num_processes)
        if batch_size // num_processes == 0:
            to_pad = num_processes - batch_size
        else:
            to_pad = num_processes - (batch_size // num_processes)
        if last_inputs == 0:
            return tensor
        old_size = tensor.shape
        new_size = list(old_size)
        new_size[0] = batch_size + to_pad
        new_tensor = tensor.new_zeros(tuple(new_size))
        indices = tuple(slice(0, old_size[dim]) if i == dim else slice(None) for i in range(len(new_size)))
        new_tensor[indices] = tensor
        return new_tensor

    return recursively_apply(
        _pad_input_tensors, tensor, error_on_other_type=True, batch_size=batch_size, num_processes=num_processes, dim=dim
    )


@verify_operation
def reduce(tensor, reduction="mean", scale=1.0):
    """All-reduce a (nested) tensor across ranks (reference: operations.py:846-886)."""

    def _reduce_across_processes(tensor, reduction="mean", scale=1.0):
        state = PartialState()
        cloned_tensor = tensor.clone()
        if not state.use_distributed:
            return cloned_tensor
        # reference parity: only "max" maps to MAX; "sum"/"mean"/"none"
        # all-reduce SUM (mean divides afterwards, none leaves the sum)
        op = torch.distributed.ReduceOp.MAX if reduction == "max" else torch.distributed.ReduceOp.SUM
        torch.distributed.all_reduce(cloned_tensor, op)
        if reduction == "mean":
            cloned_tensor /= state.num_processes
        if scale != 1.0:
            cloned_tensor *= scale
        return cloned_tensor

    return recursively_apply(
        _reduce_across_processes, tensor, error_on_other_type=True, reduction=reduction, scale=scale
    )


def gather_tensor_shape(tensor):
    """All ranks learn the shape of a tensor that only one rank holds."""
    state = PartialState()
    max_dims = 16
    shape_tensor = torch.zeros(max_dims + 1, dtype=torch.int64, device=state.device)
    if tensor is not None:
        shape = tensor.shape
        shape_tensor[0] = len(shape)
        for i, s in enumerate(shape):
            shape_tensor[i + 1] = s
    reduced = reduce(shape_tensor, reduction="max")
    ndim = int(reduced[0].item())
    return torch.Size(int(reduced[i + 1].item()) for i in range(ndim))


def copy_tensor_to_devices(tensor=None):
    """Broadcast-via-reduce a tensor that exists on one rank to all ranks
    (used by pipeline output fan-out, reference: operations.py:583)."""
    state = PartialState()
    shape = gather_tensor_shape(tensor)
    if tensor is None:
        tensor = torch.zeros(shape, device=state.device)
    else:
        tensor = tensor.to(state.device)
    return reduce(tensor, reduction="sum")


def drop_last_batches(tensor, num_batches):
    return tensor[:num_batches]


def find_device(data):
    """Find the device of the first tensor leaf."""
    if isinstance(data, Mapping):
        for obj in data.values():
            device = find_device(obj)
            if device is not None:
                return device
    elif isinstance(data, (tuple, list)):
        for obj in data:
            device = find_device(obj)
            if device is not None:
                return device
    elif isinstance(data, torch.Tensor):
        return data.device
    return None
