"""Memory utilities: OOM-retry batch-size finder, cache release
(reference: utils/memory.py)."""

import functools
import gc
import inspect

import torch


def clear_device_cache(garbage_collection: bool = False):
    if garbage_collection:
        gc.collect()
    if torch.cuda.is_available():
        torch.cuda.empty_cache()


def release_memory(*objects):
    """del the passed objects and empty the HIP caching allocator
    (reference: memory.py:40-68)."""
    if not isinstance(objects, list):
        objects = list(objects)
    for i in range(len(objects)):
        objects[i] = None
    clear_device_cache(garbage_collection=True)
    return objects


def should_reduce_batch_size(exception: Exception) -> bool:
    """Heuristic OOM detection (reference: memory.py:100-116). ROCm raises
    torch.OutOfMemoryError / 'HIP out of memory' through the same
    caching-allocator path as CUDA."""
    _statements = [
        "CUDA out of memory.",
        "HIP out of memory.",
        "cudaErrorMemoryAllocation",
        "hipErrorOutOfMemory",
        "DefaultCPUAllocator: can't allocate memory",
    ]
    if isinstance(exception, torch.cuda.OutOfMemoryError):
        return True
    if isinstance(exception, RuntimeError) and len(exception.args) == 1:
        return any(err in exception.args[0] for err in _statements)
    return False


def find_executable_batch_size(function=None, starting_batch_size: int = 128, reduce_batch_size_fn=None):
    """Retry decorator that halves... well, shrinks ×0.9 the batch size on OOM
    (reference: memory.py:119-187)."""
    if function is None:
        return functools.partial(
            find_executable_batch_size, starting_batch_size=starting_batch_size, reduce_batch_size_fn=reduce_batch_size_fn
        )
    if reduce_batch_size_fn is None:

        def reduce_batch_size_fn(batch_size):
            return int(batch_size * 0.9)

    batch_size = starting_batch_size

    def decorator(*args, **kwargs):
        nonlocal batch_size
        clear_device_cache(garbage_collection=True)
        params = list(inspect.signature(function).parameters.keys())
        # Guard against user error
        if len(params) < (len(args) + 1):
            arg_str = ", ".join([f"{arg}={value}" for arg, value in zip(params[1:], args[1:])])
            raise TypeError(
                f"Batch size was passed into `{function.__name__}` as the first argument when called."
                f"Remove this as the decorator already does so: `{function.__name__}({arg_str})`"
            )
        while True:
            if batch_size == 0:
                raise RuntimeError("No executable batch size found, reached zero.")
            try:
                return function(batch_size, *args, **kwargs)
            except Exception as e:
                if should_reduce_batch_size(e):
                    clear_device_cache(garbage_collection=True)
                    batch_size = reduce_batch_size_fn(batch_size)
                else:
                    raise

    return decorator


def get_xpu_available_memory(*args, **kwargs):  # pragma: no cover
    raise NotImplementedError("XPU is not a target of the MI355X-native framework.")
