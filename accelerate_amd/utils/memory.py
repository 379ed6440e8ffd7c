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
    """OOM-retry decorator: call ``function(batch_size, ...)`` and shrink the
    batch on every allocator failure until a size fits (behavior parity with
    reference memory.py:119-187; default shrink ×0.9 per retry).

    The wrapped function's FIRST positional parameter is the batch size and
    is supplied by the decorator — callers pass only the remaining args.
    The chosen size is sticky across calls (a training re-run starts at the
    last size that worked, not back at ``starting_batch_size``).
    """
    if function is None:
        return functools.partial(
            find_executable_batch_size, starting_batch_size=starting_batch_size, reduce_batch_size_fn=reduce_batch_size_fn
        )
    shrink = reduce_batch_size_fn or (lambda bs: int(bs * 0.9))
    state = {"bs": starting_batch_size}

    def _reject_caller_supplied_batch_size(args):
        declared = list(inspect.signature(function).parameters)
        if len(declared) < len(args) + 1:
            shown = ", ".join(f"{name}={val}" for name, val in zip(declared[1:], args[1:]))
            raise TypeError(
                f"`{function.__name__}` received a batch size from its caller, but the "
                f"find_executable_batch_size decorator injects it. Call it as "
                f"`{function.__name__}({shown})` instead."
            )

    @functools.wraps(function)
    def attempt_until_it_fits(*args, **kwargs):
        clear_device_cache(garbage_collection=True)
        _reject_caller_supplied_batch_size(args)
        while state["bs"] > 0:
            try:
                return function(state["bs"], *args, **kwargs)
            except Exception as oom:
                if not should_reduce_batch_size(oom):
                    raise
                clear_device_cache(garbage_collection=True)
                state["bs"] = shrink(state["bs"])
        raise RuntimeError("No executable batch size found, reached zero.")

    return attempt_until_it_fits


def get_xpu_available_memory(*args, **kwargs):  # pragma: no cover
    raise NotImplementedError("XPU is not a target of the MI355X-native framework.")
