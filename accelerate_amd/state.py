"""Process/device state singletons (reference: state.py).

`PartialState` bootstraps the distributed world: one process per MI355X GPU,
``torch.distributed`` over RCCL (backend "nccl" IS RCCL on ROCm) for GPU
worlds, gloo for CPU worlds. `AcceleratorState` layers mixed-precision and
strategy resolution on top. `GradientState` links dataloader iteration to
gradient synchronization.

All three share state through a class-level dict so that every instance in a
process observes the same world (reference: state.py:91-120 SharedDict).
"""

import logging
import os
import threading
import warnings
import weakref
from contextlib import contextmanager
from datetime import timedelta
from functools import wraps
from typing import Any, Callable, Optional

import torch

from .utils.dataclasses import DistributedType, GradientAccumulationPlugin
from .utils.environment import (
    get_cpu_distributed_information,
    parse_choice_from_env,
    parse_flag_from_env,
    set_numa_affinity,
)
from .utils.imports import is_hip_available

logger = logging.getLogger(__name__)


def is_initialized() -> bool:
    """Whether a PartialState has been created in this process."""
    return len(PartialState._shared_state) > 0


def do_nothing(*args, **kwargs):
    return None


class ThreadLocalSharedDict(threading.local):
    """Descriptor holding the singleton state dict (thread-local so notebook
    launchers that re-enter on worker threads get fresh state;
    reference: state.py:91-120)."""

    def __init__(self, thread_local: bool = False):
        self._storage = {}

    def __get__(self, obj, objtype=None):
        return self._storage

    def __set__(self, obj, value):
        self._storage = value


SharedDict = dict


class PartialState:
    """Singleton: process-group bootstrap and rank/device facts.

    (reference: state.py:123-866)
    """

    _shared_state = SharedDict()
    _known_attrs = [
        "_cpu",
        "backend",
        "device",
        "debug",
        "distributed_type",
        "fork_launched",
        "local_process_index",
        "num_processes",
        "process_index",
    ]

    def __init__(self, cpu: bool = False, **kwargs):
        self.__dict__ = self._shared_state
        if self.initialized:
            return

        self._cpu = cpu
        self.backend = None
        self.debug = parse_flag_from_env("ACCELERATE_DEBUG_MODE")
        env_device = os.environ.get("ACCELERATE_TORCH_DEVICE", None)
        self.device = torch.device(env_device) if env_device is not None else None
        self.fork_launched = parse_flag_from_env("FORK_LAUNCHED", False)

        use_distributed = int(os.environ.get("WORLD_SIZE", "1")) > 1 or int(os.environ.get("LOCAL_RANK", "-1")) >= 0
        gpu_world = is_hip_available() and not cpu

        if use_distributed:
            backend = kwargs.pop("backend", None)
            if backend is None:
                backend = "nccl" if gpu_world else "gloo"
            self.backend = backend
            if not torch.distributed.is_initialized():
                if "timeout" not in kwargs:
                    kwargs["timeout"] = timedelta(seconds=int(os.environ.get("ACCELERATE_NCCL_TIMEOUT", "1800")))
                # one process per GPU; pin the device BEFORE init so RCCL
                # communicator setup lands on the right HIP device.
                rank = int(os.environ.get("RANK", "0"))
                world_size = int(os.environ.get("WORLD_SIZE", "1"))
                local_rank = int(os.environ.get("LOCAL_RANK", rank))
                if gpu_world:
                    device_index = local_rank % torch.cuda.device_count()
                    torch.cuda.set_device(device_index)
                torch.distributed.init_process_group(backend=backend, rank=rank, world_size=world_size, **kwargs)
            self.num_processes = torch.distributed.get_world_size()
            self.process_index = torch.distributed.get_rank()
            self.local_process_index = int(os.environ.get("LOCAL_RANK", self.process_index))
            if gpu_world:
                self.distributed_type = DistributedType.MULTI_GPU
                self.device = torch.device("cuda", self.local_process_index % torch.cuda.device_count())
                torch.cuda.set_device(self.device)
                if parse_flag_from_env("ACCELERATE_CPU_AFFINITY", False):
                    set_numa_affinity(self.local_process_index)
            else:
                self.distributed_type = DistributedType.MULTI_CPU
                if self.device is None:
                    self.device = torch.device("cpu")
        else:
            self.distributed_type = DistributedType.NO
            self.num_processes = 1
            self.process_index = 0
            self.local_process_index = 0
            if self.device is None:
                self.device = self.default_device

    def __repr__(self) -> str:
        return (
            f"Distributed environment: {self.distributed_type}{('  Backend: ' + self.backend) if self.backend else ''}\n"
            f"Num processes: {self.num_processes}\n"
            f"Process index: {self.process_index}\n"
            f"Local process index: {self.local_process_index}\n"
            f"Device: {self.device}\n"
        )

    @staticmethod
    def _reset_state():
        """Reset the shared state (for tests; reference: state.py:848)."""
        PartialState._shared_state.clear()

    def destroy_process_group(self, group=None):
        if self.fork_launched and group is None:
            return
        if torch.distributed.is_initialized():
            torch.distributed.destroy_process_group(group)

    @property
    def initialized(self) -> bool:
        return self._shared_state.get("_cpu") is not None or len(self._shared_state) > 0

    @property
    def use_distributed(self) -> bool:
        return self.distributed_type != DistributedType.NO and self.num_processes > 1

    @property
    def is_last_process(self) -> bool:
        return self.process_index == self.num_processes - 1

    @property
    def is_main_process(self) -> bool:
        return self.process_index == 0

    @property
    def is_local_main_process(self) -> bool:
        return self.local_process_index == 0

    @property
    def default_device(self) -> torch.device:
        return torch.device("cuda") if is_hip_available() else torch.device("cpu")

    def wait_for_everyone(self):
        """Barrier across all processes (reference: state.py:377-414)."""
        if self.use_distributed:
            if self.backend == "nccl":
                torch.distributed.barrier(device_ids=[self.local_process_index])
            else:
                torch.distributed.barrier()

    def _goes_first(self, is_main: bool):
        if not is_main:
            self.wait_for_everyone()
        yield
        if is_main:
            self.wait_for_everyone()

    @contextmanager
    def main_process_first(self):
        yield from self._goes_first(self.is_main_process)

    @contextmanager
    def local_main_process_first(self):
        yield from self._goes_first(self.is_local_main_process)

    @contextmanager
    def split_between_processes(self, inputs, apply_padding: bool = False):
        """Split `inputs` (list/tuple/dict/tensor) between processes
        (reference: state.py:426-512)."""
        if self.num_processes == 1:
            yield inputs
            return
        length = len(inputs)
        if isinstance(inputs, dict):
            length = len(inputs[list(inputs.keys())[0]])
            if not all(len(v) == length for v in inputs.values()):
                raise ValueError("All values in the dictionary must have the same length")
        num_samples_per_process, num_extras = divmod(length, self.num_processes)
        start_index = self.process_index * num_samples_per_process + min(self.process_index, num_extras)
        end_index = start_index + num_samples_per_process + (1 if self.process_index < num_extras else 0)

        def _split_values(inputs, start_index, end_index):
            if isinstance(inputs, (list, tuple, torch.Tensor)):
                if start_index >= len(inputs):
                    result = inputs[-1:]
                else:
                    result = inputs[start_index:end_index]
                if apply_padding:
                    if isinstance(result, torch.Tensor):
                        from .utils.operations import pad_across_processes, send_to_device

                        tensorized_result = send_to_device(result, self.device)
                        result = pad_across_processes(tensorized_result, pad_index=inputs[-1])
                    else:
                        result += [result[-1]] * (num_samples_per_process + (1 if num_extras > 0 else 0) - len(result))
                return result
            elif isinstance(inputs, dict):
                for key in inputs.keys():
                    inputs[key] = _split_values(inputs[key], start_index, end_index)
                return inputs
            else:
                return inputs

        yield _split_values(inputs, start_index, end_index)

    def on_main_process(self, function: Callable[..., Any] = None):
        if not self.initialized:
            raise ValueError("The `PartialState` or `Accelerator` must be initialized before calling this function.")
        if self.is_main_process or not self.use_distributed:
            return function
        return do_nothing

    def on_local_main_process(self, function: Callable[..., Any] = None):
        if self.is_local_main_process or not self.use_distributed:
            return function
        return do_nothing

    def on_last_process(self, function: Callable[..., Any]):
        if self.is_last_process or not self.use_distributed:
            return function
        return do_nothing

    def on_process(self, function: Callable[..., Any] = None, process_index: int = None):
        if function is None:
            return lambda func: self.on_process(func, process_index)
        if (self.process_index == process_index) or (not self.use_distributed):
            return function
        return do_nothing

    def on_local_process(self, function: Callable[..., Any] = None, local_process_index: int = None):
        if function is None:
            return lambda func: self.on_local_process(func, local_process_index)
        if (self.local_process_index == local_process_index) or (not self.use_distributed):
            return function
        return do_nothing

    def print(self, *args, **kwargs):
        if self.is_local_main_process:
            print(*args, **kwargs)

    def set_device(self):
        if self.device is not None and self.device.type == "cuda":
            torch.cuda.set_device(self.device)


class AcceleratorState:
    """Adds mixed precision + strategy resolution on top of PartialState
    (reference: state.py:868-1228)."""

    _shared_state = SharedDict()

    def __init__(
        self,
        mixed_precision: str = None,
        cpu: bool = False,
        fsdp_plugin=None,
        _from_accelerator: bool = False,
        **kwargs,
    ):
        self.__dict__ = self._shared_state
        if parse_flag_from_env("ACCELERATE_USE_CPU"):
            cpu = True
        if self.initialized:
            if mixed_precision is not None and mixed_precision != self._mixed_precision:
                raise ValueError(
                    "AcceleratorState already initialized with a different mixed_precision; "
                    "call AcceleratorState._reset_state() first (tests) or create the Accelerator once."
                )
            return

        if PartialState._shared_state == {}:
            PartialState(cpu, **kwargs)
        self.__dict__.update(PartialState._shared_state)

        if mixed_precision is None:
            mixed_precision = parse_choice_from_env("ACCELERATE_MIXED_PRECISION", "no")
        mixed_precision = str(mixed_precision).lower()
        if mixed_precision not in ("no", "fp16", "bf16", "fp8"):
            raise ValueError(f"Unknown mixed_precision mode: {mixed_precision}")
        self._mixed_precision = mixed_precision

        self.use_fsdp = fsdp_plugin is not None or parse_flag_from_env("ACCELERATE_USE_FSDP")
        self.fsdp_plugin = fsdp_plugin
        # sharded engine works at any world size (world 1 = no collectives,
        # still bf16-compute/fp32-master semantics + sharded checkpoints)
        if self.use_fsdp and self.distributed_type in (
            DistributedType.MULTI_GPU,
            DistributedType.MULTI_CPU,
            DistributedType.NO,
        ):
            self.distributed_type = DistributedType.FSDP
            if self.fsdp_plugin is None:
                from .utils.dataclasses import FullyShardedDataParallelPlugin

                self.fsdp_plugin = FullyShardedDataParallelPlugin()
        self.dynamo_plugin = None

    def __repr__(self):
        return PartialState().__repr__() + f"Mixed precision type: {self.mixed_precision}\n"

    @property
    def initialized(self) -> bool:
        return "_mixed_precision" in self._shared_state

    @property
    def mixed_precision(self) -> str:
        return self._mixed_precision

    @staticmethod
    def _reset_state(reset_partial_state: bool = False):
        AcceleratorState._shared_state.clear()
        if reset_partial_state:
            PartialState._reset_state()

    def __getattr__(self, name: str):
        # proxy rank facts to PartialState
        if name in PartialState._shared_state:
            return PartialState._shared_state[name]
        raise AttributeError(f"`AcceleratorState` object has no attribute `{name}`")

    # convenience passthroughs
    @property
    def use_distributed(self):
        return PartialState().use_distributed

    @property
    def is_main_process(self):
        return PartialState().is_main_process

    @property
    def is_local_main_process(self):
        return PartialState().is_local_main_process

    @property
    def is_last_process(self):
        return PartialState().is_last_process

    def wait_for_everyone(self):
        PartialState().wait_for_everyone()

    @contextmanager
    def main_process_first(self):
        with PartialState().main_process_first():
            yield

    @contextmanager
    def local_main_process_first(self):
        with PartialState().local_main_process_first():
            yield

    @contextmanager
    def split_between_processes(self, inputs, apply_padding: bool = False):
        with PartialState().split_between_processes(inputs, apply_padding=apply_padding) as result:
            yield result

    def print(self, *args, **kwargs):
        PartialState().print(*args, **kwargs)

    def destroy_process_group(self, group=None):
        PartialState().destroy_process_group(group)


class GradientState:
    """Singleton linking dataloader iteration to gradient sync
    (reference: state.py:1231-1371)."""

    _shared_state = SharedDict()

    def __init__(self, gradient_accumulation_plugin: Optional[GradientAccumulationPlugin] = None):
        self.__dict__ = self._shared_state
        if not self.initialized:
            self.sync_gradients = True
            self.dataloader_references = [None]
            self.plugin_kwargs = (
                gradient_accumulation_plugin.to_kwargs() if gradient_accumulation_plugin is not None else {}
            )
            self._is_xla_gradients_synced = False
        if gradient_accumulation_plugin is not None and self.plugin_kwargs != gradient_accumulation_plugin.to_kwargs():
            self.plugin_kwargs = gradient_accumulation_plugin.to_kwargs()

    @property
    def num_steps(self) -> int:
        return self.plugin_kwargs.get("num_steps", 1)

    @property
    def adjust_scheduler(self) -> bool:
        return self.plugin_kwargs.get("adjust_scheduler", False)

    @property
    def sync_with_dataloader(self) -> bool:
        return self.plugin_kwargs.get("sync_with_dataloader", True)

    @property
    def initialized(self) -> bool:
        return GradientState._shared_state != {}

    @property
    def end_of_dataloader(self) -> bool:
        if not self.in_dataloader:
            return False
        return self.active_dataloader.end_of_dataloader

    @property
    def remainder(self) -> int:
        if not self.in_dataloader:
            return -1
        return self.active_dataloader.remainder

    def __repr__(self):
        return (
            f"Sync Gradients: {self.sync_gradients}\n"
            f"At end of current dataloader: {self.end_of_dataloader}\n"
            f"Extra samples added: {self.remainder}\n"
            f"Gradient accumulation plugin: {self.plugin_kwargs}\n"
        )

    def _set_sync_gradients(self, sync_gradients: bool):
        self.sync_gradients = sync_gradients

    @property
    def active_dataloader(self):
        # weakref-only registry: the GradientState singleton must never keep
        # a dataloader alive (reference: state.py:1335-1366; a strong ref
        # here leaks loaders abandoned mid-iteration)
        last = self.dataloader_references[-1]
        return last() if last is not None else None

    def _add_dataloader(self, dataloader):
        self.dataloader_references.append(weakref.ref(dataloader))

    def _remove_dataloader(self, dataloader):
        self.dataloader_references = [
            ref for ref in self.dataloader_references if ref is None or ref() is not dataloader
        ] or [None]

    @property
    def in_dataloader(self) -> bool:
        return self.active_dataloader is not None

    @staticmethod
    def _reset_state():
        GradientState._shared_state.clear()
