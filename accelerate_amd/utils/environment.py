"""Environment-variable parsing and platform probes.

The env-var plane is the ABI between the launcher and the library
(reference: utils/environment.py, utils/launch.py:100-427). All knobs are
independently settable via ``ACCELERATE_*`` variables so a launcher can fully
configure a worker without touching Python.
"""

import os
from contextlib import contextmanager
from functools import lru_cache

_TRUE = {"1", "true", "yes", "on", "y", "t"}
_FALSE = {"0", "false", "no", "off", "n", "f", ""}


def str_to_bool(value: str) -> int:
    """Convert a string to 1/0 (reference: utils/environment.py:59-70)."""
    value = str(value).lower().strip()
    if value in _TRUE:
        return 1
    if value in _FALSE:
        return 0
    raise ValueError(f"invalid truth value {value!r}")


def get_int_from_env(env_keys, default):
    """Return the first positive int found among ``env_keys``."""
    for key in env_keys:
        val = int(os.environ.get(key, -1))
        if val >= 0:
            return val
    return default


def parse_flag_from_env(key: str, default: bool = False) -> bool:
    value = os.environ.get(key, str(default))
    return bool(str_to_bool(value))


def parse_choice_from_env(key: str, default: str = "no") -> str:
    return os.environ.get(key, str(default))


def are_libraries_initialized(*library_names) -> list:
    import sys

    return [lib for lib in library_names if lib in sys.modules.keys()]


@lru_cache
def get_gpu_count() -> int:
    """Number of visible MI355X devices, without initializing the HIP runtime.

    Uses amdsmi if importable, else falls back to torch (which on ROCm probes
    via the driver without creating a context for ``device_count``).
    """
    try:
        import amdsmi

        try:
            amdsmi.amdsmi_init()
            n = len(amdsmi.amdsmi_get_processor_handles())
            amdsmi.amdsmi_shut_down()
            return n
        except Exception:
            return 0
    except ImportError:
        pass
    try:
        import torch

        return torch.cuda.device_count()
    except Exception:
        return 0


def get_cpu_distributed_information() -> dict:
    """Collect rank info for CPU (gloo/mpi) runs from common env layouts."""
    information = {}
    information["size"] = get_int_from_env(["WORLD_SIZE", "PMI_SIZE", "OMPI_COMM_WORLD_SIZE", "MV2_COMM_WORLD_SIZE"], 1)
    information["rank"] = get_int_from_env(["RANK", "PMI_RANK", "OMPI_COMM_WORLD_RANK", "MV2_COMM_WORLD_RANK"], 0)
    information["local_size"] = get_int_from_env(
        ["LOCAL_WORLD_SIZE", "MPI_LOCALNRANKS", "OMPI_COMM_WORLD_LOCAL_SIZE", "MV2_COMM_WORLD_LOCAL_SIZE"], 1
    )
    information["local_rank"] = get_int_from_env(
        ["LOCAL_RANK", "MPI_LOCALRANKID", "OMPI_COMM_WORLD_LOCAL_RANK", "MV2_COMM_WORLD_LOCAL_RANK"], 0
    )
    return information


def set_numa_affinity(local_process_index: int, verbose: bool = False) -> None:
    """Pin the process to the NUMA node closest to its GPU.

    On an 8×MI355X node GPUs are split across NUMA domains; binding the
    dataloader workers and RCCL proxy threads to the local domain avoids
    cross-socket HBM staging traffic. Best effort: no-op when the sysfs
    topology is unavailable (e.g. in CI containers).
    """
    try:
        import ctypes

        libnuma = ctypes.CDLL("libnuma.so.1")
        if libnuma.numa_available() < 0:
            return
        num_nodes = libnuma.numa_max_node() + 1
        node = local_process_index % max(num_nodes, 1)
        libnuma.numa_run_on_node(ctypes.c_int(node))
        if verbose:
            print(f"[accelerate_amd] pinned process {local_process_index} to NUMA node {node}")
    except Exception:
        return


@contextmanager
def clear_environment():
    """Temporarily clear os.environ (restored on exit)."""
    backup = os.environ.copy()
    os.environ.clear()
    try:
        yield
    finally:
        os.environ.clear()
        os.environ.update(backup)


@contextmanager
def patch_environment(**kwargs):
    """Temporarily set environment variables (reference: utils/other.py pattern)."""
    existing = {}
    for key, value in kwargs.items():
        key = key.upper()
        if key in os.environ:
            existing[key] = os.environ[key]
        os.environ[key] = str(value)
    try:
        yield
    finally:
        for key in kwargs:
            key = key.upper()
            if key in existing:
                os.environ[key] = existing[key]
            else:
                os.environ.pop(key, None)


def purge_accelerate_environment(func):
    """Decorator that runs ``func`` with all ACCELERATE_* env vars scrubbed."""
    import functools

    @functools.wraps(func)
    def wrapper(*args, **kwargs):
        backup = os.environ.copy()
        for key in list(os.environ):
            if key.startswith("ACCELERATE_"):
                del os.environ[key]
        try:
            return func(*args, **kwargs)
        finally:
            os.environ.clear()
            os.environ.update(backup)

    return wrapper
