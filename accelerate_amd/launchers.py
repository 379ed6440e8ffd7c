"""In-process launchers (reference: launchers.py).

``notebook_launcher`` spawns one process per GPU with ``start_method="spawn"``
— mandatory on ROCm: a HIP runtime initialized in the parent breaks forked
children (reference: launchers.py:211-213 makes the same choice for ROCm).
``debug_launcher`` runs N CPU processes over gloo for tests.
"""

import os
import sys
import tempfile

import torch

from .state import AcceleratorState
from .utils.environment import patch_environment


class PrepareForLaunch:
    """Callable wrapper setting rank env vars inside each spawned process
    (reference: utils/launch.py:783-827)."""

    def __init__(self, launcher, distributed_type="NO", debug=False):
        self.launcher = launcher
        self.distributed_type = distributed_type
        self.debug = debug

    def __call__(self, index, *args):
        if self.debug:
            world_size = int(os.environ.get("WORLD_SIZE"))
            rdv_file = os.environ.get("ACCELERATE_DEBUG_RDV_FILE")
            torch.distributed.init_process_group(
                "gloo",
                rank=index,
                store=torch.distributed.FileStore(rdv_file, world_size),
                world_size=world_size,
            )
        elif self.distributed_type in ("MULTI_GPU", "MULTI_CPU"):
            os.environ["LOCAL_RANK"] = str(index)
            os.environ["RANK"] = str(index)
        os.environ["FORK_LAUNCHED"] = str(1)
        self.launcher(*args)


def notebook_launcher(
    function,
    args=(),
    num_processes=None,
    mixed_precision="no",
    use_port="29500",
    master_addr="127.0.0.1",
    node_rank=0,
    num_nodes=1,
    rdzv_backend="static",
    rdzv_endpoint="",
    rdzv_conf=None,
    rdzv_id="none",
    max_restarts=0,
    monitor_interval=0.1,
):
    """Launch ``function(*args)`` on N processes from a notebook
    (reference: launchers.py:43)."""
    in_colab = "google.colab" in sys.modules
    if num_processes is None:
        num_processes = torch.cuda.device_count() if torch.cuda.is_available() else 1

    if num_processes > 1 or num_nodes > 1:
        if len(AcceleratorState._shared_state) > 0:
            raise ValueError(
                "To launch a multi-GPU training from your notebook, the `Accelerator` should only be initialized "
                "inside your training function."
            )
        if torch.cuda.is_initialized():
            raise ValueError(
                "To launch a multi-GPU training from your notebook, you need to avoid running any instruction "
                "using `torch.cuda` before starting it: the HIP runtime initialized in the parent process "
                "cannot be forked/spawned into working children."
            )
        from torch.multiprocessing import ProcessRaisedException, start_processes

        with tempfile.NamedTemporaryFile() as _:
            with patch_environment(
                nproc=num_processes,
                node_rank=node_rank,
                world_size=num_nodes * num_processes,
                master_addr=master_addr,
                master_port=use_port,
                mixed_precision=mixed_precision,
                accelerate_mixed_precision=mixed_precision,
            ):
                launcher = PrepareForLaunch(function, distributed_type="MULTI_GPU" if torch.cuda.is_available() else "MULTI_CPU")
                print(f"Launching training on {num_processes} GPUs.")
                try:
                    start_processes(launcher, args=args, nprocs=num_processes, start_method="spawn")
                except ProcessRaisedException as e:
                    if "Cannot re-initialize CUDA in forked subprocess" in e.args[0]:
                        raise RuntimeError(
                            "HIP was initialized before launching; restart the notebook and initialize "
                            "everything inside the training function."
                        ) from e
                    raise
    else:
        # single process
        if torch.cuda.is_available():
            os.environ["ACCELERATE_MIXED_PRECISION"] = mixed_precision
        function(*args)


def debug_launcher(function, args=(), num_processes=2):
    """Run ``function`` on N CPU processes over a gloo FileStore rendezvous —
    the lowest-rung distributed test harness (reference: launchers.py:287)."""
    from torch.multiprocessing import start_processes

    with tempfile.NamedTemporaryFile() as tmp_file:
        with patch_environment(
            world_size=num_processes,
            master_addr="127.0.0.1",
            master_port="29500",
            accelerate_mixed_precision="no",
            accelerate_debug_rdv_file=tmp_file.name,
            accelerate_use_cpu="yes",
        ):
            launcher = PrepareForLaunch(function, debug=True)
            start_processes(launcher, args=args, nprocs=num_processes, start_method="fork")
