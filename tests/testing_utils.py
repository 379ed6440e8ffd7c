"""Helpers for multi-process CPU tests: run a script under torch.distributed.run
with gloo, world_size 2 (the reference's execute_subprocess_async pattern,
test_utils/testing.py:781)."""

import os
import socket
import subprocess
import sys
from contextlib import closing
from pathlib import Path

REPO_ROOT = str(Path(__file__).parent.parent)


def get_free_port() -> int:
    with closing(socket.socket(socket.AF_INET, socket.SOCK_STREAM)) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def launch_distributed(script_path, nproc=2, timeout=180, extra_env=None, args=()):
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO_ROOT + os.pathsep + env.get("PYTHONPATH", "")
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    if extra_env:
        env.update(extra_env)
    cmd = [
        sys.executable,
        "-m",
        "torch.distributed.run",
        "--nnodes=1",
        f"--nproc-per-node={nproc}",
        "--master-addr=127.0.0.1",
        f"--master-port={get_free_port()}",
        str(script_path),
        *args,
    ]
    result = subprocess.run(cmd, env=env, capture_output=True, text=True, timeout=timeout)
    if result.returncode != 0:
        raise AssertionError(
            f"distributed script failed (rc={result.returncode})\nSTDOUT:\n{result.stdout}\nSTDERR:\n{result.stderr}"
        )
    return result.stdout
