"""debug_launcher (2 CPU procs over gloo FileStore) and notebook_launcher
single-process path (reference: tests/test_cpu.py / launchers.py:287)."""

import subprocess
import sys

from testing_utils import REPO_ROOT


def test_debug_launcher_two_procs():
    code = """
import torch
import torch.distributed as dist
from accelerate_amd import debug_launcher

def fn():
    assert dist.is_initialized()
    assert dist.get_world_size() == 2
    t = torch.ones(1) * (dist.get_rank() + 1)
    dist.all_reduce(t)
    assert t.item() == 3.0
    if dist.get_rank() == 0:
        print("DEBUG_LAUNCH_OK")

debug_launcher(fn)
"""
    result = subprocess.run(
        [sys.executable, "-c", code], capture_output=True, text=True, timeout=120,
        env={"PYTHONPATH": REPO_ROOT, "PATH": "/usr/bin:/bin:/usr/local/bin"},
    )
    assert result.returncode == 0, result.stderr
    assert "DEBUG_LAUNCH_OK" in result.stdout


def test_notebook_launcher_single_process():
    from accelerate_amd import notebook_launcher

    out = []

    def fn(x):
        out.append(x * 2)

    notebook_launcher(fn, args=(21,), num_processes=1)
    assert out == [42]
