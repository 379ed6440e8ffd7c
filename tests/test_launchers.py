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


def test_bench_contract_2proc_plumbing():
    """The driver launches bench.py via torchrun at N>1: the full
    prepare/DDP/loader plumbing must survive world>1 (regression: a
    kwargs-handler field leak crashed every SCALE run). BENCH_TINY=1
    shrinks the model so this is a plumbing smoke, not a measurement."""
    import json
    import os
    import subprocess
    import sys

    from tests.testing_utils import get_free_port

    env = dict(os.environ, BENCH_TINY="1", PYTHONPATH=os.getcwd())
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1", "--nproc-per-node", "2",
         "--master-addr", "127.0.0.1", "--master-port", str(get_free_port()),
         "bench.py", "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=280, env=env, cwd=os.getcwd(),
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    rec = json.loads(line)
    assert rec["n_gpus"] == 2 and rec["config"]["parallelism"] == "dp2"
