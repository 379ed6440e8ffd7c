"""CLI tests (reference: tests/test_cli.py pattern): config round-trip,
env, launch of the bundled test script (2-proc gloo)."""

import os
import subprocess
import sys
from pathlib import Path

import yaml

from testing_utils import REPO_ROOT, get_free_port


def run_cli(*args, timeout=240):
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO_ROOT + os.pathsep + env.get("PYTHONPATH", "")
    result = subprocess.run(
        [sys.executable, "-m", "accelerate_amd", *args], capture_output=True, text=True, timeout=timeout, env=env
    )
    return result


def test_env_command():
    r = run_cli("env")
    assert r.returncode == 0
    assert "accelerate_amd version" in r.stdout
    assert "PyTorch version" in r.stdout


def test_config_default_roundtrip(tmp_path):
    cfg_file = tmp_path / "cfg.yaml"
    r = run_cli("config", "--default", "--config_file", str(cfg_file))
    assert r.returncode == 0, r.stderr
    data = yaml.safe_load(cfg_file.read_text())
    assert "distributed_type" in data
    assert "num_processes" in data

    from accelerate_amd.commands.config import ClusterConfig

    cfg = ClusterConfig.load(cfg_file)
    assert cfg.num_processes == data["num_processes"]


def test_launch_single_process():
    script = Path(REPO_ROOT) / "accelerate_amd" / "test_utils" / "test_script.py"
    r = run_cli("launch", "--num_processes", "1", "--cpu", str(script))
    assert r.returncode == 0, r.stderr
    assert "All checks passed!" in r.stdout


def test_launch_multi_process():
    script = Path(REPO_ROOT) / "accelerate_amd" / "test_utils" / "test_script.py"
    r = run_cli(
        "launch",
        "--num_processes",
        "2",
        "--cpu",
        "--main_process_port",
        str(get_free_port()),
        str(script),
    )
    assert r.returncode == 0, r.stdout + r.stderr
    assert "All checks passed!" in r.stdout


def test_launch_respects_config_file(tmp_path):
    cfg_file = tmp_path / "cfg.yaml"
    cfg_file.write_text(
        yaml.safe_dump(
            {
                "distributed_type": "MULTI_CPU",
                "num_processes": 2,
                "use_cpu": True,
                "mixed_precision": "no",
                "main_process_port": get_free_port(),
            }
        )
    )
    script = Path(REPO_ROOT) / "accelerate_amd" / "test_utils" / "test_script.py"
    r = run_cli("launch", "--config_file", str(cfg_file), str(script))
    assert r.returncode == 0, r.stdout + r.stderr
    assert "2 process(es)" in r.stdout


def test_cv_example_runs_cpu():
    """examples/cv_example.py (reference cv_example parity): one epoch on
    synthetic images, single CPU process."""
    import subprocess
    import sys

    r = subprocess.run(
        [sys.executable, str(Path(REPO_ROOT) / "examples" / "cv_example.py"),
         "--cpu", "--epochs", "1", "--batch_size", "16"],
        capture_output=True, text=True, timeout=300,
        env={**os.environ, "PYTHONPATH": REPO_ROOT},
    )
    assert r.returncode == 0, r.stderr
    assert "eval accuracy" in r.stdout


def test_estimate_bundled_families():
    """estimate resolves the bundled model families fully offline; the
    llama3-70b bf16 figure must match the measured 131.4 GB residency
    (BENCHMARKS.md 70B demo)."""
    r = run_cli("estimate", "llama3-70b", "--dtypes", "bfloat16")
    assert r.returncode == 0, r.stderr
    assert "131.4" in r.stdout
    r2 = run_cli("estimate", "mixtral-8x7b", "--dtypes", "bfloat16")
    assert r2.returncode == 0, r2.stderr


def test_launch_flags_accept_hyphens_and_underscores():
    """Both --foo-bar and --foo_bar spellings parse to the same dest
    (reference tests/test_cli.py test_hyphen/test_underscore)."""
    import argparse

    from accelerate_amd.commands.launch import add_parser

    parser = argparse.ArgumentParser()
    add_parser(parser.add_subparsers())
    a = parser.parse_args(["launch", "--fsdp_sharding_strategy", "hybrid_shard", "s.py"])
    b = parser.parse_args(["launch", "--fsdp-sharding-strategy", "hybrid_shard", "s.py"])
    assert a.fsdp_sharding_strategy == b.fsdp_sharding_strategy == "hybrid_shard"
    c = parser.parse_args(["launch", "--parallelism-config-tp-size", "4", "s.py"])
    assert c.parallelism_config_tp_size == 4


def test_config_rejects_unknown_keys(tmp_path):
    import pytest
    import yaml

    from accelerate_amd.commands.config import ClusterConfig

    p = tmp_path / "cfg.yaml"
    p.write_text(yaml.safe_dump({"num_processes": 2, "not_a_real_key": 1}))
    with pytest.raises(ValueError, match="not_a_real_key"):
        ClusterConfig.load(p)


def test_config_rejects_invalid_values(tmp_path):
    import pytest
    import yaml

    from accelerate_amd.commands.config import ClusterConfig

    p = tmp_path / "cfg.yaml"
    p.write_text(yaml.safe_dump({"distributed_type": "XPU_TPU"}))
    with pytest.raises(ValueError, match="distributed_type"):
        ClusterConfig.load(p)
    p.write_text(yaml.safe_dump({"mixed_precision": "int3"}))
    with pytest.raises(ValueError, match="mixed_precision"):
        ClusterConfig.load(p)


def test_launch_rccl_debug_plumb():
    """--rccl_debug sets the NCCL_* plane (RCCL on ROCm) for workers
    (SURVEY §5.2 MI355X note)."""
    import argparse

    import accelerate_amd.commands.launch as launch_mod
    from accelerate_amd.commands.config import ClusterConfig

    parser = argparse.ArgumentParser()
    launch_mod.add_parser(parser.add_subparsers())
    args = parser.parse_args(["launch", "--rccl_debug", "INFO", "--cpu", "script.py"])
    env = launch_mod.build_env(args, ClusterConfig())
    assert env["NCCL_DEBUG"] == "INFO"
