import json
import os
import tempfile

import pytest
import torch
import torch.nn as nn

from accelerate_amd import find_executable_batch_size
from accelerate_amd.tracking import JSONLTracker
from accelerate_amd.utils.environment import patch_environment, str_to_bool
from accelerate_amd.utils.memory import should_reduce_batch_size
from accelerate_amd.utils.offload import OffloadedWeightsLoader, offload_state_dict


def test_find_executable_batch_size():
    calls = []

    @find_executable_batch_size(starting_batch_size=128)
    def train(batch_size):
        calls.append(batch_size)
        if batch_size > 16:
            raise RuntimeError("CUDA out of memory.")
        return batch_size

    result = train()
    assert result <= 16
    assert calls[0] == 128
    assert all(calls[i + 1] < calls[i] for i in range(len(calls) - 1))


def test_non_oom_error_propagates():
    @find_executable_batch_size(starting_batch_size=8)
    def train(batch_size):
        raise ValueError("unrelated")

    with pytest.raises(ValueError):
        train()


def test_should_reduce_batch_size():
    assert should_reduce_batch_size(RuntimeError("HIP out of memory."))
    assert should_reduce_batch_size(RuntimeError("CUDA out of memory."))
    assert not should_reduce_batch_size(RuntimeError("something else"))


def test_str_to_bool():
    assert str_to_bool("yes") == 1
    assert str_to_bool("FALSE") == 0
    with pytest.raises(ValueError):
        str_to_bool("maybe")


def test_patch_environment():
    with patch_environment(test_var_xyz="1"):
        assert os.environ["TEST_VAR_XYZ"] == "1"
    assert "TEST_VAR_XYZ" not in os.environ


def test_offload_state_dict_roundtrip():
    sd = {
        "a": torch.randn(4, 4),
        "b": torch.randn(3).to(torch.bfloat16),
        "scalar": torch.tensor(3.5),
    }
    with tempfile.TemporaryDirectory() as d:
        offload_state_dict(d, sd)
        loader = OffloadedWeightsLoader(save_folder=d)
        assert set(loader.keys()) == set(sd.keys())
        for k in sd:
            assert torch.equal(loader[k], sd[k]), k
        assert loader["b"].dtype == torch.bfloat16


def test_jsonl_tracker():
    from accelerate_amd.state import PartialState

    PartialState()
    with tempfile.TemporaryDirectory() as d:
        tracker = JSONLTracker("run1", logging_dir=d)
        tracker.store_init_configuration({"lr": 0.1})
        tracker.log({"loss": 1.5}, step=0)
        tracker.log({"loss": 1.0}, step=1)
        tracker.finish()
        lines = [json.loads(l) for l in open(os.path.join(d, "run1", "metrics.jsonl"))]
        assert lines[0]["_config"] == {"lr": 0.1}
        assert lines[1]["loss"] == 1.5
        assert lines[2]["_step"] == 1


def test_scheduler_steps_num_processes_times():
    # single process: 1 step per step
    from accelerate_amd import Accelerator

    acc = Accelerator()
    model = nn.Linear(2, 2)
    opt = torch.optim.SGD(model.parameters(), lr=1.0)
    sched = torch.optim.lr_scheduler.LambdaLR(opt, lambda s: 1.0 / (s + 1))
    model, opt, sched = acc.prepare(model, opt, sched)
    model(torch.randn(1, 2)).sum().backward()
    opt.step()
    sched.step()
    assert sched.scheduler._step_count == 2


def test_local_sgd_single_process_noop():
    from accelerate_amd import Accelerator, LocalSGD

    acc = Accelerator()
    model = nn.Linear(2, 2)
    model = acc.prepare_model(model)
    with LocalSGD(accelerator=acc, model=model, local_sgd_steps=2) as ls:
        ls.step()


def test_parallelism_config_groups_two_process():
    from tests.testing_utils import launch_distributed

    out = launch_distributed("tests/distributed_scripts/pconfig_script.py", nproc=2)
    assert "PCONFIG_PASS" in out


def test_dataloader_configuration_overrides():
    from accelerate_amd import Accelerator, DataLoaderConfiguration
    from accelerate_amd.state import PartialState

    PartialState._reset_state()
    cfg = DataLoaderConfiguration(split_batches=True, even_batches=False, non_blocking=False)
    acc = Accelerator(cpu=True, dataloader_config=cfg)
    assert acc.split_batches is True
    assert acc.even_batches is False
    assert acc.non_blocking is False
    PartialState._reset_state()


def test_compile_regions_llama_blocks():
    """Regional compilation (reference utils/other.py:106): repeated decoder
    layers become individually compiled regions; forward still matches."""
    import torch
    from accelerate_amd.models.llama import LlamaConfig, LlamaForCausalLM
    from accelerate_amd.utils import compile_regions, has_compiled_regions, is_compiled_module

    torch.manual_seed(0)
    model = LlamaForCausalLM(LlamaConfig.tiny()).eval()
    ids = torch.randint(0, 1024, (1, 8))
    with torch.no_grad():
        ref = model(ids)["logits"]
    compile_regions(model, backend="eager")  # eager backend: wrapper-only, fast
    assert has_compiled_regions(model)
    assert is_compiled_module(model.layers[0])
    with torch.no_grad():
        out = model(ids)["logits"]
    assert torch.allclose(out, ref, atol=1e-5)


def test_small_util_helpers():
    from accelerate_amd.utils import convert_dict_to_env_variables, merge_dicts

    assert merge_dicts({"a": {"b": 1}}, {"a": {"c": 2}, "d": 3}) == {"a": {"b": 1, "c": 2}, "d": 3}
    env = convert_dict_to_env_variables({"GOOD": "1", "BAD": "x\ny"})
    assert env == ["GOOD=1"]


def test_write_basic_config(tmp_path):
    import yaml

    from accelerate_amd.utils import write_basic_config

    path = write_basic_config(mixed_precision="bf16", save_location=str(tmp_path / "cfg.yaml"))
    cfg = yaml.safe_load(open(path))
    assert cfg["mixed_precision"] == "bf16"
    assert cfg["distributed_type"] in ("NO", "MULTI_GPU")


def test_tp_through_prepare_4proc():
    """TP x DP via plain Accelerator(parallelism_config=...).prepare() for
    Llama AND GPT-2 — no model-specific calls (VERDICT round-1 item 4)."""
    from tests.testing_utils import launch_distributed

    out = launch_distributed("tests/distributed_scripts/pconfig_prepare_script.py", nproc=4, timeout=300)
    assert "TP_SAVE_STATE_PASS" in out
    assert "TP_CLIP_PASS" in out
    assert "TP_PREPARE_LLAMA_PASS" in out
    assert "TP_PREPARE_GPT2_PASS" in out


def test_cp_through_prepare_2proc():
    """CP/Ulysses via prepare + maybe_context_parallel for Llama and GPT-2."""
    from tests.testing_utils import launch_distributed

    out = launch_distributed("tests/distributed_scripts/cp_prepare_script.py", nproc=2, timeout=300)
    assert "CP_PREPARE_LLAMA_ALLGATHER_PASS" in out
    assert "CP_PREPARE_LLAMA_RING_PASS" in out
    assert "CP_PREPARE_LLAMA_ULYSSES_PASS" in out
    assert "CP_PREPARE_GPT2_ALLGATHER_PASS" in out


def test_ring_cp_2proc():
    """Ring (P2P KV rotation) attention: fwd+bwd parity vs full-sequence
    flash attention, incl. causal, non-causal, GQA, and dk/dv homing."""
    from tests.testing_utils import launch_distributed

    out = launch_distributed("tests/distributed_scripts/ring_script.py", nproc=2, timeout=300)
    assert "RING_CP_PASS" in out


def test_shard_sequence_world1_passthrough():
    """No-dist world-1 path returns the tensor untouched (the non-divisible
    ValueError path is exercised by the gloo scripts at world > 1)."""
    from accelerate_amd.parallel.cp import shard_sequence

    t = torch.randn(2, 10)
    assert shard_sequence(t) is t


def test_ring_script_world3_shapes():
    """Ring CP oracle also holds at world 3 (multi-hop rotation)."""
    from tests.testing_utils import launch_distributed

    out = launch_distributed("tests/distributed_scripts/ring_script.py", nproc=3, timeout=300)
    assert "RING_CP_PASS" in out


def test_dp_cp_combined_4proc():
    """dp_replicate=2 x cp=2: DDP over the flattened dp x cp grad group +
    per-step sequence sharding, one train step == single-process reference."""
    from tests.testing_utils import launch_distributed

    out = launch_distributed("tests/distributed_scripts/dpcp_script.py", nproc=4, timeout=300)
    assert "DPCP_PREPARE_PASS" in out


def test_tp_cp_combined_4proc():
    """TP (tp=2) x CP (cp=2): tp-sharded attention heads + cp KV all-gather
    + grad averaging over the cp group — one step == single-process ref."""
    from tests.testing_utils import launch_distributed

    out = launch_distributed("tests/distributed_scripts/tp_cp_script.py", nproc=4, timeout=300)
    assert "TP_CP_PASS" in out


def test_multiprocess_logger_adapter(caplog):
    """Logging parity (reference tests/test_logging.py): get_logger wraps a
    MultiProcessAdapter honoring main_process_only / in_order kwargs, and
    warning_once emits exactly once."""
    import logging

    from accelerate_amd.logging import get_logger

    logger = get_logger("accelerate_amd.test_logger")
    with caplog.at_level(logging.INFO, logger="accelerate_amd.test_logger"):
        logger.info("hello-main", main_process_only=True)
        logger.info("hello-all", main_process_only=False)
        logger.info("hello-order", in_order=True)
        logger.warning_once("only-once")
        logger.warning_once("only-once")
    messages = [r.message for r in caplog.records]
    assert "hello-main" in messages and "hello-all" in messages and "hello-order" in messages
    assert messages.count("only-once") == 1


def test_lazy_import_hygiene():
    """Importing the package must not drag in heavy optional deps
    (reference tests/test_imports.py): trackers and model libs load lazily."""
    import subprocess
    import sys

    code = (
        "import sys; import accelerate_amd; "
        "heavy=[m for m in ('wandb','mlflow','comet_ml','aim','clearml','dvclive',"
        "'swanlab','transformers','datasets','pandas') if m in sys.modules]; "
        "assert not heavy, heavy; print('ok')"
    )
    out = subprocess.run([sys.executable, "-c", code], capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr[-1500:]


def test_examples_parse():
    """Every bundled example is at least syntactically valid and imports
    only names the package exports (guards against API drift)."""
    import ast
    import pathlib

    root = pathlib.Path(__file__).resolve().parent.parent / "examples"
    assert root.is_dir()
    files = sorted(root.glob("*.py"))
    assert len(files) >= 6, files
    import accelerate_amd

    for f in files:
        tree = ast.parse(f.read_text())
        for node in ast.walk(tree):
            if isinstance(node, ast.ImportFrom) and node.module == "accelerate_amd":
                for alias in node.names:
                    assert hasattr(accelerate_amd, alias.name), f"{f.name}: accelerate_amd.{alias.name} missing"


def test_top_level_export_parity():
    """Every public name the reference exports at package top level exists
    here (migration criterion: 'switch and find everything')."""
    import accelerate_amd as A

    for name in (
        "Accelerator", "DeepSpeedPlugin", "FullyShardedDataParallelPlugin",
        "DistributedDataParallelKwargs", "InitProcessGroupKwargs", "GradScalerKwargs",
        "ParallelismConfig", "PartialState", "DistributedType", "notebook_launcher",
        "debug_launcher", "init_empty_weights", "dispatch_model", "infer_auto_device_map",
        "load_checkpoint_and_dispatch", "cpu_offload", "disk_offload",
        "find_executable_batch_size", "skip_first_batches", "LocalSGD", "prepare_pippy",
        "ProfileKwargs", "DataLoaderConfiguration", "AutocastKwargs",
    ):
        assert hasattr(A, name), name
