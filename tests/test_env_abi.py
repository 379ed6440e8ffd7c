"""Launcher↔library env-var ABI round trips (mirrors the reference's
tests/fsdp/test_fsdp.py:90-423 plugin-field→env suite, extended to the
PARALLELISM_CONFIG_*, ACCELERATE_DYNAMO_* and ACCELERATE_FP8_* planes)."""

import argparse
import os

import pytest

from accelerate_amd.commands.launch import add_parser, build_env
from accelerate_amd.commands.config import ClusterConfig
from accelerate_amd.parallelism_config import ParallelismConfig
from accelerate_amd.utils.dataclasses import (
    FP8RecipeKwargs,
    FullyShardedDataParallelPlugin,
    TorchDynamoPlugin,
)


@pytest.fixture(autouse=True)
def clean_env():
    saved = dict(os.environ)
    for k in list(os.environ):
        if k.startswith(("FSDP_", "PARALLELISM_CONFIG_", "ACCELERATE_")):
            del os.environ[k]
    yield
    os.environ.clear()
    os.environ.update(saved)


def parse_launch(argv):
    parser = argparse.ArgumentParser()
    sub = parser.add_subparsers()
    add_parser(sub)
    return parser.parse_args(["launch"] + argv + ["script.py"])


# ---- FSDP_* plane: env -> plugin field, one test per field --------------

FSDP_ENV_CASES = [
    ("FSDP_SHARDING_STRATEGY", "hybrid_shard", "sharding_strategy", "hybrid_shard"),
    ("FSDP_RESHARD_AFTER_FORWARD", "false", "reshard_after_forward", False),
    ("FSDP_OFFLOAD_PARAMS", "true", "cpu_offload", True),
    ("FSDP_ACTIVATION_CHECKPOINTING", "true", "activation_checkpointing", True),
    ("FSDP_STATE_DICT_TYPE", "SHARDED_STATE_DICT", "state_dict_type", "sharded_state_dict"),
    ("FSDP_SYNC_MODULE_STATES", "false", "sync_module_states", False),
    ("FSDP_AUTO_WRAP_POLICY", "transformer_based_wrap", "auto_wrap_policy", "transformer_based_wrap"),
    ("FSDP_TRANSFORMER_CLS_TO_WRAP", "LlamaDecoderLayer,BertLayer", "transformer_cls_names_to_wrap",
     ["LlamaDecoderLayer", "BertLayer"]),
    ("FSDP_MIN_NUM_PARAMS", "123456", "min_num_params", 123456),
    ("FSDP_VERSION", "2", "fsdp_version", 2),
    ("FSDP_FORWARD_PREFETCH", "false", "forward_prefetch", False),
    ("FSDP_BACKWARD_PREFETCH", "BACKWARD_POST", "backward_prefetch", "backward_post"),
    ("FSDP_CPU_RAM_EFFICIENT_LOADING", "true", "cpu_ram_efficient_loading", True),
]


@pytest.mark.parametrize("env_key,env_val,field,expected", FSDP_ENV_CASES)
def test_fsdp_env_to_plugin(env_key, env_val, field, expected):
    os.environ[env_key] = env_val
    plugin = FullyShardedDataParallelPlugin()
    assert getattr(plugin, field) == expected, (field, getattr(plugin, field))


def test_fsdp_plugin_defaults():
    plugin = FullyShardedDataParallelPlugin()
    assert plugin.sharding_strategy == "full_shard"
    assert plugin.reshard_after_forward is True
    assert plugin.fsdp_version == 2
    assert plugin.forward_prefetch is True
    assert plugin.cpu_ram_efficient_loading is False


# ---- launch CLI -> env round trip ---------------------------------------

def test_launch_cli_fsdp_roundtrip():
    args = parse_launch(
        [
            "--use_fsdp",
            "--fsdp_sharding_strategy", "hybrid_shard",
            "--fsdp_offload_params", "true",
            "--fsdp_min_num_params", "7777",
            "--fsdp_backward_prefetch", "backward_post",
            "--fsdp_cpu_ram_efficient_loading", "true",
            "--fsdp_state_dict_type", "SHARDED_STATE_DICT",
            "--mixed_precision", "bf16",
        ]
    )
    args.gradient_accumulation_steps = 1
    env = build_env(args, ClusterConfig())
    assert env["ACCELERATE_USE_FSDP"] == "1"
    assert env["FSDP_SHARDING_STRATEGY"] == "hybrid_shard"
    assert env["FSDP_OFFLOAD_PARAMS"] == "true"
    assert env["FSDP_MIN_NUM_PARAMS"] == "7777"
    assert env["FSDP_BACKWARD_PREFETCH"] == "backward_post"
    assert env["FSDP_CPU_RAM_EFFICIENT_LOADING"] == "true"
    assert env["ACCELERATE_MIXED_PRECISION"] == "bf16"
    # ...and the plugin a worker would construct from that env
    os.environ.update({k: v for k, v in env.items() if k.startswith("FSDP_")})
    plugin = FullyShardedDataParallelPlugin()
    assert plugin.sharding_strategy == "hybrid_shard"
    assert plugin.cpu_offload is True
    assert plugin.min_num_params == 7777
    assert plugin.cpu_ram_efficient_loading is True


def test_launch_cli_parallelism_roundtrip():
    args = parse_launch(
        [
            "--parallelism_config_dp_replicate_size", "2",
            "--parallelism_config_tp_size", "2",
            "--parallelism_config_cp_size", "2",
            "--parallelism_config_cp_comm_strategy", "alltoall",
        ]
    )
    args.mixed_precision = "no"
    args.gradient_accumulation_steps = 1
    env = build_env(args, ClusterConfig())
    assert env["ACCELERATE_USE_PARALLELISM_CONFIG"] == "true"
    os.environ.update({k: v for k, v in env.items() if k.startswith("PARALLELISM_CONFIG_")})
    pc = ParallelismConfig()
    assert pc.dp_replicate_size == 2
    assert pc.tp_size == 2
    assert pc.cp_size == 2
    assert pc.cp_impl == "ulysses"  # alltoall spelling maps to ulysses
    assert pc.total_size == 8


def test_launch_cli_dynamo_roundtrip():
    args = parse_launch(["--dynamo_backend", "inductor", "--dynamo_use_regional_compilation"])
    args.mixed_precision = "no"
    args.gradient_accumulation_steps = 1
    env = build_env(args, ClusterConfig())
    assert env["ACCELERATE_DYNAMO_BACKEND"] == "INDUCTOR"
    os.environ.update({k: v for k, v in env.items() if k.startswith("ACCELERATE_DYNAMO_")})
    plugin = TorchDynamoPlugin()
    assert plugin.enabled
    assert plugin.backend == "INDUCTOR"
    assert plugin.use_regional_compilation is True
    assert plugin.compile_kwargs()["backend"] == "inductor"


def test_launch_cli_fp8_roundtrip():
    args = parse_launch(["--fp8_format", "E4M3", "--fp8_amax_history_len", "32", "--fp8_margin", "1"])
    args.mixed_precision = "fp8"
    args.gradient_accumulation_steps = 1
    env = build_env(args, ClusterConfig())
    os.environ.update({k: v for k, v in env.items() if k.startswith("ACCELERATE_FP8_")})
    recipe = FP8RecipeKwargs()
    assert recipe.format == "E4M3"
    assert recipe.amax_history_len == 32
    assert recipe.margin == 1


def test_dynamo_plugin_defaults_disabled():
    plugin = TorchDynamoPlugin()
    assert not plugin.enabled


def test_parallelism_config_env_defaults():
    pc = ParallelismConfig()
    assert pc.total_size == 1 and pc.cp_impl == "allgather"


def test_parallelism_config_invalid_sizes():
    with pytest.raises(ValueError):
        ParallelismConfig(tp_size=0)
    with pytest.raises(ValueError):
        ParallelismConfig(cp_impl="bogus").validate(1)


def test_accelerator_picks_up_parallelism_env():
    os.environ["ACCELERATE_USE_PARALLELISM_CONFIG"] = "true"
    os.environ["PARALLELISM_CONFIG_TP_SIZE"] = "1"
    from accelerate_amd import Accelerator
    from accelerate_amd.state import AcceleratorState

    AcceleratorState._reset_state()
    acc = Accelerator(cpu=True)
    assert acc.parallelism_config is not None
    assert acc.parallelism_config.tp_size == 1
    AcceleratorState._reset_state()
