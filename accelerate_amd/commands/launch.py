"""`accelerate-amd launch` (reference: commands/launch.py + utils/launch.py).

Merges YAML config + CLI flags, builds the ACCELERATE_* env ABI and
dispatches:
- multi-process (one per MI355X GPU): torch.distributed.run in-process
  (the elastic agent spawns N local workers over RCCL)
- single process: plain subprocess
"""

import argparse
import os
import subprocess
import sys
from pathlib import Path

from .config import DEFAULT_CONFIG_FILE, ClusterConfig


class _DashAliasParserProxy:
    """Registers every ``--foo_bar`` flag with a ``--foo-bar`` alias so both
    spellings work (reference: launch flags accept hyphens AND underscores,
    tests/test_cli.py test_hyphen/test_underscore)."""

    def __init__(self, parser):
        self._parser = parser

    def add_argument(self, name, *args, **kwargs):
        names = [name]
        if name.startswith("--") and "_" in name:
            names.append(name.replace("_", "-"))
            kwargs.setdefault("dest", name[2:])
        return self._parser.add_argument(*names, *args, **kwargs)

    def __getattr__(self, item):
        return getattr(self._parser, item)


def add_parser(subparsers):
    real_parser = subparsers.add_parser("launch", help="Launch a training script on MI355X GPUs")
    parser = _DashAliasParserProxy(real_parser)
    parser.add_argument("--config_file", default=None)
    parser.add_argument("--num_processes", type=int, default=None, help="Total number of processes (one per GPU)")
    parser.add_argument("--num_machines", type=int, default=None)
    parser.add_argument("--machine_rank", type=int, default=None)
    parser.add_argument("--main_process_ip", default=None)
    parser.add_argument("--main_process_port", type=int, default=None)
    parser.add_argument("--mixed_precision", default=None, choices=["no", "fp16", "bf16", "fp8"])
    parser.add_argument("--cpu", action="store_true", help="Force CPU-only")
    parser.add_argument("--multi_gpu", action="store_true", help="Force multi-GPU even with config absent")
    parser.add_argument("--use_fsdp", action="store_true", help="Use the sharded-parameter engine")
    parser.add_argument("--fsdp_sharding_strategy", default=None,
                        help="full_shard | hybrid_shard (reference --fsdp_sharding_strategy)")
    parser.add_argument("--fsdp_transformer_layer_cls_to_wrap", default=None,
                        help="Comma-separated module class names forming shard units")
    parser.add_argument("--fsdp_shard_group_size", type=int, default=None,
                        help="HSDP: ranks per shard group (replicas = world/size)")
    parser.add_argument("--fsdp_state_dict_type", default=None,
                        help="FULL_STATE_DICT | SHARDED_STATE_DICT")
    parser.add_argument("--fsdp_activation_checkpointing", default=None,
                        help="true|false: checkpoint each wrapped unit")
    parser.add_argument("--fsdp_offload_params", default=None, help="true|false")
    parser.add_argument("--fsdp_min_num_params", type=int, default=None)
    parser.add_argument("--fsdp_auto_wrap_policy", default=None)
    parser.add_argument("--fsdp_reshard_after_forward", default=None, help="true|false")
    parser.add_argument("--fsdp_forward_prefetch", default=None, help="true|false")
    parser.add_argument("--fsdp_backward_prefetch", default=None,
                        help="backward_pre | backward_post | no_prefetch")
    parser.add_argument("--fsdp_cpu_ram_efficient_loading", default=None,
                        help="true|false: meta-init + per-rank sliced checkpoint load")
    parser.add_argument("--fsdp_sync_module_states", default=None, help="true|false")
    parser.add_argument("--fsdp_use_orig_params", default=None, help="true|false")
    parser.add_argument("--fsdp_version", type=int, default=None)
    # multi-dimensional parallelism plane (reference: utils/launch.py:400-425)
    parser.add_argument("--parallelism_config_dp_replicate_size", type=int, default=None)
    parser.add_argument("--parallelism_config_dp_shard_size", type=int, default=None)
    parser.add_argument("--parallelism_config_tp_size", type=int, default=None)
    parser.add_argument("--parallelism_config_cp_size", type=int, default=None)
    parser.add_argument("--parallelism_config_cp_comm_strategy", default=None,
                        help="allgather | ulysses (alltoall accepted as ulysses)")
    # torch.compile plane (reference: utils/launch.py:189-193)
    parser.add_argument("--dynamo_backend", default=None, help="e.g. inductor; NO disables")
    parser.add_argument("--dynamo_mode", default=None,
                        help="default | reduce-overhead | max-autotune")
    parser.add_argument("--dynamo_use_fullgraph", action="store_true")
    parser.add_argument("--dynamo_use_dynamic", action="store_true")
    parser.add_argument("--dynamo_use_regional_compilation", action="store_true")
    # fp8 recipe plane (our CDNA4 recipe, utils/dataclasses.py FP8RecipeKwargs)
    parser.add_argument("--fp8_format", default=None, help="E4M3 | HYBRID")
    parser.add_argument("--fp8_amax_history_len", type=int, default=None)
    parser.add_argument("--fp8_margin", type=int, default=None)
    parser.add_argument("--num_cpu_threads_per_process", type=int, default=None,
                        help="Sets OMP_NUM_THREADS per worker")
    parser.add_argument("--gpu_ids", default=None, help="Comma-separated HIP device ids to use")
    parser.add_argument("--gradient_accumulation_steps", type=int, default=None)
    parser.add_argument("--debug", action="store_true")
    parser.add_argument(
        "--rccl_debug", default=None,
        help="Set NCCL_DEBUG (=RCCL's debug plane on ROCm) for every worker, e.g. INFO or TRACE",
    )
    parser.add_argument("--max_restarts", type=int, default=0)
    parser.add_argument("--monitor_interval", type=float, default=0.1)
    parser.add_argument("--rdzv_backend", default="static")
    parser.add_argument("--rdzv_conf", default="")
    parser.add_argument("--module", action="store_true", help="Treat script as a python module")
    parser.add_argument("--no_python", action="store_true")
    parser.add_argument("training_script", help="The script to launch")
    parser.add_argument("training_script_args", nargs=argparse.REMAINDER, help="Script arguments")
    parser.set_defaults(func=launch_command)
    return parser


def _load_config(args) -> ClusterConfig:
    path = args.config_file or (DEFAULT_CONFIG_FILE if DEFAULT_CONFIG_FILE.exists() else None)
    if path is not None and Path(path).exists():
        return ClusterConfig.load(path)
    return ClusterConfig()


def _resolve(args, config: ClusterConfig):
    """CLI > config-file > detection (reference: launch.py:1196-1381)."""
    import torch

    if args.num_machines is None:
        args.num_machines = config.num_machines
    if args.machine_rank is None:
        args.machine_rank = config.machine_rank
    if args.main_process_ip is None:
        args.main_process_ip = config.main_process_ip or "127.0.0.1"
    if args.main_process_port is None:
        args.main_process_port = config.main_process_port or 29500
    if args.mixed_precision is None:
        args.mixed_precision = config.mixed_precision
    if args.gradient_accumulation_steps is None:
        args.gradient_accumulation_steps = config.gradient_accumulation_steps
    if args.gpu_ids is None:
        args.gpu_ids = config.gpu_ids or "all"
    if not args.cpu:
        args.cpu = config.use_cpu or config.distributed_type == "MULTI_CPU"
    if not args.use_fsdp:
        args.use_fsdp = config.distributed_type == "FSDP"
    if args.num_processes is None:
        if config.distributed_type in ("MULTI_GPU", "FSDP", "MULTI_CPU"):
            args.num_processes = config.num_processes
        elif not args.cpu and torch.cuda.is_available() and (args.multi_gpu or args.use_fsdp):
            args.num_processes = torch.cuda.device_count()
        else:
            args.num_processes = 1
    return args


def build_env(args, config: ClusterConfig) -> dict:
    """The env-var ABI consumed by Accelerator/PartialState in workers
    (reference: utils/launch.py:201-427)."""
    env = os.environ.copy()
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    env["ACCELERATE_MIXED_PRECISION"] = str(args.mixed_precision)
    env["ACCELERATE_GRADIENT_ACCUMULATION_STEPS"] = str(args.gradient_accumulation_steps)
    if args.debug or config.debug:
        env["ACCELERATE_DEBUG_MODE"] = "1"
    if args.rccl_debug:
        # RCCL reads the NCCL_* env plane on ROCm (SURVEY §5.2 plumb-through)
        env["NCCL_DEBUG"] = str(args.rccl_debug)
    if args.cpu:
        env["ACCELERATE_USE_CPU"] = "1"
    if args.use_fsdp:
        env["ACCELERATE_USE_FSDP"] = "1"
        for key, value in (config.fsdp_config or {}).items():
            env[key.replace("fsdp_", "FSDP_").upper()] = str(value)
        # CLI flags override the config file (reference flag precedence)
        cli_fsdp = {
            "FSDP_SHARDING_STRATEGY": args.fsdp_sharding_strategy,
            "FSDP_TRANSFORMER_CLS_TO_WRAP": args.fsdp_transformer_layer_cls_to_wrap,
            "FSDP_SHARD_GROUP_SIZE": args.fsdp_shard_group_size,
            "FSDP_STATE_DICT_TYPE": args.fsdp_state_dict_type,
            "FSDP_ACTIVATION_CHECKPOINTING": args.fsdp_activation_checkpointing,
            "FSDP_OFFLOAD_PARAMS": args.fsdp_offload_params,
            "FSDP_MIN_NUM_PARAMS": args.fsdp_min_num_params,
            "FSDP_AUTO_WRAP_POLICY": args.fsdp_auto_wrap_policy,
            "FSDP_RESHARD_AFTER_FORWARD": args.fsdp_reshard_after_forward,
            "FSDP_FORWARD_PREFETCH": args.fsdp_forward_prefetch,
            "FSDP_BACKWARD_PREFETCH": args.fsdp_backward_prefetch,
            "FSDP_CPU_RAM_EFFICIENT_LOADING": args.fsdp_cpu_ram_efficient_loading,
            "FSDP_SYNC_MODULE_STATES": args.fsdp_sync_module_states,
            "FSDP_USE_ORIG_PARAMS": args.fsdp_use_orig_params,
            "FSDP_VERSION": args.fsdp_version,
        }
        for key, value in cli_fsdp.items():
            if value is not None:
                env[key] = str(value)
    # parallelism plane (reference: utils/launch.py:400-425)
    pc_cli = {
        "PARALLELISM_CONFIG_DP_REPLICATE_SIZE": args.parallelism_config_dp_replicate_size,
        "PARALLELISM_CONFIG_DP_SHARD_SIZE": args.parallelism_config_dp_shard_size,
        "PARALLELISM_CONFIG_TP_SIZE": args.parallelism_config_tp_size,
        "PARALLELISM_CONFIG_CP_SIZE": args.parallelism_config_cp_size,
        "PARALLELISM_CONFIG_CP_COMM_STRATEGY": args.parallelism_config_cp_comm_strategy,
    }
    if any(v is not None for v in pc_cli.values()):
        env["ACCELERATE_USE_PARALLELISM_CONFIG"] = "true"
        for key, value in pc_cli.items():
            if value is not None:
                env[key] = str(value)
    # torch.compile plane
    if args.dynamo_backend is not None:
        env["ACCELERATE_DYNAMO_BACKEND"] = str(args.dynamo_backend).upper()
        if args.dynamo_mode is not None:
            env["ACCELERATE_DYNAMO_MODE"] = str(args.dynamo_mode)
        env["ACCELERATE_DYNAMO_USE_FULLGRAPH"] = str(args.dynamo_use_fullgraph)
        env["ACCELERATE_DYNAMO_USE_DYNAMIC"] = str(args.dynamo_use_dynamic)
        env["ACCELERATE_DYNAMO_USE_REGIONAL_COMPILATION"] = str(args.dynamo_use_regional_compilation)
    # fp8 recipe plane
    for key, value in {
        "ACCELERATE_FP8_FORMAT": args.fp8_format,
        "ACCELERATE_FP8_AMAX_HISTORY_LEN": args.fp8_amax_history_len,
        "ACCELERATE_FP8_MARGIN": args.fp8_margin,
    }.items():
        if value is not None:
            env[key] = str(value)
    if args.num_cpu_threads_per_process is not None:
        env["OMP_NUM_THREADS"] = str(args.num_cpu_threads_per_process)
    if args.gpu_ids not in (None, "all"):
        env["HIP_VISIBLE_DEVICES"] = str(args.gpu_ids)
    if config.enable_cpu_affinity:
        env["ACCELERATE_CPU_AFFINITY"] = "1"
    return env


def multi_process_launcher(args, config):
    from torch.distributed.run import run as distrib_run
    from torch.distributed.run import get_args_parser

    env = build_env(args, config)
    os.environ.update(env)
    nproc_per_node = args.num_processes // args.num_machines
    cmd_args = [
        f"--nproc-per-node={nproc_per_node}",
        f"--nnodes={args.num_machines}",
        f"--max-restarts={args.max_restarts}",
        f"--monitor-interval={args.monitor_interval}",
    ]
    if args.num_machines > 1 or args.rdzv_backend != "static":
        cmd_args += [
            f"--node-rank={args.machine_rank}",
            f"--rdzv-backend={args.rdzv_backend or 'c10d'}",
            f"--rdzv-endpoint={args.main_process_ip}:{args.main_process_port}",
        ]
        if args.rdzv_conf:
            cmd_args.append(f"--rdzv-conf={args.rdzv_conf}")
    else:
        cmd_args += [
            "--node-rank=0",
            f"--master-addr={args.main_process_ip}",
            f"--master-port={args.main_process_port}",
        ]
    if args.module:
        cmd_args.append("--module")
    elif args.no_python:
        cmd_args.append("--no-python")
    cmd_args.append(args.training_script)
    cmd_args += args.training_script_args
    parsed = get_args_parser().parse_args(cmd_args)
    distrib_run(parsed)


def simple_launcher(args, config):
    env = build_env(args, config)
    cmd = []
    if not args.no_python:
        cmd.append(sys.executable)
        if args.module:
            cmd.append("-m")
    cmd.append(args.training_script)
    cmd.extend(args.training_script_args)
    process = subprocess.run(cmd, env=env)
    if process.returncode != 0:
        sys.exit(process.returncode)


def launch_command(args):
    config = _load_config(args)
    args = _resolve(args, config)
    if args.num_processes > 1:
        multi_process_launcher(args, config)
    else:
        simple_launcher(args, config)
