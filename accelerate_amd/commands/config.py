"""`accelerate-amd config` — interactive questionnaire → YAML
(reference: commands/config/). The YAML is one of three config planes:
file → launch flags → ACCELERATE_* env vars (the launcher↔library ABI).
"""

import argparse
import os
from dataclasses import asdict, dataclass, field
from pathlib import Path
from typing import List, Optional

import yaml

DEFAULT_CONFIG_DIR = Path(os.environ.get("ACCELERATE_CONFIG_DIR", Path.home() / ".cache" / "accelerate_amd"))
DEFAULT_CONFIG_FILE = DEFAULT_CONFIG_DIR / "default_config.yaml"


@dataclass
class ClusterConfig:
    """(reference: commands/config/config_args.py:179)"""

    compute_environment: str = "LOCAL_MACHINE"
    distributed_type: str = "NO"  # NO | MULTI_GPU | MULTI_CPU | FSDP
    num_processes: int = 1
    num_machines: int = 1
    machine_rank: int = 0
    main_process_ip: Optional[str] = None
    main_process_port: Optional[int] = None
    mixed_precision: str = "no"  # no | fp16 | bf16 | fp8
    gpu_ids: Optional[str] = "all"
    gradient_accumulation_steps: int = 1
    use_cpu: bool = False
    debug: bool = False
    fsdp_config: dict = field(default_factory=dict)
    dynamo_config: dict = field(default_factory=dict)
    downcast_bf16: bool = False
    enable_cpu_affinity: bool = False

    def to_dict(self):
        result = asdict(self)
        return {k: v for k, v in result.items() if v is not None}

    def save(self, path=None):
        path = Path(path or DEFAULT_CONFIG_FILE)
        path.parent.mkdir(parents=True, exist_ok=True)
        with open(path, "w") as f:
            yaml.safe_dump(self.to_dict(), f)
        return path

    _VALID_DISTRIBUTED = ("NO", "MULTI_GPU", "MULTI_CPU", "FSDP")
    _VALID_PRECISION = ("no", "fp16", "bf16", "fp8")

    @classmethod
    def load(cls, path=None):
        path = Path(path or DEFAULT_CONFIG_FILE)
        with open(path) as f:
            data = yaml.safe_load(f) or {}
        known = {f_.name for f_ in cls.__dataclass_fields__.values()}
        unknown = sorted(set(data) - known)
        if unknown:
            raise ValueError(
                f"Config file {path} contains unknown key(s): {', '.join(unknown)}. "
                f"Valid keys: {', '.join(sorted(known))}."
            )
        cfg = cls(**data)
        if cfg.distributed_type not in cls._VALID_DISTRIBUTED:
            raise ValueError(
                f"distributed_type {cfg.distributed_type!r} is not one of {cls._VALID_DISTRIBUTED}"
            )
        if cfg.mixed_precision not in cls._VALID_PRECISION:
            raise ValueError(f"mixed_precision {cfg.mixed_precision!r} is not one of {cls._VALID_PRECISION}")
        return cfg


def _ask(prompt, default=None, cast=str, choices=None):
    suffix = f" [{default}]" if default is not None else ""
    while True:
        raw = input(f"{prompt}{suffix}: ").strip()
        if raw == "" and default is not None:
            return default
        try:
            val = cast(raw)
        except (TypeError, ValueError):
            print("Invalid value, try again.")
            continue
        if choices is not None and val not in choices:
            print(f"Choose one of {choices}.")
            continue
        return val


def _ask_bool(prompt, default=False):
    val = _ask(prompt + " (yes/NO)" if not default else prompt + " (YES/no)", "yes" if default else "no")
    return str(val).lower() in ("1", "yes", "y", "true")


def get_cluster_input() -> ClusterConfig:
    """Interactive Q&A (reference: commands/config/cluster.py:59)."""
    import torch

    n_gpus = torch.cuda.device_count() if torch.cuda.is_available() else 0
    dist = _ask(
        "Which type of machine are you using? (NO = single process, MULTI_GPU, MULTI_CPU, FSDP)",
        "MULTI_GPU" if n_gpus > 1 else "NO",
        str,
        ["NO", "MULTI_GPU", "MULTI_CPU", "FSDP"],
    )
    cfg = ClusterConfig(distributed_type=dist)
    if dist != "NO":
        cfg.num_machines = _ask("How many different machines will you use (in total)?", 1, int)
        if cfg.num_machines > 1:
            cfg.machine_rank = _ask("What is the rank of this machine?", 0, int)
            cfg.main_process_ip = _ask("What is the IP address of the machine that hosts rank 0?", "127.0.0.1")
            cfg.main_process_port = _ask("What is the port of the machine that hosts rank 0?", 29500, int)
        cfg.num_processes = _ask(
            "How many processes should be launched in total (one per MI355X GPU)?",
            max(n_gpus, 1) * cfg.num_machines,
            int,
        )
    if dist == "FSDP":
        cfg.fsdp_config = {
            "fsdp_sharding_strategy": _ask("Sharding strategy?", "full_shard", str, ["full_shard", "hybrid_shard", "no_shard"]),
            "fsdp_reshard_after_forward": _ask_bool("Reshard parameters after forward?", True),
            "fsdp_activation_checkpointing": _ask_bool("Use activation checkpointing?", False),
            "fsdp_state_dict_type": _ask("State-dict type?", "full_state_dict", str, ["full_state_dict", "sharded_state_dict"]),
        }
    cfg.mixed_precision = _ask("Mixed precision? (no/fp16/bf16/fp8)", "bf16" if n_gpus else "no", str, ["no", "fp16", "bf16", "fp8"])
    cfg.gradient_accumulation_steps = _ask("Gradient accumulation steps?", 1, int)
    return cfg


def config_command(args):
    if args.default:
        import torch

        n = torch.cuda.device_count() if torch.cuda.is_available() else 1
        cfg = ClusterConfig(
            distributed_type="MULTI_GPU" if n > 1 else "NO",
            num_processes=n,
            mixed_precision="bf16" if torch.cuda.is_available() else "no",
        )
    else:
        cfg = get_cluster_input()
    path = cfg.save(args.config_file)
    print(f"accelerate_amd configuration saved at {path}")


def add_parser(subparsers: argparse._SubParsersAction):
    parser = subparsers.add_parser("config", help="Create the launch configuration")
    parser.add_argument("--config_file", default=None, help="Where to save the config file")
    parser.add_argument("--default", action="store_true", help="Write a sensible default config without questions")
    parser.set_defaults(func=config_command)
    return parser
