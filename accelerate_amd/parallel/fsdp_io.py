"""Sharded-checkpoint IO for the MI355X sharded engine
(reference: fsdp_utils.py:103-277 save/load + merge_fsdp_weights :462).

Layout:
    {dir}/shard_rank{r}.bin   # ShardedModel.sharded_state_dict() per rank
Merging needs no process group — it is a pure file operation.
"""

import os
import re
from pathlib import Path

import torch


def save_fsdp_sharded_checkpoint(model, directory):
    """Every rank writes its own shard file (no communication)."""
    import torch.distributed as dist

    directory = Path(directory)
    directory.mkdir(parents=True, exist_ok=True)
    rank = dist.get_rank() if dist.is_initialized() else 0
    torch.save(model.sharded_state_dict(), directory / f"shard_rank{rank}.bin")


def load_fsdp_sharded_checkpoint(model, directory):
    import torch.distributed as dist

    rank = dist.get_rank() if dist.is_initialized() else 0
    sd = torch.load(Path(directory) / f"shard_rank{rank}.bin", weights_only=False)
    model.load_sharded_state_dict(sd)


def merge_fsdp_weights(checkpoint_directory, output_path, safe_serialization: bool = True):
    """Merge per-rank shard files into one full state dict
    (reference: fsdp_utils.py:462, `accelerate merge-weights`)."""
    directory = Path(checkpoint_directory)
    files = sorted(directory.glob("shard_rank*.bin"), key=lambda p: int(re.findall(r"\d+", p.stem)[-1]))
    if not files:
        raise FileNotFoundError(f"No shard_rank*.bin files in {directory}")
    shards = [torch.load(f, weights_only=False) for f in files]
    world = shards[0]["world_size"]
    if len(shards) != world:
        raise ValueError(f"Expected {world} shard files, found {len(shards)}")

    full_sd = {}
    for unit_name, meta in shards[0]["units"].items():
        flat = torch.cat([s["units"][unit_name]["shard"] for s in shards])
        assert flat.numel() == meta["padded"], (unit_name, flat.numel(), meta["padded"])
        off = 0
        for pname, shape in zip(meta["param_names"], meta["shapes"]):
            n = 1
            for d in shape:
                n *= d
            full_sd[pname] = flat[off : off + n].view(shape).clone()
            off += n
    output_path = Path(output_path)
    if output_path.suffix in (".bin", ".pt", ".safetensors"):
        output_path.parent.mkdir(parents=True, exist_ok=True)
        target = output_path
    else:
        output_path.mkdir(parents=True, exist_ok=True)
        target = output_path / ("model.safetensors" if safe_serialization else "pytorch_model.bin")
    if safe_serialization and str(target).endswith(".safetensors"):
        import safetensors.torch

        safetensors.torch.save_file(full_sd, target, metadata={"format": "pt"})
    else:
        torch.save(full_sd, target)
    return str(target)
