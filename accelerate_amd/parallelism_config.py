"""Unified multi-dimensional parallelism descriptor (reference:
parallelism_config.py:34 ``ParallelismConfig``).

Declares the degree of every dimension and builds the RCCL/gloo process
groups our engines take as ``group=`` arguments:

- dp_replicate × dp_shard — DDP replicas of sharded (FSDP/HSDP) groups
- tp — tensor parallel (parallel/tp.py Column/RowParallelLinear)
- cp — context parallel (parallel/cp.py)
- ep — expert parallel (parallel/ep.py); EP shares ranks with dp_shard in
  the usual MoE layout, so it is validated against that product rather than
  multiplied into the world size.

Rank layout is megatron-style innermost-first: tp varies fastest, then cp,
then dp_shard, then dp_replicate — so TP groups are intra-node neighbors
(xGMI-adjacent) when launched one-rank-per-GPU.
"""

import os
from dataclasses import dataclass, field
from typing import Dict, Optional

import torch.distributed as dist


@dataclass
class ParallelismConfig:
    """Fields left ``None`` fall back to the ``PARALLELISM_CONFIG_*`` env
    plane set by the launcher (reference: parallelism_config.py:274-341 —
    this is the launcher↔library ABI SURVEY.md §2.7 calls load-bearing)."""

    dp_replicate_size: Optional[int] = None
    dp_shard_size: Optional[int] = None
    tp_size: Optional[int] = None
    cp_size: Optional[int] = None
    ep_size: Optional[int] = None
    # sequence-parallel collective pattern for the cp dimension:
    # 'allgather' = KV all-gather CP (reference torch CP default rotate),
    # 'ring'      = P2P KV rotation (sequence-local memory, xGMI send/recv),
    # 'ulysses'   = dual all-to-all head resharding (reference DeepSpeed SP);
    # the reference's 'alltoall' rotate spelling is accepted as 'ulysses'
    cp_impl: Optional[str] = None

    _groups: Dict[str, Optional[object]] = field(default_factory=dict, repr=False)

    def __post_init__(self):
        env = os.environ
        if self.dp_replicate_size is None:
            self.dp_replicate_size = int(env.get("PARALLELISM_CONFIG_DP_REPLICATE_SIZE", "1"))
        if self.dp_shard_size is None:
            self.dp_shard_size = int(env.get("PARALLELISM_CONFIG_DP_SHARD_SIZE", "1"))
        if self.tp_size is None:
            self.tp_size = int(env.get("PARALLELISM_CONFIG_TP_SIZE", "1"))
        if self.cp_size is None:
            self.cp_size = int(env.get("PARALLELISM_CONFIG_CP_SIZE", "1"))
        if self.ep_size is None:
            self.ep_size = int(env.get("PARALLELISM_CONFIG_EP_SIZE", "1"))
        if self.cp_impl is None:
            self.cp_impl = env.get("PARALLELISM_CONFIG_CP_COMM_STRATEGY", "allgather")
        if self.cp_impl == "alltoall":
            self.cp_impl = "ulysses"
        for name in ("dp_replicate_size", "dp_shard_size", "tp_size", "cp_size", "ep_size"):
            if getattr(self, name) < 1:
                raise ValueError(f"{name} must be at least 1, got {getattr(self, name)}")

    @property
    def total_size(self) -> int:
        return self.dp_replicate_size * self.dp_shard_size * self.tp_size * self.cp_size

    @property
    def dp_size(self) -> int:
        return self.dp_replicate_size * self.dp_shard_size

    def validate(self, world_size: int):
        if self.cp_impl not in ("allgather", "ring", "ulysses"):
            raise ValueError(
                f"cp_impl must be 'allgather', 'ring' or 'ulysses', got {self.cp_impl!r}"
            )
        if self.total_size != world_size:
            raise ValueError(
                f"ParallelismConfig total_size ({self.total_size}) != world size ({world_size}); "
                f"dims: dp_replicate={self.dp_replicate_size} dp_shard={self.dp_shard_size} "
                f"tp={self.tp_size} cp={self.cp_size}"
            )
        if self.ep_size > 1 and self.dp_shard_size % self.ep_size != 0:
            raise ValueError(
                f"ep_size ({self.ep_size}) must divide dp_shard_size ({self.dp_shard_size}) — "
                "experts are sharded across (a subset of) the data-parallel shard group"
            )

    # -- rank coordinates (innermost-first: tp, cp, dp_shard, dp_replicate) --
    def coords(self, rank: int):
        tp = rank % self.tp_size
        cp = (rank // self.tp_size) % self.cp_size
        shard = (rank // (self.tp_size * self.cp_size)) % self.dp_shard_size
        repl = rank // (self.tp_size * self.cp_size * self.dp_shard_size)
        return {"tp": tp, "cp": cp, "dp_shard": shard, "dp_replicate": repl}

    def build_groups(self) -> Dict[str, Optional[object]]:
        """Create one process group per dimension containing THIS rank.

        Every rank must call this (new_group is collective). Returns
        {"tp": pg, "cp": pg, "dp_shard": pg, "dp_replicate": pg, "dp": pg};
        a size-1 dimension maps to None (engines treat None as 'world' or
        'off' appropriately — pass the group explicitly).
        """
        if not dist.is_initialized():
            raise RuntimeError("ParallelismConfig.build_groups needs torch.distributed initialized")
        world = dist.get_world_size()
        self.validate(world)
        rank = dist.get_rank()

        def ranks_varying(varying):
            """All rank lists where `varying` sweeps and other dims are fixed."""
            out = []
            for r in range(world):
                c = self.coords(r)
                key = tuple(c[d] for d in ("tp", "cp", "dp_shard", "dp_replicate") if d != varying)
                out.append((key, r))
            groups = {}
            for key, r in out:
                groups.setdefault(key, []).append(r)
            return list(groups.values())

        result: Dict[str, Optional[object]] = {}
        for dim, size in (
            ("tp", self.tp_size),
            ("cp", self.cp_size),
            ("dp_shard", self.dp_shard_size),
            ("dp_replicate", self.dp_replicate_size),
        ):
            if size == 1:
                result[dim] = None
                continue
            mine = None
            for ranks in ranks_varying(dim):
                pg = dist.new_group(ranks)
                if rank in ranks:
                    mine = pg
            result[dim] = mine
        # combined DP group (replicate x shard): the batch-sharding domain
        if self.dp_size in (world, 1):
            result["dp"] = None  # whole world / off — use the default group
        else:
            mine = None
            tpcp = self.tp_size * self.cp_size
            lists = {}
            for r in range(world):
                lists.setdefault(r % tpcp, []).append(r)
            for ranks in lists.values():
                pg = dist.new_group(ranks)
                if rank in ranks:
                    mine = pg
            result["dp"] = mine
        # gradient-averaging domain = dp x cp (cp ranks hold REPLICATED
        # params over different sequence shards; a mean-reduced loss means
        # the true gradient is the cp-average too — the reference's
        # dp_shard_cp mesh flattening, parallelism_config.py:136-143)
        grad_size = self.dp_size * self.cp_size
        if grad_size in (world, 1):
            result["grad"] = None
        elif self.cp_size == 1:
            result["grad"] = result["dp"]
        else:
            mine = None
            lists = {}
            for r in range(world):
                lists.setdefault(r % self.tp_size, []).append(r)
            for ranks in lists.values():
                pg = dist.new_group(ranks)
                if rank in ranks:
                    mine = pg
            result["grad"] = mine
        self._groups = result
        return result
