"""2-process gloo oracle for ParallelismConfig.build_groups: group membership
and sizes per dimension for tp=2 and dp_shard=2 layouts."""

import torch.distributed as dist

from accelerate_amd import Accelerator, ParallelismConfig


def main():
    acc = Accelerator(cpu=True)
    n, r = acc.num_processes, acc.process_index
    assert n == 2

    pc = ParallelismConfig(tp_size=2)
    groups = pc.build_groups()
    assert groups["dp_shard"] is None and groups["cp"] is None
    assert dist.get_world_size(groups["tp"]) == 2
    assert groups["dp"] is None or dist.get_world_size(groups["dp"]) == 1

    pc2 = ParallelismConfig(dp_shard_size=2)
    g2 = pc2.build_groups()
    assert g2["tp"] is None
    assert g2["dp_shard"] is not None and dist.get_world_size(g2["dp_shard"]) == 2
    assert g2["dp"] is None  # dp == world -> default group

    if acc.is_main_process:
        print("PCONFIG_PASS")
    acc.end_training()


if __name__ == "__main__":
    main()
