"""4-process gloo oracle: TP (tp=2) x FSDP (dp_shard=2) — the reference's
2D FSDP2+TP layout. The tp plan shards each Linear's weights; the
flat-shard engine must then shard/reduce ONLY over the dp group for this
tp coordinate (the "grad" group) — sharding over the world group would
average the two DIFFERENT tp shards together.

Checks: one training step through plain prepare(), then forward parity of
the stepped model against a single-process reference stepped on the mean
of the two dp halves' losses (row-parallel outputs are tp-replicated, so
logit parity proves the weights updated correctly on every rank).
"""

import torch
import torch.distributed as dist

from accelerate_amd import Accelerator, ParallelismConfig, set_seed
from accelerate_amd.models.llama import LlamaConfig, LlamaForCausalLM
from accelerate_amd.utils import FullyShardedDataParallelPlugin


def main():
    pc = ParallelismConfig(tp_size=2, dp_shard_size=2)
    plugin = FullyShardedDataParallelPlugin()
    acc = Accelerator(cpu=True, parallelism_config=pc, fsdp_plugin=plugin)
    assert acc.num_processes == 4
    me = pc.coords(acc.process_index)
    dp = me["dp_shard"]

    set_seed(0)
    model = LlamaForCausalLM(LlamaConfig.tiny(num_hidden_layers=2))
    set_seed(0)
    ref = LlamaForCausalLM(LlamaConfig.tiny(num_hidden_layers=2))

    opt = torch.optim.SGD(model.parameters(), lr=0.05)
    ref_opt = torch.optim.SGD(ref.parameters(), lr=0.05)
    model, opt = acc.prepare(model, opt)

    S = 16
    g = torch.Generator().manual_seed(11)
    X = torch.randint(0, 1024, (2, 2, S), generator=g)  # [dp half][batch][S]
    ids = X[dp].clone()

    opt.zero_grad()
    loss = model(ids)["logits"].float().pow(2).mean()
    acc.backward(loss)
    opt.step()

    ref_opt.zero_grad()
    (0.5 * (ref(X[0])["logits"].float().pow(2).mean() + ref(X[1])["logits"].float().pow(2).mean())).backward()
    ref_opt.step()

    with torch.no_grad():
        out = model(X[0])["logits"]
        want = ref(X[0])["logits"]
    assert torch.allclose(out, want, atol=1e-4), f"tp x fsdp step diverges: {(out - want).abs().max()}"
    dist.barrier()
    if acc.is_main_process:
        print("TP_FSDP_PASS")
    acc.end_training()


if __name__ == "__main__":
    main()
