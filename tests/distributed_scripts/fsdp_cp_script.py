"""4-process gloo oracle: FSDP (dp_shard=2) x CP (cp=2) — the reference's
flagship pairing (FSDP2 + context_parallel; its mesh flattens the shard
dim to dp_shard x cp). Our ShardedModel shards flat-params over the
DEFAULT group (all 4 ranks = the flattened dp_shard x cp domain) while
`maybe_context_parallel` shards each batch's sequence over the cp group:
the reduce-scattered gradient is then the mean over (batch shard x
sequence shard) — one training step must match a single-process reference
trained on the mean of the two dp halves' full-sequence losses."""

import torch
import torch.distributed as dist

from accelerate_amd import Accelerator, ParallelismConfig, set_seed
from accelerate_amd.models.llama import LlamaConfig, LlamaForCausalLM
from accelerate_amd.utils import FullyShardedDataParallelPlugin


def main():
    pc = ParallelismConfig(dp_shard_size=2, cp_size=2, cp_impl="allgather")
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
    X = torch.randint(0, 1024, (2, 2, S), generator=g)  # [dp shard][batch 2][S]
    ids = X[dp].clone()

    opt.zero_grad()
    shard = [ids]
    with acc.maybe_context_parallel(buffers=shard, buffer_seq_dims=[1]):
        local = shard[0]
        assert local.shape[1] == S // 2
        loss = model(local)["logits"].float().pow(2).mean()
        acc.backward(loss)
    opt.step()

    ref_opt.zero_grad()
    (0.5 * (ref(X[0])["logits"].float().pow(2).mean() + ref(X[1])["logits"].float().pow(2).mean())).backward()
    ref_opt.step()

    # compare via the gathered full state dict
    from accelerate_amd.parallel.fsdp import gather_full_state_dict

    full = gather_full_state_dict(model)
    ref_sd = ref.state_dict()
    for k, v in ref_sd.items():
        assert torch.allclose(full[k], v, atol=1e-5), f"fsdp x cp mismatch {k}: {(full[k] - v).abs().max()}"
    dist.barrier()
    if acc.is_main_process:
        print("FSDP_CP_PASS")
    acc.end_training()


if __name__ == "__main__":
    main()
