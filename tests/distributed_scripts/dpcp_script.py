"""4-process gloo oracle: dp_replicate=2 x cp=2 through plain
`Accelerator(parallelism_config=...).prepare()` — the DDP reducer must run
over the flattened dp x cp "grad" group (reference: grads average over
every rank that holds a data/sequence shard), and `maybe_context_parallel`
shards each dp replica's batch along the sequence.

One training step must reproduce a single-process reference trained on the
mean of the two dp halves' full-sequence losses: with equal sequence
shards, averaging the four (dp, cp) shard-mean grads equals that
reference gradient exactly.
"""

import torch
import torch.distributed as dist

from accelerate_amd import Accelerator, ParallelismConfig, set_seed
from accelerate_amd.models.llama import LlamaConfig, LlamaForCausalLM


def main():
    pc = ParallelismConfig(dp_replicate_size=2, cp_size=2, cp_impl="allgather")
    acc = Accelerator(cpu=True, parallelism_config=pc)
    assert acc.num_processes == 4
    me = pc.coords(acc.process_index)
    dp = me["dp_replicate"]

    set_seed(0)
    model = LlamaForCausalLM(LlamaConfig.tiny(num_hidden_layers=2))
    set_seed(0)
    ref = LlamaForCausalLM(LlamaConfig.tiny(num_hidden_layers=2))

    opt = torch.optim.SGD(model.parameters(), lr=0.05)
    ref_opt = torch.optim.SGD(ref.parameters(), lr=0.05)
    model, opt = acc.prepare(model, opt)

    S = 16
    g = torch.Generator().manual_seed(11)
    X = torch.randint(0, 1024, (2, 2, S), generator=g)  # [dp half][batch 2][S]
    ids = X[dp].clone()

    opt.zero_grad()
    shard = [ids]
    with acc.maybe_context_parallel(buffers=shard, buffer_seq_dims=[1]):
        local = shard[0]
        assert local.shape[1] == S // 2  # cp=2 sequence shard
        loss = model(local)["logits"].float().pow(2).mean()
        acc.backward(loss)
    opt.step()

    # reference: mean of the two dp halves' full-sequence losses
    ref_opt.zero_grad()
    (0.5 * (ref(X[0])["logits"].float().pow(2).mean() + ref(X[1])["logits"].float().pow(2).mean())).backward()
    ref_opt.step()

    inner = acc.unwrap_model(model)
    for (pn, p), (_, rp) in zip(inner.named_parameters(), ref.named_parameters()):
        assert torch.allclose(p, rp, atol=1e-5), f"dpxcp mismatch {pn}: {(p - rp).abs().max()}"

    # all four ranks end with identical weights (grad group = dp x cp)
    w = inner.layers[0].self_attn.q_proj.weight.detach()
    ws = [torch.empty_like(w) for _ in range(4)]
    dist.all_gather(ws, w)
    for r in range(1, 4):
        assert torch.equal(ws[0], ws[r])
    if acc.is_main_process:
        print("DPCP_PREPARE_PASS")
    acc.end_training()


if __name__ == "__main__":
    main()
