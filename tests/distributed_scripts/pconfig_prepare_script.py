"""4-process gloo oracle: TP x DP routed ENTIRELY through
`Accelerator(parallelism_config=...).prepare(model, optimizer, dataloader)`
with NO model-specific calls — the model's class `tp_plan` drives sharding
(VERDICT round-1 item 4; reference flow: accelerator.py:1531-1560).

Covers BOTH in-repo decoder families (Llama incl. GQA head split, GPT-2
incl. fused-qkv colwise_fused3) and checks:
- dataloader shards over the dp dimension (tp peers see the SAME batch)
- forward parity vs a single-process reference model
- one optimizer step: updated TP shards still match the reference's
  correspondingly-updated full weights (grad all-reduce on dp only)
"""

import torch
import torch.distributed as dist
from torch.utils.data import DataLoader, TensorDataset

from accelerate_amd import Accelerator, ParallelismConfig, set_seed
from accelerate_amd.models.gpt2 import GPT2Config, GPT2LMHeadModel
from accelerate_amd.models.llama import LlamaConfig, LlamaForCausalLM


def run_family(acc, pc, name, make_model, vocab):
    set_seed(0)
    ref = make_model()
    set_seed(0)
    model = make_model()

    opt = torch.optim.SGD(model.parameters(), lr=0.05)
    ref_opt = torch.optim.SGD(ref.parameters(), lr=0.05)

    g = torch.Generator().manual_seed(7)  # SAME dataset on every rank
    X = torch.randint(0, vocab, (8, 16), generator=g)
    dl = DataLoader(TensorDataset(X), batch_size=2)

    model, opt, dl = acc.prepare(model, opt, dl)

    me = pc.coords(acc.process_index)
    tp, dp = me["tp"], me["dp_replicate"]

    # 1. tp peers must receive identical batches
    batches = [b[0] for b in dl]
    assert len(batches) == 2, f"{name}: expected 2 dp-sharded batches, got {len(batches)}"
    flat = torch.cat([b.reshape(-1) for b in batches]).to(torch.float64)
    sig = torch.stack([flat.sum(), (flat * torch.arange(1, flat.numel() + 1)).sum()])
    sigs = [torch.empty_like(sig) for _ in range(acc.num_processes)]
    dist.all_gather(sigs, sig)
    for r in range(acc.num_processes):
        if pc.coords(r)["dp_replicate"] == dp:
            assert torch.equal(sigs[r], sig), f"{name}: tp peer rank {r} saw a different batch"

    # 2. forward parity (row-parallel outputs are replicated across tp)
    xb = batches[0]
    with torch.no_grad():
        out = model(xb)
        out = out["logits"] if isinstance(out, dict) else out
        ref_out = ref(xb)
        ref_out = ref_out["logits"] if isinstance(ref_out, dict) else ref_out
    assert torch.allclose(out, ref_out, atol=1e-4), f"{name}: forward diverges {(out - ref_out).abs().max()}"

    # 3. one training step: grads averaged over dp replicas only
    opt.zero_grad()
    loss = model(xb, labels=xb)["loss"]
    acc.backward(loss)
    opt.step()

    all_first = [torch.empty_like(xb) for _ in range(acc.num_processes)]
    dist.all_gather(all_first, xb)
    replica_rank = {}
    for r in range(acc.num_processes):
        replica_rank.setdefault(pc.coords(r)["dp_replicate"], r)
    uniq = [all_first[r] for r in sorted(replica_rank.values())]
    ref_opt.zero_grad()
    ref_loss = sum(ref(b, labels=b)["loss"] for b in uniq) / len(uniq)
    ref_loss.backward()
    ref_opt.step()

    m = acc.unwrap_model(model)
    if name == "LLAMA":
        for i, layer in enumerate(m.layers):
            rl = ref.layers[i]
            q = rl.self_attn.q_proj.weight
            per = q.shape[0] // 2
            got, want = layer.self_attn.q_proj.weight, q[tp * per : (tp + 1) * per]
            assert torch.allclose(got, want, atol=1e-5), f"{name} q_proj step mismatch layer {i}"
            d = rl.mlp.down_proj.weight
            perc = d.shape[1] // 2
            got, want = layer.mlp.down_proj.weight, d[:, tp * perc : (tp + 1) * perc]
            assert torch.allclose(got, want, atol=1e-5), f"{name} down_proj step mismatch layer {i}"
    else:
        for i, block in enumerate(m.h):
            rb = ref.h[i]
            W = rb.attn.c_attn.weight  # [3H, H]
            H = W.shape[0] // 3
            per = H // 2
            rows = torch.cat([torch.arange(j * H + tp * per, j * H + (tp + 1) * per) for j in range(3)])
            assert torch.allclose(block.attn.c_attn.weight, W[rows], atol=1e-5), f"{name} c_attn mismatch {i}"
            C = rb.mlp.c_proj.weight
            perc = C.shape[1] // 2
            assert torch.allclose(
                block.mlp.c_proj.weight, C[:, tp * perc : (tp + 1) * perc], atol=1e-5
            ), f"{name} mlp.c_proj mismatch {i}"
    # 4. tp-degree-agnostic checkpoints: state_dict() gathers FULL weights
    # (collective — every rank calls it), and loading a full dict re-slices
    # this rank's shard, so save_state/load_state round-trips under TP
    sd = m.state_dict()
    ref_sd = ref.state_dict()
    for k, v in ref_sd.items():
        assert k in sd and sd[k].shape == v.shape, f"{name}: state_dict shape {k}: {sd.get(k, None) is not None and sd[k].shape} vs {v.shape}"
        assert torch.allclose(sd[k], v, atol=1e-5), f"{name}: state_dict value {k}: {(sd[k] - v).abs().max()}"
    m.load_state_dict(sd)  # full dict loads back into the sharded module
    with torch.no_grad():
        out_rt = model(xb)
        out_rt = out_rt["logits"] if isinstance(out_rt, dict) else out_rt
        ref_rt = ref(xb)
        ref_rt = ref_rt["logits"] if isinstance(ref_rt, dict) else ref_rt
    assert torch.allclose(out_rt, ref_rt, atol=1e-4), f"{name}: post-roundtrip forward diverges"
    if acc.is_main_process:
        print(f"TP_PREPARE_{name}_PASS")


def check_tp_clip(acc, pc):
    """clip_grad_norm_ under TP: global norm sums sharded contributions over
    the tp group + replicated once; returned norm and the clipped update
    must match the single-process reference exactly."""
    set_seed(0)
    model = LlamaForCausalLM(LlamaConfig.tiny(num_hidden_layers=2))
    set_seed(0)
    ref = LlamaForCausalLM(LlamaConfig.tiny(num_hidden_layers=2))
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    ref_opt = torch.optim.SGD(ref.parameters(), lr=0.1)
    model, opt = acc.prepare(model, opt)

    ids = torch.randint(0, 1024, (2, 16), generator=torch.Generator().manual_seed(5))
    opt.zero_grad()
    loss = model(ids)["logits"].float().pow(2).mean()
    acc.backward(loss)
    norm = acc.clip_grad_norm_([p for p in acc.unwrap_model(model).parameters()], max_norm=0.05)
    opt.step()

    ref_opt.zero_grad()
    ref(ids)["logits"].float().pow(2).mean().backward()
    ref_norm = torch.nn.utils.clip_grad_norm_(ref.parameters(), max_norm=0.05)
    ref_opt.step()
    assert abs(float(norm) - float(ref_norm)) < 1e-4 * max(1.0, float(ref_norm)), (
        f"tp clip norm {float(norm)} vs ref {float(ref_norm)}"
    )
    with torch.no_grad():
        out = model(ids)["logits"]
        want = ref(ids)["logits"]
    assert torch.allclose(out, want, atol=1e-4), f"post-clip step diverges {(out - want).abs().max()}"
    dist.barrier()
    if acc.is_main_process:
        print("TP_CLIP_PASS")


def check_save_state_roundtrip(acc, pc):
    """save_state under TP writes PER-RANK optimizer files (moments live on
    tp shards); load_state restores THIS rank's moments exactly."""
    import tempfile

    import torch as _torch

    # fresh registries: the earlier run_family models would otherwise be
    # re-saved (and their per-rank tp gathers re-run) as models 0..1
    acc._models.clear()
    acc._optimizers.clear()
    acc._schedulers.clear()
    acc._dataloaders.clear()
    set_seed(0)
    model = LlamaForCausalLM(LlamaConfig.tiny(num_hidden_layers=2))
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    model, opt = acc.prepare(model, opt)
    ids = torch.randint(0, 1024, (2, 16), generator=torch.Generator().manual_seed(5))
    opt.zero_grad()
    loss = model(ids)["logits"].float().pow(2).mean()
    acc.backward(loss)
    opt.step()  # materialize per-shard Adam moments

    tmp = tempfile.mkdtemp(prefix=f"tp_ckpt_{acc.process_index}_")
    # shared dir decided by rank 0
    holder = [tmp]
    dist.broadcast_object_list(holder, src=0)
    tmp = holder[0]
    acc.save_state(tmp)
    snap = {
        id(p): {k: v.clone() for k, v in st.items() if _torch.is_tensor(v)}
        for p, st in opt.optimizer.state.items()
    }
    with _torch.no_grad():
        for st in opt.optimizer.state.values():
            for v in st.values():
                if _torch.is_tensor(v) and v.is_floating_point():
                    v.add_(1.0)
    acc.load_state(tmp)
    for p, st in opt.optimizer.state.items():
        for k, v in st.items():
            if _torch.is_tensor(v) and id(p) in snap and k in snap[id(p)]:
                assert _torch.allclose(v, snap[id(p)][k]), f"optimizer state {k} not restored per-rank"
    dist.barrier()
    if acc.is_main_process:
        print("TP_SAVE_STATE_PASS")


def main():
    pc = ParallelismConfig(dp_replicate_size=2, tp_size=2)
    acc = Accelerator(cpu=True, parallelism_config=pc)
    assert acc.num_processes == 4

    run_family(acc, pc, "LLAMA", lambda: LlamaForCausalLM(LlamaConfig.tiny(num_hidden_layers=2)), 1024)
    run_family(acc, pc, "GPT2", lambda: GPT2LMHeadModel(GPT2Config.tiny()), 1024)
    check_save_state_roundtrip(acc, pc)
    acc._models.clear(); acc._optimizers.clear()
    check_tp_clip(acc, pc)

    acc.end_training()


if __name__ == "__main__":
    main()
