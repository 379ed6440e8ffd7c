"""T5 encoder-decoder family (reference parity: Megatron t5 config parsing +
the big-model table's T0pp-11B entry; see models/t5.py)."""

import torch

from accelerate_amd.models.t5 import (
    T5Config,
    T5ForConditionalGeneration,
    relative_position_bucket,
)


def test_bucket_function_properties():
    pos = torch.arange(-40, 41)
    b_bi = relative_position_bucket(pos[None], True, 32, 128)[0]
    assert b_bi.min() >= 0 and b_bi.max() < 32
    assert b_bi[40] == 0  # distance 0 -> bucket 0
    # symmetric directions land in disjoint halves
    assert b_bi[39] != b_bi[41]
    b_causal = relative_position_bucket(pos[None], False, 32, 128)[0]
    assert (b_causal[41:] == 0).all()  # future positions collapse to 0 (masked anyway)
    # monotone non-decreasing with distance into the past
    past = b_causal[:41].flip(0)
    assert (past[1:] >= past[:-1]).all()


def test_forward_loss_and_shapes():
    torch.manual_seed(0)
    model = T5ForConditionalGeneration(T5Config.tiny())
    src = torch.randint(0, 512, (2, 12))
    tgt = torch.randint(0, 512, (2, 8))
    out = model(src, labels=tgt)
    assert out["logits"].shape == (2, 8, 512)
    assert torch.isfinite(out["loss"])


def test_decoder_is_causal():
    """Changing a FUTURE target token must not change earlier logits."""
    torch.manual_seed(0)
    model = T5ForConditionalGeneration(T5Config.tiny()).eval()
    src = torch.randint(0, 512, (1, 10))
    dec = torch.randint(0, 512, (1, 6))
    with torch.no_grad():
        a = model(src, decoder_input_ids=dec)["logits"]
        dec2 = dec.clone()
        dec2[0, -1] = (dec2[0, -1] + 1) % 512
        b = model(src, decoder_input_ids=dec2)["logits"]
    assert torch.allclose(a[:, :-1], b[:, :-1], atol=1e-5)
    assert not torch.allclose(a[:, -1], b[:, -1])


def test_encoder_conditions_decoder():
    torch.manual_seed(0)
    model = T5ForConditionalGeneration(T5Config.tiny()).eval()
    dec = torch.randint(0, 512, (1, 4))
    with torch.no_grad():
        a = model(torch.randint(0, 512, (1, 8)), decoder_input_ids=dec)["logits"]
        b = model(torch.randint(0, 512, (1, 8)), decoder_input_ids=dec)["logits"]
    assert not torch.allclose(a, b)


def test_trains_and_generates():
    torch.manual_seed(0)
    model = T5ForConditionalGeneration(T5Config.tiny())
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    src = torch.randint(0, 512, (4, 12))
    tgt = torch.randint(0, 512, (4, 8))
    first = None
    for _ in range(8):
        opt.zero_grad()
        loss = model(src, labels=tgt)["loss"]
        loss.backward()
        opt.step()
        first = first or loss.item()
    assert loss.item() < first
    out = model.generate(src[:1], max_new_tokens=5)
    assert out.shape == (1, 6)


def test_t5_11b_geometry():
    c = T5Config.t5_11b()
    assert c.d_ff == 65536 and c.num_heads == 128  # the T0pp table entry
