"""Big-model inference: meta init → auto device map sized for 288 GB HBM3E →
load_checkpoint_and_dispatch with CPU/disk offload hooks → generate
(the reference's examples/big_model_inference pattern, offline weights).

  python examples/big_model_inference.py --model tiny        # full end-to-end
  python examples/big_model_inference.py --model llama3-70b  # plan the map
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import tempfile

import torch

from accelerate_amd import init_empty_weights, load_checkpoint_and_dispatch, infer_auto_device_map
from accelerate_amd.models.llama import LlamaConfig, LlamaForCausalLM
from accelerate_amd.utils.modeling import save_model_weights

CONFIGS = {
    "tiny": LlamaConfig.tiny,
    "llama3-8b": LlamaConfig.llama3_8b,
    "llama3-70b": LlamaConfig.llama3_70b,
}


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--model", default="tiny", choices=list(CONFIGS))
    parser.add_argument("--max_new_tokens", type=int, default=8)
    args = parser.parse_args()

    config = CONFIGS[args.model]()
    with init_empty_weights():
        model = LlamaForCausalLM(config)
    n_params = sum(p.numel() for p in model.parameters())
    print(f"meta-initialized {args.model}: {n_params/1e9:.2f}B params (zero host RAM)")

    device_map = infer_auto_device_map(
        model, no_split_module_classes=["LlamaDecoderLayer"], dtype=torch.bfloat16
    )
    placements = {}
    for dev in device_map.values():
        placements[str(dev)] = placements.get(str(dev), 0) + 1
    print(f"auto device map (blocks per device): {placements}")

    if args.model != "tiny":
        print("(checkpoint loading skipped: no pretrained weights offline — "
              "use load_checkpoint_and_dispatch(model, ckpt_dir, device_map='auto'))")
        return

    # end-to-end with a real (random-init) checkpoint on disk
    with tempfile.TemporaryDirectory() as ckpt_dir:
        torch.manual_seed(0)
        source = LlamaForCausalLM(config)
        save_model_weights(source.state_dict(), ckpt_dir, max_shard_size="200MB")
        with init_empty_weights():
            model = LlamaForCausalLM(config)
        model = load_checkpoint_and_dispatch(
            model, ckpt_dir, device_map="auto", no_split_module_classes=["LlamaDecoderLayer"]
        )
        model.eval()
        device = next(p.device for p in model.parameters() if p.device.type != "meta")
        ids = torch.randint(0, config.vocab_size, (1, 8), device=device)
        out = model.generate(ids, max_new_tokens=args.max_new_tokens)
        print(f"generated {out.shape[1] - ids.shape[1]} tokens on device map above: {out[0, -5:].tolist()}")


if __name__ == "__main__":
    main()
