"""`accelerate-amd estimate` — per-dtype memory table for a model
(reference: commands/estimate.py). Meta-loads via init_empty_weights (zero
RAM); works offline for local checkpoints/configs and for transformers
models when their config is cached locally.
"""

from ..big_modeling import init_empty_weights
from ..utils.modeling import compute_module_sizes
from ..utils.other import convert_bytes


# bundled model families resolvable fully offline (models/):
#   name -> (config factory, model class path)
_BUILTIN = {
    "bert-base": ("accelerate_amd.models", "BertConfig", "bert_base", "BertForSequenceClassification"),
    "gpt2-large": ("accelerate_amd.models", "GPT2Config", "gpt2_large", "GPT2LMHeadModel"),
    "llama3-8b": ("accelerate_amd.models", "LlamaConfig", "llama3_8b", "LlamaForCausalLM"),
    "llama3-70b": ("accelerate_amd.models", "LlamaConfig", "llama3_70b", "LlamaForCausalLM"),
    "llama3-405b": ("accelerate_amd.models", "LlamaConfig", "llama3_405b", "LlamaForCausalLM"),
    "mixtral-8x7b": ("accelerate_amd.models", "LlamaMoEConfig", "mixtral_8x7b_shape", "LlamaMoEForCausalLM"),
    "t5-11b": ("accelerate_amd.models", "T5Config", "t5_11b", "T5ForConditionalGeneration"),
}


def create_empty_model(model_name: str, trust_remote_code: bool = False):
    if model_name in _BUILTIN:
        import importlib

        mod_name, cfg_cls, factory, model_cls = _BUILTIN[model_name]
        mod = importlib.import_module(mod_name)
        config = getattr(getattr(mod, cfg_cls), factory)()
        with init_empty_weights():
            return getattr(mod, model_cls)(config)
    import transformers

    config = transformers.AutoConfig.from_pretrained(model_name, trust_remote_code=trust_remote_code)
    with init_empty_weights():
        model = transformers.AutoModel.from_config(config, trust_remote_code=trust_remote_code)
    return model


def estimate_command(args):
    import torch

    model = create_empty_model(args.model_name, trust_remote_code=args.trust_remote_code)
    sizes = {}
    for dtype in args.dtypes:
        torch_dtype = getattr(torch, {"float32": "float32", "float16": "float16", "bfloat16": "bfloat16", "int8": "int8", "fp8": "float8_e4m3fn"}[dtype])
        total = compute_module_sizes(model, dtype=torch_dtype)[""]
        # training ≈ params + grads + 2× Adam state (fp32) + activations headroom
        train_total = total * (4 if dtype in ("float32",) else 6)
        sizes[dtype] = (total, train_total)
    largest = max(len(d) for d in sizes)
    print(f"Memory estimate for {args.model_name} (MI355X: 288 GB HBM3E per GPU):")
    print(f"{'dtype':<{largest+2}} {'inference':>12} {'training(Adam)':>16}")
    for dtype, (inf, train) in sizes.items():
        print(f"{dtype:<{largest+2}} {convert_bytes(inf):>12} {convert_bytes(train):>16}")


def add_parser(subparsers):
    parser = subparsers.add_parser("estimate", help="Estimate model memory usage")
    parser.add_argument(
        "model_name",
        help="bundled family (bert-base, gpt2-large, llama3-8b, llama3-70b, llama3-405b, "
        "mixtral-8x7b, t5-11b), transformers model name, or local path",
    )
    parser.add_argument("--dtypes", nargs="+", default=["float32", "float16", "int8"],
                        choices=["float32", "float16", "bfloat16", "int8", "fp8"])
    parser.add_argument("--trust_remote_code", action="store_true")
    parser.set_defaults(func=estimate_command)
    return parser
