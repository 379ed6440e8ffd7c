"""Model-level weight-only quantization with big-model dispatch.

API parity with the reference's bitsandbytes layer (utils/bnb.py:44-199
``load_and_quantize_model``, :280 ``replace_with_bnb_layers``,
``BnbQuantizationConfig`` dataclasses.py:3057) — bitsandbytes is CUDA-only,
so the storage formats and kernels here are our own gfx950 ones
(``ops/quant.py`` / ``ops/csrc/quant_kernels.hip``).
"""

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Union

import torch
import torch.nn as nn


@dataclass
class QuantizationConfig:
    """Weight-only quantization recipe (reference: BnbQuantizationConfig).

    int8 uses one symmetric scale per output channel; int4 uses group-wise
    scales along in-features. ``skip_modules`` defaults to the lm_head-style
    output projection (quantizing it measurably hurts logits, same policy as
    the reference's ``llm_int8_skip_modules``).
    """

    load_in_8bit: bool = False
    load_in_4bit: bool = False
    group_size: int = 128
    compute_dtype: Union[str, torch.dtype] = torch.bfloat16
    skip_modules: Optional[List[str]] = None
    keep_in_fp32_modules: List[str] = field(default_factory=list)

    def __post_init__(self):
        if self.load_in_8bit and self.load_in_4bit:
            raise ValueError("pick ONE of load_in_8bit / load_in_4bit")
        if not (self.load_in_8bit or self.load_in_4bit):
            raise ValueError("one of load_in_8bit / load_in_4bit must be True")
        if isinstance(self.compute_dtype, str):
            self.compute_dtype = getattr(torch, self.compute_dtype)

    @property
    def bits(self) -> int:
        return 8 if self.load_in_8bit else 4


def _default_skip_modules(model: nn.Module) -> List[str]:
    # the last Linear in execution order is (almost always) the output head
    last = None
    for name, module in model.named_modules():
        if isinstance(module, nn.Linear):
            last = name
    return [last] if last is not None else []


def replace_with_quantized_layers(
    model: nn.Module,
    config: QuantizationConfig,
    modules_to_not_convert: Optional[List[str]] = None,
) -> nn.Module:
    """Swap every eligible nn.Linear for a QuantLinear (in place).

    int4 layers whose in_features don't divide 2*group_size are left in
    the original dtype (and reported via ``model._quant_skipped``).
    """
    from ..ops.quant import QuantLinear

    if modules_to_not_convert is None:
        modules_to_not_convert = (
            config.skip_modules if config.skip_modules is not None else _default_skip_modules(model)
        )
    skipped = []
    for name, module in list(model.named_modules()):
        if not isinstance(module, nn.Linear) or isinstance(module, QuantLinear):
            continue
        if any(name == s or name.endswith("." + s) or s in name.split(".") for s in modules_to_not_convert):
            skipped.append(name)
            continue
        if config.bits == 4 and (
            module.in_features % 2 != 0 or module.in_features % config.group_size != 0
        ):
            skipped.append(name)
            continue
        qlin = QuantLinear.from_linear(
            module, bits=config.bits, group_size=config.group_size, compute_dtype=config.compute_dtype
        )
        parent_name, _, child = name.rpartition(".")
        parent = model.get_submodule(parent_name) if parent_name else model
        setattr(parent, child, qlin)
    model._quant_skipped = skipped
    return model


def load_and_quantize_model(
    model: nn.Module,
    quantization_config: QuantizationConfig,
    weights_location: Optional[str] = None,
    device_map: Optional[Union[str, Dict[str, Union[int, str]]]] = None,
    no_split_module_classes: Optional[List[str]] = None,
    max_memory: Optional[Dict] = None,
    offload_dir: Optional[str] = None,
) -> nn.Module:
    """Load weights (optional), quantize the Linears, dispatch across devices.

    Mirrors reference utils/bnb.py:44-199: quantization happens BEFORE
    dispatch so the device map is computed on the quantized footprint
    (an 8B model in int4 plans at ~4.5 GB, not 16 GB).
    """
    from ..big_modeling import dispatch_model
    from .modeling import infer_auto_device_map, load_checkpoint_in_model

    from ..ops.quant import QuantLinear

    if weights_location is not None:
        load_checkpoint_in_model(model, weights_location)
    model = replace_with_quantized_layers(model, quantization_config)
    # the rest of the model (embeddings, norms, head) computes in
    # compute_dtype so QuantLinear outputs flow without dtype mismatches
    model.to(quantization_config.compute_dtype)
    for mod in model.modules():
        if isinstance(mod, QuantLinear):
            mod.scales.data = mod.scales.data.float()  # .to() downcast undone
    for name in quantization_config.keep_in_fp32_modules:
        for mod_name, mod in model.named_modules():
            if mod_name == name or mod_name.endswith("." + name):
                mod.float()
    model.eval()
    for p in model.parameters():
        p.requires_grad_(False)
    if device_map is not None:
        if device_map == "auto":
            device_map = infer_auto_device_map(
                model, max_memory=max_memory, no_split_module_classes=no_split_module_classes
            )
        model = dispatch_model(model, device_map, offload_dir=offload_dir)
    return model
