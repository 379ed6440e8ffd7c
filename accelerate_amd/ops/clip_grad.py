"""Gradient clipping via the CDNA4 multi-tensor L2-norm kernel
(reference: SURVEY.md §2.9 N8 — torch.nn.utils.clip_grad_norm_ foreach path).

Fully on-device: no host sync; returns the total norm as a 0-dim device
tensor (like torch's). Falls back to torch's implementation on CPU or for
non-fp32 grads.
"""

from typing import Iterable, Union

import torch

from . import _load_extension


def get_grad_norm(parameters, norm_type: float = 2.0) -> torch.Tensor:
    if isinstance(parameters, torch.Tensor):
        parameters = [parameters]
    grads = [p.grad for p in parameters if p.grad is not None]
    if len(grads) == 0:
        return torch.tensor(0.0)
    if norm_type == 2.0 and all(g.is_cuda and g.dtype == torch.float32 and g.is_contiguous() for g in grads):
        ext = _load_extension(required=True)
        return ext.l2norm_squared(grads).sqrt().squeeze()
    return torch.nn.utils.get_total_norm(grads, norm_type=norm_type)


def clip_grad_norm_(
    parameters: Union[torch.Tensor, Iterable[torch.Tensor]],
    max_norm: float,
    norm_type: float = 2.0,
) -> torch.Tensor:
    if isinstance(parameters, torch.Tensor):
        parameters = [parameters]
    parameters = list(parameters)
    grads = [p.grad for p in parameters if p.grad is not None]
    if len(grads) == 0:
        return torch.tensor(0.0)
    if norm_type == 2.0 and all(g.is_cuda and g.dtype == torch.float32 and g.is_contiguous() for g in grads):
        ext = _load_extension(required=True)
        return ext.clip_grad_norm(grads, float(max_norm))
    return torch.nn.utils.clip_grad_norm_(parameters, max_norm, norm_type=norm_type)
