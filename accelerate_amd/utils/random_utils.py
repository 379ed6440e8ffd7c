"""Seeding and cross-process RNG synchronization (reference: utils/random.py)."""

import random
from typing import List, Optional, Union

import numpy as np
import torch

from ..state import PartialState
from .dataclasses import DistributedType, RNGType


def set_seed(seed: int, device_specific: bool = False, deterministic: bool = False):
    """Seed python/numpy/torch (+HIP) RNGs (reference: random.py:40-76)."""
    if device_specific:
        seed += PartialState().process_index
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)
    if deterministic:
        torch.use_deterministic_algorithms(True)


def synchronize_rng_state(rng_type: Optional[RNGType] = None, generator: Optional[torch.Generator] = None):
    """Broadcast one RNG state from rank 0 to all ranks
    (reference: random.py:79-160)."""
    # Get the proper rng state
    if rng_type == RNGType.TORCH:
        rng_state = torch.get_rng_state()
    elif rng_type == RNGType.CUDA:
        rng_state = torch.cuda.get_rng_state()
    elif rng_type == RNGType.GENERATOR:
        if generator is None:
            raise ValueError("Need a generator to synchronize its seed.")
        rng_state = generator.get_state()
    else:
        raise ValueError(f"Unknown RNG type {rng_type}")

    # Broadcast the rng state from device 0 to other devices
    state = PartialState()
    if state.use_distributed:
        rng_state = rng_state.to(state.device)
        torch.distributed.broadcast(rng_state, 0)
        rng_state = rng_state.cpu()

    # Set the broadcast rng state
    if rng_type == RNGType.TORCH:
        torch.set_rng_state(rng_state)
    elif rng_type == RNGType.CUDA:
        torch.cuda.set_rng_state(rng_state)
    elif rng_type == RNGType.GENERATOR:
        generator.set_state(rng_state)


def synchronize_rng_states(rng_types: List[Union[str, RNGType]], generator: Optional[torch.Generator] = None):
    for rng_type in rng_types:
        synchronize_rng_state(RNGType(rng_type), generator=generator)
