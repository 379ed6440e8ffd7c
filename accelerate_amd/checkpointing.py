"""Checkpoint save/load (reference: checkpointing.py).

On-disk layout matches the reference contract:

    {project_dir}/checkpoints/checkpoint_{iter}/
      model.safetensors (or pytorch_model.bin)
      optimizer.bin, scheduler.bin, sampler.bin, scaler.pt
      random_states_{rank}.pkl
      custom_checkpoint_{i}.pkl
"""

import random
from pathlib import Path

import numpy as np
import torch

from .logging import get_logger
from .utils.constants import (
    MODEL_NAME,
    OPTIMIZER_NAME,
    RNG_STATE_NAME,
    SAFE_MODEL_NAME,
    SAMPLER_NAME,
    SCALER_NAME,
    SCHEDULER_NAME,
)
from .utils.imports import is_safetensors_available

logger = get_logger(__name__)


from .utils.other import save  # noqa: F401  (canonical home: utils.other)


def save_accelerator_state(
    output_dir,
    model_states,
    optimizers,
    schedulers,
    dataloaders,
    process_index,
    step,
    scaler=None,
    save_on_each_node=False,
    safe_serialization=True,
):
    """(reference: checkpointing.py:63-180)"""
    output_dir = Path(output_dir)
    # Model states
    for i, state in enumerate(model_states):
        if isinstance(state, dict) and state.get("__fsdp_sharded__"):
            continue  # written per-rank by Accelerator.save_state (sharded)
        if safe_serialization and is_safetensors_available():
            import safetensors.torch

            # safetensors cannot handle shared storage: clone-contiguize
            state = {k: v.contiguous().clone() if isinstance(v, torch.Tensor) else v for k, v in state.items()}
            weights_name = f"{SAFE_MODEL_NAME}.safetensors" if i == 0 else f"{SAFE_MODEL_NAME}_{i}.safetensors"
            if PartialStateIsMain(save_on_each_node):
                safetensors.torch.save_file(state, output_dir / weights_name, metadata={"format": "pt"})
        else:
            weights_name = f"{MODEL_NAME}.bin" if i == 0 else f"{MODEL_NAME}_{i}.bin"
            if PartialStateIsMain(save_on_each_node):
                torch.save(state, output_dir / weights_name)
        logger.info(f"Model weights saved in {output_dir / weights_name}")
    # Optimizer states
    for i, opt in enumerate(optimizers):
        if getattr(opt, "_sharded", False):
            # sharded world: the optimizer state IS per-rank (master shards);
            # every rank writes its own file, no cross-rank communication
            # (reference rationale: fsdp_utils.py:107-118)
            suffix = f"_{i}" if i > 0 else ""
            optimizer_name = f"{OPTIMIZER_NAME}{suffix}_rank{process_index}.bin"
            torch.save(opt.state_dict(), output_dir / optimizer_name)
        else:
            optimizer_name = f"{OPTIMIZER_NAME}.bin" if i == 0 else f"{OPTIMIZER_NAME}_{i}.bin"
            if PartialStateIsMain(save_on_each_node):
                torch.save(opt.state_dict(), output_dir / optimizer_name)
        logger.info(f"Optimizer state saved in {output_dir / optimizer_name}")
    # Scheduler states
    for i, scheduler in enumerate(schedulers):
        scheduler_name = f"{SCHEDULER_NAME}.bin" if i == 0 else f"{SCHEDULER_NAME}_{i}.bin"
        if PartialStateIsMain(save_on_each_node):
            torch.save(scheduler.state_dict(), output_dir / scheduler_name)
        logger.info(f"Scheduler state saved in {output_dir / scheduler_name}")
    # DataLoader states (sampler epoch bookkeeping / stateful state_dicts)
    for i, dataloader in enumerate(dataloaders):
        sampler_name = f"{SAMPLER_NAME}.bin" if i == 0 else f"{SAMPLER_NAME}_{i}.bin"
        from .data_loader import SeedableRandomSampler

        sampler = None
        if hasattr(dataloader, "get_sampler"):
            sampler = dataloader.get_sampler()
        if isinstance(sampler, SeedableRandomSampler) and PartialStateIsMain(save_on_each_node):
            torch.save(sampler, output_dir / sampler_name)
        if getattr(dataloader, "use_stateful_dataloader", False) and PartialStateIsMain(save_on_each_node):
            dataloader_state_dict_name = "dl_state_dict.bin" if i == 0 else f"dl_state_dict_{i}.bin"
            torch.save(dataloader.state_dict(), output_dir / dataloader_state_dict_name)
        logger.info(f"Sampler state for dataloader {i} saved in {output_dir / sampler_name}")
    # GradScaler state
    if scaler is not None:
        if PartialStateIsMain(save_on_each_node):
            torch.save(scaler.state_dict(), output_dir / f"{SCALER_NAME}.pt")
        logger.info(f"Gradient scaler state saved in {output_dir / f'{SCALER_NAME}.pt'}")
    # Per-rank RNG states
    states = {}
    states["step"] = step
    states["random_state"] = random.getstate()
    states["numpy_random_seed"] = np.random.get_state()
    states["torch_manual_seed"] = torch.get_rng_state()
    if torch.cuda.is_available():
        states["torch_cuda_manual_seed"] = torch.cuda.get_rng_state_all()
    torch.save(states, output_dir / f"{RNG_STATE_NAME}_{process_index}.pkl")
    logger.info(f"Random states saved in {output_dir / f'{RNG_STATE_NAME}_{process_index}.pkl'}")
    return output_dir


def PartialStateIsMain(save_on_each_node: bool) -> bool:
    from .state import PartialState

    state = PartialState()
    return state.is_local_main_process if save_on_each_node else state.is_main_process


def load_accelerator_state(
    input_dir,
    models,
    optimizers,
    schedulers,
    dataloaders,
    process_index,
    scaler=None,
    map_location=None,
    load_model_func_kwargs=None,
    load_kwargs=None,
):
    """(reference: checkpointing.py:183-320). Returns override_attributes
    (currently just ``step``)."""
    load_model_func_kwargs = load_model_func_kwargs or {}
    override_attributes = {}
    if map_location not in [None, "cpu", "on_device"]:
        raise TypeError("Unsupported optimizer map location passed, please choose one of `None`, `'cpu'`, or `'on_device'`")
    from .state import PartialState

    state = PartialState()
    if map_location is None:
        map_location = "cpu"
    elif map_location == "on_device":
        map_location = state.device
    input_dir = Path(input_dir)

    # Models
    for i, model in enumerate(models):
        if model is None:
            continue  # sharded: loaded per-rank by Accelerator.load_state
        ending = f"_{i}" if i > 0 else ""
        safe_path = input_dir / f"{SAFE_MODEL_NAME}{ending}.safetensors"
        bin_path = input_dir / f"{MODEL_NAME}{ending}.bin"
        if safe_path.exists():
            import safetensors.torch

            state_dict = safetensors.torch.load_file(safe_path, device=str(map_location))
        else:
            state_dict = torch.load(bin_path, map_location=map_location, weights_only=True)
        model.load_state_dict(state_dict, **load_model_func_kwargs)
        logger.info("All model weights loaded successfully")

    # Optimizers
    for i, opt in enumerate(optimizers):
        if getattr(opt, "_sharded", False):
            suffix = f"_{i}" if i > 0 else ""
            optimizer_name = f"{OPTIMIZER_NAME}{suffix}_rank{process_index}.bin"
        else:
            optimizer_name = f"{OPTIMIZER_NAME}.bin" if i == 0 else f"{OPTIMIZER_NAME}_{i}.bin"
        optimizer_state = torch.load(input_dir / optimizer_name, map_location=map_location, weights_only=True)
        optimizers[i].load_state_dict(optimizer_state)
    logger.info("All optimizer states loaded successfully")

    # Schedulers
    for i, scheduler in enumerate(schedulers):
        scheduler_name = f"{SCHEDULER_NAME}.bin" if i == 0 else f"{SCHEDULER_NAME}_{i}.bin"
        scheduler_state = torch.load(input_dir / scheduler_name, weights_only=True)
        scheduler.load_state_dict(scheduler_state)
    logger.info("All scheduler states loaded successfully")

    # DataLoaders
    for i, dataloader in enumerate(dataloaders):
        sampler_name = f"{SAMPLER_NAME}.bin" if i == 0 else f"{SAMPLER_NAME}_{i}.bin"
        from .data_loader import SeedableRandomSampler

        if (input_dir / sampler_name).exists():
            sampler = torch.load(input_dir / sampler_name, weights_only=False)
            if isinstance(sampler, SeedableRandomSampler) and hasattr(dataloader, "set_sampler"):
                dataloader.set_sampler(sampler)
        if getattr(dataloader, "use_stateful_dataloader", False):
            dataloader_state_dict_name = "dl_state_dict.bin" if i == 0 else f"dl_state_dict_{i}.bin"
            if (input_dir / dataloader_state_dict_name).exists():
                dataloader.load_state_dict(torch.load(input_dir / dataloader_state_dict_name, weights_only=False))
    logger.info("All dataloader sampler states loaded successfully")

    # GradScaler
    if scaler is not None:
        scaler.load_state_dict(torch.load(input_dir / f"{SCALER_NAME}.pt", weights_only=False))
        logger.info("GradScaler state loaded successfully")

    # Per-rank RNG states
    try:
        states = torch.load(input_dir / f"{RNG_STATE_NAME}_{process_index}.pkl", weights_only=False)
        if "step" in states:
            override_attributes["step"] = states["step"]
        random.setstate(states["random_state"])
        np.random.set_state(states["numpy_random_seed"])
        torch.set_rng_state(states["torch_manual_seed"])
        if torch.cuda.is_available() and "torch_cuda_manual_seed" in states:
            torch.cuda.set_rng_state_all(states["torch_cuda_manual_seed"])
        logger.info("All random states loaded successfully")
    except Exception:
        logger.info("Could not load random states")

    return override_attributes


def save_custom_state(obj, path, index: int = 0, save_on_each_node: bool = False):
    """Pickle a registered object's state (reference: checkpointing.py:323)."""
    load_location = Path(path) / f"custom_checkpoint_{index}.pkl"
    logger.info(f"Saving the state of {get_pretty_name(obj)} to {load_location}")
    save(obj.state_dict(), load_location, save_on_each_node=save_on_each_node)


def load_custom_state(obj, path, index: int = 0):
    load_location = f"{path}/custom_checkpoint_{index}.pkl"
    logger.info(f"Loading the state of {get_pretty_name(obj)} from {load_location}")
    obj.load_state_dict(torch.load(load_location, map_location="cpu", weights_only=False))


def get_pretty_name(obj):
    if not hasattr(obj, "__qualname__") and not hasattr(obj, "__name__"):
        obj = getattr(obj, "__class__", obj)
    if hasattr(obj, "__qualname__"):
        return obj.__qualname__
    if hasattr(obj, "__name__"):
        return obj.__name__
    return str(obj)
