"""Disk offload store: one numpy memmap per tensor plus a JSON index.

Same on-disk contract as the reference (utils/offload.py — ``index.json``
with dtype/shape entries next to ``<name>.dat`` memmaps, safetensors
entries resolved lazily), implemented around a small dtype-codec:
numpy has no bf16, so bf16 tensors travel through the store bit-cast to
int16 and are re-viewed on load. Memmaps keep host RAM flat no matter how
large the offloaded model is — only the tensors a forward actually touches
are paged in.
"""

import json
import os
from collections.abc import Mapping
from typing import Dict, List, Optional, Union

import numpy as np
import torch

_INDEX_FILE = "index.json"


def _encode_for_numpy(t: torch.Tensor):
    """-> (numpy array, logical dtype tag). bf16 is bit-cast to int16."""
    if t.dtype == torch.bfloat16:
        return t.view(torch.int16).cpu().numpy(), "bfloat16"
    arr = t.cpu().numpy()
    return arr, str(arr.dtype)


def _memmap_dtype(tag: str) -> str:
    return "int16" if tag == "bfloat16" else tag


def offload_weight(weight, weight_name, offload_folder, index: dict = None):
    array, tag = _encode_for_numpy(weight)
    if array.ndim == 0:
        array = array[None]  # memmaps cannot be 0-d; shape in the index is
    if index is not None:
        index[weight_name] = {"dtype": tag, "shape": list(weight.shape)}
    mm = np.memmap(
        os.path.join(offload_folder, f"{weight_name}.dat"),
        dtype=array.dtype,
        mode="w+",
        shape=array.shape,
    )
    mm[:] = array[:]
    mm.flush()
    return index


def load_offloaded_weight(weight_file, weight_info: dict) -> torch.Tensor:
    logical_shape = tuple(weight_info["shape"])
    mm = np.memmap(
        weight_file,
        dtype=_memmap_dtype(weight_info["dtype"]),
        shape=logical_shape or (1,),
        mode="r",
    )
    t = torch.tensor(mm if logical_shape else mm[0])
    if weight_info["dtype"] == "bfloat16":
        t = t.view(torch.bfloat16)
    return t


def save_offload_index(index: dict, offload_folder):
    if not index:
        return
    path = os.path.join(offload_folder, _INDEX_FILE)
    merged = {}
    if os.path.isfile(path):
        with open(path, encoding="utf-8") as f:
            merged = json.load(f)
    merged.update(index)
    with open(path, "w", encoding="utf-8") as f:
        json.dump(merged, f, indent=2)


def offload_state_dict(save_dir, state_dict: Dict[str, torch.Tensor]) -> dict:
    """Offload a whole state dict; returns (and writes) its index."""
    os.makedirs(save_dir, exist_ok=True)
    index: dict = {}
    for name, tensor in state_dict.items():
        offload_weight(tensor, name, save_dir, index=index)
    save_offload_index(index, save_dir)
    return index


class PrefixedDataset(Mapping):
    """View of a mapping restricted to keys under ``prefix`` (stripped)."""

    def __init__(self, dataset: Mapping, prefix: str):
        self.dataset = dataset
        self.prefix = prefix

    def _matching(self):
        return [k for k in self.dataset if k.startswith(self.prefix)]

    def __getitem__(self, key):
        return self.dataset[self.prefix + key]

    def __iter__(self):
        return iter(self._matching())

    def __len__(self):
        return len(self._matching())


class OffloadedWeightsLoader(Mapping):
    """Lazy mapping over (in-memory state dict) ∪ (offload folder / index).

    In-memory entries win; index entries resolve to memmap files or to
    safetensors files when the index says so. Iteration order: state-dict
    keys first, then index-only keys.
    """

    def __init__(self, state_dict=None, save_folder=None, index: Mapping = None, device=None):
        if state_dict is None and save_folder is None and index is None:
            raise ValueError("OffloadedWeightsLoader needs a state_dict, save_folder, or index")
        self.state_dict = dict(state_dict) if state_dict else {}
        self.save_folder = save_folder
        if index is None and save_folder is not None:
            with open(os.path.join(save_folder, _INDEX_FILE)) as f:
                index = json.load(f)
        self.index = dict(index) if index else {}
        self.device = device
        self.all_keys = list(self.state_dict)
        self.all_keys += [k for k in self.index if k not in self.state_dict]

    def _from_safetensors(self, key, info):
        import safetensors.torch

        loaded = safetensors.torch.load_file(
            info["safetensors_file"], device=str(self.device or "cpu")
        )
        tensor = loaded[info.get("weight_name", key)]
        if info.get("dtype") is not None:
            tensor = tensor.to(getattr(torch, info["dtype"].replace("torch.", "")))
        return tensor

    def __getitem__(self, key: str):
        if key in self.state_dict:
            return self.state_dict[key]
        info = self.index[key]
        if info.get("safetensors_file") is not None:
            return self._from_safetensors(key, info)
        return load_offloaded_weight(os.path.join(self.save_folder, f"{key}.dat"), info)

    def __iter__(self):
        return iter(self.all_keys)

    def __len__(self):
        return len(self.all_keys)


def extract_submodules_state_dict(state_dict: Dict[str, torch.Tensor], submodule_names: List[str]):
    """Entries belonging to the named submodules (exact or dotted-prefix)."""
    picked = {}
    for name in submodule_names:
        for key, value in state_dict.items():
            if key == name or key.startswith(name + "."):
                picked[key] = value
    return picked
