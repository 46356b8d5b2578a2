"""Checkpoint helpers: HF-compatible naming, sharding, safetensors I/O.

Formats match the reference exactly (colossalai/checkpoint_io/utils.py:33-41)
so checkpoints interoperate: ``pytorch_model.bin`` / ``model.safetensors``,
sharded ``*-00001-of-00005.*`` + ``*.index.json``, optimizer
``pytorch_optim.bin`` + ``pytorch_optim_group.bin``.
"""

import os
import re
from collections import OrderedDict
from typing import Iterator, Mapping, Optional, Tuple

import torch

__all__ = [
    "WEIGHTS_NAME",
    "SAFE_WEIGHTS_NAME",
    "WEIGHTS_INDEX_NAME",
    "SAFE_WEIGHTS_INDEX_NAME",
    "OPTIM_NAME",
    "OPTIM_GROUP_NAME",
    "OPTIM_INDEX_NAME",
    "StateDictSharder",
    "calculate_tensor_size",
    "is_safetensors_available",
    "load_state_dict",
    "save_state_dict",
    "shard_model_checkpoint",
    "get_model_base_filenames",
    "get_optimizer_base_filenames",
    "load_state_dict_into_model",
]

WEIGHTS_NAME = "pytorch_model.bin"
SAFE_WEIGHTS_NAME = "model.safetensors"
WEIGHTS_INDEX_NAME = "pytorch_model.bin.index.json"
SAFE_WEIGHTS_INDEX_NAME = "model.safetensors.index.json"
OPTIM_NAME = "pytorch_optim.bin"
OPTIM_GROUP_NAME = "pytorch_optim_group.bin"
OPTIM_INDEX_NAME = "pytorch_optim.bin.index.json"


def is_safetensors_available() -> bool:
    try:
        import safetensors  # noqa: F401

        return True
    except ImportError:
        return False


def calculate_tensor_size(tensor: torch.Tensor) -> float:
    """Tensor size in MB."""
    return tensor.numel() * tensor.element_size() / 1024 / 1024


class StateDictSharder:
    """Accumulate tensors into shards no larger than ``size_per_shard`` MB
    (reference: colossalai/checkpoint_io/utils.py:149)."""

    def __init__(self, size_per_shard: int):
        self.max_shard_size = size_per_shard
        self.current_block: "OrderedDict[str, torch.Tensor]" = OrderedDict()
        self.current_block_size = 0.0

    def append_param(self, name: str, tensor: torch.Tensor) -> Tuple[Optional[OrderedDict], int]:
        tensor_size = calculate_tensor_size(tensor)
        ret_block, ret_size = None, 0
        if self.current_block_size + tensor_size > self.max_shard_size and self.current_block_size > 0:
            ret_block, ret_size = self.current_block, self.current_block_size
            self.current_block = OrderedDict()
            self.current_block_size = 0
        self.current_block[name] = tensor
        self.current_block_size += tensor_size
        return ret_block, ret_size

    def append_optim_state(self, param_id: int, state: OrderedDict) -> Tuple[Optional[OrderedDict], int]:
        state_size = sum(calculate_tensor_size(v) for v in state.values() if isinstance(v, torch.Tensor))
        ret_block, ret_size = None, 0
        if self.current_block_size + state_size > self.max_shard_size and self.current_block_size > 0:
            ret_block, ret_size = self.current_block, self.current_block_size
            self.current_block = OrderedDict()
            self.current_block_size = 0
        self.current_block[param_id] = state
        self.current_block_size += state_size
        return ret_block, ret_size


def shard_model_checkpoint(
    state_dict: Mapping[str, torch.Tensor], max_shard_size: int = 1024
) -> Iterator[Tuple[OrderedDict, int]]:
    sharder = StateDictSharder(max_shard_size)
    for key, weight in state_dict.items():
        block, size = sharder.append_param(key, weight)
        if block is not None:
            yield block, size
    yield sharder.current_block, sharder.current_block_size


def get_model_base_filenames(prefix: Optional[str] = None, use_safetensors: bool = True):
    weights_name = SAFE_WEIGHTS_NAME if use_safetensors else WEIGHTS_NAME
    weights_name = add_prefix(weights_name, prefix)
    index_name = SAFE_WEIGHTS_INDEX_NAME if use_safetensors else WEIGHTS_INDEX_NAME
    index_name = add_prefix(index_name, prefix)
    return weights_name, index_name


def get_optimizer_base_filenames(prefix: Optional[str] = None, use_safetensors: bool = False):
    states_name = OPTIM_NAME
    if use_safetensors:
        states_name = states_name.replace(".bin", ".safetensors")
    states_name = add_prefix(states_name, prefix)
    param_group_name = add_prefix(OPTIM_GROUP_NAME, prefix)
    index_name = add_prefix(OPTIM_INDEX_NAME, prefix)
    return states_name, param_group_name, index_name


def add_prefix(filename: str, prefix: Optional[str] = None) -> str:
    return f"{prefix}.{filename}" if prefix else filename


def generate_checkpoint_shard_file_name(index: int, total: int, use_safetensors: bool, prefix: str = None) -> str:
    basename = SAFE_WEIGHTS_NAME if use_safetensors else WEIGHTS_NAME
    if prefix:
        basename = f"{prefix}.{basename}"
    root, ext = os.path.splitext(basename)
    return f"{root}-{index:05d}-of-{total:05d}{ext}"


def save_state_dict(state_dict: Mapping, checkpoint_file_path: str, use_safetensors: bool) -> None:
    # safetensors requires contiguous CPU tensors and refuses shared storage
    state_dict = {k: v.contiguous().cpu() if isinstance(v, torch.Tensor) else v for k, v in state_dict.items()}
    if use_safetensors:
        from safetensors.torch import save_file

        save_file(state_dict, checkpoint_file_path, metadata={"format": "pt"})
    else:
        torch.save(state_dict, checkpoint_file_path)


def load_state_dict(checkpoint_file_path: str) -> dict:
    if checkpoint_file_path.endswith(".safetensors"):
        from safetensors.torch import load_file

        return load_file(checkpoint_file_path)
    return torch.load(checkpoint_file_path, map_location="cpu", weights_only=False)


def load_state_dict_into_model(
    model: torch.nn.Module, state_dict: Mapping, missing_keys=None, strict: bool = False
) -> None:
    incompat = model.load_state_dict(state_dict, strict=False)
    if missing_keys is not None:
        # intersect missing across shards: a key is missing only if no shard had it
        if not missing_keys:
            missing_keys.extend(incompat.missing_keys)
        else:
            still_missing = set(missing_keys) & set(incompat.missing_keys)
            missing_keys.clear()
            missing_keys.extend(sorted(still_missing))
    if strict and incompat.unexpected_keys:
        raise RuntimeError(f"Unexpected keys: {incompat.unexpected_keys}")
