"""Single-process checkpoint IO
(reference: colossalai/checkpoint_io/general_checkpoint_io.py:37)."""

import os
from pathlib import Path

import torch
import torch.nn as nn
from torch.optim import Optimizer

from ..interface import OptimizerWrapper
from .checkpoint_io_base import CheckpointIO
from .index_file import CheckpointIndexFile
from .utils import (
    StateDictSharder,
    calculate_tensor_size,
    generate_checkpoint_shard_file_name,
    get_model_base_filenames,
    get_optimizer_base_filenames,
    load_state_dict,
    load_state_dict_into_model,
    save_state_dict,
    shard_model_checkpoint,
)

__all__ = ["GeneralCheckpointIO"]


def _unwrap_optim(optimizer):
    return optimizer.unwrap() if isinstance(optimizer, OptimizerWrapper) else optimizer


class GeneralCheckpointIO(CheckpointIO):
    def load_unsharded_model(self, model: nn.Module, checkpoint: str, strict: bool):
        state_dict = load_state_dict(checkpoint)
        model.load_state_dict(state_dict, strict=strict)

    def save_unsharded_model(
        self, model: nn.Module, checkpoint: str, gather_dtensor: bool, use_safetensors: bool, use_async: bool = False
    ):
        state_dict = model.state_dict()
        if use_async:
            cpu_sd = {k: v.detach().to("cpu", non_blocking=False).contiguous() for k, v in state_dict.items()}
            self._submit_async(save_state_dict, cpu_sd, checkpoint, use_safetensors)
        else:
            save_state_dict(state_dict, checkpoint, use_safetensors)

    def load_sharded_model(self, model: nn.Module, index_file_path: str, strict: bool = False):
        index = CheckpointIndexFile.from_file(str(index_file_path))
        ckpt_root = Path(index_file_path).parent
        missing_keys = []
        for shard_file in index.get_checkpoint_filenames():
            state_dict = load_state_dict(str(ckpt_root / shard_file))
            load_state_dict_into_model(model, state_dict, missing_keys=missing_keys, strict=strict)
        if strict and missing_keys:
            raise RuntimeError(f"Missing keys when loading sharded checkpoint: {missing_keys}")

    def save_sharded_model(
        self,
        model: nn.Module,
        checkpoint_path: str,
        gather_dtensor: bool = False,
        prefix: str = None,
        max_shard_size: int = 1024,
        use_safetensors: bool = False,
        use_async: bool = False,
    ):
        os.makedirs(checkpoint_path, exist_ok=True)
        state_dict = model.state_dict()
        weights_name, save_index_file = get_model_base_filenames(prefix, use_safetensors)
        index_file = CheckpointIndexFile(checkpoint_path)

        shards = list(shard_model_checkpoint(state_dict, max_shard_size=max_shard_size))
        total = len(shards)
        total_size = 0
        for idx, (shard, shard_size) in enumerate(shards):
            shard_file = generate_checkpoint_shard_file_name(idx + 1, total, use_safetensors, prefix)
            for key in shard.keys():
                index_file.append_weight_map(key, shard_file)
            file_path = os.path.join(checkpoint_path, shard_file)
            if use_async:
                cpu_shard = {k: v.detach().to("cpu").contiguous() for k, v in shard.items()}
                self._submit_async(save_state_dict, cpu_shard, file_path, use_safetensors)
            else:
                save_state_dict(shard, file_path, use_safetensors)
            total_size += shard_size
        index_file.append_meta_data("total_size", total_size)
        index_file.write_index_file(save_index_file)

    def load_unsharded_optimizer(self, optimizer: Optimizer, checkpoint: str):
        optimizer = _unwrap_optim(optimizer)
        state_dict = load_state_dict(checkpoint)
        optimizer.load_state_dict(state_dict)

    def save_unsharded_optimizer(self, optimizer: Optimizer, checkpoint: str, gather_dtensor: bool, use_async: bool = False):
        optimizer = _unwrap_optim(optimizer)
        state = optimizer.state_dict()
        if use_async:
            self._submit_async(torch.save, _state_dict_to_cpu(state), checkpoint)
        else:
            torch.save(state, checkpoint)

    def load_sharded_optimizer(self, optimizer: Optimizer, index_file_path: str):
        optimizer = _unwrap_optim(optimizer)
        index = CheckpointIndexFile.from_file(str(index_file_path))
        ckpt_root = Path(index_file_path).parent
        # param groups live in a dedicated file
        param_group_file = index.metadata.get("param_groups")
        assert param_group_file is not None, "sharded optimizer checkpoint lacks param_groups metadata"
        param_groups = torch.load(str(ckpt_root / param_group_file), weights_only=False)
        states = {}
        for shard_file in index.get_checkpoint_filenames():
            shard = torch.load(str(ckpt_root / shard_file), weights_only=False)
            states.update(shard)
        states = {int(k): v for k, v in states.items()}
        optimizer.load_state_dict({"state": states, "param_groups": param_groups})

    def save_sharded_optimizer(
        self,
        optimizer: Optimizer,
        checkpoint: str,
        gather_dtensor: bool = False,
        prefix: str = None,
        size_per_shard: int = 1024,
        use_async: bool = False,
    ):
        optimizer = _unwrap_optim(optimizer)
        os.makedirs(checkpoint, exist_ok=True)
        full = optimizer.state_dict()
        states_name, group_name, save_index_file = get_optimizer_base_filenames(prefix)
        index_file = CheckpointIndexFile(checkpoint)

        torch.save(full["param_groups"], os.path.join(checkpoint, group_name))
        index_file.append_meta_data("param_groups", group_name)

        sharder = StateDictSharder(size_per_shard)
        blocks = []
        for pid, state in full["state"].items():
            block, size = sharder.append_optim_state(pid, state)
            if block is not None:
                blocks.append((block, size))
        blocks.append((sharder.current_block, sharder.current_block_size))

        total = len(blocks)
        total_size = 0
        root, ext = os.path.splitext(states_name)
        for idx, (block, size) in enumerate(blocks):
            shard_file = f"{root}-{idx + 1:05d}-of-{total:05d}{ext}"
            for pid in block.keys():
                index_file.append_weight_map(str(pid), shard_file)
            if use_async:
                self._submit_async(torch.save, _state_dict_to_cpu(block), os.path.join(checkpoint, shard_file))
            else:
                torch.save(block, os.path.join(checkpoint, shard_file))
            total_size += size
        index_file.append_meta_data("total_size", total_size)
        index_file.write_index_file(save_index_file)


def _state_dict_to_cpu(obj):
    if isinstance(obj, torch.Tensor):
        return obj.detach().to("cpu")
    if isinstance(obj, dict):
        return {k: _state_dict_to_cpu(v) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        return type(obj)(_state_dict_to_cpu(v) for v in obj)
    return obj
