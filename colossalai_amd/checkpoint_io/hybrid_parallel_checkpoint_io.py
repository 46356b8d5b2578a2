"""Checkpoint IO for TP/PP-sharded models
(reference: colossalai/checkpoint_io/hybrid_parallel_checkpoint_io.py:59).

Save: TP-sharded weights are gathered per module (ParallelModules expose
``gather_weight``); each pipeline stage contributes its own parameters; the
merged state dict is written by global rank 0 in the standard (HF-style)
format, so hybrid checkpoints interoperate with single-process loads.

Load: every rank reads the full file(s) and takes its own TP slice / PP
subset via the same module metadata.
"""

from typing import Dict, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from ..interface import ModelWrapper, OptimizerWrapper
from ..shardformer.layer.embedding import VocabParallelEmbedding1D
from ..shardformer.layer.linear import Linear1D_Col, Linear1D_Row, _shard_rows
from .general_checkpoint_io import GeneralCheckpointIO
from .utils import load_state_dict, save_state_dict

__all__ = ["HybridParallelCheckpointIO"]


def _full_state_dict(model: nn.Module, tp_group) -> Dict[str, torch.Tensor]:
    """State dict with TP-sharded weights gathered to full shape."""
    sd = {}
    handled_prefixes = []
    for name, module in model.named_modules():
        if isinstance(module, (Linear1D_Col, Linear1D_Row, VocabParallelEmbedding1D)):
            sd[f"{name}.weight" if name else "weight"] = module.gather_weight().cpu()
            if getattr(module, "bias", None) is not None:
                if isinstance(module, Linear1D_Col):
                    from ..shardformer.layer.linear import _gather_rows

                    sd[f"{name}.bias"] = _gather_rows(
                        module.bias.data.unsqueeze(-1), module.process_group, module.split_sizes
                    ).squeeze(-1).cpu()
                else:
                    sd[f"{name}.bias"] = module.bias.data.cpu()
            handled_prefixes.append(name + ".")
    for name, param in model.named_parameters():
        if any(name.startswith(p) for p in handled_prefixes):
            continue
        sd[name] = param.data.cpu()
    for name, buf in model.named_buffers():
        if any(name.startswith(p) for p in handled_prefixes):
            continue
        sd[name] = buf.data.cpu()
    return sd


def _load_into_sharded(model: nn.Module, full_sd: Dict[str, torch.Tensor]) -> None:
    for name, module in model.named_modules():
        wkey = f"{name}.weight" if name else "weight"
        if isinstance(module, Linear1D_Col) and wkey in full_sd:
            module.weight.data.copy_(
                _shard_rows(full_sd.pop(wkey).to(module.weight.device), module.process_group, module.split_sizes)
            )
            bkey = f"{name}.bias"
            if getattr(module, "bias", None) is not None and bkey in full_sd:
                module.bias.data.copy_(
                    _shard_rows(full_sd.pop(bkey).unsqueeze(-1).to(module.bias.device), module.process_group,
                                module.split_sizes).squeeze(-1)
                )
        elif isinstance(module, Linear1D_Row) and wkey in full_sd:
            w = full_sd.pop(wkey).to(module.weight.device)
            rank = dist.get_rank(module.process_group) if dist.is_initialized() else 0
            module.weight.data.copy_(w.chunk(module.world, dim=1)[rank])
            bkey = f"{name}.bias"
            if getattr(module, "bias", None) is not None and bkey in full_sd:
                module.bias.data.copy_(full_sd.pop(bkey).to(module.bias.device))
        elif isinstance(module, VocabParallelEmbedding1D) and wkey in full_sd:
            w = full_sd.pop(wkey).to(module.weight.device)
            module.weight.data.copy_(w[module.vocab_start : module.vocab_end])
    # remaining plain params (only those this stage holds)
    own = dict(model.named_parameters())
    own.update(dict(model.named_buffers()))
    for name, tensor in full_sd.items():
        if name in own and own[name].shape == tensor.shape:
            own[name].data.copy_(tensor.to(own[name].device))


class HybridParallelCheckpointIO(GeneralCheckpointIO):
    def __init__(self, dp_group, pp_group, tp_group, sp_size: int = 1):
        super().__init__()
        self.dp_group = dp_group
        self.pp_group = pp_group
        self.tp_group = tp_group
        self.dp_rank = dist.get_rank(dp_group) if dp_group is not None else 0
        self.tp_rank = dist.get_rank(tp_group) if tp_group is not None else 0
        self.pp_rank = dist.get_rank(pp_group) if pp_group is not None else 0
        self.pp_size = dist.get_world_size(pp_group) if pp_group is not None else 1
        self.global_rank = dist.get_rank()

    def _local_state_dict(self, model) -> dict:
        """Collective per-rank full-weight state dict (TP gathered); MoE IO
        extends this with the EP expert gather."""
        return _full_state_dict(model, self.tp_group)

    def _pre_load(self, model, full_sd: dict) -> dict:
        """Hook: adapt a full checkpoint to this rank before sharded load."""
        return full_sd

    def save_unsharded_model(self, model, checkpoint: str, gather_dtensor: bool = True,
                             use_safetensors: bool = False, use_async: bool = False):
        if isinstance(model, ModelWrapper):
            model = model.unwrap()
        # TP weight gathers are collective: every rank participates
        local_sd = self._local_state_dict(model)
        if self.pp_size > 1:
            gathered = [None] * self.pp_size
            dist.all_gather_object(gathered, local_sd, group=self.pp_group)
            merged = {}
            for sd in gathered:
                merged.update(sd)
            local_sd = merged
        if self.global_rank == 0:
            save_state_dict(local_sd, checkpoint, use_safetensors)
        dist.barrier()

    def load_unsharded_model(self, model, checkpoint: str, strict: bool = False):
        if isinstance(model, ModelWrapper):
            model = model.unwrap()
        full_sd = self._pre_load(model, dict(load_state_dict(checkpoint)))
        _load_into_sharded(model, full_sd)
        if dist.is_initialized():
            dist.barrier()

    def save_sharded_model(self, model, checkpoint_path: str, gather_dtensor: bool = True, prefix: str = None,
                           max_shard_size: int = 1024, use_safetensors: bool = False, use_async: bool = False):
        # v1: gather to full on rank 0 then reuse the general sharded writer
        if isinstance(model, ModelWrapper):
            model = model.unwrap()
        local_sd = self._local_state_dict(model)
        if self.pp_size > 1:
            gathered = [None] * self.pp_size
            dist.all_gather_object(gathered, local_sd, group=self.pp_group)
            merged = {}
            for sd in gathered:
                merged.update(sd)
            local_sd = merged
        if self.global_rank == 0:
            class _Holder(nn.Module):
                def state_dict(self_inner):  # noqa: N805
                    return local_sd

            super().save_sharded_model(_Holder(), checkpoint_path, False, prefix, max_shard_size,
                                       use_safetensors, use_async)
        dist.barrier()

    def load_sharded_model(self, model, index_file_path: str, strict: bool = False):
        if isinstance(model, ModelWrapper):
            model = model.unwrap()
        from pathlib import Path

        from .index_file import CheckpointIndexFile

        index = CheckpointIndexFile.from_file(str(index_file_path))
        root = Path(index_file_path).parent
        full_sd = {}
        for shard_file in index.get_checkpoint_filenames():
            full_sd.update(load_state_dict(str(root / shard_file)))
        _load_into_sharded(model, full_sd)
        if dist.is_initialized():
            dist.barrier()

    def save_unsharded_optimizer(self, optimizer, checkpoint: str, gather_dtensor: bool = True, use_async: bool = False):
        # rank-local shard states (ZeRO/hybrid): one file per rank
        state = optimizer.state_dict()
        path = f"{checkpoint}.rank{self.global_rank}" if dist.get_world_size() > 1 else checkpoint
        torch.save(state, path)
        dist.barrier()

    def load_unsharded_optimizer(self, optimizer, checkpoint: str):
        import os

        path = f"{checkpoint}.rank{self.global_rank}" if dist.get_world_size() > 1 else checkpoint
        if not os.path.exists(path):
            path = checkpoint
        optimizer.load_state_dict(torch.load(path, weights_only=False))
        dist.barrier()
