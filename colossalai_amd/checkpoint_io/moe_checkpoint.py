"""EP-aware checkpoint IO (reference:
colossalai/checkpoint_io/moe_checkpoint.py MoECheckpointIO).

Expert-parallel weights live as the LOCAL expert slice of the 3-D packed
tensors ([E_local, 2I, H] / [E_local, H, I]); checkpoints store the FULL
expert dim. Save all-gathers each MoE block's expert tensors over the ep
group (one collective per tensor — the xGMI crossbar makes this cheap);
load narrows the full tensors back to this rank's ``expert_start`` slice.
TP/PP handling is inherited from HybridParallelCheckpointIO.
"""

from typing import Dict

import torch
import torch.distributed as dist

from .hybrid_parallel_checkpoint_io import HybridParallelCheckpointIO

__all__ = ["MoECheckpointIO"]


def _moe_blocks(model):
    from ..models.mixtral import MixtralSparseMoeBlock

    for name, mod in model.named_modules():
        if isinstance(mod, MixtralSparseMoeBlock) and mod.ep_size > 1:
            yield name, mod


class MoECheckpointIO(HybridParallelCheckpointIO):
    def __init__(self, dp_group, pp_group, tp_group, ep_group, sp_size: int = 1):
        super().__init__(dp_group, pp_group, tp_group, sp_size)
        self.ep_group = ep_group

    def _local_state_dict(self, model) -> Dict[str, torch.Tensor]:
        sd = super()._local_state_dict(model)
        for name, mod in _moe_blocks(model):
            for pname in ("w_gate_up", "w_down"):
                local = getattr(mod, pname).data
                parts = [torch.empty_like(local) for _ in range(mod.ep_size)]
                dist.all_gather(parts, local.contiguous(), group=self.ep_group)
                sd[f"{name}.{pname}" if name else pname] = torch.cat(parts, dim=0).cpu()
        return sd

    def _pre_load(self, model, full_sd: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
        for name, mod in _moe_blocks(model):
            for pname in ("w_gate_up", "w_down"):
                key = f"{name}.{pname}" if name else pname
                if key in full_sd:
                    full_sd[key] = full_sd[key].narrow(
                        0, mod.expert_start, mod.num_local_experts
                    ).contiguous()
        return full_sd
