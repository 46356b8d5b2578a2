"""Policy machinery (reference: colossalai/shardformer/policies/base_policy.py).

A policy maps module classes to ``ModulePolicyDescription``s:
- ``attribute_replacement``: dotted attr path -> new value (e.g. head counts / tp)
- ``sub_module_replacement``: replace a child (by suffix) with a ParallelModule
  built via ``target_module.from_native_module``
- ``method_replacement``: bind replacement forwards (pipeline-stage forwards)
"""

from abc import ABC, abstractmethod
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, List, Optional, Type, Union

import torch.nn as nn

__all__ = ["Policy", "ModulePolicyDescription", "SubModuleReplacementDescription"]


@dataclass
class SubModuleReplacementDescription:
    suffix: str
    target_module: Any  # ParallelModule subclass
    kwargs: Dict[str, Any] = field(default_factory=dict)
    ignore_if_not_exist: bool = False


@dataclass
class ModulePolicyDescription:
    attribute_replacement: Optional[Dict[str, Any]] = None
    param_replacement: Optional[List[Callable]] = None
    sub_module_replacement: Optional[List[SubModuleReplacementDescription]] = None
    method_replacement: Optional[Dict[str, Callable]] = None


class Policy(ABC):
    def __init__(self):
        self.model: Optional[nn.Module] = None
        self.shard_config = None

    def set_model(self, model: nn.Module) -> None:
        self.model = model

    def set_shard_config(self, shard_config) -> None:
        self.shard_config = shard_config
        self.config_sanity_check()

    def config_sanity_check(self):
        pass

    @abstractmethod
    def preprocess(self) -> nn.Module: ...

    @abstractmethod
    def module_policy(self) -> Dict[Union[str, Type[nn.Module]], ModulePolicyDescription]: ...

    @abstractmethod
    def postprocess(self) -> nn.Module: ...

    def append_or_create_submodule_replacement(
        self, description, policy: Dict, target_key
    ) -> Dict:
        if isinstance(description, SubModuleReplacementDescription):
            description = [description]
        if target_key in policy:
            if policy[target_key].sub_module_replacement is None:
                policy[target_key].sub_module_replacement = []
            policy[target_key].sub_module_replacement.extend(description)
        else:
            policy[target_key] = ModulePolicyDescription(sub_module_replacement=description)
        return policy

    # ---- pipeline helpers (used by stage-aware policies) --------------------
    @staticmethod
    def distribute_layers(num_layers: int, num_stages: int) -> List[int]:
        """Even layer split; remainder layers go to the middle stages
        (first/last stages also carry embedding / head)."""
        quotient, remainder = divmod(num_layers, num_stages)
        layers_per_stage = [quotient] * num_stages
        for i in range(remainder):
            layers_per_stage[num_stages - 2 - (i % max(num_stages - 1, 1))] += 1
        return layers_per_stage

    @staticmethod
    def get_stage_index(layers_per_stage: List[int], stage: int):
        start = sum(layers_per_stage[:stage])
        return (start, start + layers_per_stage[stage])
