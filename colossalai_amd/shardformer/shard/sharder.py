"""Policy-driven in-place model surgery
(reference: colossalai/shardformer/shard/sharder.py:33)."""

from typing import Any, Dict, List, Optional, Set, Union

import torch.nn as nn

from ..policies.base_policy import ModulePolicyDescription, Policy, SubModuleReplacementDescription
from .shard_config import ShardConfig

__all__ = ["ModelSharder"]


def _set_dotted_attr(obj: Any, path: str, value: Any) -> None:
    parts = path.split(".")
    for p in parts[:-1]:
        obj = getattr(obj, p)
    setattr(obj, parts[-1], value)


def _get_child(module: nn.Module, suffix: str) -> Optional[nn.Module]:
    obj = module
    for p in suffix.split("."):
        if not hasattr(obj, p):
            return None
        obj = getattr(obj, p)
    return obj


def _set_child(module: nn.Module, suffix: str, child: nn.Module) -> None:
    parts = suffix.split(".")
    obj = module
    for p in parts[:-1]:
        obj = getattr(obj, p)
    setattr(obj, parts[-1], child)


class ModelSharder:
    def __init__(self, model: nn.Module, policy: Policy, shard_config: ShardConfig):
        self.model = model
        self.policy = policy
        self.shard_config = shard_config

    def shard(self) -> List[Dict[int, nn.Parameter]]:
        self.policy.set_model(self.model)
        self.policy.set_shard_config(self.shard_config)
        self.policy.preprocess()
        module_policies = self.policy.module_policy()
        self._replace_modules(self.model, module_policies)
        self.policy.postprocess()
        return []

    def _replace_modules(self, root: nn.Module, module_policies: Dict) -> None:
        # match both by class object and by class name string
        for module in list(root.modules()):
            desc = module_policies.get(type(module)) or module_policies.get(type(module).__name__)
            if desc is None:
                continue
            self._apply_description(module, desc)

    def _apply_description(self, module: nn.Module, desc: ModulePolicyDescription) -> None:
        if desc.attribute_replacement:
            for path, value in desc.attribute_replacement.items():
                _set_dotted_attr(module, path, value)
        if desc.method_replacement:
            for name, fn in desc.method_replacement.items():
                bound = fn.__get__(module, module.__class__)
                setattr(module, name, bound)
        if desc.param_replacement:
            for fn in desc.param_replacement:
                fn(module)
        if desc.sub_module_replacement:
            for rep in desc.sub_module_replacement:
                child = _get_child(module, rep.suffix)
                if child is None:
                    if rep.ignore_if_not_exist:
                        continue
                    raise AttributeError(f"{type(module).__name__} has no submodule {rep.suffix}")
                if type(child).__name__ == "_StageStub":
                    continue  # owned by another pipeline stage
                new_child = rep.target_module.from_native_module(
                    child, process_group=self.shard_config.tensor_parallel_process_group, **rep.kwargs
                )
                _set_child(module, rep.suffix, new_child)
