from .auto_policy import get_autopolicy, register_policy
from .base_policy import ModulePolicyDescription, Policy, SubModuleReplacementDescription

__all__ = [
    "get_autopolicy",
    "register_policy",
    "Policy",
    "ModulePolicyDescription",
    "SubModuleReplacementDescription",
]
