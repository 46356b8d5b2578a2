"""Policy for the native Falcon family
(reference: colossalai/shardformer/policies/falcon.py).

Falcon-7B is multi-query (one shared KV head): the packed q|k|v column
split across tp ranks would have to split a single KV head, so TP here
requires KV replication — deferred. DP/ZeRO/PP paths need no surgery.
"""

from typing import Dict

from .base_policy import Policy

__all__ = ["FalconPolicy", "FalconForCausalLMPolicy"]


class FalconPolicy(Policy):
    def preprocess(self):
        return self.model

    def module_policy(self) -> Dict:
        tp = self.shard_config.tensor_parallel_size
        if self.shard_config.enable_tensor_parallelism and tp > 1:
            raise NotImplementedError(
                "Falcon TP needs KV-head replication under multi-query attention; "
                "use dp/zero/pp for Falcon this round"
            )
        return {}

    def postprocess(self):
        return self.model


class FalconForCausalLMPolicy(FalconPolicy):
    pass
