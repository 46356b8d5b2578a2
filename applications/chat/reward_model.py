"""Reward model + pairwise ranking loss (reference:
applications/ColossalChat/coati/models/reward_model.py).

A causal-LM backbone with a scalar value head; the reward of a sequence is
the head's output at the last non-pad token. Trained with the Bradley-
Terry pairwise loss -log sigmoid(r_chosen - r_rejected).
"""

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

__all__ = ["RewardModel", "reward_pairwise_loss"]


class RewardModel(nn.Module):
    def __init__(self, backbone: nn.Module, hidden_size: Optional[int] = None):
        """backbone: a native causal LM (e.g. LlamaForCausalLM) — its
        ``model`` submodule maps input_ids -> final hidden states."""
        super().__init__()
        self.model = backbone.model  # the decoder stack
        hidden_size = hidden_size or backbone.config.hidden_size
        self.value_head = nn.Linear(hidden_size, 1, bias=False)
        self.value_head.weight.data.normal_(0.0, 1.0 / (hidden_size + 1) ** 0.5)

    def forward(self, input_ids: torch.Tensor, attention_mask: Optional[torch.Tensor] = None) -> torch.Tensor:
        """-> rewards [B] (score at each sequence's last real token)."""
        hidden = self.model(input_ids)  # [B, S, H]
        values = self.value_head(hidden).squeeze(-1)  # [B, S]
        if attention_mask is None:
            return values[:, -1]
        last = attention_mask.long().sum(dim=1) - 1
        return values.gather(1, last.unsqueeze(1)).squeeze(1)


def reward_pairwise_loss(r_chosen: torch.Tensor, r_rejected: torch.Tensor) -> torch.Tensor:
    return -F.logsigmoid(r_chosen - r_rejected).mean()
