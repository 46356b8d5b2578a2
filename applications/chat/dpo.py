"""Direct Preference Optimization trainer (reference:
applications/ColossalChat/coati/trainer/dpo.py; Rafailov et al. 2023).

loss = -log sigmoid(beta * ((logp_pi(chosen) - logp_ref(chosen))
                          - (logp_pi(rejected) - logp_ref(rejected))))

summed over response tokens (``loss_mask`` = 1 on response positions).
The reference policy is a frozen copy evaluated under no_grad; any
Booster plugin drives the policy's distributed training.
"""

import copy
from typing import Iterable, Optional

import torch
import torch.nn.functional as F

from colossalai_amd import Booster

__all__ = ["DPOTrainer", "dpo_loss", "sequence_log_probs"]


def sequence_log_probs(logits: torch.Tensor, labels: torch.Tensor, loss_mask: torch.Tensor) -> torch.Tensor:
    """Sum of next-token log-probs over masked positions.
    logits [B, S, V]; labels/loss_mask [B, S] (label = token at that
    position; position i is predicted from logits at i-1)."""
    lp = F.log_softmax(logits[:, :-1].float(), dim=-1)
    tok = labels[:, 1:].clamp(min=0)
    picked = lp.gather(-1, tok.unsqueeze(-1)).squeeze(-1)  # [B, S-1]
    return (picked * loss_mask[:, 1:].float()).sum(dim=-1)


def dpo_loss(pi_chosen, pi_rejected, ref_chosen, ref_rejected, beta: float = 0.1):
    margin = (pi_chosen - ref_chosen) - (pi_rejected - ref_rejected)
    loss = -F.logsigmoid(beta * margin).mean()
    with torch.no_grad():
        reward_acc = (margin > 0).float().mean()
    return loss, reward_acc


class DPOTrainer:
    def __init__(self, policy, optimizer, booster: Booster, beta: float = 0.1, lr_scheduler=None):
        self.ref = copy.deepcopy(policy).eval()
        for p in self.ref.parameters():
            p.requires_grad_(False)
        criterion = lambda out, batch: out["loss"]
        self.model, self.optimizer, _, _, self.lr_scheduler = booster.boost(
            policy, optimizer, criterion, lr_scheduler=lr_scheduler
        )
        self.booster = booster
        self.beta = beta
        dev = next(self.model.parameters()).device
        self.ref = self.ref.to(dev)
        if next(self.model.parameters()).dtype != next(self.ref.parameters()).dtype:
            self.ref = self.ref.to(next(self.model.parameters()).dtype)

    def _logp(self, model, ids, mask, grad: bool):
        ctx = torch.enable_grad() if grad else torch.no_grad()
        with ctx:
            logits = model(input_ids=ids)["logits"]
            return sequence_log_probs(logits, ids, mask)

    def train_step(self, batch: dict):
        """batch: chosen_ids / rejected_ids [B, S] and chosen_mask /
        rejected_mask [B, S] (1 on response tokens)."""
        self.model.train()
        pi_c = self._logp(self.model, batch["chosen_ids"], batch["chosen_mask"], grad=True)
        pi_r = self._logp(self.model, batch["rejected_ids"], batch["rejected_mask"], grad=True)
        ref_c = self._logp(self.ref, batch["chosen_ids"], batch["chosen_mask"], grad=False)
        ref_r = self._logp(self.ref, batch["rejected_ids"], batch["rejected_mask"], grad=False)
        loss, acc = dpo_loss(pi_c, pi_r, ref_c, ref_r, self.beta)
        self.booster.backward(loss, self.optimizer)
        self.optimizer.step()
        self.optimizer.zero_grad()
        if self.lr_scheduler is not None:
            self.lr_scheduler.step()
        return float(loss.detach()), float(acc)

    def fit(self, dataloader: Iterable, epochs: int = 1, max_steps: Optional[int] = None):
        hist = []
        step = 0
        for _ in range(epochs):
            for batch in dataloader:
                hist.append(self.train_step(batch))
                step += 1
                if max_steps is not None and step >= max_steps:
                    return hist
        return hist
