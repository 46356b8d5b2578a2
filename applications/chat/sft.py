"""Supervised fine-tuning trainer (reference:
applications/ColossalChat/coati/trainer/sft.py — lean rewrite over the
Booster API).

Batches are dicts with ``input_ids`` and ``labels`` where prompt tokens
are masked to -100 so only response tokens contribute to the loss; any
plugin (DDP / ZeRO / Gemini / Hybrid) works underneath.
"""

from typing import Iterable, Optional

import torch

from colossalai_amd import Booster

__all__ = ["SFTTrainer"]


class SFTTrainer:
    def __init__(self, model, optimizer, booster: Booster, lr_scheduler=None, max_norm: float = 0.0):
        criterion = lambda out, batch: out["loss"]
        self.model, self.optimizer, self.criterion, _, self.lr_scheduler = booster.boost(
            model, optimizer, criterion, lr_scheduler=lr_scheduler
        )
        self.booster = booster
        self.max_norm = max_norm

    def train_step(self, batch: dict) -> float:
        self.model.train()
        # right-padded SFT batches: the mask flows into the flash kernels
        # (pad rows are excluded from attention, not just from the loss)
        kwargs = {}
        if batch.get("attention_mask") is not None:
            kwargs["attention_mask"] = batch["attention_mask"]
        if batch.get("cu_seqlens") is not None:
            # packed ragged batch from applications/chat/packing.py
            kwargs["cu_seqlens"] = batch["cu_seqlens"]
        out = self.model(input_ids=batch["input_ids"], labels=batch["labels"], **kwargs)
        loss = self.criterion(out, batch)
        self.booster.backward(loss, self.optimizer)
        if self.max_norm > 0:
            self.optimizer.clip_grad_by_norm(self.max_norm)
        self.optimizer.step()
        self.optimizer.zero_grad()
        if self.lr_scheduler is not None:
            self.lr_scheduler.step()
        return float(loss.detach())

    def fit(self, dataloader: Iterable, epochs: int = 1, max_steps: Optional[int] = None):
        losses = []
        step = 0
        for _ in range(epochs):
            for batch in dataloader:
                losses.append(self.train_step(batch))
                step += 1
                if max_steps is not None and step >= max_steps:
                    return losses
        return losses
