"""Continued pretraining (reference: applications/Colossal-LLaMA — domain
adaptation of a pretrained base model with packed corpora, checkpointed
resume and token-budget accounting)."""

import os
import sys
from typing import Iterable, Optional

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))

import torch
import torch.distributed as dist

from colossalai_amd import Booster


class ContinuedPretrainTrainer:
    """Booster-based causal-LM trainer with packed varlen batches, periodic
    topology-independent checkpoints, and exact resume (model + optimizer +
    scheduler + step counter)."""

    def __init__(self, model, optimizer, booster: Booster, lr_scheduler=None,
                 save_dir: Optional[str] = None, save_interval: int = 1000):
        criterion = lambda out, batch: out["loss"]
        self.model, self.optimizer, self.criterion, _, self.lr_scheduler = booster.boost(
            model, optimizer, criterion, lr_scheduler=lr_scheduler
        )
        self.booster = booster
        self.save_dir = save_dir
        self.save_interval = save_interval
        self.step_count = 0
        self.tokens_seen = 0

    def train_step(self, batch: dict) -> float:
        self.model.train()
        kwargs = {}
        if batch.get("cu_seqlens") is not None:
            kwargs["cu_seqlens"] = batch["cu_seqlens"]
        if batch.get("attention_mask") is not None:
            kwargs["attention_mask"] = batch["attention_mask"]
        out = self.model(input_ids=batch["input_ids"], labels=batch["labels"], **kwargs)
        loss = self.criterion(out, batch)
        self.booster.backward(loss, self.optimizer)
        self.optimizer.step()
        self.optimizer.zero_grad()
        if self.lr_scheduler is not None:
            self.lr_scheduler.step()
        self.step_count += 1
        self.tokens_seen += int(batch["input_ids"].numel())
        if self.save_dir and self.step_count % self.save_interval == 0:
            self.save(self.save_dir)
        return float(loss.detach())

    def train(self, data_iter: Iterable[dict], max_steps: int, log_interval: int = 10) -> float:
        last = 0.0
        for batch in data_iter:
            last = self.train_step(batch)
            if self.step_count % log_interval == 0 and (not dist.is_initialized() or dist.get_rank() == 0):
                lr = self.optimizer.param_groups[0]["lr"]
                print(f"step {self.step_count}: loss {last:.4f} lr {lr:.2e} tokens {self.tokens_seen}")
            if self.step_count >= max_steps:
                break
        return last

    # ------------------------------------------------------------ checkpoint
    def save(self, path: str) -> None:
        os.makedirs(path, exist_ok=True)
        self.booster.save_model(self.model, os.path.join(path, "model.bin"))
        self.booster.save_optimizer(self.optimizer, os.path.join(path, "optim"))
        if self.lr_scheduler is not None:
            self.booster.save_lr_scheduler(self.lr_scheduler, os.path.join(path, "lr_sched.bin"))
        if not dist.is_initialized() or dist.get_rank() == 0:
            torch.save({"step_count": self.step_count, "tokens_seen": self.tokens_seen},
                       os.path.join(path, "trainer_state.bin"))
        if dist.is_initialized():
            dist.barrier()

    def load(self, path: str) -> None:
        self.booster.load_model(self.model, os.path.join(path, "model.bin"))
        if hasattr(self.optimizer, "update_master_params"):
            self.optimizer.update_master_params(self.model.unwrap())
        self.booster.load_optimizer(self.optimizer, os.path.join(path, "optim"))
        if self.lr_scheduler is not None and os.path.exists(os.path.join(path, "lr_sched.bin")):
            self.booster.load_lr_scheduler(self.lr_scheduler, os.path.join(path, "lr_sched.bin"))
        state = torch.load(os.path.join(path, "trainer_state.bin"), weights_only=False)
        self.step_count = state["step_count"]
        self.tokens_seen = state["tokens_seen"]
