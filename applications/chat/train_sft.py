"""SFT training launcher (synthetic data; swap in your dataset).

    colossalai_amd run --nproc_per_node 8 applications/chat/train_sft.py --model llama-7b
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))

import torch
import torch.distributed as dist

import colossalai_amd
from applications.chat import SFTTrainer
from colossalai_amd import Booster
from colossalai_amd.booster.plugin import LowLevelZeroPlugin
from colossalai_amd.models import LLAMA_CONFIGS, LlamaForCausalLM
from colossalai_amd.nn import FusedAdam
from colossalai_amd.nn.lr_scheduler import CosineAnnealingWarmupLR


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama-7b", choices=list(LLAMA_CONFIGS))
    p.add_argument("--lr", type=float, default=2e-5)
    p.add_argument("--batch", type=int, default=8)
    p.add_argument("--seq", type=int, default=2048)
    p.add_argument("--steps", type=int, default=100)
    args = p.parse_args()

    colossalai_amd.launch_from_torch()
    cfg = LLAMA_CONFIGS[args.model]
    model = LlamaForCausalLM(cfg)
    model.gradient_checkpointing_enable()
    opt = FusedAdam(model.parameters(), lr=args.lr, weight_decay=0.0)
    sched = CosineAnnealingWarmupLR(opt, total_steps=args.steps, warmup_steps=max(args.steps // 20, 1))
    trainer = SFTTrainer(model, opt, Booster(plugin=LowLevelZeroPlugin(stage=2, precision="bf16")),
                         lr_scheduler=sched)

    device = "cuda" if torch.cuda.is_available() else "cpu"
    for step in range(args.steps):
        x = torch.randint(0, cfg.vocab_size, (args.batch, args.seq), device=device)
        labels = x.clone()
        labels[:, : args.seq // 4] = -100  # prompt masked
        loss = trainer.train_step({"input_ids": x, "labels": labels})
        if step % 10 == 0 and dist.get_rank() == 0:
            print(f"step {step}: sft loss {loss:.4f}")


if __name__ == "__main__":
    main()
