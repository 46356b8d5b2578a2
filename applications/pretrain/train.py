"""Continued-pretraining launcher (synthetic corpus; swap in your data).

    colossalai_amd run --nproc_per_node 8 applications/pretrain/train.py --model llama-7b
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))

import torch

import colossalai_amd
from applications.pretrain import ContinuedPretrainTrainer
from colossalai_amd import Booster
from colossalai_amd.booster.plugin import LowLevelZeroPlugin
from colossalai_amd.models import LLAMA_CONFIGS, LlamaForCausalLM
from colossalai_amd.nn import FusedAdam
from colossalai_amd.nn.lr_scheduler import CosineAnnealingWarmupLR


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama-7b", choices=list(LLAMA_CONFIGS))
    p.add_argument("--lr", type=float, default=1e-4)
    p.add_argument("--batch", type=int, default=8)
    p.add_argument("--seq", type=int, default=4096)
    p.add_argument("--steps", type=int, default=1000)
    p.add_argument("--save-dir", default=None)
    args = p.parse_args()

    colossalai_amd.launch_from_torch()
    cfg = LLAMA_CONFIGS[args.model]
    model = LlamaForCausalLM(cfg)
    model.gradient_checkpointing_enable()
    opt = FusedAdam(model.parameters(), lr=args.lr, weight_decay=0.1)
    sched = CosineAnnealingWarmupLR(opt, total_steps=args.steps, warmup_steps=max(args.steps // 50, 1))
    trainer = ContinuedPretrainTrainer(
        model, opt, Booster(plugin=LowLevelZeroPlugin(stage=2, precision="bf16")),
        lr_scheduler=sched, save_dir=args.save_dir)

    device = "cuda" if torch.cuda.is_available() else "cpu"

    def corpus():
        while True:
            x = torch.randint(0, cfg.vocab_size, (args.batch, args.seq), device=device)
            yield {"input_ids": x, "labels": x.clone()}

    trainer.train(corpus(), max_steps=args.steps)


if __name__ == "__main__":
    main()
