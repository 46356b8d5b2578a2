"""DPO training launcher (synthetic preference data; swap in your dataset).

    colossalai_amd run --nproc_per_node 8 applications/chat/train_dpo.py --model llama-7b
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))

import torch

import colossalai_amd
from applications.chat import DPOTrainer
from colossalai_amd import Booster
from colossalai_amd.booster.plugin import LowLevelZeroPlugin
from colossalai_amd.models import LLAMA_CONFIGS, LlamaForCausalLM
from colossalai_amd.nn import FusedAdam


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama-7b", choices=list(LLAMA_CONFIGS))
    p.add_argument("--beta", type=float, default=0.1)
    p.add_argument("--lr", type=float, default=5e-7)
    p.add_argument("--batch", type=int, default=4)
    p.add_argument("--seq", type=int, default=1024)
    p.add_argument("--steps", type=int, default=50)
    args = p.parse_args()

    colossalai_amd.launch_from_torch()
    cfg = LLAMA_CONFIGS[args.model]
    policy = LlamaForCausalLM(cfg)
    policy.gradient_checkpointing_enable()
    trainer = DPOTrainer(policy, FusedAdam(policy.parameters(), lr=args.lr),
                         Booster(plugin=LowLevelZeroPlugin(stage=2, precision="bf16")),
                         beta=args.beta)

    device = "cuda" if torch.cuda.is_available() else "cpu"
    import torch.distributed as dist

    for step in range(args.steps):
        chosen = torch.randint(0, cfg.vocab_size, (args.batch, args.seq), device=device)
        rejected = torch.randint(0, cfg.vocab_size, (args.batch, args.seq), device=device)
        mask = torch.ones_like(chosen)
        mask[:, : args.seq // 2] = 0  # prompt half
        loss, acc = trainer.train_step({"chosen_ids": chosen, "chosen_mask": mask,
                                        "rejected_ids": rejected, "rejected_mask": mask})
        if step % 10 == 0 and dist.get_rank() == 0:
            print(f"step {step}: dpo loss {loss:.4f} reward-acc {acc:.2f}")


if __name__ == "__main__":
    main()
