"""GPT-2 pretraining example (synthetic data).

    colossalai_amd run --nproc_per_node 8 examples/language/gpt2/train.py --model gpt2 --plugin zero2
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", "..", ".."))

import torch
import torch.distributed as dist

import colossalai_amd
from colossalai_amd import Booster
from colossalai_amd.booster.plugin import HybridParallelPlugin, LowLevelZeroPlugin
from colossalai_amd.models.gpt2 import GPT2_CONFIGS, GPT2LMHeadModel
from colossalai_amd.nn import FusedAdam


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="gpt2", choices=list(GPT2_CONFIGS))
    p.add_argument("--plugin", default="zero2", choices=["zero1", "zero2", "hybrid"])
    p.add_argument("--tp", type=int, default=1)
    p.add_argument("--batch", type=int, default=32)
    p.add_argument("--seq", type=int, default=1024)
    p.add_argument("--steps", type=int, default=50)
    args = p.parse_args()

    colossalai_amd.launch_from_torch()
    cfg = GPT2_CONFIGS[args.model]
    model = GPT2LMHeadModel(cfg)
    model.gradient_checkpointing_enable()
    if args.plugin == "hybrid":
        plugin = HybridParallelPlugin(tp_size=args.tp, precision="bf16", zero_stage=1)
    else:
        plugin = LowLevelZeroPlugin(stage=2 if args.plugin == "zero2" else 1, precision="bf16")
    optimizer = FusedAdam(model.parameters(), lr=3e-4, weight_decay=0.1)
    booster = Booster(plugin=plugin)
    model, optimizer, *_ = booster.boost(model, optimizer)

    device = "cuda" if torch.cuda.is_available() else "cpu"
    for step in range(args.steps):
        x = torch.randint(0, cfg.vocab_size, (args.batch, args.seq), device=device)
        out = model(x, labels=x)
        optimizer.backward(out["loss"])
        optimizer.step()
        optimizer.zero_grad()
        if step % 10 == 0 and dist.get_rank() == 0:
            print(f"step {step}: loss {out['loss'].item():.4f}")


if __name__ == "__main__":
    main()
