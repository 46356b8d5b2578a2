"""Mixtral (sparse MoE) expert-parallel training example (synthetic data).

    colossalai_amd run --nproc_per_node 8 examples/language/mixtral/train.py --ep 8
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", "..", ".."))

import torch
import torch.distributed as dist

import colossalai_amd
from colossalai_amd import Booster
from colossalai_amd.booster.plugin import MoeHybridParallelPlugin
from colossalai_amd.models.mixtral import MIXTRAL_CONFIGS, MixtralForCausalLM
from colossalai_amd.nn import FusedAdam


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="mixtral-small", choices=list(MIXTRAL_CONFIGS))
    p.add_argument("--ep", type=int, default=1)
    p.add_argument("--zero", type=int, default=1)
    p.add_argument("--batch", type=int, default=8)
    p.add_argument("--seq", type=int, default=4096)
    p.add_argument("--steps", type=int, default=50)
    args = p.parse_args()

    colossalai_amd.launch_from_torch()
    cfg = MIXTRAL_CONFIGS[args.model]
    model = MixtralForCausalLM(cfg)
    model.gradient_checkpointing_enable()
    plugin = MoeHybridParallelPlugin(ep_size=args.ep, zero_stage=args.zero, precision="bf16")
    optimizer = FusedAdam(model.parameters(), lr=3e-4, weight_decay=0.1)
    booster = Booster(plugin=plugin)
    model, optimizer, *_ = booster.boost(model, optimizer)

    device = "cuda" if torch.cuda.is_available() else "cpu"
    for step in range(args.steps):
        x = torch.randint(0, cfg.vocab_size, (args.batch, args.seq), device=device)
        out = model(input_ids=x, labels=x)
        optimizer.backward(out["loss"])
        optimizer.step()
        optimizer.zero_grad()
        if step % 10 == 0 and dist.get_rank() == 0:
            print(f"step {step}: loss {out['loss'].item():.4f}")


if __name__ == "__main__":
    main()
