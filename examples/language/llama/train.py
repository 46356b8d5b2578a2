"""Llama pretraining example on MI355X (mirrors the reference's
examples/language/llama/benchmark.py usage shape).

Launch:
    colossalai_amd run --nproc_per_node 8 examples/language/llama/train.py \
        --model llama-7b --plugin zero2 --batch 36 --seq 4096 --steps 50
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", "..", ".."))

import torch
import torch.distributed as dist

import colossalai_amd
from colossalai_amd import Booster
from colossalai_amd.booster.plugin import GeminiPlugin, HybridParallelPlugin, LowLevelZeroPlugin
from colossalai_amd.models import LLAMA_CONFIGS, LlamaForCausalLM
from colossalai_amd.nn import FusedAdam, HybridAdam
from colossalai_amd.nn.lr_scheduler import CosineAnnealingWarmupLR
from colossalai_amd.utils import MultiTimer, report_memory_usage


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama-7b")
    p.add_argument("--plugin", default="zero2", choices=["zero1", "zero2", "gemini", "hybrid"])
    p.add_argument("--tp", type=int, default=1)
    p.add_argument("--pp", type=int, default=1)
    p.add_argument("--batch", type=int, default=36)
    p.add_argument("--seq", type=int, default=4096)
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--lr", type=float, default=3e-4)
    p.add_argument("--grad-ckpt", action="store_true", default=True)
    p.add_argument("--save", type=str, default=None)
    args = p.parse_args()

    colossalai_amd.launch_from_torch()
    cfg = LLAMA_CONFIGS[args.model]
    model = LlamaForCausalLM(cfg)
    if args.grad_ckpt:
        model.gradient_checkpointing_enable()

    if args.plugin == "gemini":
        plugin = GeminiPlugin(precision="bf16")
        optimizer = HybridAdam(model.parameters(), lr=args.lr, weight_decay=0.1)
    elif args.plugin == "hybrid":
        plugin = HybridParallelPlugin(tp_size=args.tp, pp_size=args.pp, precision="bf16", zero_stage=1,
                                      num_microbatches=max(args.pp * 2, 1) if args.pp > 1 else None)
        optimizer = FusedAdam(model.parameters(), lr=args.lr, weight_decay=0.1)
    else:
        plugin = LowLevelZeroPlugin(stage=2 if args.plugin == "zero2" else 1, precision="bf16")
        optimizer = FusedAdam(model.parameters(), lr=args.lr, weight_decay=0.1)

    lr_sched = CosineAnnealingWarmupLR(optimizer, total_steps=args.steps, warmup_steps=max(args.steps // 20, 1))
    booster = Booster(plugin=plugin)
    criterion = lambda out, batch: out["loss"]
    model, optimizer, criterion, _, lr_sched = booster.boost(model, optimizer, criterion, lr_scheduler=lr_sched)

    timer = MultiTimer()
    device = "cuda" if torch.cuda.is_available() else "cpu"
    for step in range(args.steps):
        data = torch.randint(0, cfg.vocab_size, (args.batch, args.seq), device=device)
        timer.start("step")
        if args.pp > 1:
            result = booster.execute_pipeline(iter([{"input_ids": data, "labels": data}]),
                                              model, criterion, optimizer, return_loss=True)
            loss = result["loss"]
        else:
            out = model(data, labels=data)
            loss = out["loss"]
            booster.backward(loss, optimizer)
        optimizer.step()
        optimizer.zero_grad()
        lr_sched.step()
        dt = timer.stop("step")
        if dist.get_rank() == 0 and loss is not None:
            tput = args.batch * dist.get_world_size() / dt
            print(f"step {step}: loss {float(loss):.4f}  {dt*1000:.0f} ms  {tput:.2f} samples/s")
    if dist.get_rank() == 0:
        report_memory_usage("final")
    if args.save:
        booster.save_model(model, args.save, use_safetensors=args.save.endswith(".safetensors"))
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
