"""Flagship benchmark: Llama-7B, seq 4096, bf16, synthetic data
(BASELINE.json metric: samples/sec + TFLOPS/GPU, ZeRO-2 dp8).

Run (single GPU):   python bench.py --steps 10 --warmup 3
Run (N GPUs):       torchrun --nproc-per-node N bench.py --gpus N ...

Prints ONE JSON line from rank 0 with the whole-job aggregate.
"""

import argparse
import json
import os
import time

import torch
import torch.distributed as dist


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=5)
    p.add_argument("--warmup", type=int, default=2)
    p.add_argument("--model", type=str, default="llama-7b")
    p.add_argument("--batch", type=int, default=36, help="per-DP-rank batch size (reference: bs/DP=36)")
    p.add_argument("--seq", type=int, default=4096)
    p.add_argument("--plugin", type=str, default="zero2", choices=["ddp", "zero2", "zero1"])
    p.add_argument("--grad-ckpt", dest="grad_ckpt", action="store_true", default=True)
    p.add_argument("--ckpt-ratio", type=float, default=1.0, help="fraction of layers checkpointed")
    p.add_argument("--no-grad-ckpt", dest="grad_ckpt", action="store_false")
    return p.parse_args()


def main():
    args = parse_args()
    import colossalai_amd
    from colossalai_amd.models import LLAMA_CONFIGS, LlamaForCausalLM, llama_flops_per_token

    distributed = "RANK" in os.environ
    if distributed:
        colossalai_amd.launch_from_torch(verbose=False)
        rank, world = dist.get_rank(), dist.get_world_size()
    else:
        rank, world = 0, 1
    device = "cuda" if torch.cuda.is_available() else "cpu"
    if device == "cpu":
        raise SystemExit("bench.py requires an MI355X GPU")

    cfg = LLAMA_CONFIGS[args.model]
    torch.manual_seed(42)
    with torch.device("meta"):
        model = LlamaForCausalLM(cfg)
    model = model.to_empty(device=device).to(torch.bfloat16)
    # cheap random init (init on meta then re-init materialized weights)
    with torch.no_grad():
        for p in model.parameters():
            p.normal_(0.0, cfg.initializer_range)
    if args.grad_ckpt:
        model.gradient_checkpointing_enable(args.ckpt_ratio)

    numel = model.num_parameters

    if args.plugin == "ddp":
        from colossalai_amd import Booster
        from colossalai_amd.booster.plugin import TorchDDPPlugin
        from colossalai_amd.nn import FusedAdam

        optimizer = FusedAdam(model.parameters(), lr=1e-5, weight_decay=0.1)
        if distributed:
            booster = Booster(plugin=TorchDDPPlugin(bucket_cap_mb=128))
            model, optimizer, *_ = booster.boost(model, optimizer)
            backward = lambda loss: booster.backward(loss, optimizer)
        else:
            backward = lambda loss: loss.backward()
    else:
        from colossalai_amd import Booster
        from colossalai_amd.booster.plugin import LowLevelZeroPlugin
        from colossalai_amd.nn import FusedAdam

        optimizer = FusedAdam(model.parameters(), lr=1e-5, weight_decay=0.1)
        plugin = LowLevelZeroPlugin(stage=2 if args.plugin == "zero2" else 1, precision="bf16")
        booster = Booster(plugin=plugin)
        model, optimizer, *_ = booster.boost(model, optimizer)
        backward = lambda loss: booster.backward(loss, optimizer)

    B, S = args.batch, args.seq
    data = torch.randint(0, cfg.vocab_size, (B, S), device=device)

    def step():
        out = model(data, labels=data)
        loss = out["loss"]
        backward(loss)
        optimizer.step()
        optimizer.zero_grad()
        return loss

    for _ in range(args.warmup):
        step()

    if distributed:
        dist.barrier()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    if distributed:
        dist.barrier()
    torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0

    if distributed:
        t = torch.tensor([elapsed], device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    ms_per_step = elapsed / args.steps * 1000
    samples_per_sec = args.steps * B * world / elapsed
    flop_per_token = llama_flops_per_token(cfg, S, args.grad_ckpt)  # reference formula (full-ckpt factor)
    tflops_per_gpu = flop_per_token * B * S * args.steps / elapsed / 1e12  # per GPU (weak scaling)

    if rank == 0:
        result = {
            "metric": "samples/sec (Llama-7B seq4096 bf16)",
            "value": round(samples_per_sec, 3),
            "unit": "samples/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 1),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": round(samples_per_sec / 17.13, 4),
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": B * world,
                "seq_len": S,
                "parallelism": f"{args.plugin}(dp{world})",
                "grad_ckpt": args.grad_ckpt,
                "tflops_per_gpu": round(tflops_per_gpu, 1),
                "params": numel,
                "peak_mem_gib": round(torch.cuda.max_memory_allocated() / 2**30, 1),
            },
        }
        print(json.dumps(result))

    if distributed:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
