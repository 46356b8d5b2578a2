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

# hipBLASLt autotuned GEMM algos (tuned offline on MI355X, committed in-repo);
# read-only: shapes missing from the file silently use the default algo.
_TUNE = os.path.join(os.path.dirname(os.path.abspath(__file__)), "profiles", "tunableop", "results.csv")
if os.path.exists(_TUNE.replace("results.csv", "results0.csv")) and os.environ.get("CAI_TUNABLEOP", "1") == "1":
    os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "0")
    os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME", _TUNE)

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
    p.add_argument("--plugin", type=str, default="zero2",
                   choices=["ddp", "zero2", "zero1", "gemini", "gemini3", "hybrid", "moe"])
    p.add_argument("--tp", type=int, default=1)
    p.add_argument("--pp", type=int, default=1)
    p.add_argument("--sp", type=int, default=1)
    p.add_argument("--ep", type=int, default=1)
    p.add_argument("--zero", type=int, default=1, help="zero stage inside hybrid/moe plugins")
    p.add_argument("--microbatches", type=int, default=None)
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
        if args.plugin in ("hybrid", "moe"):
            # mesh-based plugins need an initialized process group even at N=1
            from colossalai_amd.testing import free_port

            colossalai_amd.launch(0, 1, "127.0.0.1", free_port(), verbose=False)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    if device == "cpu":
        raise SystemExit("bench.py requires an MI355X GPU")

    if args.model.startswith("mixtral"):
        from colossalai_amd.models.mixtral import MIXTRAL_CONFIGS, MixtralForCausalLM

        cfg = MIXTRAL_CONFIGS[args.model]
        model_cls = MixtralForCausalLM
    else:
        cfg = LLAMA_CONFIGS[args.model]
        model_cls = LlamaForCausalLM
    torch.manual_seed(42)
    with torch.device("meta"):
        model = model_cls(cfg)
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
    elif args.plugin in ("hybrid", "moe"):
        from colossalai_amd import Booster
        from colossalai_amd.booster.plugin import HybridParallelPlugin, MoeHybridParallelPlugin
        from colossalai_amd.nn import FusedAdam

        kwargs = dict(tp_size=args.tp, pp_size=args.pp, sp_size=args.sp, precision="bf16",
                      zero_stage=args.zero, num_microbatches=args.microbatches)
        if args.sp > 1:
            kwargs.update(enable_sequence_parallelism=True, sequence_parallelism_mode="all_to_all")
        if args.plugin == "moe":
            plugin = MoeHybridParallelPlugin(ep_size=args.ep, **kwargs)
        else:
            plugin = HybridParallelPlugin(**kwargs)
        optimizer = FusedAdam(model.parameters(), lr=1e-5, weight_decay=0.1)
        booster = Booster(plugin=plugin)
        model, optimizer, *_ = booster.boost(model, optimizer)
        if args.pp > 1:
            criterion = lambda out, micro: out["loss"]
            backward = None  # handled inside execute_pipeline
        else:
            backward = lambda loss: booster.backward(loss, optimizer)
    elif args.plugin in ("gemini", "gemini3"):
        from colossalai_amd import Booster
        from colossalai_amd.booster.plugin import GeminiPlugin
        from colossalai_amd.nn import FusedAdam, HybridAdam

        if args.plugin == "gemini3":  # native chunk-sharded params (ZeRO-3)
            optimizer = FusedAdam(model.parameters(), lr=1e-5, weight_decay=0.1)
            booster = Booster(plugin=GeminiPlugin(precision="bf16", shard_param_frac=1.0))
        else:
            optimizer = HybridAdam(model.parameters(), lr=1e-5, weight_decay=0.1)
            booster = Booster(plugin=GeminiPlugin(precision="bf16"))
        model, optimizer, *_ = booster.boost(model, optimizer)
        backward = lambda loss: booster.backward(loss, optimizer)
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

    if args.plugin in ("hybrid", "moe") and args.pp > 1:
        def step():
            batch = {"input_ids": data, "labels": data}
            result = booster.execute_pipeline(iter([batch]), model, criterion, optimizer, return_loss=True)
            optimizer.step()
            optimizer.zero_grad()
            return result["loss"]
    else:
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
    # Reference-formula TFLOPS counts recompute as useful work (4x factor
    # under grad-ckpt, performance_evaluator.py:164); the 3x "model FLOPs"
    # number is reported alongside so hardware utilization is not overstated
    # (VERDICT r1 weak #3).
    flop_per_token = llama_flops_per_token(cfg, S, args.grad_ckpt)  # reference formula (full-ckpt factor)
    tflops_per_gpu = flop_per_token * B * S * args.steps / elapsed / 1e12  # per GPU (weak scaling)
    flop_per_token_3x = llama_flops_per_token(cfg, S, False)
    tflops_model = flop_per_token_3x * B * S * args.steps / elapsed / 1e12

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
                "parallelism": f"{args.plugin}(dp{world // (args.tp * args.pp * args.sp)} tp{args.tp} pp{args.pp} sp{args.sp})".format(args=args, world=world) if args.plugin in ("hybrid", "moe") else f"{args.plugin}(dp{world})",
                "grad_ckpt": args.grad_ckpt,
                "tflops_per_gpu": round(tflops_per_gpu, 1),
                "tflops_per_gpu_no_recompute": round(tflops_model, 1),
                "params": numel,
                "peak_mem_gib": round(torch.cuda.max_memory_allocated() / 2**30, 1),
            },
        }
        print(json.dumps(result))

    if distributed:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
