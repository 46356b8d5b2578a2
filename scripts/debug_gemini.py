import copy
import faulthandler

import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def run(rank):
    faulthandler.enable()
    import os
    import sys

    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    import colossalai_amd
    from colossalai_amd.models import LlamaConfig, LlamaForCausalLM
    from colossalai_amd.nn import FusedAdam
    from colossalai_amd.zero import GeminiDDP, GeminiOptimizer

    colossalai_amd.launch(rank, 2, "127.0.0.1", 29531, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)
    model = LlamaForCausalLM(cfg)
    gm = GeminiDDP(copy.deepcopy(model), precision="fp32", chunk_size_m=1)
    opt = GeminiOptimizer(FusedAdam(gm.parameters(), lr=1e-2), gm)
    if rank == 0:
        print("built", flush=True)
    x = torch.randint(0, 128, (2, 16))
    out = gm(input_ids=x, labels=x)
    if rank == 0:
        print("fwd", out["loss"].item(), flush=True)
    opt.backward(out["loss"])
    if rank == 0:
        print("bwd done", flush=True)
    opt.step()
    if rank == 0:
        print("step done", flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    mp.spawn(run, nprocs=2)
