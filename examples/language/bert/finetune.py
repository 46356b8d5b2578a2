"""BERT sequence-classification fine-tune example (synthetic data).

    colossalai_amd run --nproc_per_node 8 examples/language/bert/finetune.py --plugin ddp
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", "..", ".."))

import torch
import torch.distributed as dist

import colossalai_amd
from colossalai_amd import Booster
from colossalai_amd.booster.plugin import LowLevelZeroPlugin, TorchDDPPlugin
from colossalai_amd.models.bert import BERT_CONFIGS, BertForSequenceClassification
from colossalai_amd.nn import FusedAdam


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="bert-base", choices=list(BERT_CONFIGS))
    p.add_argument("--plugin", default="ddp", choices=["ddp", "zero1"])
    p.add_argument("--batch", type=int, default=64)
    p.add_argument("--seq", type=int, default=128)
    p.add_argument("--steps", type=int, default=50)
    args = p.parse_args()

    colossalai_amd.launch_from_torch()
    cfg = BERT_CONFIGS[args.model]
    model = BertForSequenceClassification(cfg)
    plugin = TorchDDPPlugin() if args.plugin == "ddp" else LowLevelZeroPlugin(stage=1, precision="bf16")
    optimizer = FusedAdam(model.parameters(), lr=2e-5)
    booster = Booster(plugin=plugin)
    model, optimizer, *_ = booster.boost(model, optimizer)

    device = "cuda" if torch.cuda.is_available() else "cpu"
    for step in range(args.steps):
        x = torch.randint(0, cfg.vocab_size, (args.batch, args.seq), device=device)
        y = torch.randint(0, cfg.num_labels, (args.batch,), device=device)
        out = model(x, labels=y)
        optimizer.backward(out["loss"])
        optimizer.step()
        optimizer.zero_grad()
        if step % 10 == 0 and dist.get_rank() == 0:
            print(f"step {step}: loss {out['loss'].item():.4f}")


if __name__ == "__main__":
    main()
