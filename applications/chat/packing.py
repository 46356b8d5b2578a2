"""Sequence packing for SFT: concatenate ragged samples into dense
cu_seqlens batches (reference behavior: the varlen path of
colossalai/shardformer/layer/attn.py:139 + ColossalChat's packed dataset).

Packed batches eliminate pad-token compute entirely: the model consumes
them through the varlen flash kernels via ``cu_seqlens``, and each
sequence's first label is masked (-100) so the global causal shift never
produces a cross-boundary target.
"""

from typing import Dict, Iterable, List

import torch

__all__ = ["pack_sft_samples"]


def pack_sft_samples(samples: Iterable[Dict[str, torch.Tensor]], max_tokens: int) -> List[Dict]:
    """samples: dicts with 1-D ``input_ids`` and ``labels``. Greedy first-fit
    packing into batches of at most ``max_tokens`` tokens. Returns dicts of
    {"input_ids" [1, total], "labels" [1, total], "cu_seqlens" [n+1] int32}.
    """
    batches: List[Dict] = []
    cur_ids: List[torch.Tensor] = []
    cur_labels: List[torch.Tensor] = []
    cur_lens: List[int] = []

    def flush():
        if not cur_ids:
            return
        cu = torch.zeros(len(cur_lens) + 1, dtype=torch.int32)
        cu[1:] = torch.tensor(cur_lens, dtype=torch.int32).cumsum(0)
        batches.append({
            "input_ids": torch.cat(cur_ids).unsqueeze(0),
            "labels": torch.cat(cur_labels).unsqueeze(0),
            "cu_seqlens": cu,
        })
        cur_ids.clear()
        cur_labels.clear()
        cur_lens.clear()

    for s in samples:
        ids = s["input_ids"].reshape(-1)
        labels = s["labels"].reshape(-1).clone()
        # after the GLOBAL causal shift, the prediction at the previous
        # sequence's last token targets THIS sequence's first label — mask it
        labels[0] = -100
        if sum(cur_lens) + ids.numel() > max_tokens and cur_lens:
            flush()
        assert ids.numel() <= max_tokens, "sample longer than max_tokens"
        cur_ids.append(ids)
        cur_labels.append(labels)
        cur_lens.append(ids.numel())
    flush()
    return batches
