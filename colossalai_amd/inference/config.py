"""Inference configuration (reference: colossalai/inference/config.py)."""

from dataclasses import dataclass
from typing import Optional

import torch

__all__ = ["InferenceConfig", "GenerationConfig"]


@dataclass
class InferenceConfig:
    max_batch_size: int = 8
    max_input_len: int = 2048
    max_output_len: int = 256
    dtype: torch.dtype = torch.bfloat16
    pad_token_id: int = 0
    eos_token_id: Optional[int] = None
    # capture the decode step in a hipGraph (one graph per batch size):
    # removes per-layer launch gaps in the latency-bound decode loop
    use_hip_graph: bool = False

    @property
    def max_seq_len(self) -> int:
        return self.max_input_len + self.max_output_len


@dataclass
class GenerationConfig:
    max_new_tokens: int = 128
    do_sample: bool = False
    temperature: float = 1.0
    top_k: int = 0
    top_p: float = 1.0
