"""Lazy (meta-device) model initialization
(reference: colossalai/lazy/lazy_init.py:134,474 — re-designed).

The reference records constructor ops on a LazyTensor and replays them at
materialization. On MI355X the simpler meta-device approach covers the same
use case (build a 70B model skeleton instantly, shard it, then materialize
only this rank's shard): inside ``LazyInitContext`` every ``nn.Module``
constructor allocates parameters on the meta device; ``materialize``
re-runs ``reset_parameters``/init on the real device for whatever modules
survived sharding/stage release.
"""

from contextlib import contextmanager
from typing import Callable, Optional

import torch
import torch.nn as nn

__all__ = ["LazyInitContext"]


class LazyInitContext:
    """
    Usage::

        with LazyInitContext():
            model = LlamaForCausalLM(cfg)      # params on meta device, instant
        shardformer.optimize(model)            # surgery on meta tensors
        LazyInitContext.materialize(model, device="cuda", init_fn=model_init)
    """

    def __init__(self, default_device: str = "meta"):
        self.default_device = default_device
        self._ctx = None

    def __enter__(self):
        self._ctx = torch.device(self.default_device)
        self._ctx.__enter__()
        return self

    def __exit__(self, *args):
        self._ctx.__exit__(*args)
        self._ctx = None

    @staticmethod
    def materialize(module: nn.Module, device: str = "cuda", dtype: Optional[torch.dtype] = None,
                    init_fn: Optional[Callable[[nn.Module], None]] = None) -> nn.Module:
        """Allocate real storage for all meta params/buffers and initialize."""
        def is_meta(m: nn.Module) -> bool:
            return any(t.is_meta for t in list(m.parameters(recurse=False)) + list(m.buffers(recurse=False)))

        module = module.to_empty(device=device) if any(p.is_meta for p in module.parameters()) else module
        if dtype is not None:
            module = module.to(dtype)
        if init_fn is not None:
            init_fn(module)
        else:
            for m in module.modules():
                if hasattr(m, "reset_parameters"):
                    m.reset_parameters()
        return module

    @staticmethod
    def materialize_from_state_dict(module: nn.Module, state_dict, device: str = "cuda",
                                    dtype: Optional[torch.dtype] = None, strict: bool = False) -> nn.Module:
        """from_pretrained integration (reference: colossalai/lazy/pretrained.py):
        allocate real storage and fill it straight from a checkpoint state
        dict — the model never exists twice in memory (meta skeleton ->
        storage -> weights). Entries may be a dict or an iterable of dicts
        (HF-style sharded files loaded one at a time)."""
        if any(p.is_meta for p in module.parameters()):
            module = module.to_empty(device=device)
        shards = [state_dict] if isinstance(state_dict, dict) else list(state_dict)
        loaded = set()
        for shard in shards:
            cast = {}
            for k, v in shard.items():
                if torch.is_tensor(v):
                    v = v.to(device=device)
                    if dtype is not None and v.is_floating_point():
                        v = v.to(dtype)
                cast[k] = v
            module.load_state_dict(cast, strict=False)
            loaded.update(cast.keys())
        if strict:
            missing = set(k for k, _ in module.named_parameters()) - loaded
            if missing:
                raise RuntimeError(f"materialize_from_state_dict: missing keys {sorted(missing)[:8]}...")
        if dtype is not None:
            module = module.to(dtype)
        return module
