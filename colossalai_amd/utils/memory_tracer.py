"""Module-level memory tracer (reference:
colossalai/zero/gemini/memory_tracer/ — MemStats + param/runtime hooks,
re-designed as plain module hooks over the HIP caching allocator).

Records, for every module in a model, the allocator delta across its
forward and backward and the bytes of its outputs/parameters, plus the
global peak. On CPU the allocator numbers are 0 but tensor-byte accounting
still works, so the API (and its test) is device-neutral.

Use it to size chunk budgets / activation-checkpoint ratios:

    tracer = MemoryTracer(model)
    with tracer:
        loss = model(**batch)["loss"]; loss.backward()
    print(tracer.report())
"""

from dataclasses import dataclass, field
from typing import Dict, List, Optional

import torch
import torch.nn as nn

__all__ = ["MemoryTracer", "ModuleMemStats"]


def _alloc() -> int:
    return torch.cuda.memory_allocated() if torch.cuda.is_available() else 0


def _tensor_bytes(obj) -> int:
    if isinstance(obj, torch.Tensor):
        return obj.numel() * obj.element_size()
    if isinstance(obj, (list, tuple)):
        return sum(_tensor_bytes(o) for o in obj)
    if isinstance(obj, dict):
        return sum(_tensor_bytes(o) for o in obj.values())
    return 0


@dataclass
class ModuleMemStats:
    name: str
    param_bytes: int = 0
    fwd_alloc_delta: int = 0  # allocator growth across forward (activations)
    fwd_out_bytes: int = 0
    bwd_alloc_delta: int = 0
    calls: int = 0


class MemoryTracer:
    """Attach with ``with tracer:`` (or ``start()``/``stop()``) around one
    or more fwd+bwd iterations; read ``stats`` / ``report()`` after."""

    def __init__(self, model: nn.Module, leaf_only: bool = True):
        self.model = model
        self.leaf_only = leaf_only
        self.stats: Dict[str, ModuleMemStats] = {}
        self.peak_bytes = 0
        self._handles: List = []
        self._fwd_start: Dict[int, int] = {}

    # ------------------------------------------------------------- lifecycle
    def start(self):
        if torch.cuda.is_available():
            torch.cuda.reset_peak_memory_stats()
        for name, mod in self.model.named_modules():
            if self.leaf_only and any(True for _ in mod.children()):
                continue
            st = self.stats.setdefault(
                name or "<root>",
                ModuleMemStats(name or "<root>",
                               param_bytes=sum(_tensor_bytes(p) for p in mod.parameters(recurse=False))),
            )
            self._handles.append(mod.register_forward_pre_hook(self._fwd_pre(st, mod)))
            self._handles.append(mod.register_forward_hook(self._fwd_post(st, mod)))
            self._handles.append(mod.register_full_backward_pre_hook(self._bwd_pre(st, mod)))
            self._handles.append(mod.register_full_backward_hook(self._bwd_post(st, mod)))
        return self

    def stop(self):
        for h in self._handles:
            h.remove()
        self._handles.clear()
        if torch.cuda.is_available():
            self.peak_bytes = max(self.peak_bytes, torch.cuda.max_memory_allocated())

    def __enter__(self):
        return self.start()

    def __exit__(self, *exc):
        self.stop()

    # ----------------------------------------------------------------- hooks
    def _fwd_pre(self, st, mod):
        def hook(module, args):
            self._fwd_start[id(mod)] = _alloc()
        return hook

    def _fwd_post(self, st, mod):
        def hook(module, args, output):
            st.fwd_alloc_delta += _alloc() - self._fwd_start.pop(id(mod), _alloc())
            st.fwd_out_bytes += _tensor_bytes(output)
            st.calls += 1
        return hook

    def _bwd_pre(self, st, mod):
        def hook(module, grad_output):
            self._fwd_start[id(mod)] = _alloc()
        return hook

    def _bwd_post(self, st, mod):
        def hook(module, grad_input, grad_output):
            st.bwd_alloc_delta += _alloc() - self._fwd_start.pop(id(mod), _alloc())
        return hook

    # ---------------------------------------------------------------- output
    def report(self, top: int = 30) -> str:
        rows = sorted(self.stats.values(), key=lambda s: -(s.fwd_alloc_delta + s.fwd_out_bytes))
        lines = [f"{'module':<52} {'params':>10} {'fwd Δalloc':>12} {'fwd out':>10} {'bwd Δalloc':>12}"]
        mb = 1024 * 1024
        for s in rows[:top]:
            if s.calls == 0:
                continue
            lines.append(
                f"{s.name[:52]:<52} {s.param_bytes/mb:>9.1f}M {s.fwd_alloc_delta/mb:>11.1f}M "
                f"{s.fwd_out_bytes/mb:>9.1f}M {s.bwd_alloc_delta/mb:>11.1f}M"
            )
        lines.append(f"peak allocator bytes: {self.peak_bytes/mb:.1f} MiB")
        return "\n".join(lines)
