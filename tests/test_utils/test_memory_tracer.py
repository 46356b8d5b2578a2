"""MemoryTracer API test (CPU: allocator deltas are 0 but byte accounting
and the hook lifecycle must work)."""

import torch
import torch.nn as nn

from colossalai_amd.utils import MemoryTracer


def test_memory_tracer_records():
    model = nn.Sequential(nn.Linear(16, 32), nn.ReLU(), nn.Linear(32, 4))
    tracer = MemoryTracer(model)
    with tracer:
        out = model(torch.randn(8, 16)).sum()
        out.backward()
    stats = {s.name: s for s in tracer.stats.values() if s.calls > 0}
    assert len(stats) == 3
    lin0 = stats["0"]
    assert lin0.param_bytes == (16 * 32 + 32) * 4
    assert lin0.fwd_out_bytes == 8 * 32 * 4
    report = tracer.report()
    assert "peak allocator" in report
    # hooks must be removed after stop()
    n_hooks = len(model[0]._forward_hooks) + len(model[0]._forward_pre_hooks)
    assert n_hooks == 0
