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


def test_watchdog_fires_and_disarms():
    import time

    from colossalai_amd.utils import Watchdog

    hits = []
    wd = Watchdog(timeout_s=0.1, on_timeout=lambda: hits.append(1))
    with wd.step():
        time.sleep(0.25)
    assert wd.fired == 1 and hits == [1]
    # disarmed after the step: no further fires
    with wd.step():
        pass
    time.sleep(0.2)
    assert wd.fired == 1


def test_cast_to_distributed():
    import torch

    from colossalai_amd.nn import DistributedLamb, Lamb
    from colossalai_amd.nn.optimizer import cast_to_distributed

    p = torch.nn.Parameter(torch.randn(4))
    opt = Lamb([p], lr=1e-2)
    d = cast_to_distributed(opt)
    assert isinstance(d, DistributedLamb)
    adam = torch.optim.Adam([p])
    assert cast_to_distributed(adam) is adam
