"""FP8 collectives vs fp32 reference (CPU/gloo; fp8 payloads as uint8 views)."""

import pytest
import torch
import torch.distributed as dist

import colossalai_amd
from colossalai_amd.quantization import (
    all_gather_fp8,
    all_reduce_fp8,
    all_to_all_single_fp8,
    cast_from_fp8,
    cast_to_fp8,
    reduce_scatter_fp8,
)
from colossalai_amd.testing import rerun_if_address_is_in_use, spawn


def test_cast_roundtrip():
    torch.manual_seed(0)
    x = torch.randn(1000) * 5
    for fmt in ("e4m3", "e5m2"):
        f, s = cast_to_fp8(x, fmt)
        y = cast_from_fp8(f, s, torch.float32)
        rel = ((x - y).abs() / x.abs().clamp(min=1e-3)).median()
        assert rel < 0.08, f"{fmt}: median rel err {rel}"


def _run(rank, world_size, port):
    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    g = dist.group.WORLD
    torch.manual_seed(rank)
    x = torch.randn(64)
    ref = x.clone()
    dist.all_reduce(ref, group=g)
    y = x.clone()
    all_reduce_fp8(y, group=g)
    assert ((y - ref).abs() / ref.abs().clamp(min=1e-2)).median() < 0.1

    shard = torch.randn(32)
    out = all_gather_fp8(shard, group=g)
    refs = [torch.empty_like(shard) for _ in range(world_size)]
    dist.all_gather(refs, shard, group=g)
    ref_full = torch.cat(refs)
    assert ((out - ref_full).abs() / ref_full.abs().clamp(min=1e-2)).median() < 0.1

    a2a_in = torch.randn(world_size * 4, 8)
    out2 = all_to_all_single_fp8(a2a_in, group=g)
    assert out2.shape == a2a_in.shape
    dist.destroy_process_group()


@rerun_if_address_is_in_use()
def test_fp8_collectives():
    spawn(_run, 2)
