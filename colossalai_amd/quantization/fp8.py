"""FP8 communication compression (reference: colossalai/quantization/fp8.py).

gfx950 uses OCP fp8 formats (e4m3fn / e5m2 — NOT the MI300X fnuz variants);
torch's float8_e4m3fn/e5m2 dtypes map onto them directly. Collectives move
the fp8 payload as uint8 views (RCCL/gloo agnostic) plus a per-rank fp32
amax scale, halving (bf16) the bytes on the 7-link xGMI ring at the cost of
one cast each side.
"""

from typing import List, Tuple

import torch
import torch.distributed as dist

__all__ = [
    "cast_to_fp8",
    "cast_from_fp8",
    "all_reduce_fp8",
    "all_gather_fp8",
    "reduce_scatter_fp8",
    "all_to_all_single_fp8",
]

_FP8_DTYPES = {"e4m3": torch.float8_e4m3fn, "e5m2": torch.float8_e5m2}
_FP8_MAX = {"e4m3": 448.0, "e5m2": 57344.0}

def _a2a_list(outputs: List[torch.Tensor], inputs: List[torch.Tensor], group) -> None:
    """dist.all_to_all with a gloo (CPU test) fallback via all_gather."""
    if dist.get_backend(group) != "gloo":
        dist.all_to_all(outputs, inputs, group=group)
        return
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    # emulate: for each destination d, gather every rank's chunk-for-d; keep ours
    for d in range(world):
        gathered = [torch.empty_like(inputs[d]) for _ in range(world)]
        dist.all_gather(gathered, inputs[d].contiguous(), group=group)
        if d == rank:
            for src in range(world):
                outputs[src].copy_(gathered[src])



def cast_to_fp8(x: torch.Tensor, fp8_format: str = "e4m3") -> Tuple[torch.Tensor, torch.Tensor]:
    """-> (fp8 tensor, fp32 scale) with per-tensor amax scaling."""
    fmax = _FP8_MAX[fp8_format]
    amax = x.abs().max().float().clamp(min=1e-12)
    scale = fmax / amax
    fp8 = (x.float() * scale).clamp(-fmax, fmax).to(_FP8_DTYPES[fp8_format])
    return fp8, scale


def cast_from_fp8(fp8: torch.Tensor, scale: torch.Tensor, dtype: torch.dtype) -> torch.Tensor:
    return (fp8.float() / scale).to(dtype)


def all_reduce_fp8(x: torch.Tensor, fp8_format: str = "e4m3", group=None, async_op: bool = False):
    """In-place all-reduce (sum) with fp8 wire format: reduce-scatter the fp8
    shards with fp32 local accumulation, then all-gather fp8 — the same
    two-phase shape as the reference (fp8.py:187)."""
    world = dist.get_world_size(group)
    if world == 1:
        return None
    n = x.numel()
    pad = (n + world - 1) // world * world
    flat = torch.zeros(pad, dtype=x.dtype, device=x.device)
    flat[:n] = x.reshape(-1)
    rs = reduce_scatter_fp8(flat, fp8_format=fp8_format, group=group)
    gathered = all_gather_fp8(rs, fp8_format=fp8_format, group=group)
    x.reshape(-1).copy_(gathered[:n].to(x.dtype))
    return None


def reduce_scatter_fp8(flat: torch.Tensor, fp8_format: str = "e4m3", group=None) -> torch.Tensor:
    """flat [world*chunk] -> this rank's reduced chunk (fp32 accumulate)."""
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    chunk = flat.numel() // world
    inp = flat.view(world, chunk)
    fp8s = []
    scales = []
    for r in range(world):
        f, s = cast_to_fp8(inp[r], fp8_format)
        fp8s.append(f.view(torch.uint8))
        scales.append(s.reshape(1))
    recv = [torch.empty(chunk, dtype=torch.uint8, device=flat.device) for _ in range(world)]
    _a2a_list(recv, fp8s, group)
    recv_scales = [torch.empty(1, dtype=torch.float32, device=flat.device) for _ in range(world)]
    _a2a_list(recv_scales, scales, group)
    acc = torch.zeros(chunk, dtype=torch.float32, device=flat.device)
    for r in range(world):
        acc += recv[r].view(_FP8_DTYPES[fp8_format]).float() / recv_scales[r]
    return acc.to(flat.dtype)


def all_gather_fp8(shard: torch.Tensor, fp8_format: str = "e4m3", group=None) -> torch.Tensor:
    world = dist.get_world_size(group)
    fp8, scale = cast_to_fp8(shard, fp8_format)
    payload = fp8.view(torch.uint8).contiguous()
    out = torch.empty(world * payload.numel(), dtype=torch.uint8, device=shard.device)
    dist.all_gather_into_tensor(out, payload, group=group)
    scales = torch.empty(world, dtype=torch.float32, device=shard.device)
    dist.all_gather_into_tensor(scales, scale.reshape(1).float(), group=group)
    parts = out.view(world, -1)
    res = torch.empty(world, shard.numel(), dtype=shard.dtype, device=shard.device)
    for r in range(world):
        res[r] = (parts[r].view(_FP8_DTYPES[fp8_format]).float() / scales[r]).to(shard.dtype)
    return res.reshape(-1)


def all_to_all_single_fp8(x: torch.Tensor, fp8_format: str = "e4m3", group=None) -> torch.Tensor:
    """all_to_all_single with fp8 payload; x's dim0 divisible by world."""
    world = dist.get_world_size(group)
    chunks = x.chunk(world, dim=0)
    fp8s, scales = [], []
    for c in chunks:
        f, s = cast_to_fp8(c.contiguous(), fp8_format)
        fp8s.append(f.view(torch.uint8).reshape(-1))
        scales.append(s.reshape(1))
    recv = [torch.empty_like(fp8s[0]) for _ in range(world)]
    _a2a_list(recv, fp8s, group)
    recv_scales = [torch.empty(1, dtype=torch.float32, device=x.device) for _ in range(world)]
    _a2a_list(recv_scales, scales, group)
    outs = []
    for r in range(world):
        outs.append((recv[r].view(_FP8_DTYPES[fp8_format]).float() / recv_scales[r]).to(x.dtype).view(chunks[0].shape))
    return torch.cat(outs, dim=0)
