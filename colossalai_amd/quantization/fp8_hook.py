"""FP8 gradient-communication hooks (reference:
colossalai/quantization/fp8_hook.py + utils.py's fp8_compress_ddp_grad_comm_hook).

DDP comm hook: each 32–64 MB gradient bucket is cast to OCP e4m3 with a
per-bucket amax scale and summed via the all_gather-based fp8 all-reduce in
``quantization.fp8`` — halving (vs bf16) the bytes each of the 7 xGMI links
carries during backward overlap. Accumulation happens in fp32 after
decode, so only the wire format is 8-bit.
"""

from typing import Any

import torch
import torch.distributed as dist

from .fp8 import all_reduce_fp8

__all__ = ["fp8_compress_ddp_grad_comm_hook"]


def fp8_compress_ddp_grad_comm_hook(
    process_group: Any, bucket: dist.GradBucket
) -> torch.futures.Future[torch.Tensor]:
    """torch DDP communication hook: fp8 all-reduce (mean) of the bucket."""
    group = process_group if isinstance(process_group, dist.ProcessGroup) or process_group is None else None
    world = dist.get_world_size(group) if dist.is_initialized() else 1
    buf = bucket.buffer()
    all_reduce_fp8(buf, fp8_format="e4m3", group=group)
    buf.div_(world)
    fut: torch.futures.Future = torch.futures.Future()
    fut.set_result(buf)
    return fut
