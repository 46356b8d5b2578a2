"""Memory reporting (reference: colossalai/utils/memory.py)."""

import torch

__all__ = ["report_memory_usage"]


def report_memory_usage(message: str = "", logger=None, report_cpu: bool = False) -> str:
    parts = [message]
    if torch.cuda.is_available():
        alloc = torch.cuda.memory_allocated() / 2**30
        peak = torch.cuda.max_memory_allocated() / 2**30
        reserved = torch.cuda.memory_reserved() / 2**30
        parts.append(f"GPU alloc {alloc:.2f} GiB (peak {peak:.2f}, reserved {reserved:.2f}) of 288 GiB HBM3E")
    if report_cpu:
        import psutil

        vm = psutil.virtual_memory()
        parts.append(f"CPU used {vm.used / 2**30:.1f}/{vm.total / 2**30:.1f} GiB")
    msg = " | ".join(p for p in parts if p)
    if logger is not None:
        logger.info(msg)
    else:
        print(msg, flush=True)
    return msg
