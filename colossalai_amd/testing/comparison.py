"""Tensor / state-dict comparison helpers (reference: colossalai/testing/comparison.py)."""

from typing import Any, OrderedDict

import torch
from torch.testing import assert_close

__all__ = ["assert_close", "assert_close_loose", "assert_equal", "check_state_dict_equal"]


def assert_close_loose(a: torch.Tensor, b: torch.Tensor, rtol: float = 1e-3, atol: float = 1e-3, msg=None):
    assert_close(a, b, rtol=rtol, atol=atol, msg=msg, check_dtype=False, check_device=False)


def assert_equal(a: torch.Tensor, b: torch.Tensor):
    assert torch.all(a == b), f"expected a and b to be equal but they are not, {a} vs {b}"


def _to_cpu(x: Any) -> Any:
    return x.to("cpu") if isinstance(x, torch.Tensor) else x


def check_state_dict_equal(d1: "OrderedDict[str, Any]", d2: "OrderedDict[str, Any]", ignore_device: bool = True):
    assert set(d1.keys()) == set(d2.keys()), f"state dict keys differ: {set(d1.keys()) ^ set(d2.keys())}"
    for k, v1 in d1.items():
        v2 = d2[k]
        if isinstance(v1, dict):
            check_state_dict_equal(v1, v2, ignore_device)
        elif isinstance(v1, (list, tuple)):
            for e1, e2 in zip(v1, v2):
                if isinstance(e1, torch.Tensor):
                    e1, e2 = (_to_cpu(e1), _to_cpu(e2)) if ignore_device else (e1, e2)
                    assert_close(e1, e2, check_dtype=False)
                else:
                    assert e1 == e2, f"{k}: {e1} != {e2}"
        elif isinstance(v1, torch.Tensor):
            a, b = (_to_cpu(v1), _to_cpu(v2)) if ignore_device else (v1, v2)
            assert_close(a, b, check_dtype=False, msg=lambda m: f"key {k}: {m}")
        else:
            assert v1 == v2, f"{k}: {v1} != {v2}"
