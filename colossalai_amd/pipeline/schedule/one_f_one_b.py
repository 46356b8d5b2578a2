"""1F1B pipeline schedule (reference: colossalai/pipeline/schedule/one_f_one_b.py:28).

Classic warmup → steady 1F1B → cooldown. Stage IO protocol: one tensor per
boundary (the model's residual stream); models boosted for pipeline accept
``hidden_states=`` and return ``{"hidden_states": t}`` on non-last stages.
"""

from typing import Any, Callable, Dict, Iterable, List, Optional

import torch
import torch.nn as nn

from ...interface import OptimizerWrapper
from ..p2p import PipelineP2PCommunication
from ..stage_manager import PipelineStageManager
from .base import PipelineSchedule

__all__ = ["OneForwardOneBackwardSchedule"]


def _split_batch(batch: Dict[str, torch.Tensor], num_microbatches: int) -> List[Dict[str, torch.Tensor]]:
    keys = list(batch.keys())
    first = batch[keys[0]]
    assert first.shape[0] % num_microbatches == 0, (
        f"batch size {first.shape[0]} must divide num_microbatches {num_microbatches}"
    )
    chunks = {k: v.chunk(num_microbatches, dim=0) for k, v in batch.items()}
    return [{k: chunks[k][i] for k in keys} for i in range(num_microbatches)]


class OneForwardOneBackwardSchedule(PipelineSchedule):
    def __init__(
        self,
        stage_manager: PipelineStageManager,
        num_microbatches: Optional[int] = None,
        microbatch_size: Optional[int] = None,
    ):
        super().__init__(stage_manager)
        assert num_microbatches is not None or microbatch_size is not None, (
            "either num_microbatches or microbatch_size must be set"
        )
        self.num_microbatches = num_microbatches
        self.microbatch_size = microbatch_size
        self.comm = PipelineP2PCommunication(stage_manager)

    def _num_microbatches(self, batch_size: int) -> int:
        if self.num_microbatches is not None:
            return self.num_microbatches
        assert batch_size % self.microbatch_size == 0
        return batch_size // self.microbatch_size

    def forward_backward_step(
        self,
        model: nn.Module,
        data_iter: Iterable,
        criterion: Callable,
        optimizer: Optional[OptimizerWrapper] = None,
        return_loss: bool = False,
        return_outputs: bool = False,
    ) -> dict:
        sm = self.stage_manager
        batch = next(data_iter)
        if isinstance(batch, (list, tuple)):
            batch = batch[0]
        assert isinstance(batch, dict), "pipeline schedule expects dict batches"
        device = next(model.parameters()).device
        batch = {k: v.to(device) if isinstance(v, torch.Tensor) else v for k, v in batch.items()}
        bs = next(iter(batch.values())).shape[0]
        M = self._num_microbatches(bs)
        micros = _split_batch(batch, M)

        is_first = sm.is_first_stage()
        is_last = sm.is_last_stage()
        forward_only = optimizer is None

        num_warmup = min(sm.num_stages - sm.stage - 1, M)
        num_steady = M - num_warmup

        input_objs: List[Optional[torch.Tensor]] = []
        output_objs: List[torch.Tensor] = []
        accum_loss = torch.zeros(1, device=device) if (return_loss and is_last) else None
        outputs = [] if (return_outputs and is_last) else None
        micro_idx = 0

        def forward_step(input_obj: Optional[torch.Tensor]):
            nonlocal micro_idx
            micro = micros[micro_idx]
            micro_idx += 1
            if is_first:
                out = model(**micro)
            else:
                input_obj.requires_grad_(True)
                out = model(**micro, hidden_states=input_obj)
            if is_last:
                loss = criterion(out, micro) / M
                if accum_loss is not None:
                    accum_loss.add_(loss.detach())
                if outputs is not None:
                    outputs.append({k: v.detach() if isinstance(v, torch.Tensor) else v for k, v in out.items()})
                return loss
            return out["hidden_states"]

        def backward_step(input_obj, output_obj, output_grad):
            if forward_only:
                return None
            return self._backward_step(optimizer, input_obj, output_obj, output_grad)

        # ---- warmup: forwards only
        for _ in range(num_warmup):
            input_obj = None if is_first else self.comm.recv_forward()
            output_obj = forward_step(input_obj)
            if not is_last:
                self.comm.send_forward(output_obj)
            input_objs.append(input_obj)
            output_objs.append(output_obj)

        # ---- steady 1F1B
        input_obj = None
        if num_steady > 0:
            input_obj = None if is_first else self.comm.recv_forward()
        for i in range(num_steady):
            output_obj = forward_step(input_obj)
            if is_last:
                output_grad = None
            else:
                output_grad = self.comm.send_forward_recv_backward(output_obj)
            input_objs.append(input_obj)
            output_objs.append(output_obj)

            # FIFO: the grad received in steady iteration i belongs to the
            # oldest outstanding forward (microbatch i) — standard 1F1B.
            in_obj = input_objs.pop(0)
            out_obj = output_objs.pop(0)
            input_grad = backward_step(in_obj, out_obj, output_grad)
            last_iteration = i == num_steady - 1
            if is_first:
                if not last_iteration:
                    input_obj = None
            else:
                if last_iteration:
                    if input_grad is not None:
                        self.comm.send_backward(input_grad)
                else:
                    input_obj = self.comm.send_backward_recv_forward(input_grad)

        # ---- cooldown: drain remaining backwards
        for _ in range(num_warmup):
            in_obj = input_objs.pop(0)
            out_obj = output_objs.pop(0)
            grad = None if is_last else self.comm.recv_backward()
            input_grad = backward_step(in_obj, out_obj, grad)
            if not is_first and input_grad is not None:
                self.comm.send_backward(input_grad)
            self._on_cooldown_backward()

        self.comm.flush_sends()
        if not forward_only:
            self._finalize_backward()
        result = {"loss": accum_loss.squeeze() if accum_loss is not None else None}
        if outputs is not None:
            result["outputs"] = outputs
        return result

    # ---- hooks for schedule variants (zero-bubble overrides these) -------
    def _backward_step(self, optimizer, input_obj, output_obj, output_grad):
        if output_grad is None:  # last stage: output_obj is the loss
            optimizer.backward(output_obj, retain_graph=False)
        else:
            optimizer.backward_by_grad(output_obj, output_grad, retain_graph=False)
        return input_obj.grad if input_obj is not None else None

    def _on_cooldown_backward(self):
        pass

    def _finalize_backward(self):
        pass
