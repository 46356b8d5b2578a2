"""Interleaved (virtual-stage) 1F1B schedule
(reference: colossalai/pipeline/schedule/interleaved_pp.py:26).

Each rank holds ``num_model_chunks`` model chunks; global virtual stage of
chunk c on rank r is ``c * pp_size + r``. Microbatches cycle through chunks
in groups of ``pp_size``, shrinking the pipeline bubble by ~1/num_chunks.
This implementation follows Megatron's all-warmup ordering: warmup fills
the pipeline with forwards, steady runs 1F1B, cooldown drains backwards.
"""

from typing import Callable, Iterable, List, Optional

import torch
import torch.nn as nn

from ...interface import OptimizerWrapper
from ..p2p import PipelineP2PCommunication
from ..stage_manager import PipelineStageManager
from .base import PipelineSchedule
from .one_f_one_b import _split_batch

__all__ = ["InterleavedSchedule"]


class InterleavedSchedule(PipelineSchedule):
    def __init__(
        self,
        stage_manager: PipelineStageManager,
        num_model_chunks: int,
        num_microbatches: int,
        microbatch_size: Optional[int] = None,
    ):
        super().__init__(stage_manager)
        self.num_model_chunks = num_model_chunks
        self.num_microbatches = num_microbatches
        assert num_microbatches % stage_manager.num_stages == 0, (
            "interleaved schedule requires num_microbatches % pp_size == 0"
        )
        self.comm = PipelineP2PCommunication(stage_manager)

    # virtual stage of (chunk) on this rank
    def _vstage(self, chunk: int) -> int:
        return chunk * self.stage_manager.num_stages + self.stage_manager.stage

    def _chunk_of_step(self, step: int) -> int:
        """Which model chunk handles the ``step``-th local forward."""
        pp = self.stage_manager.num_stages
        return (step // pp) % self.num_model_chunks

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
        pp = sm.num_stages
        V = self.num_model_chunks
        M = self.num_microbatches

        batch = next(data_iter)
        if isinstance(batch, (list, tuple)):
            batch = batch[0]
        device = next(model.parameters()).device
        batch = {k: v.to(device) if isinstance(v, torch.Tensor) else v for k, v in batch.items()}
        micros = _split_batch(batch, M)

        total_steps = M * V
        # Megatron warmup count for interleaved 1F1B
        if M == pp:
            num_warmup = total_steps
        else:
            num_warmup = min((pp - sm.stage - 1) * 2 + (V - 1) * pp, total_steps)
        num_steady = total_steps - num_warmup

        accum_loss = torch.zeros(1, device=device) if (return_loss and sm.stage == pp - 1) else None

        input_objs: List[List[Optional[torch.Tensor]]] = [[] for _ in range(V)]
        output_objs: List[List[torch.Tensor]] = [[] for _ in range(V)]
        fwd_counts = [0] * V  # microbatch index per chunk (forward)
        bwd_counts = [0] * V

        def vstage_is_first(chunk):  # first virtual stage overall
            return sm.stage == 0 and chunk == 0

        def vstage_is_last(chunk):
            return sm.stage == pp - 1 and chunk == V - 1

        def forward_chunk(chunk, input_obj):
            # with the groups-of-pp chunk cycling, the k-th forward of every
            # chunk handles global microbatch k
            mb = fwd_counts[chunk]
            fwd_counts[chunk] += 1
            micro = micros[mb]
            if vstage_is_first(chunk):
                out = model(**micro, pp_chunk=chunk)
            else:
                input_obj.requires_grad_(True)
                out = model(**micro, hidden_states=input_obj, pp_chunk=chunk)
            if vstage_is_last(chunk):
                loss = criterion(out, micro) / M
                if accum_loss is not None:
                    accum_loss.add_(loss.detach())
                return loss
            return out["hidden_states"]

        def backward_chunk(chunk, grad):
            in_obj = input_objs[chunk].pop(0)
            out_obj = output_objs[chunk].pop(0)
            bwd_counts[chunk] += 1
            if optimizer is None:
                return None
            if grad is None:
                optimizer.backward(out_obj)
            else:
                optimizer.backward_by_grad(out_obj, grad)
            return in_obj.grad if in_obj is not None else None

        # peers: forward comes from prev rank (chunk same) unless rank 0, where
        # it comes from chunk-1's last rank (prev rank with wraparound).
        def recv_fwd(chunk):
            if vstage_is_first(chunk):
                return None
            return self.comm.recv_forward(self.stage_manager.get_prev_rank())

        def send_fwd(chunk, obj):
            if not vstage_is_last(chunk):
                self.comm.send_forward(obj, self.stage_manager.get_next_rank())

        def recv_bwd(chunk):
            if vstage_is_last(chunk):
                return None
            return self.comm.recv_backward(self.stage_manager.get_next_rank())

        def send_bwd(chunk, grad):
            if not vstage_is_first(chunk) and grad is not None:
                self.comm.send_backward(grad, self.stage_manager.get_prev_rank())

        # ---- warmup forwards
        fstep = 0
        bstep = 0
        for _ in range(num_warmup):
            chunk = self._chunk_of_step(fstep)
            inp = recv_fwd(chunk)
            out = forward_chunk(chunk, inp)
            input_objs[chunk].append(inp)
            output_objs[chunk].append(out)
            send_fwd(chunk, out)
            fstep += 1

        # ---- steady 1F1B
        for _ in range(num_steady):
            chunk = self._chunk_of_step(fstep)
            inp = recv_fwd(chunk)
            out = forward_chunk(chunk, inp)
            input_objs[chunk].append(inp)
            output_objs[chunk].append(out)
            send_fwd(chunk, out)
            fstep += 1

            bchunk = V - 1 - self._chunk_of_step(bstep)
            grad = recv_bwd(bchunk)
            gin = backward_chunk(bchunk, grad)
            send_bwd(bchunk, gin)
            bstep += 1

        # ---- cooldown backwards
        for _ in range(total_steps - bstep):
            bchunk = V - 1 - self._chunk_of_step(bstep)
            grad = recv_bwd(bchunk)
            gin = backward_chunk(bchunk, grad)
            send_bwd(bchunk, gin)
            bstep += 1

        self.comm.flush_sends()
        return {"loss": accum_loss.squeeze() if accum_loss is not None else None}
