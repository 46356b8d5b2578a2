"""ZB-V zero-bubble pipeline executor
(reference: colossalai/pipeline/schedule/zero_bubble_pp.py:40
ZeroBubbleVPipeScheduler — rebuilt on the precomputed node list of
v_schedule.build_zbv_schedule).

Every rank walks its static (F | B | W) node list. F/B payloads cross rank
boundaries over the cached-metadata P2P channels in the exact order the
schedule simulation fixed per directed pair; out-of-order needs are served
from a small per-channel buffer. The V turnaround edges (vstage pp-1 -> pp
on rank pp-1, and the loss at vstage 2·pp-1 on rank 0) are rank-local.
Weight gradients are queued per (vstage, micro) through WeightGradStore and
run in the schedule's W slots — the zero-bubble property.
"""

from typing import Callable, Dict, Iterable, Optional, Tuple

import torch
import torch.nn as nn

from ...interface import OptimizerWrapper
from ..p2p import PipelineP2PCommunication
from ..stage_manager import PipelineStageManager
from ..weight_grad_store import WeightGradStore
from .base import PipelineSchedule
from .one_f_one_b import _split_batch
from .v_schedule import build_zbv_schedule, owner_of_vstage

__all__ = ["ZeroBubbleVSchedule"]


class ZeroBubbleVSchedule(PipelineSchedule):
    def __init__(self, stage_manager: PipelineStageManager, num_microbatches: int,
                 microbatch_size: Optional[int] = None):
        super().__init__(stage_manager)
        self.num_microbatches = num_microbatches
        self.comm = PipelineP2PCommunication(stage_manager)
        self._plan_cache: Dict[Tuple[int, int], tuple] = {}

    def _plan(self, pp: int, M: int):
        key = (pp, M)
        if key not in self._plan_cache:
            self._plan_cache[key] = build_zbv_schedule(pp, M)
        return self._plan_cache[key]

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
        r = sm.stage
        M = self.num_microbatches
        V = 2 * pp
        nodes, chan_orders = self._plan(pp, M)
        my_nodes = nodes[r]

        batch = next(data_iter)
        if isinstance(batch, (list, tuple)):
            batch = batch[0]
        device = next(model.parameters()).device
        batch = {k: v.to(device) if isinstance(v, torch.Tensor) else v for k, v in batch.items()}
        micros = _split_batch(batch, M)

        accum_loss = torch.zeros(1, device=device) if (return_loss and r == 0) else None

        # per-incoming-channel planned orders + reorder buffers
        rank_of = sm._pp_ranks  # pipeline-local index -> global rank
        in_orders = {src: list(order) for (src, dst), order in chan_orders.items() if dst == r}
        in_bufs: Dict[int, Dict[tuple, torch.Tensor]] = {src: {} for src in in_orders}
        local_fwd: Dict[tuple, torch.Tensor] = {}   # V-turn hidden states
        local_bwd: Dict[tuple, torch.Tensor] = {}   # V-turn grads

        def fetch(src: int, key: tuple) -> torch.Tensor:
            buf = in_bufs[src]
            while key not in buf:
                nxt = in_orders[src].pop(0)
                buf[nxt] = self.comm.recv_forward(peer=rank_of[src])
            return buf.pop(key)

        inputs: Dict[tuple, Optional[torch.Tensor]] = {}
        outputs: Dict[tuple, torch.Tensor] = {}

        for node in my_nodes:
            v, m = node.vstage, node.micro
            chunk = 0 if v < pp else 1
            if node.type == "F":
                if v == 0:
                    inp = None
                else:
                    src = owner_of_vstage(v - 1, pp)
                    inp = local_fwd.pop(("F", v - 1, m)) if src == r else fetch(src, ("F", v - 1, m))
                    inp.requires_grad_(True)
                micro = micros[m]
                if inp is None:
                    out = model(**micro, pp_chunk=chunk)
                else:
                    out = model(**micro, hidden_states=inp, pp_chunk=chunk)
                if v == V - 1:
                    loss = criterion(out, micro) / M
                    if accum_loss is not None:
                        accum_loss.add_(loss.detach())
                    outputs[(v, m)] = loss
                else:
                    hid = out["hidden_states"]
                    outputs[(v, m)] = hid
                    dst = owner_of_vstage(v + 1, pp)
                    if dst == r:
                        # V-turn: next chunk's input must be a fresh leaf
                        local_fwd[("F", v, m)] = hid.detach()
                    else:
                        self.comm.send_forward(hid, peer=rank_of[dst])
                inputs[(v, m)] = inp
            elif node.type == "B":
                out_obj = outputs.pop((v, m))
                in_obj = inputs.pop((v, m))
                if v == V - 1:
                    grad = None
                else:
                    src = owner_of_vstage(v + 1, pp)
                    grad = local_bwd.pop(("B", v + 1, m)) if src == r else fetch(src, ("B", v + 1, m))
                WeightGradStore.enabled = True
                try:
                    if optimizer is None:
                        torch.autograd.backward(out_obj, grad)
                    elif grad is None:
                        optimizer.backward(out_obj)
                    else:
                        optimizer.backward_by_grad(out_obj, grad)
                finally:
                    WeightGradStore.enabled = False
                WeightGradStore.commit_key((v, m))
                if v > 0:
                    gin = in_obj.grad if in_obj is not None else None
                    dst = owner_of_vstage(v - 1, pp)
                    if dst == r:
                        local_bwd[("B", v, m)] = gin
                    else:
                        self.comm.send_backward(gin, peer=rank_of[dst])
            else:  # W
                WeightGradStore.pop_key((v, m))

        WeightGradStore.flush()
        self.comm.flush_sends()
        return {"loss": accum_loss.squeeze() if accum_loss is not None else None}
