"""Zero-bubble (ZB-H1) pipeline schedule
(reference: colossalai/pipeline/schedule/zero_bubble_pp.py — the ZB-H1
memory-neutral variant, rebuilt on our 1F1B skeleton).

Backward is split per Qi et al.: B (input grads — the inter-stage critical
path) runs inside the schedule's backward slots; W (weight grads — needed
only before the optimizer step) is deferred through ``WeightGradStore``.
Each cooldown B is followed by one microbatch's W so activation memory
stays 1F1B-shaped; the remainder drains after the last send, filling the
tail bubble that 1F1B spends idle. Requires the model's linears to be
converted with ``convert_to_zb_linears`` (done by the hybrid plugin for
``pp_style='zb'``); unconverted layers simply compute their weight grads
in B — correct, just less deferral.
"""

import torch.nn as nn

from ..weight_grad_store import WeightGradStore
from .one_f_one_b import OneForwardOneBackwardSchedule

__all__ = ["ZeroBubbleSchedule"]


class ZeroBubbleSchedule(OneForwardOneBackwardSchedule):
    def _backward_step(self, optimizer, input_obj, output_obj, output_grad):
        WeightGradStore.enabled = True
        try:
            ret = super()._backward_step(optimizer, input_obj, output_obj, output_grad)
        finally:
            WeightGradStore.enabled = False
        WeightGradStore.commit()
        return ret

    def _on_cooldown_backward(self):
        # one W-batch per cooldown B keeps memory 1F1B-shaped
        WeightGradStore.pop()

    def _finalize_backward(self):
        WeightGradStore.flush()
