"""HybridParallelPlugin — dp × pp × tp × sp on one mesh
(reference: colossalai/booster/plugin/hybrid_parallel_plugin.py:928).

Mesh axes (dp, pp, tp, sp). Shardformer applies TP/SP; pipeline stages own a
contiguous layer range and exchange the residual stream via RCCL P2P; the dp
dimension runs ZeRO-0/1/2 (ZeRO via LowLevelZeroOptimizer with the dp group).
"""

from contextlib import contextmanager, nullcontext
from typing import Any, Callable, Iterator, List, Optional, Tuple

import torch
import torch.distributed as dist
import torch.nn as nn
from torch.optim import Optimizer
from torch.optim.lr_scheduler import _LRScheduler as LRScheduler
from torch.utils.data import DataLoader

from ...checkpoint_io import CheckpointIO
from ...cluster import ProcessGroupMesh
from ...interface import ModelWrapper, OptimizerWrapper
from ...amp import MixedPrecisionOptimizer
from ...pipeline import PipelineStageManager
from ...pipeline.schedule import OneForwardOneBackwardSchedule
from ...shardformer import ShardConfig, ShardFormer
from ...zero import LowLevelZeroOptimizer
from .plugin_base import Plugin
from .torch_ddp_plugin import TorchDDPCheckpointIO

__all__ = ["HybridParallelPlugin", "HybridParallelModule"]

DP_AXIS, PP_AXIS, TP_AXIS, SP_AXIS = 0, 1, 2, 3

_PRECISION_DTYPE = {"fp16": torch.float16, "bf16": torch.bfloat16, "fp32": torch.float32}


class HybridParallelModule(ModelWrapper):
    def __init__(self, module: nn.Module, dtype: torch.dtype, dp_group, tp_group, sp_group,
                 embed_group=None, tied_param: Optional[torch.Tensor] = None):
        super().__init__(module)
        self.dtype = dtype
        self.dp_group = dp_group
        self.tp_group = tp_group
        self.sp_group = sp_group
        # tied embed/lm_head under pp>1: first and last stage each hold a copy;
        # grads are summed over the 2-rank embed group every step
        # (reference: hybrid_parallel_plugin.py:131 sync_shared_params)
        self.embed_group = embed_group
        self.tied_param = tied_param
        self.require_grad_sync = True

    def forward(self, *args, **kwargs):
        args = [a.to(self.dtype) if isinstance(a, torch.Tensor) and a.is_floating_point() else a for a in args]
        kwargs = {
            k: (v.to(self.dtype) if isinstance(v, torch.Tensor) and v.is_floating_point() else v)
            for k, v in kwargs.items()
        }
        return self.module(*args, **kwargs)

    @contextmanager
    def no_sync(self):
        old = self.require_grad_sync
        self.require_grad_sync = False
        try:
            yield
        finally:
            self.require_grad_sync = old

    def sync_shared_params(self):
        """Sum grads of the tied embed/lm_head weight across the pipeline's
        first and last stage so both copies take the same update."""
        if self.tied_param is None or self.embed_group is None:
            return
        if dist.get_world_size(self.embed_group) == 1:
            return
        if self.tied_param.grad is not None:
            dist.all_reduce(self.tied_param.grad, group=self.embed_group)

    def sync_partial_sp_grads(self):
        """split_gather SP: norm-weight grads are partial over seq shards —
        sum them over the tp/sp group so every rank holds the full grad
        (reference: SeqParallelUtils.allreduce_partial_data_grad)."""
        group = self.tp_group
        if group is None or dist.get_world_size(group) == 1:
            return
        for p in self.module.parameters():
            if getattr(p, "_sp_partial_grad", False) and p.grad is not None:
                dist.all_reduce(p.grad, group=group)

    def sync_dp_grads(self):
        """ZeRO-0 path: average grads over the dp group (bucket-coalesced)."""
        if self.dp_group is None or dist.get_world_size(self.dp_group) == 1:
            return
        world = dist.get_world_size(self.dp_group)
        bucket: List[torch.Tensor] = []
        size = 0
        BUCKET_BYTES = 128 * 1024 * 1024  # xGMI-sized
        for p in self.module.parameters():
            if p.grad is None:
                continue
            bucket.append(p.grad)
            size += p.grad.numel() * p.grad.element_size()
            if size >= BUCKET_BYTES:
                self._flush(bucket, world)
                bucket, size = [], 0
        if bucket:
            self._flush(bucket, world)

    def _flush(self, grads: List[torch.Tensor], world: int):
        flat = torch.cat([g.reshape(-1) for g in grads])
        dist.all_reduce(flat, group=self.dp_group)
        flat /= world
        off = 0
        for g in grads:
            g.copy_(flat[off : off + g.numel()].view_as(g))
            off += g.numel()


class HybridParallelNaiveOptimizer(MixedPrecisionOptimizer):
    """ZeRO-0 optimizer: bf16/fp16 masters + manual dp grad sync in backward."""

    def __init__(self, optim: Optimizer, model: HybridParallelModule, precision: str, max_norm: float = 0.0, **kw):
        self.model_wrapper = model
        super().__init__(optim, precision=precision, max_norm=max_norm, **kw)

    def backward(self, loss, inputs=None, retain_graph=False, **kwargs):
        super().backward(loss, inputs=inputs, retain_graph=retain_graph, **kwargs)
        if self.model_wrapper.require_grad_sync:
            self.model_wrapper.sync_partial_sp_grads()
            self.model_wrapper.sync_dp_grads()

    def backward_by_grad(self, tensor, grad, inputs=None, retain_graph=False):
        super().backward_by_grad(tensor, grad, inputs=inputs, retain_graph=retain_graph)
        if self.model_wrapper.require_grad_sync:
            self.model_wrapper.sync_partial_sp_grads()
            self.model_wrapper.sync_dp_grads()


class HybridParallelFP32Optimizer(OptimizerWrapper):
    """ZeRO-0 fp32: plain optimizer + dp×sp grad averaging after backward."""

    def __init__(self, optim: Optimizer, model: HybridParallelModule, max_norm: float = 0.0,
                 tp_process_group=None, pp_process_group=None):
        super().__init__(optim)
        self.model_wrapper = model
        self.max_norm = max_norm
        self.tp_pg = tp_process_group
        self.pp_pg = pp_process_group

    def step(self, *args, **kwargs):
        if self.max_norm > 0.0:
            from ...amp.mixed_precision_optimizer import compute_global_grad_norm

            pairs = [(p, p.grad) for g in self.optim.param_groups for p in g["params"] if p.grad is not None]
            total_norm = compute_global_grad_norm(pairs, self.tp_pg, self.pp_pg)
            if total_norm > self.max_norm:
                scale = self.max_norm / total_norm
                for _, g in pairs:
                    g.mul_(scale)
        return self.optim.step(*args, **kwargs)

    # ------------------------------------------------------------ checkpoint
    def get_param_states(self, names):
        out = {}
        for gi, group in enumerate(self.optim.param_groups):
            for p in group["params"]:
                name = names.get(id(p))
                if name is None:
                    continue
                st = {"_group": gi}
                for k, v in self.optim.state.get(p, {}).items():
                    st[k] = v.cpu().clone() if isinstance(v, torch.Tensor) else v
                out[name] = st
        return out

    def set_param_states(self, states, names):
        for group in self.optim.param_groups:
            for p in group["params"]:
                name = names.get(id(p))
                if name is None or name not in states:
                    continue
                st = dict(states[name])
                st.pop("_group", None)
                st.pop("master", None)
                inner = {}
                for k, v in st.items():
                    if isinstance(v, torch.Tensor) and v.shape == p.shape:
                        inner[k] = v.to(p.device)
                    elif k == "step" and not isinstance(v, torch.Tensor):
                        inner[k] = torch.tensor(float(v))
                    else:
                        inner[k] = v
                self.optim.state[p] = inner

    def backward(self, loss, inputs=None, retain_graph=False, **kwargs):
        loss.backward(inputs=inputs, retain_graph=retain_graph, **kwargs)
        if self.model_wrapper.require_grad_sync:
            self.model_wrapper.sync_partial_sp_grads()
            self.model_wrapper.sync_dp_grads()

    def backward_by_grad(self, tensor, grad, inputs=None, retain_graph=False):
        torch.autograd.backward(tensor, grad, inputs=inputs, retain_graph=retain_graph)
        if self.model_wrapper.require_grad_sync:
            self.model_wrapper.sync_partial_sp_grads()
            self.model_wrapper.sync_dp_grads()


class HybridParallelPlugin(Plugin):
    def __init__(
        self,
        tp_size: int = 1,
        pp_size: int = 1,
        sp_size: Optional[int] = None,
        precision: str = "bf16",
        zero_stage: int = 0,
        enable_flash_attention: bool = True,
        enable_fused_normalization: bool = True,
        enable_sequence_parallelism: bool = False,
        sequence_parallelism_mode: Optional[str] = None,
        num_microbatches: Optional[int] = None,
        microbatch_size: Optional[int] = None,
        initial_scale: float = 2**16,
        min_scale: float = 1,
        growth_factor: float = 2,
        backoff_factor: float = 0.5,
        growth_interval: int = 1000,
        hysteresis: int = 2,
        max_scale: float = 2**32,
        max_norm: float = 0.0,
        zero_bucket_size_in_m: int = 32,
        overlap_communication: bool = True,
        parallel_output: bool = True,
        pp_style: str = "1f1b",
        num_model_chunks: int = 1,
        sp_zigzag: bool = False,
        **kwargs,
    ):
        assert dist.is_initialized(), "launch colossalai_amd before creating HybridParallelPlugin"
        assert zero_stage in (0, 1, 2)
        assert pp_style in ("1f1b", "interleaved", "zb", "zbv"), f"unsupported pp_style {pp_style}"
        if pp_style == "interleaved":
            assert num_model_chunks > 1, "interleaved pipeline needs num_model_chunks > 1"
        if pp_style in ("zb", "zbv"):
            assert zero_stage == 0, (
                "zero-bubble defers weight grads past AccumulateGrad, which ZeRO's "
                "bucket hooks rely on — use pp_style='zb'/'zbv' with zero_stage=0"
            )
        world = dist.get_world_size()
        if sp_size is None:
            sp_size = 1
        assert world % (tp_size * pp_size * sp_size) == 0, (
            f"world {world} not divisible by tp{tp_size}*pp{pp_size}*sp{sp_size}"
        )
        dp_size = world // (tp_size * pp_size * sp_size)
        self.tp_size, self.pp_size, self.dp_size, self.sp_size = tp_size, pp_size, dp_size, sp_size
        self.precision = precision
        self.zero_stage = zero_stage
        self.max_norm = max_norm
        self.amp_kwargs = dict(
            initial_scale=initial_scale, min_scale=min_scale, growth_factor=growth_factor,
            backoff_factor=backoff_factor, growth_interval=growth_interval, hysteresis=hysteresis,
            max_scale=max_scale,
        )
        self.zero_kwargs = dict(
            reduce_bucket_size=zero_bucket_size_in_m * 1024 * 1024,
            overlap_communication=overlap_communication,
            partition_grad=(zero_stage == 2),
            clip_grad_norm=max_norm,
        )

        self.pg_mesh = ProcessGroupMesh(dp_size, pp_size, tp_size, sp_size)
        self.dp_group = self.pg_mesh.get_group_along_axis(DP_AXIS)
        self.pp_group = self.pg_mesh.get_group_along_axis(PP_AXIS)
        self.tp_group = self.pg_mesh.get_group_along_axis(TP_AXIS)
        self.sp_group = self.pg_mesh.get_group_along_axis(SP_AXIS)
        # grads of dp×sp-replicated params sync over the flattened group
        self.dp_sp_group = self.pg_mesh.get_group_along_axis([DP_AXIS, SP_AXIS])
        # tied embed/lm_head sync group: first + last pipeline stage
        # (None on middle stages; the groups are still created collectively)
        self.embed_group = None
        if pp_size > 1:
            self.embed_group = self.pg_mesh.get_group_along_axis(PP_AXIS, indices_at_axis=[0, pp_size - 1])

        self.stage_manager = None
        self.scheduler = None
        self.pp_style = pp_style
        self.num_model_chunks = num_model_chunks
        if pp_size > 1:
            assert num_microbatches is not None or microbatch_size is not None, (
                "pipeline parallelism requires num_microbatches or microbatch_size"
            )
            if pp_style == "interleaved":
                from ...pipeline.schedule.interleaved_pp import InterleavedSchedule

                self.stage_manager = PipelineStageManager(
                    self.pg_mesh, PP_AXIS, enable_interleave=True, num_model_chunks=num_model_chunks
                )
                self.scheduler = InterleavedSchedule(
                    self.stage_manager, num_model_chunks=num_model_chunks, num_microbatches=num_microbatches
                )
            elif pp_style == "zb":
                from ...pipeline.schedule.zero_bubble import ZeroBubbleSchedule

                self.stage_manager = PipelineStageManager(self.pg_mesh, PP_AXIS)
                self.scheduler = ZeroBubbleSchedule(
                    self.stage_manager, num_microbatches=num_microbatches, microbatch_size=microbatch_size
                )
            elif pp_style == "zbv":
                from ...pipeline.schedule.zbv_schedule import ZeroBubbleVSchedule

                self.stage_manager = PipelineStageManager(self.pg_mesh, PP_AXIS)
                self.scheduler = ZeroBubbleVSchedule(
                    self.stage_manager, num_microbatches=num_microbatches, microbatch_size=microbatch_size
                )
                self.num_model_chunks = 2
            else:
                self.stage_manager = PipelineStageManager(self.pg_mesh, PP_AXIS)
                self.scheduler = OneForwardOneBackwardSchedule(
                    self.stage_manager, num_microbatches=num_microbatches, microbatch_size=microbatch_size
                )

        self.shard_config = ShardConfig(
            tensor_parallel_process_group=self.tp_group if tp_size > 1 else None,
            sequence_parallel_process_group=self.sp_group if sp_size > 1 else None,
            pipeline_stage_manager=self.stage_manager,
            enable_tensor_parallelism=tp_size > 1,
            enable_sequence_parallelism=enable_sequence_parallelism,
            sequence_parallelism_mode=sequence_parallelism_mode,
            sp_zigzag=sp_zigzag,
            enable_flash_attention=enable_flash_attention,
            enable_fused_normalization=enable_fused_normalization,
            parallel_output=parallel_output,
        )

    # -------------------------------------------------------------- metadata
    def supported_devices(self) -> List[str]:
        return ["cuda", "cpu"]

    def supported_precisions(self) -> List[str]:
        return ["fp16", "bf16", "fp32"]

    def control_device(self) -> bool:
        return True

    def control_precision(self) -> bool:
        return True

    def support_no_sync(self) -> bool:
        return True

    def support_lora(self) -> bool:
        return False

    def control_checkpoint_io(self) -> bool:
        return True

    def get_checkpoint_io(self) -> CheckpointIO:
        from ...checkpoint_io.hybrid_parallel_checkpoint_io import HybridParallelCheckpointIO

        return HybridParallelCheckpointIO(self.dp_group, self.pp_group, self.tp_group, self.sp_size)

    # -------------------------------------------------------------- configure
    def configure(
        self,
        model: nn.Module,
        optimizer: Optional[Optimizer] = None,
        criterion: Optional[Callable] = None,
        dataloader: Optional[DataLoader] = None,
        lr_scheduler: Optional[LRScheduler] = None,
    ) -> Tuple[nn.Module, OptimizerWrapper, Callable, DataLoader, LRScheduler]:
        dtype = _PRECISION_DTYPE[self.precision]

        pre_shard_names: Optional[dict] = None
        if not isinstance(model, ModelWrapper):
            # record param identities before surgery so multi-group optimizers
            # can be re-pointed at the sharded replacements by name
            pre_shard_names = {id(p): n for n, p in model.named_parameters()}
            # pipeline layer assignment BEFORE sharding (policies may use it)
            if self.stage_manager is not None:
                self._assign_pipeline_stage(model)
            if self.tp_size > 1 or self.shard_config.enable_sequence_parallelism:
                shardformer = ShardFormer(self.shard_config)
                model, _ = shardformer.optimize(model)
            model = model.to(dtype)
            if torch.cuda.is_available():
                model = model.to("cuda")
            if self.pp_style in ("zb", "zbv"):
                from ...ops.zb_linear import convert_to_zb_linears

                convert_to_zb_linears(model)
            tied = getattr(model, "_tied_embed_param", None)
            if tied is not None and self.embed_group is not None and dist.get_world_size(self.embed_group) > 1:
                # both copies must start identical: broadcast from the first stage
                src = min(self.pg_mesh.get_ranks_in_group(self.embed_group))
                dist.broadcast(tied.data, src=src, group=self.embed_group)
            model = HybridParallelModule(
                model, dtype, self.dp_sp_group, self.tp_group, self.sp_group,
                embed_group=self.embed_group, tied_param=tied,
            )

        if optimizer is not None and not isinstance(optimizer, OptimizerWrapper):
            self._rebuild_param_groups(optimizer, model.module, pre_shard_names)
            if self.zero_stage == 0:
                if self.precision == "fp32":
                    optimizer = HybridParallelFP32Optimizer(
                        optimizer, model, max_norm=self.max_norm,
                        tp_process_group=self.tp_group if self.tp_size > 1 else None,
                        pp_process_group=self.pp_group if self.pp_size > 1 else None,
                    )
                else:
                    optimizer = HybridParallelNaiveOptimizer(
                        optimizer, model, precision=self.precision, max_norm=self.max_norm,
                        tp_process_group=self.tp_group if self.tp_size > 1 else None,
                        pp_process_group=self.pp_group if self.pp_size > 1 else None,
                        **self.amp_kwargs,
                    )
            else:
                optimizer = LowLevelZeroOptimizer(
                    optimizer,
                    dp_process_group=self.dp_sp_group,
                    forced_dtype=dtype if self.precision != "fp32" else None,
                    tp_process_group=self.tp_group if self.tp_size > 1 else None,
                    pp_process_group=self.pp_group if self.pp_size > 1 else None,
                    **self.zero_kwargs,
                    **({} if self.precision != "fp16" else self.amp_kwargs),
                )
        if optimizer is not None and isinstance(optimizer, OptimizerWrapper):
            # topology-independent optimizer checkpoints need param names +
            # TP shard metadata (checkpoint_io/hybrid_parallel_checkpoint_io.py)
            from ...checkpoint_io.param_meta import build_tp_shard_map

            optimizer.ckpt_param_names = {id(p): n for n, p in model.module.named_parameters()}
            optimizer.ckpt_local_params = dict(model.module.named_parameters())
            optimizer.ckpt_tp_map = build_tp_shard_map(model.module)
        return model, optimizer, criterion, dataloader, lr_scheduler

    def _assign_pipeline_stage(self, model: nn.Module) -> None:
        """Set model.stage_range and free layers outside this stage.

        Convention: the model exposes ``.model.layers`` (decoder stack) — the
        native model family contract; HF models get stage-aware policies.
        """
        inner, layers = self._find_layer_stack(model)
        n_layers = len(layers)
        # tied embed/lm_head detection BEFORE stubbing: first stage keeps the
        # weight via the embedding, last via lm_head; grads sync over embed_group
        head = getattr(model, "lm_head", None)
        embed = None
        for attr in ("embed_tokens", "wte", "word_embeddings"):
            if hasattr(inner, attr):
                embed = getattr(inner, attr)
                break
        is_tied = (
            head is not None and embed is not None
            and getattr(head, "weight", None) is not None and head.weight is embed.weight
        )
        model._tied_embed_param = None
        if self.pp_style == "zbv":
            # V placement: rank r holds vstages r (descending arm) and
            # 2*pp-1-r (ascending arm); embeddings, LM head and the loss all
            # live on rank 0 — tied weights need no cross-stage sync.
            pp = self.pp_size
            V = 2 * pp
            per = [n_layers // V] * V
            for i in range(n_layers % V):
                per[i] += 1
            starts = [sum(per[:i]) for i in range(V)]
            vs = [self.stage_manager.stage, V - 1 - self.stage_manager.stage]
            model.chunk_ranges = [(starts[v], starts[v] + per[v]) for v in vs]
            model.stage_range = model.chunk_ranges[0]
            held = set()
            for a, b in model.chunk_ranges:
                held.update(range(a, b))
            for i in range(n_layers):
                if i not in held:
                    layers[i] = _StageStub()
            if self.stage_manager.stage != 0:
                self._stub_embeddings(inner)
                if head is not None:
                    model.lm_head = _StageStub()
            return
        if is_tied:
            last_stage = (
                self.stage_manager.is_last_stage(self.num_model_chunks - 1)
                if self.pp_style == "interleaved" else self.stage_manager.is_last_stage()
            )
            if self.stage_manager.is_first_stage():
                model._tied_embed_param = embed.weight
            elif last_stage:
                model._tied_embed_param = head.weight
                # full grad lives on both stages after sync — count it once
                head.weight._grad_norm_skip = True
        if self.pp_style == "interleaved":
            V, pp = self.num_model_chunks, self.pp_size
            per = [n_layers // (pp * V)] * (pp * V)
            for i in range(n_layers % (pp * V)):
                per[i] += 1
            starts = [sum(per[:i]) for i in range(pp * V)]
            held = []
            model.chunk_ranges = []
            for c in range(V):
                vs = c * pp + self.stage_manager.stage
                rng = (starts[vs], starts[vs] + per[vs])
                model.chunk_ranges.append(rng)
                held.extend(range(*rng))
            model.stage_range = model.chunk_ranges[0]
            for i in range(n_layers):
                if i not in held:
                    layers[i] = _StageStub()
            if not self.stage_manager.is_first_stage():
                self._stub_embeddings(inner)
            if not self.stage_manager.is_last_stage(self.num_model_chunks - 1) and hasattr(model, "lm_head"):
                model.lm_head = _StageStub()
            return
        start, end = self.stage_manager.stage_index(n_layers)
        model.stage_range = (start, end)
        for i in range(n_layers):
            if not (start <= i < end):
                layers[i] = _StageStub()
        if not self.stage_manager.is_first_stage():
            self._stub_embeddings(inner)
        if not self.stage_manager.is_last_stage() and hasattr(model, "lm_head"):
            model.lm_head = _StageStub()

    @staticmethod
    def _find_layer_stack(model):
        """Locate the decoder stack: .model.layers (llama-family),
        .transformer.layers (gpt2) or .transformer.h (gptj/bloom)."""
        for path, lattr in (("model", "layers"), ("transformer", "layers"),
                            ("transformer", "h"), (None, "layers"), (None, "h")):
            base = getattr(model, path, None) if path else model
            if base is not None and hasattr(base, lattr):
                return base, getattr(base, lattr)
        raise AssertionError("pipeline parallelism needs a decoder stack "
                             "(.model.layers / .transformer.layers / .transformer.h)")

    @staticmethod
    def _stub_embeddings(inner) -> None:
        for attr in ("embed_tokens", "wte", "wpe", "word_embeddings"):
            if hasattr(inner, attr) and isinstance(getattr(inner, attr), nn.Module)                     and not isinstance(getattr(inner, attr), _StageStub):
                setattr(inner, attr, _StageStub())

    @staticmethod
    def _rebuild_param_groups(optimizer: Optimizer, model: nn.Module,
                              pre_shard_names: Optional[dict] = None) -> None:
        """Sharding/stage-release replaced parameter objects; re-point the
        optimizer's param groups at the live model parameters.

        Multi-group optimizers (e.g. weight-decay splits) are re-pointed by
        matching each old param's NAME (recorded before surgery) to the
        sharded replacement; params released to other pipeline stages are
        dropped from their group."""
        live = list(model.parameters())
        current = [p for g in optimizer.param_groups for p in g["params"]]
        if len(current) == len(live) and all(a is b for a, b in zip(current, live)):
            return
        if len(optimizer.param_groups) == 1:
            optimizer.param_groups[0]["params"] = live
            optimizer.state.clear()
            return
        assert pre_shard_names is not None, (
            "multi-group optimizer re-pointing needs the pre-shard name map; "
            "construct the optimizer after booster.boost, or pass the bare model"
        )
        live_by_name = dict(model.named_parameters())
        seen = set()
        for group in optimizer.param_groups:
            new_params = []
            for p in group["params"]:
                name = pre_shard_names.get(id(p))
                if name is None:
                    raise ValueError(
                        "optimizer contains a parameter that was not in the model passed to boost()"
                    )
                new_p = live_by_name.get(name)
                if new_p is None or id(new_p) in seen:
                    continue  # released to another pipeline stage / tied duplicate
                seen.add(id(new_p))
                new_params.append(new_p)
            group["params"] = new_params
        optimizer.state.clear()

    # ---------------------------------------------------------------- runtime
    def execute_pipeline(
        self,
        data_iter: Iterator,
        model: HybridParallelModule,
        criterion: Callable[[Any, Any], torch.Tensor],
        optimizer: Optional[OptimizerWrapper] = None,
        return_loss: bool = True,
        return_outputs: bool = False,
    ) -> dict:
        assert self.scheduler is not None, "execute_pipeline requires pp_size > 1"
        if isinstance(optimizer, LowLevelZeroOptimizer):
            ctx = optimizer.no_sync()
        else:
            ctx = model.no_sync()
        with ctx:
            result = self.scheduler.forward_backward_step(
                model, data_iter, criterion, optimizer, return_loss, return_outputs
            )
        # grad sync after all microbatches: tied embed/head first (so the dp
        # reduction sees the summed grad), then sp partials, then dp
        model.sync_shared_params()
        model.sync_partial_sp_grads()
        if isinstance(optimizer, LowLevelZeroOptimizer):
            optimizer.sync_dp_grads()
        else:
            model.sync_dp_grads()
        return result

    def no_sync(self, model: nn.Module, optimizer: OptimizerWrapper = None) -> Iterator[None]:
        if isinstance(optimizer, LowLevelZeroOptimizer):
            return optimizer.no_sync()
        return model.no_sync()

    def prepare_dataloader(self, dataset, batch_size, shuffle=False, seed=1024, drop_last=False,
                           pin_memory=False, num_workers=0, **kwargs):
        import numpy as np
        import random
        from torch.utils.data import DataLoader
        from torch.utils.data.distributed import DistributedSampler

        sampler = DistributedSampler(
            dataset,
            num_replicas=self.dp_size,
            rank=self.pg_mesh.coordinate(DP_AXIS),
            shuffle=shuffle,
        )

        def seed_worker(worker_id):
            np.random.seed(seed)
            random.seed(seed)

        return DataLoader(dataset, batch_size=batch_size, sampler=sampler, worker_init_fn=seed_worker,
                          drop_last=drop_last, pin_memory=pin_memory, num_workers=num_workers, **kwargs)


class _StageStub(nn.Module):
    """Placeholder for modules owned by other pipeline stages."""

    def forward(self, *args, **kwargs):
        raise RuntimeError("This module belongs to another pipeline stage")
