"""Pipeline stage manager (reference: colossalai/pipeline/stage_manager.py:11)."""

from typing import List, Optional, Tuple

import torch.distributed as dist

from ..cluster import ProcessGroupMesh

__all__ = ["PipelineStageManager"]


class PipelineStageManager:
    def __init__(
        self,
        pg_mesh: ProcessGroupMesh,
        pipeline_axis: int,
        enable_interleave: bool = False,
        num_model_chunks: int = 1,
    ):
        self.pg_mesh = pg_mesh
        self.pipeline_axis = pipeline_axis
        self.num_stages = pg_mesh.size(pipeline_axis)
        self.stage = pg_mesh.coordinate(pipeline_axis)
        self.num_model_chunks = num_model_chunks
        self.enable_interleave = enable_interleave
        # group along the pp axis (creates deterministic RCCL communicators)
        self.pp_group = pg_mesh.get_group_along_axis(pipeline_axis)
        self._pp_ranks = pg_mesh.get_ranks_in_group(self.pp_group)

    def is_first_stage(self, model_chunk_id: int = 0) -> bool:
        if self.enable_interleave:
            return self.stage == 0 and model_chunk_id == 0
        return self.stage == 0

    def is_last_stage(self, model_chunk_id: int = 0) -> bool:
        if self.enable_interleave:
            return self.stage == self.num_stages - 1 and model_chunk_id == self.num_model_chunks - 1
        return self.stage == self.num_stages - 1

    def get_rank(self) -> int:
        return dist.get_rank()

    def get_prev_rank(self) -> int:
        return self._pp_ranks[(self.stage - 1) % self.num_stages]

    def get_next_rank(self) -> int:
        return self._pp_ranks[(self.stage + 1) % self.num_stages]

    def get_stage_of_rank(self, rank: int) -> int:
        return self._pp_ranks.index(rank)

    @staticmethod
    def distribute_layers(num_layers: int, num_stages: int) -> List[int]:
        quotient, remainder = divmod(num_layers, num_stages)
        layers = [quotient] * num_stages
        # give the spare layers to middle stages (first/last also run embed/head)
        for i in range(remainder):
            layers[(num_stages // 2 + i) % num_stages] += 1
        return layers

    def stage_index(self, num_layers: int, stage: Optional[int] = None) -> Tuple[int, int]:
        stage = self.stage if stage is None else stage
        layers = self.distribute_layers(num_layers, self.num_stages)
        start = sum(layers[:stage])
        return start, start + layers[stage]
