"""N-D cartesian process-group mesh over torch.distributed.

Re-designed equivalent of the reference's ProcessGroupMesh
(colossalai/cluster/process_group_mesh.py:25): ranks are laid out in an
N-D grid; sub-``ProcessGroup``s are created along any axis (or flattened
set of axes) and cached so each unique rank-tuple is built exactly once
on every rank (RCCL communicator creation is collective — all ranks must
create groups in the same order, which the sorted iteration guarantees).

On an 8×MI355X node the mesh axes map onto the xGMI point-to-point fabric:
every GPU pair is one hop, so group placement is topology-free intra-node;
the mesh's job is bookkeeping + deterministic communicator creation.
"""

import itertools
from typing import Dict, List, Optional, Tuple, Union

import numpy as np
import torch.distributed as dist
from torch.distributed import ProcessGroup

__all__ = ["ProcessGroupMesh"]


def prod(xs) -> int:
    out = 1
    for x in xs:
        out *= int(x)
    return out


class ProcessGroupMesh:
    """Cartesian mesh of ranks with cached per-axis process groups.

    Args:
        *size: size of each mesh dimension; ``prod(size)`` must equal the
            world size.

    Example::

        mesh = ProcessGroupMesh(2, 2, 2)   # (dp, pp, tp) on 8 ranks
        tp_group = mesh.get_group_along_axis(2)
    """

    def __init__(self, *size: int):
        assert dist.is_initialized(), "Please initialize torch.distributed first (colossalai_amd.launch)."
        world_size = dist.get_world_size()
        assert prod(size) == world_size, (
            f"mesh size {size} (prod={prod(size)}) must multiply to world size {world_size}"
        )
        self._shape: Tuple[int, ...] = tuple(int(s) for s in size)
        self._rank: int = dist.get_rank()
        self._coord: Tuple[int, ...] = self.unravel(self._rank, self._shape)
        # (ranks-tuple, backend) -> ProcessGroup
        self._group_cache: Dict[Tuple[Tuple[int, ...], Optional[str]], ProcessGroup] = {}
        self._group_to_ranks: Dict[ProcessGroup, Tuple[int, ...]] = {}

    def __del__(self):
        # Groups are destroyed by torch at process teardown; nothing to do.
        pass

    @property
    def shape(self) -> Tuple[int, ...]:
        return self._shape

    @property
    def rank(self) -> int:
        return self._rank

    def size(self, dim: Optional[int] = None) -> int:
        return prod(self._shape) if dim is None else self._shape[dim]

    def coordinate(self, dim: Optional[int] = None) -> Union[int, Tuple[int, ...]]:
        return self._coord if dim is None else self._coord[dim]

    # -- coordinate math ---------------------------------------------------
    @staticmethod
    def unravel(rank: int, shape: Tuple[int, ...]) -> Tuple[int, ...]:
        return tuple(int(x) for x in np.unravel_index(rank, shape))

    @staticmethod
    def ravel(coord: Tuple[int, ...], shape: Tuple[int, ...], mode: str = "raise") -> int:
        return int(np.ravel_multi_index(coord, shape, mode=mode))

    # -- group creation ----------------------------------------------------
    def _get_group(self, ranks_in_group: Tuple[int, ...], backend: Optional[str] = None) -> ProcessGroup:
        key = (ranks_in_group, backend)
        if key not in self._group_cache:
            group = dist.new_group(ranks=list(ranks_in_group), backend=backend)
            self._group_cache[key] = group
            self._group_to_ranks[group] = ranks_in_group
        return self._group_cache[key]

    def get_ranks_in_group(self, group: ProcessGroup) -> List[int]:
        return list(self._group_to_ranks[group])

    def create_group_along_axis(
        self,
        axis: Union[int, List[int]],
        indices_at_axis: Optional[Union[List[int], List[List[int]]]] = None,
        backend: Optional[str] = None,
        return_ranks_by_group: bool = False,
    ) -> Union[ProcessGroup, List[Tuple[int, ...]]]:
        """Create (on every rank, deterministically) the groups partitioning the
        mesh along ``axis``; return this rank's group.

        ``axis`` may be a list of axes — the returned groups then span the
        flattened product of those axes (e.g. dp×sp grad-sync groups).
        """
        axes = [axis] if isinstance(axis, int) else list(axis)
        if indices_at_axis is None:
            indices = [list(range(self._shape[a])) for a in axes]
        else:
            if isinstance(axis, int):
                indices = [list(indices_at_axis)]
            else:
                indices = [list(ix) for ix in indices_at_axis]

        reduced_shape = list(self._shape)
        for a in axes:
            reduced_shape[a] = 1  # iterate over the other axes

        target_group = None
        all_groups: List[Tuple[int, ...]] = []
        # Iterate base coordinates in sorted order => same creation order on
        # all ranks => RCCL communicator init cannot deadlock.
        for base in itertools.product(*[range(s) for s in reduced_shape]):
            ranks_in_group: List[int] = []
            for combo in itertools.product(*indices):
                coord = list(base)
                for a, idx in zip(axes, combo):
                    coord[a] = idx
                ranks_in_group.append(self.ravel(tuple(coord), self._shape))
            ranks_tuple = tuple(ranks_in_group)
            all_groups.append(ranks_tuple)
            group = self._get_group(ranks_tuple, backend=backend)
            if self._rank in ranks_in_group:
                target_group = group
        if return_ranks_by_group:
            return all_groups
        return target_group

    def get_group_along_axis(
        self,
        axis: Union[int, List[int]],
        indices_at_axis: Optional[List[int]] = None,
        backend: Optional[str] = None,
    ) -> ProcessGroup:
        """Get (or lazily create) the group along ``axis`` containing this rank."""
        return self.create_group_along_axis(axis, indices_at_axis, backend=backend)

    def get_coords_along_axis(
        self, base_coord: Tuple[int, ...], axis: Union[int, List[int]], indices_at_axis: Union[List[int], List[List[int]]]
    ) -> List[Tuple[int, ...]]:
        axes = [axis] if isinstance(axis, int) else list(axis)
        if isinstance(axis, int):
            indices = [list(indices_at_axis)]
        else:
            indices = [list(ix) for ix in indices_at_axis]
        coords = []
        for combo in itertools.product(*indices):
            coord = list(base_coord)
            for a, idx in zip(axes, combo):
                coord[a] = idx
            coords.append(tuple(coord))
        return coords
