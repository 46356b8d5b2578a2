"""Logical device mesh with a per-axis communication cost model
(reference: colossalai/device/device_mesh.py:500-518 — re-sized for the
MI355X xGMI topology).

The mesh arranges the world's ranks into an N-D grid and exposes process
groups along each axis plus α-β collective cost estimates
(cost = α · steps + β · bytes_on_wire). Defaults model one 8-GPU MI355X
node: 7 xGMI links/GPU at ≈153 GB/s each — ring collectives are per-link
bound, so β is 1/153e9 s/B regardless of group size, and α ≈ 10 µs per
ring step. Feed measured values from ``AlphaBetaProfiler`` to override.
"""

import itertools
from typing import Dict, List, Optional, Tuple

import torch.distributed as dist

__all__ = ["DeviceMesh"]

# xGMI defaults (per-link, one MI355X node)
DEFAULT_ALPHA = 10e-6  # s per collective step
DEFAULT_BETA = 1.0 / 153e9  # s per byte per link


class DeviceMesh:
    def __init__(self, mesh_shape: Tuple[int, ...], alpha: float = DEFAULT_ALPHA,
                 beta: float = DEFAULT_BETA):
        self.shape = tuple(mesh_shape)
        self.alpha = alpha
        self.beta = beta
        self._groups: Dict[int, List] = {}
        if dist.is_initialized():
            world = dist.get_world_size()
            total = 1
            for s in self.shape:
                total *= s
            assert total == world, f"mesh {self.shape} != world {world}"
            self._build_groups()

    # ------------------------------------------------------------- topology
    def _coords(self, rank: int) -> Tuple[int, ...]:
        c = []
        for s in reversed(self.shape):
            c.append(rank % s)
            rank //= s
        return tuple(reversed(c))

    def _rank(self, coords: Tuple[int, ...]) -> int:
        r = 0
        for c, s in zip(coords, self.shape):
            r = r * s + c
        return r

    def _build_groups(self):
        my = self._coords(dist.get_rank())
        for axis, size in enumerate(self.shape):
            groups = []
            other_axes = [range(s) for i, s in enumerate(self.shape) if i != axis]
            for rest in itertools.product(*other_axes):
                ranks = []
                for v in range(size):
                    coords = list(rest)
                    coords.insert(axis, v)
                    ranks.append(self._rank(tuple(coords)))
                g = dist.new_group(ranks)
                groups.append((ranks, g))
            self._groups[axis] = groups

    def get_process_group(self, axis: int):
        """This rank's group along `axis`."""
        me = dist.get_rank()
        for ranks, g in self._groups[axis]:
            if me in ranks:
                return g
        raise RuntimeError("rank not in any group (mesh inconsistent)")

    # ----------------------------------------------------------- cost model
    def _n(self, axis: int) -> int:
        return self.shape[axis]

    def all_reduce_cost(self, nbytes: int, axis: int) -> float:
        n = self._n(axis)
        if n == 1:
            return 0.0
        # ring: 2(n-1) steps, 2·(n-1)/n of the data over the slowest link
        return 2 * (n - 1) * self.alpha + 2 * (n - 1) / n * nbytes * self.beta

    def all_gather_cost(self, nbytes: int, axis: int) -> float:
        n = self._n(axis)
        if n == 1:
            return 0.0
        return (n - 1) * self.alpha + (n - 1) / n * nbytes * self.beta

    def reduce_scatter_cost(self, nbytes: int, axis: int) -> float:
        return self.all_gather_cost(nbytes, axis)

    def all_to_all_cost(self, nbytes: int, axis: int) -> float:
        n = self._n(axis)
        if n == 1:
            return 0.0
        # xGMI is point-to-point: each rank ships (n-1)/n of its data directly
        return self.alpha + (n - 1) / n * nbytes * self.beta
