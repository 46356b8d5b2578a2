from ._operation import (
    all_to_all_comm,
    gather_forward_reduce_scatter_backward,
    gather_forward_split_backward,
    linear_with_async_comm,
    reduce_backward,
    reduce_forward,
    reduce_scatter_forward_gather_backward,
    split_forward_gather_backward,
)
from .embedding import VocabParallelEmbedding1D
from .linear import Linear1D_Col, Linear1D_Row
from .loss import DistCrossEntropy, DistLogProb, dist_cross_entropy, dist_log_prob
from .parallel_module import ParallelModule
from .randomizer import Randomizer

__all__ = [
    "Linear1D_Col",
    "Linear1D_Row",
    "VocabParallelEmbedding1D",
    "DistCrossEntropy",
    "dist_cross_entropy",
    "ParallelModule",
    "Randomizer",
    "DistLogProb",
    "dist_log_prob",
    "linear_with_async_comm",
    "reduce_forward",
    "reduce_backward",
    "gather_forward_split_backward",
    "split_forward_gather_backward",
    "gather_forward_reduce_scatter_backward",
    "reduce_scatter_forward_gather_backward",
    "all_to_all_comm",
]
