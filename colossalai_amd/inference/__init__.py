from .config import GenerationConfig, InferenceConfig
from .async_engine import AsyncInferenceEngine
from .diffusion import DiffusionEngine, PatchParallelDiT, ddim_sample
from .engine import LLMEngine
from .kv_cache import KVCacheManager
from .paged_engine import ContinuousBatchEngine
from .request_manager import Request, RequestManager, RequestStatus
from .spec_decode import BatchedSpeculativeEngine, GlideCrossAttention, GlideSpeculativeEngine, SpeculativeEngine

__all__ = ["InferenceConfig", "GenerationConfig", "LLMEngine", "ContinuousBatchEngine", "DiffusionEngine", "AsyncInferenceEngine", "PatchParallelDiT", "ddim_sample",
           "KVCacheManager", "SpeculativeEngine", "BatchedSpeculativeEngine", "GlideSpeculativeEngine", "GlideCrossAttention", "RequestManager", "Request", "RequestStatus"]
