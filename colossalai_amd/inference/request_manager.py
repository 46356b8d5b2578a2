"""Continuous-batching request scheduler (reference:
colossalai/inference/core/request_handler.py — RunningList/WaitingList).

Requests join a FIFO waiting queue; each engine step admits as many
waiting requests as fit (batch slots AND KV blocks for prompt +
max_new_tokens worst case — admission control instead of preemption,
which 288 GB of KV pool makes the right trade), prefills them, then
decodes the whole running set one token. Finished sequences free their
blocks immediately, so short requests drain and new ones stream in
without waiting for the longest member of a static batch.
"""

from dataclasses import dataclass, field
from enum import Enum
from typing import Dict, List, Optional

from .kv_cache import KVCacheManager

__all__ = ["RequestStatus", "Request", "RequestManager"]


class RequestStatus(Enum):
    WAITING = "waiting"
    RUNNING = "running"
    FINISHED = "finished"


@dataclass
class Request:
    request_id: int
    prompt: List[int]
    max_new_tokens: int
    output: List[int] = field(default_factory=list)
    status: RequestStatus = RequestStatus.WAITING

    @property
    def seq_len(self) -> int:
        return len(self.prompt) + len(self.output)

    @property
    def tokens(self) -> List[int]:
        return self.prompt + self.output


class RequestManager:
    def __init__(self, kv: KVCacheManager, max_batch_size: int = 32):
        self.kv = kv
        self.max_batch_size = max_batch_size
        self.waiting: List[Request] = []
        self.running: List[Request] = []
        self._next_id = 0

    def add_request(self, prompt: List[int], max_new_tokens: int) -> int:
        rid = self._next_id
        self._next_id += 1
        self.waiting.append(Request(rid, list(prompt), max_new_tokens))
        return rid

    def schedule(self) -> List[Request]:
        """Admit waiting requests while slots + worst-case KV blocks allow;
        returns the newly admitted (to be prefilled) requests."""
        admitted = []
        while self.waiting and len(self.running) < self.max_batch_size:
            req = self.waiting[0]
            worst = len(req.prompt) + req.max_new_tokens
            if not self.kv.can_allocate(worst):
                break
            self.waiting.pop(0)
            self.kv.allocate(req.request_id, len(req.prompt))
            req.status = RequestStatus.RUNNING
            self.running.append(req)
            admitted.append(req)
        return admitted

    def append_token(self, req: Request, token: int, eos_token_id: Optional[int], max_seq_len: int) -> None:
        req.output.append(token)
        self.kv.extend(req.request_id, req.seq_len)
        if (
            (eos_token_id is not None and token == eos_token_id)
            or len(req.output) >= req.max_new_tokens
            or req.seq_len >= max_seq_len
        ):
            self.finish(req)

    def finish(self, req: Request) -> None:
        req.status = RequestStatus.FINISHED
        self.running.remove(req)
        self.kv.free(req.request_id)

    @property
    def has_work(self) -> bool:
        return bool(self.waiting or self.running)
