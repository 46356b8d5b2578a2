"""Asyncio front-end over the continuous-batching engine
(reference: colossalai/inference/core/async_engine.py — Tracer +
AsyncInferenceEngine).

Concurrent callers ``await submit(prompt)``; a single background task
drives ``engine.step()`` whenever requests are in flight, resolving each
caller's future as its sequence finishes. New requests join the running
batch at the next engine iteration (continuous batching), so overlapping
awaits share prefill/decode steps instead of serializing.
"""

import asyncio
from typing import Dict, List, Optional

from .paged_engine import ContinuousBatchEngine

__all__ = ["AsyncInferenceEngine"]


class AsyncInferenceEngine:
    def __init__(self, engine: ContinuousBatchEngine, step_sleep: float = 0.0):
        self.engine = engine
        self.step_sleep = step_sleep
        self._futures: Dict[int, asyncio.Future] = {}
        self._driver: Optional[asyncio.Task] = None

    async def submit(self, prompt: List[int], max_new_tokens: Optional[int] = None) -> List[int]:
        """Queue a prompt; resolves with the FULL token sequence."""
        loop = asyncio.get_running_loop()
        rid = self.engine.add_request(prompt, max_new_tokens)
        fut: asyncio.Future = loop.create_future()
        self._futures[rid] = fut
        self._ensure_driver()
        return await fut

    def _ensure_driver(self):
        if self._driver is None or self._driver.done():
            self._driver = asyncio.get_running_loop().create_task(self._drive())

    async def _drive(self):
        try:
            while self._futures:
                finished = await asyncio.to_thread(self.engine.step)
                for rid, tokens in finished.items():
                    fut = self._futures.pop(rid, None)
                    if fut is not None and not fut.done():
                        fut.set_result(tokens)
                if self.step_sleep:
                    await asyncio.sleep(self.step_sleep)
                else:
                    await asyncio.sleep(0)  # yield so new submits can join
        except Exception as exc:  # propagate to every waiter (reference Tracer)
            for fut in self._futures.values():
                if not fut.done():
                    fut.set_exception(exc)
            self._futures.clear()
            raise

    def abort(self, request_id: int):
        fut = self._futures.pop(request_id, None)
        if fut is not None and not fut.done():
            fut.cancel()
