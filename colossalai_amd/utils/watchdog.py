"""Training-step watchdog (failure detection).

The reference has no hang detector; distributed runs die silently when one
rank stalls in a collective. This watchdog arms a timer around each step:
if a step exceeds ``timeout_s``, it dumps every Python thread's stack to
stderr (and optionally calls a user hook) so the stuck collective is
identifiable from any single rank's log.

    wd = Watchdog(timeout_s=120)
    for batch in loader:
        with wd.step():
            train_step(batch)
"""

import contextlib
import faulthandler
import sys
import threading
import time
from typing import Callable, Optional

__all__ = ["Watchdog"]


class Watchdog:
    def __init__(self, timeout_s: float = 300.0, on_timeout: Optional[Callable] = None,
                 fire_once: bool = True):
        self.timeout_s = timeout_s
        self.on_timeout = on_timeout
        self.fire_once = fire_once
        self.fired = 0
        self._timer: Optional[threading.Timer] = None

    def _fire(self):
        self.fired += 1
        sys.stderr.write(
            f"[colossalai_amd watchdog] step exceeded {self.timeout_s:.0f}s — dumping stacks\n"
        )
        faulthandler.dump_traceback(file=sys.stderr)
        if self.on_timeout is not None:
            self.on_timeout()

    def arm(self):
        self.disarm()
        if self.fire_once and self.fired:
            return
        self._timer = threading.Timer(self.timeout_s, self._fire)
        self._timer.daemon = True
        self._timer.start()

    def disarm(self):
        if self._timer is not None:
            self._timer.cancel()
            self._timer = None

    @contextlib.contextmanager
    def step(self):
        self.arm()
        try:
            yield
        finally:
            self.disarm()
