"""Rank-aware distributed logger (reference: colossalai/logging/logger.py:12)."""

import logging
from typing import List, Optional

import torch.distributed as dist

__all__ = ["get_dist_logger", "DistributedLogger", "disable_existing_loggers"]

_LOGGERS = {}


class DistributedLogger:
    """A logger that tags records with the global rank and can filter by rank.

    Usage::

        logger = get_dist_logger()
        logger.info("hello", ranks=[0])   # only rank 0 emits
    """

    def __init__(self, name: str = "colossalai_amd"):
        self.name = name
        self._logger = logging.getLogger(name)
        if not self._logger.handlers:
            handler = logging.StreamHandler()
            handler.setFormatter(
                logging.Formatter("%(asctime)s %(name)s [rank%(rank)s] %(levelname)s: %(message)s")
            )
            self._logger.addHandler(handler)
            self._logger.setLevel(logging.INFO)
            self._logger.propagate = False

    @staticmethod
    def _rank() -> int:
        if dist.is_available() and dist.is_initialized():
            return dist.get_rank()
        return 0

    def _should_log(self, ranks: Optional[List[int]]) -> bool:
        return ranks is None or self._rank() in ranks

    def _log(self, level: int, message: str, ranks: Optional[List[int]] = None) -> None:
        if self._should_log(ranks):
            self._logger.log(level, message, extra={"rank": self._rank()})

    def info(self, message: str, ranks: Optional[List[int]] = None) -> None:
        self._log(logging.INFO, message, ranks)

    def warning(self, message: str, ranks: Optional[List[int]] = None) -> None:
        self._log(logging.WARNING, message, ranks)

    def error(self, message: str, ranks: Optional[List[int]] = None) -> None:
        self._log(logging.ERROR, message, ranks)

    def debug(self, message: str, ranks: Optional[List[int]] = None) -> None:
        self._log(logging.DEBUG, message, ranks)

    def set_level(self, level: str) -> None:
        self._logger.setLevel(getattr(logging, level.upper()))


def get_dist_logger(name: str = "colossalai_amd") -> DistributedLogger:
    if name not in _LOGGERS:
        _LOGGERS[name] = DistributedLogger(name)
    return _LOGGERS[name]


def disable_existing_loggers(include: Optional[List[str]] = None) -> None:
    for name in list(logging.root.manager.loggerDict.keys()):
        if include is None or name in include:
            logging.getLogger(name).setLevel(logging.WARNING)
