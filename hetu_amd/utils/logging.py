"""Leveled logging (reference python/hetu/logger.py + the C++
HT_LOG_{TRACE..FATAL} streams, common/logging.h): one package logger,
level from HETU_AMD_LOG_LEVEL (TRACE|DEBUG|INFO|WARN|ERROR, default
INFO), rank-tagged when torch.distributed is initialized."""
from __future__ import annotations

import logging
import os

_LEVELS = {"TRACE": logging.DEBUG - 5, "DEBUG": logging.DEBUG,
           "INFO": logging.INFO, "WARN": logging.WARNING,
           "ERROR": logging.ERROR, "FATAL": logging.CRITICAL}
logging.addLevelName(_LEVELS["TRACE"], "TRACE")


class _RankFilter(logging.Filter):
    def filter(self, record):
        try:
            import torch.distributed as dist
            record.rank = dist.get_rank() if dist.is_initialized() else 0
        except Exception:  # noqa: BLE001
            record.rank = 0
        return True


def get_logger(name: str = "hetu_amd") -> logging.Logger:
    log = logging.getLogger(name)
    if not getattr(log, "_hetu_configured", False):
        h = logging.StreamHandler()
        h.setFormatter(logging.Formatter(
            "[%(levelname)s r%(rank)s %(name)s] %(message)s"))
        h.addFilter(_RankFilter())
        log.addHandler(h)
        log.setLevel(_LEVELS.get(
            os.environ.get("HETU_AMD_LOG_LEVEL", "INFO").upper(),
            logging.INFO))
        log.propagate = False
        log._hetu_configured = True
    return log
