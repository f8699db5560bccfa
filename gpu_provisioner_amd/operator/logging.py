"""Structured JSON logging, zap-equivalent.

Spec: reference vendor/sigs.k8s.io/karpenter/pkg/operator/logging/
logging.go:42-124 — JSON log lines with level/ts/logger/message, env-driven
level, and a noise filter for chatty event/lease messages.
"""
from __future__ import annotations

import json
import logging
import sys
import time

_NOISY_SUBSTRINGS = (
    "successfully acquired lease",
    "attempting to acquire leader lease",
    "event publish failed",
)


class JSONFormatter(logging.Formatter):
    def format(self, record: logging.LogRecord) -> str:
        entry = {
            "level": record.levelname.lower(),
            "ts": round(time.time(), 3),
            "logger": record.name,
            "message": record.getMessage(),
        }
        if record.exc_info and record.exc_info[0] is not None:
            entry["error"] = self.formatException(record.exc_info)
        return json.dumps(entry)


class NoiseFilter(logging.Filter):
    def filter(self, record: logging.LogRecord) -> bool:
        msg = record.getMessage()
        return not any(s in msg for s in _NOISY_SUBSTRINGS)


_LEVELS = {
    "debug": logging.DEBUG,
    "info": logging.INFO,
    "warn": logging.WARNING,
    "warning": logging.WARNING,
    "error": logging.ERROR,
}


def setup_logging(level: str = "info", stream=None) -> None:
    root = logging.getLogger()
    root.setLevel(_LEVELS.get(level.lower(), logging.INFO))
    handler = logging.StreamHandler(stream or sys.stderr)
    handler.setFormatter(JSONFormatter())
    handler.addFilter(NoiseFilter())
    root.handlers = [handler]
    # keep asyncio / urllib noise down
    logging.getLogger("asyncio").setLevel(logging.WARNING)
    logging.getLogger("httpx").setLevel(logging.WARNING)
