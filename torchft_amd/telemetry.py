"""Structured observability for the FT control plane.

Reference parity: torchft/otel.py — structured loggers ``torchft_quorums``,
``torchft_commits``, ``torchft_errors`` emitted at quorum change, commit and
PG abort. This environment has no OTLP egress, so when
``TORCHFT_USE_OTEL=true`` the same records are emitted as JSON lines to
stderr and (optionally) ``TORCHFT_AMD_TELEMETRY_FILE``; swapping the
handler for an OTLP exporter is a one-line change at the deployment site.
"""

from __future__ import annotations

import json
import logging
import os
import sys
import time
from typing import Optional

USE_OTEL_ENV = "TORCHFT_USE_OTEL"
TELEMETRY_FILE_ENV = "TORCHFT_AMD_TELEMETRY_FILE"

STRUCTURED_LOGGERS = ("torchft_quorums", "torchft_commits", "torchft_errors")

_RESERVED = {
    "name", "msg", "args", "levelname", "levelno", "pathname", "filename",
    "module", "exc_info", "exc_text", "stack_info", "lineno", "funcName",
    "created", "msecs", "relativeCreated", "thread", "threadName",
    "processName", "process", "message", "taskName",
}


class JSONLineFormatter(logging.Formatter):
    def format(self, record: logging.LogRecord) -> str:
        payload = {
            "ts": time.time(),
            "logger": record.name,
            "level": record.levelname,
        }
        for k, v in record.__dict__.items():
            if k not in _RESERVED and not k.startswith("_"):
                try:
                    json.dumps(v)
                    payload[k] = v
                except TypeError:
                    payload[k] = repr(v)
        if record.getMessage():
            payload["msg"] = record.getMessage()
        return json.dumps(payload)


def setup_telemetry(path: Optional[str] = None) -> None:
    """Attach JSON-line handlers to the structured FT loggers.

    Called automatically on package import when TORCHFT_USE_OTEL=true.
    """
    path = path or os.environ.get(TELEMETRY_FILE_ENV)
    handlers: list[logging.Handler] = [logging.StreamHandler(sys.stderr)]
    if path:
        handlers.append(logging.FileHandler(path))
    fmt = JSONLineFormatter()
    for h in handlers:
        h.setFormatter(fmt)
    for name in STRUCTURED_LOGGERS:
        logger = logging.getLogger(name)
        logger.setLevel(logging.INFO)
        for h in handlers:
            logger.addHandler(h)
        logger.propagate = False


if os.environ.get(USE_OTEL_ENV, "false").lower() == "true":  # pragma: no cover
    setup_telemetry()
