"""Structured JSON logging with the reference's event taxonomy.

Mirrors the Winston setup (reference: server/src/utils/logger.ts:70-126):
rotating file + console, circular-safe serialization, and the
worker/job/performance tagged helper methods.
"""

from __future__ import annotations

import json
import logging
import logging.handlers
import os
import sys
import time
from typing import Any

_LEVELS = {"debug": logging.DEBUG, "info": logging.INFO,
           "warn": logging.WARNING, "error": logging.ERROR}


class JsonFormatter(logging.Formatter):
    def format(self, record: logging.LogRecord) -> str:
        out: dict[str, Any] = {
            "ts": time.strftime("%Y-%m-%dT%H:%M:%S", time.gmtime(record.created)),
            "level": record.levelname.lower(),
            "msg": record.getMessage(),
            "logger": record.name,
        }
        extra = getattr(record, "gridllm", None)
        if extra:
            try:
                json.dumps(extra)
                out.update(extra)
            except (TypeError, ValueError):
                out["extra"] = repr(extra)
        if record.exc_info:
            out["exc"] = self.formatException(record.exc_info)
        return json.dumps(out, default=repr)


class GridLogger(logging.LoggerAdapter):
    """Adds the reference's type-tagged helpers (logger.ts:113-126)."""

    def _tagged(self, level: int, msg: str, tag: str, fields: dict) -> None:
        self.logger.log(level, msg, extra={"gridllm": {"type": tag, **fields}})

    def worker(self, msg: str, **fields: Any) -> None:
        self._tagged(logging.INFO, msg, "worker", fields)

    def job(self, msg: str, **fields: Any) -> None:
        self._tagged(logging.INFO, msg, "job", fields)

    def performance(self, msg: str, **fields: Any) -> None:
        self._tagged(logging.INFO, msg, "performance", fields)

    def process(self, msg, kwargs):
        return msg, kwargs


_configured = False


def get_logger(name: str, level: str = "", log_dir: str = "") -> GridLogger:
    global _configured
    root = logging.getLogger("gridllm")
    if not _configured:
        lvl = _LEVELS.get(level or os.environ.get("LOG_LEVEL", "info"), logging.INFO)
        root.setLevel(lvl)
        con = logging.StreamHandler(sys.stderr)
        con.setFormatter(JsonFormatter())
        root.addHandler(con)
        d = log_dir or os.environ.get("GRIDLLM_LOG_DIR", "")
        if d:
            os.makedirs(d, exist_ok=True)
            # 5 MB x 5 rotation, like the reference (logger.ts:70-83)
            fh = logging.handlers.RotatingFileHandler(
                os.path.join(d, "gridllm.log"), maxBytes=5 * 2**20, backupCount=5)
            fh.setFormatter(JsonFormatter())
            root.addHandler(fh)
            eh = logging.handlers.RotatingFileHandler(
                os.path.join(d, "error.log"), maxBytes=5 * 2**20, backupCount=5)
            eh.setLevel(logging.ERROR)
            eh.setFormatter(JsonFormatter())
            root.addHandler(eh)
        _configured = True
    return GridLogger(logging.getLogger(f"gridllm.{name}"), {})
