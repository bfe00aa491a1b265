"""Structured logging (reference parity: C28 zerolog / P20 logger.py).

One-line JSON records with level/ts/component, switchable to plain text via
AGENTFIELD_LOG_FORMAT=text; level from AGENTFIELD_LOG_LEVEL.
"""
from __future__ import annotations

import json
import logging
import os
import sys
import time


class JsonFormatter(logging.Formatter):
    def format(self, record: logging.LogRecord) -> str:
        out = {
            "ts": round(time.time(), 3),
            "level": record.levelname.lower(),
            "component": record.name,
            "msg": record.getMessage(),
        }
        if record.exc_info:
            out["exc"] = self.formatException(record.exc_info)
        for k, v in getattr(record, "fields", {}).items():
            out[k] = v
        return json.dumps(out)


def get_logger(component: str) -> logging.Logger:
    log = logging.getLogger(f"agentfield.{component}")
    if not log.handlers:
        h = logging.StreamHandler(sys.stderr)
        if os.environ.get("AGENTFIELD_LOG_FORMAT", "json") == "text":
            h.setFormatter(logging.Formatter(
                "%(asctime)s %(levelname)s %(name)s %(message)s"))
        else:
            h.setFormatter(JsonFormatter())
        log.addHandler(h)
        log.setLevel(os.environ.get("AGENTFIELD_LOG_LEVEL", "INFO").upper())
        log.propagate = False
    return log


def log_event(log: logging.Logger, msg: str, **fields):
    log.info(msg, extra={"fields": fields})
