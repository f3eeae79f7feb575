"""Structured logging (capability parity: /root/reference/CHANGELOG.md:497).

``setup_logging(json_lines=True)`` turns every log record into one JSON
line with stable keys; the step ledger and engine logs ride the same
stream, so a node's whole story is greppable.
"""

from __future__ import annotations

import json
import logging
import sys
import time

__all__ = ["setup_logging", "JsonFormatter"]


class JsonFormatter(logging.Formatter):
    def format(self, record: logging.LogRecord) -> str:
        out = {
            "ts": round(time.time(), 3),
            "level": record.levelname.lower(),
            "logger": record.name,
            "msg": record.getMessage(),
        }
        if record.exc_info and record.exc_info[0] is not None:
            out["exc"] = self.formatException(record.exc_info)
        for key in ("node", "thread_id", "hop"):
            v = getattr(record, key, None)
            if v is not None:
                out[key] = v
        return json.dumps(out, default=str)


def setup_logging(level: int = logging.INFO, *, json_lines: bool = False,
                  stream=None) -> None:
    handler = logging.StreamHandler(stream or sys.stderr)
    if json_lines:
        handler.setFormatter(JsonFormatter())
    else:
        handler.setFormatter(logging.Formatter(
            "%(asctime)s %(levelname)s %(name)s %(message)s"))
    root = logging.getLogger("calfkit")
    root.handlers = [handler]
    root.setLevel(level)
    root.propagate = False
