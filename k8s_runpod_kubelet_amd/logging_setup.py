"""Structured logging with handler fan-out.

Counterpart of the reference's slog setup (cmd/virtual_kubelet/main.go:111-144)
and its multiHandler (cmd/virtual_kubelet/loghandler.go:7-54): key=value
structured records fanned out to stdout plus an optional JSON file sink
(standing in for the reference's optional Sentry sink — there is no network
egress here). Unlike the reference, the ``--log-level`` flag is actually
applied (reference parses it at main.go:69 but never installs it).
"""

from __future__ import annotations

import json
import logging
import sys
import time
from typing import List, Optional

_LEVELS = {
    "debug": logging.DEBUG,
    "info": logging.INFO,
    "warn": logging.WARNING,
    "warning": logging.WARNING,
    "error": logging.ERROR,
}

_RESERVED = set(
    logging.LogRecord("", 0, "", 0, "", (), None).__dict__.keys()
) | {"message", "asctime", "taskName"}


class KVFormatter(logging.Formatter):
    """slog-TextHandler-style output: ``time=... level=... msg=... k=v``."""

    def format(self, record: logging.LogRecord) -> str:
        ts = time.strftime("%Y-%m-%dT%H:%M:%S", time.localtime(record.created))
        parts = [
            f"time={ts}.{int(record.msecs):03d}",
            f"level={record.levelname}",
            f"logger={record.name}",
            f'msg="{record.getMessage()}"',
        ]
        for key, value in sorted(record.__dict__.items()):
            if key in _RESERVED or key.startswith("_"):
                continue
            parts.append(f"{key}={value!r}")
        if record.exc_info:
            parts.append(f"exc={self.formatException(record.exc_info)!r}")
        return " ".join(parts)


class JSONFormatter(logging.Formatter):
    def format(self, record: logging.LogRecord) -> str:
        payload = {
            "ts": record.created,
            "level": record.levelname,
            "logger": record.name,
            "msg": record.getMessage(),
        }
        for key, value in record.__dict__.items():
            if key in _RESERVED or key.startswith("_"):
                continue
            try:
                json.dumps(value)
                payload[key] = value
            except (TypeError, ValueError):
                payload[key] = repr(value)
        if record.exc_info:
            payload["exc"] = self.formatException(record.exc_info)
        return json.dumps(payload)


class MultiHandler(logging.Handler):
    """Fan a record out to N handlers (reference loghandler.go:7-54 semantics:
    enabled if any child is enabled; handle dispatches to all)."""

    def __init__(self, handlers: List[logging.Handler]):
        super().__init__(level=min((h.level for h in handlers), default=logging.INFO))
        self.handlers = handlers

    def emit(self, record: logging.LogRecord) -> None:
        for handler in self.handlers:
            if record.levelno >= handler.level:
                handler.handle(record)

    def close(self) -> None:
        for handler in self.handlers:
            handler.close()
        super().close()


def initialize_logger(level: str = "info", json_log_path: Optional[str] = None) -> logging.Logger:
    root = logging.getLogger()
    for handler in list(root.handlers):
        root.removeHandler(handler)

    stdout_handler = logging.StreamHandler(sys.stdout)
    stdout_handler.setFormatter(KVFormatter())
    handlers: List[logging.Handler] = [stdout_handler]

    if json_log_path:
        file_handler = logging.FileHandler(json_log_path)
        file_handler.setFormatter(JSONFormatter())
        handlers.append(file_handler)

    root.addHandler(MultiHandler(handlers) if len(handlers) > 1 else stdout_handler)
    root.setLevel(_LEVELS.get(level.lower(), logging.INFO))
    return root
