"""Logging configuration from `logging.{level,format}` — the same
config surface as the reference's logrus setup (cmd/taskhandler/
cfg.go:28-60): levels panic|fatal|error|warning|info|debug|trace,
formats text|json."""
from __future__ import annotations

import json
import logging
import sys

_LEVELS = {
    "panic": logging.CRITICAL, "fatal": logging.CRITICAL,
    "error": logging.ERROR, "warning": logging.WARNING,
    "warn": logging.WARNING, "info": logging.INFO,
    "debug": logging.DEBUG, "trace": logging.DEBUG,
}


class JsonFormatter(logging.Formatter):
    """logrus JSONFormatter-shaped lines: time/level/msg (+ exc)."""

    def format(self, record: logging.LogRecord) -> str:
        entry = {
            "time": self.formatTime(record, "%Y-%m-%dT%H:%M:%S%z"),
            "level": record.levelname.lower(),
            "msg": record.getMessage(),
            "logger": record.name,
        }
        if record.exc_info:
            entry["error"] = self.formatException(record.exc_info)
        return json.dumps(entry)


def setup_logging(cfg) -> None:
    level = _LEVELS.get(
        (cfg.get_string("logging.level") or "info").lower(),
        logging.INFO)
    fmt = (cfg.get_string("logging.format") or "text").lower()
    handler = logging.StreamHandler(sys.stderr)
    if fmt == "json":
        handler.setFormatter(JsonFormatter())
    else:
        handler.setFormatter(logging.Formatter(
            "%(asctime)s %(levelname)-7s %(name)s: %(message)s"))
    root = logging.getLogger()
    root.handlers = [handler]
    root.setLevel(level)
