"""Logging bootstrap.

Capability parity with /root/reference/pkg/utils/logger.go:31-234: configurable
level/format/output from config, dual sink (JSON file + colored console),
daily file naming (`opsagent-YYYYMMDD.log`) with size-based rotation, and a
date-rollover check on access. Uses stdlib logging + RotatingFileHandler.
"""

from __future__ import annotations

import datetime as _dt
import json
import logging
import logging.handlers
import os
import sys
import threading
from typing import Optional


class JsonFormatter(logging.Formatter):
    def format(self, record: logging.LogRecord) -> str:
        payload = {
            "ts": _dt.datetime.fromtimestamp(record.created).isoformat(),
            "level": record.levelname.lower(),
            "logger": record.name,
            "msg": record.getMessage(),
        }
        if record.exc_info:
            payload["exc"] = self.formatException(record.exc_info)
        return json.dumps(payload, ensure_ascii=False)


_COLORS = {"DEBUG": "\033[36m", "INFO": "\033[32m", "WARNING": "\033[33m", "ERROR": "\033[31m", "CRITICAL": "\033[35m"}
_RESET = "\033[0m"


class ConsoleFormatter(logging.Formatter):
    def __init__(self, color: bool = True):
        super().__init__()
        self.color = color

    def format(self, record: logging.LogRecord) -> str:
        lvl = record.levelname
        if self.color and sys.stderr.isatty():
            lvl = f"{_COLORS.get(lvl, '')}{lvl}{_RESET}"
        ts = _dt.datetime.fromtimestamp(record.created).strftime("%H:%M:%S.%f")[:-3]
        base = f"{ts} {lvl:<18s} {record.name}: {record.getMessage()}"
        if record.exc_info:
            base += "\n" + self.formatException(record.exc_info)
        return base


_lock = threading.Lock()
_configured_date: Optional[str] = None
_file_handler: Optional[logging.Handler] = None


def init_logging(
    level: str = "info",
    fmt: str = "console",
    output: str = "stderr",
    log_dir: str = "logs",
    max_bytes: int = 10 * 1024 * 1024,
    backups: int = 10,
) -> None:
    """Configure root logger. fmt: console|json; output: stderr|file|both."""
    global _configured_date, _file_handler
    with _lock:
        root = logging.getLogger("opsagent")
        root.setLevel(getattr(logging, level.upper(), logging.INFO))
        root.handlers.clear()
        root.propagate = False
        if output in ("stderr", "both"):
            h = logging.StreamHandler(sys.stderr)
            h.setFormatter(JsonFormatter() if fmt == "json" else ConsoleFormatter())
            root.addHandler(h)
        if output in ("file", "both"):
            os.makedirs(log_dir, exist_ok=True)
            today = _dt.date.today().strftime("%Y%m%d")
            path = os.path.join(log_dir, f"opsagent-{today}.log")
            fh = logging.handlers.RotatingFileHandler(path, maxBytes=max_bytes, backupCount=backups)
            fh.setFormatter(JsonFormatter())
            root.addHandler(fh)
            _file_handler = fh
            _configured_date = today


def get_logger(name: str = "opsagent") -> logging.Logger:
    """Get a logger; re-inits the file sink on date rollover (ref logger.go:70-98)."""
    global _configured_date
    if _configured_date is not None:
        today = _dt.date.today().strftime("%Y%m%d")
        if today != _configured_date:
            init_logging(output="file")
    lg = logging.getLogger(name if name.startswith("opsagent") else f"opsagent.{name}")
    if not logging.getLogger("opsagent").handlers:
        # lazy default config
        init_logging()
    return lg
