"""Structured log formatting (reference parity: the slog ReformatHandler,
internal/logging/handler.go:27-49 — `ts LEVEL "msg" k=v` lines).

Python's logging lacks slog's key/value args, so structured fields ride in
``extra={"kv": {...}}`` and everything else formats identically:

    2026-09-13T22:10:11.123Z INFO  "cell started" cell=dev-a1b2c3 pid=412
"""
from __future__ import annotations

import logging
import time


class ReformatFormatter(logging.Formatter):
    def format(self, record: logging.LogRecord) -> str:
        ts = time.strftime("%Y-%m-%dT%H:%M:%S",
                           time.gmtime(record.created))
        ms = int(record.msecs)
        msg = record.getMessage().replace('"', "'")
        line = f'{ts}.{ms:03d}Z {record.levelname:<5} "{msg}"'
        kv = getattr(record, "kv", None)
        if isinstance(kv, dict):
            line += "".join(f" {k}={v}" for k, v in kv.items())
        if record.name not in ("root", "kukeon"):
            line += f" logger={record.name}"
        if record.exc_info:
            line += " exc=" + self.formatException(
                record.exc_info).replace("\n", " | ")
        return line


def setup(level: int = logging.INFO) -> None:
    """Install the structured formatter on the root logger (idempotent)."""
    root = logging.getLogger()
    root.setLevel(level)
    for h in root.handlers:
        if isinstance(h.formatter, ReformatFormatter):
            return
    h = logging.StreamHandler()
    h.setFormatter(ReformatFormatter())
    root.handlers = [h]
