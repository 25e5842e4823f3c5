"""Structured logging for transfers and the proxy.

The reference logs with two fmt.Println hooks (start.go:197-204); here every
transfer gets a structured record (also the bench's measurement source).
"""

from __future__ import annotations

import logging
import os
import sys
import time

_LEVEL = os.environ.get("DEMODEL_LOG", "info").upper()


def get_logger(name: str) -> logging.Logger:
    logger = logging.getLogger(f"demodel.{name}")
    if not logging.getLogger("demodel").handlers:
        root = logging.getLogger("demodel")
        h = logging.StreamHandler(sys.stderr)
        h.setFormatter(logging.Formatter(
            "%(asctime)s %(name)s %(levelname)s %(message)s"))
        root.addHandler(h)
        root.setLevel(getattr(logging, _LEVEL, logging.INFO))
        root.propagate = False
    return logger


class TransferLog:
    """Accumulates per-transfer byte counts / timings for metrics."""

    def __init__(self):
        self.records: list[dict] = []

    def record(self, **kw) -> None:
        kw.setdefault("t", time.time())
        self.records.append(kw)

    def total_bytes(self) -> int:
        return sum(r.get("bytes", 0) for r in self.records)
