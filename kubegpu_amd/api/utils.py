"""Leveled logging + small helpers (KubeDevice-API `utils` parity).

Reference call sites: utils.Logf/Errorf/Logb (gpu.go:62,107,125),
utils.SortedStringKeys (gpuplugintypes/typeutils.go:66-71).
Levels follow the reference's usage: 0 = errors/always, 3-5 = debug detail.
"""

from __future__ import annotations

import logging
import os
import sys
from typing import List, Mapping

class _JsonFormatter(logging.Formatter):
    """KUBEGPU_LOG_JSON=1: one JSON object per line (log aggregators)."""

    def format(self, record: logging.LogRecord) -> str:
        import json

        return json.dumps({
            "ts": self.formatTime(record, "%Y-%m-%dT%H:%M:%S"),
            "level": record.levelname.lower(),
            "logger": "kubegpu_amd",
            "msg": record.getMessage(),
        })


_logger = logging.getLogger("kubegpu_amd")
if not _logger.handlers:
    _h = logging.StreamHandler(sys.stderr)
    if os.environ.get("KUBEGPU_LOG_JSON"):
        _h.setFormatter(_JsonFormatter())
    else:
        _h.setFormatter(logging.Formatter("%(asctime)s kubegpu_amd %(message)s"))
    _logger.addHandler(_h)
    _logger.setLevel(logging.INFO)

# Verbosity threshold, like glog -v.  Messages with level <= verbosity print.
_verbosity = int(os.environ.get("KUBEGPU_AMD_VERBOSITY", "1"))


def set_verbosity(v: int) -> None:
    global _verbosity
    _verbosity = v


def get_verbosity() -> int:
    return _verbosity


def logf(level: int, fmt: str, *args) -> None:
    """utils.Logf(level, fmt, ...)."""
    if level <= _verbosity:
        _logger.info(fmt % args if args else fmt)


def errorf(fmt: str, *args) -> None:
    """utils.Errorf(fmt, ...)."""
    _logger.error(fmt % args if args else fmt)


def logb(level: int) -> bool:
    """utils.Logb(level): true when messages at *level* would be emitted."""
    return level <= _verbosity


def sorted_string_keys(m: Mapping[str, object]) -> List[str]:
    """utils.SortedStringKeys: deterministic iteration order over maps."""
    return sorted(m.keys())

