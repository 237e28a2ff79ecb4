"""Test logger: captures records for assertions (reference testlog/testlog.go)."""

from __future__ import annotations

from typing import Any, List, Mapping, Tuple

from .level import Level
from .logger import Logger


class TestLogger(Logger):
    """Collects log records; optionally also prints via a delegate."""

    def __init__(self, delegate: Logger = None):
        self.records: List[Tuple[Level, str, dict]] = []
        self._delegate = delegate

    def log(self, level: Level, msg: str, fields: Mapping[str, Any]) -> None:
        self.records.append((level, msg, dict(fields)))
        if self._delegate is not None:
            self._delegate.log(level, msg, fields)

    def messages(self, level: Level = None) -> List[str]:
        return [m for (l, m, _) in self.records if level is None or l == level]
