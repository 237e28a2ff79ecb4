"""Misc process/device helpers (reference pkg/oim-common/util.go,
cmdmonitor.go, logging.go)."""

from __future__ import annotations

import fcntl
import os
import struct
import subprocess
import threading
from typing import Callable, Optional

from ..log import from_context

BLKGETSIZE64 = 0x80081272  # ioctl: u64 device size in bytes


def get_blk_size64(path: str) -> int:
    """Size of a block device in bytes (reference util.go:15-30)."""
    fd = os.open(path, os.O_RDONLY)
    try:
        buf = fcntl.ioctl(fd, BLKGETSIZE64, b"\x00" * 8)
        return struct.unpack("Q", buf)[0]
    finally:
        os.close(fd)


class CmdMonitor:
    """Watches a child process and invokes a callback when it dies
    (reference cmdmonitor.go:23-51: inherited-pipe death detection;
    Python's subprocess lets us wait directly)."""

    def __init__(self, process: subprocess.Popen,
                 on_exit: Optional[Callable[[int], None]] = None):
        self.process = process
        self.on_exit = on_exit
        self._thread = threading.Thread(target=self._watch, daemon=True)
        self._thread.start()

    def _watch(self) -> None:
        code = self.process.wait()
        from_context().warn("monitored process exited",
                            pid=self.process.pid, code=code)
        if self.on_exit is not None:
            self.on_exit(code)


class LogWriter:
    """Line-buffered file-like object forwarding to a logger
    (reference logging.go:19-47); use as stderr sink for children."""

    def __init__(self, logger=None, prefix: str = ""):
        self._logger = logger
        self._prefix = prefix
        self._buffer = ""

    def write(self, text: str) -> int:
        self._buffer += text
        while "\n" in self._buffer:
            line, self._buffer = self._buffer.split("\n", 1)
            logger = self._logger or from_context()
            logger.info("%s%s", self._prefix, line)
        return len(text)

    def flush(self) -> None:
        if self._buffer:
            logger = self._logger or from_context()
            logger.info("%s%s", self._prefix, self._buffer)
            self._buffer = ""
