"""Logger core: interface, plain-text formatter, simple logger, context.

Counterpart of the reference's pkg/log:
  - Logger interface with leveled calls and field chaining (log.go:37-110)
  - global logger L() / set_logger (log.go:126-137)
  - from_context/with_logger/with_fields (log.go:163-191)
  - plain-text Formatter with special keys time/at/level (formatter.go:18-82)
  - SimpleLogger + ``-log.level`` style flag hook (simple.go:19-131)
"""

from __future__ import annotations

import contextvars
import datetime
import io
import sys
import threading
from typing import Any, Mapping, Optional, TextIO

from .level import Level, parse_level


class FatalError(SystemExit):
    """Raised by Logger.fatal (reference: os.Exit via Fatal)."""


class PanicError(RuntimeError):
    """Raised by Logger.panic."""


class Formatter:
    """Plain-text formatter.

    Renders ``time level msg key=value ...``; the special keys ``time``,
    ``at`` and ``level`` are placed first when present (reference
    formatter.go:18-82).
    """

    def __init__(self, show_time: bool = True):
        self.show_time = show_time

    def format(self, level: Level, msg: str, fields: Mapping[str, Any]) -> str:
        out = io.StringIO()
        if self.show_time:
            ts = fields.get("time") or datetime.datetime.now().strftime(
                "%Y-%m-%d %H:%M:%S.%f"
            )[:-3]
            out.write(f"{ts} ")
        at = fields.get("at")
        if at is not None:
            out.write(f"{at} ")
        out.write(f"{str(level).upper():5s} {msg}")
        for key in sorted(fields):
            if key in ("time", "at", "level"):
                continue
            value = fields[key]
            text = repr(value) if isinstance(value, str) and " " in value else str(value)
            out.write(f" {key}={text}")
        return out.getvalue()


class Logger:
    """Abstract structured logger.

    Concrete loggers implement :meth:`log`; everything else is derived
    (the reference's LoggerBase embedding helper, helper.go).
    """

    def log(self, level: Level, msg: str, fields: Mapping[str, Any]) -> None:
        raise NotImplementedError

    def with_fields(self, **fields: Any) -> "Logger":
        return _FieldLogger(self, fields)

    # Leveled helpers; *args are %-format arguments like the reference's
    # Debugf family, **fields are structured key/values (Debugw family).
    def _emit(self, level: Level, msg: str, args: tuple, fields: dict) -> None:
        if args:
            msg = msg % args
        self.log(level, msg, fields)

    def debug(self, msg: str, *args: Any, **fields: Any) -> None:
        self._emit(Level.DEBUG, msg, args, fields)

    def info(self, msg: str, *args: Any, **fields: Any) -> None:
        self._emit(Level.INFO, msg, args, fields)

    def warn(self, msg: str, *args: Any, **fields: Any) -> None:
        self._emit(Level.WARN, msg, args, fields)

    def error(self, msg: str, *args: Any, **fields: Any) -> None:
        self._emit(Level.ERROR, msg, args, fields)

    def fatal(self, msg: str, *args: Any, **fields: Any) -> None:
        self._emit(Level.FATAL, msg, args, fields)
        raise FatalError(1)

    def panic(self, msg: str, *args: Any, **fields: Any) -> None:
        self._emit(Level.PANIC, msg, args, fields)
        raise PanicError(msg % args if args else msg)


class _FieldLogger(Logger):
    def __init__(self, parent: Logger, fields: Mapping[str, Any]):
        self._parent = parent
        self._fields = dict(fields)

    def log(self, level: Level, msg: str, fields: Mapping[str, Any]) -> None:
        merged = dict(self._fields)
        merged.update(fields)
        self._parent.log(level, msg, merged)


class SimpleLogger(Logger):
    """Threshold-filtered logger writing formatted lines to a stream."""

    def __init__(
        self,
        level: Level = Level.INFO,
        output: Optional[TextIO] = None,
        formatter: Optional[Formatter] = None,
    ):
        self.level = level
        # None = "current sys.stderr", resolved per log call so stream
        # redirection (tests, daemons re-opening stderr) is honored.
        self.output = output
        self.formatter = formatter or Formatter()
        self._mutex = threading.Lock()

    def log(self, level: Level, msg: str, fields: Mapping[str, Any]) -> None:
        if level < self.level:
            return
        line = self.formatter.format(level, msg, fields)
        stream = self.output if self.output is not None else sys.stderr
        with self._mutex:
            try:
                print(line, file=stream, flush=True)
            except ValueError:
                pass  # stream closed under us (interpreter shutdown)


class NullLogger(Logger):
    def log(self, level: Level, msg: str, fields: Mapping[str, Any]) -> None:
        pass


_global_lock = threading.Lock()
_global: Logger = SimpleLogger()


def L() -> Logger:
    """The process-global logger (reference log.go:126-130)."""
    return _global


def set_logger(logger: Logger) -> Logger:
    """Replace the global logger, returning the previous one."""
    global _global
    with _global_lock:
        previous, _global = _global, logger
    return previous


_context_logger: contextvars.ContextVar[Optional[Logger]] = contextvars.ContextVar(
    "oim_amd_logger", default=None
)


def from_context() -> Logger:
    """Logger attached to the current context, else the global one."""
    logger = _context_logger.get()
    return logger if logger is not None else _global


def with_logger(logger: Logger) -> contextvars.Token:
    """Attach ``logger`` to the current context; returns a reset token."""
    return _context_logger.set(logger)


def with_fields(**fields: Any) -> contextvars.Token:
    """Attach a derived logger with extra fields to the current context."""
    return _context_logger.set(from_context().with_fields(**fields))


def add_flags(parser) -> None:
    """Register ``--log.level`` on an argparse parser (simple.go:26-41)."""
    parser.add_argument(
        "--log.level",
        dest="log_level",
        default="info",
        type=parse_level,
        help="log threshold: debug, info, warn, error",
    )


def init_from_args(args) -> None:
    level = getattr(args, "log_level", Level.INFO)
    set_logger(SimpleLogger(level=level))
