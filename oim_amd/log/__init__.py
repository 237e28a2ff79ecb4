"""Context-attached structured logging.

MI355X-native counterpart of the reference's pkg/log (log.go:37-191,
simple.go:19-131, formatter.go:18-82): a small Logger interface with
leveled calls, key/value structured fields via with_fields(), a process
global logger, and attachment of loggers to call contexts.  Python has no
implicit Context, so the "context" here is contextvars-based and flows
through asyncio/threads started with copy_context, plus explicit
``logger=`` plumbing where needed.
"""

from .level import Level, parse_level
from . import logger
from .logger import (
    Logger,
    SimpleLogger,
    Formatter,
    L,
    set_logger,
    from_context,
    with_logger,
    with_fields,
    add_flags,
    init_from_args,
)
from .testlog import TestLogger

__all__ = [
    "Level",
    "parse_level",
    "Logger",
    "SimpleLogger",
    "Formatter",
    "L",
    "set_logger",
    "from_context",
    "with_logger",
    "with_fields",
    "add_flags",
    "init_from_args",
    "TestLogger",
]
