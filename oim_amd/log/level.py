"""Log levels (reference pkg/log/level/level.go:42-70)."""

import enum


class Level(enum.IntEnum):
    DEBUG = 10
    INFO = 20
    WARN = 30
    ERROR = 40
    FATAL = 50
    PANIC = 60

    def __str__(self) -> str:  # lower-case, as the reference prints them
        return self.name


_NAMES = {l.name.lower(): l for l in Level}
# Accept common aliases.
_NAMES["warning"] = Level.WARN
_NAMES["err"] = Level.ERROR


def parse_level(text: str) -> Level:
    """Parse a level name; raises ValueError on unknown names."""
    try:
        return _NAMES[text.strip().lower()]
    except KeyError:
        raise ValueError(f"unknown log level: {text!r}") from None
