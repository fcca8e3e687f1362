"""Leveled structured logging.

Mirrors the reference logging layer (`pkg/gofr/logging/logger.go`,
`pkg/gofr/logging/level.go`): levels DEBUG < INFO < NOTICE < WARN < ERROR
< FATAL; JSON-lines output when the destination is not a terminal (ERROR
and above go to stderr, the rest to stdout — logger.go:43-72); colorized
pretty-printing with per-record-type formats when it is a terminal
(logger.go:106-131).

Typed records (RequestLog, sql Log, redis QueryLog, service Log/ErrorLog,
RPCLog) are plain dataclass-like objects with a `pretty()` method; the
logger pretty-prints them on a TTY and JSON-encodes them otherwise, the
same dual behavior as logger.go:106-131.
"""

from __future__ import annotations

import io
import json
import sys
import threading
import time
from typing import Any

# Levels — reference: pkg/gofr/logging/level.go:8-16
DEBUG = 0
INFO = 1
NOTICE = 2
WARN = 3
ERROR = 4
FATAL = 5

_LEVEL_NAMES = {
    DEBUG: "DEBUG",
    INFO: "INFO",
    NOTICE: "NOTICE",
    WARN: "WARN",
    ERROR: "ERROR",
    FATAL: "FATAL",
}

# ANSI 256-color codes per level — reference: level.go color mapping
# (DEBUG grey, INFO blue, NOTICE cyan, WARN yellow, ERROR/FATAL red).
_LEVEL_COLORS = {
    DEBUG: 8,
    INFO: 74,
    NOTICE: 45,
    WARN: 220,
    ERROR: 160,
    FATAL: 160,
}


def level_from_string(s: str) -> int:
    """Reference: logging/level.go GetLevelFromString (case-insensitive,
    unknown -> INFO)."""
    return {
        "DEBUG": DEBUG,
        "INFO": INFO,
        "NOTICE": NOTICE,
        "WARN": WARN,
        "ERROR": ERROR,
        "FATAL": FATAL,
    }.get((s or "").upper(), INFO)


def level_name(level: int) -> str:
    return _LEVEL_NAMES.get(level, "INFO")


def color_for_status_code(status: int) -> int:
    """Reference: logging/logger.go:134-151 — 2xx green(34), 4xx yellow(220),
    5xx red(160), else white(37)."""
    if 200 <= status < 300:
        return 34
    if 400 <= status < 500:
        return 220
    if 500 <= status < 600:
        return 160
    return 37


def _is_terminal(stream) -> bool:
    # Reference: logging/logger.go:176-183 checkIfTerminal
    try:
        return stream.isatty()
    except Exception:
        return False


class Logger:
    """Leveled logger with the reference's dual JSON/pretty output.

    Reference: pkg/gofr/logging/logger.go:19-72. `logf` routes ERROR+ to
    stderr and the rest to stdout, stamps epoch time and level, and either
    pretty-prints (TTY) or JSON-encodes the record.
    """

    def __init__(self, level: int = INFO, out=None, err=None,
                 force_json: bool | None = None):
        self.level = level
        self._out = out if out is not None else sys.stdout
        self._err = err if err is not None else sys.stderr
        self._lock = threading.Lock()
        if force_json is None:
            self._pretty = _is_terminal(self._out)
        else:
            self._pretty = not force_json

    # -- core ---------------------------------------------------------------
    def logf(self, level: int, fmt: str, *args: Any) -> None:
        if level < self.level:
            return
        message: Any
        if args:
            if "%" in fmt:
                try:
                    message = fmt % args
                except (TypeError, ValueError):
                    message = " ".join([fmt] + [str(a) for a in args])
            else:
                message = " ".join([fmt] + [str(a) for a in args])
        else:
            message = fmt
        self._emit(level, message)

    def log_record(self, level: int, record: Any) -> None:
        """Log a typed record (RequestLog / QueryLog / ...)."""
        if level < self.level:
            return
        self._emit(level, record)

    def _emit(self, level: int, message: Any) -> None:
        # stream selection — reference: logger.go:60-64
        stream = self._err if level >= ERROR else self._out
        now = time.time()
        if self._pretty:
            line = self._pretty_line(level, message, now)
        else:
            payload = {
                "level": level_name(level),
                "time": now,
                "message": self._jsonable(message),
            }
            line = json.dumps(payload, default=str)
        with self._lock:
            try:
                stream.write(line + "\n")
                stream.flush()
            except ValueError:
                pass  # closed stream during interpreter shutdown

    @staticmethod
    def _jsonable(message: Any) -> Any:
        if hasattr(message, "to_dict"):
            return message.to_dict()
        if hasattr(message, "__dict__") and not isinstance(message, str):
            return dict(message.__dict__)
        return message

    def _pretty_line(self, level: int, message: Any, now: float) -> str:
        # Reference: logger.go:106-131 prettyPrint — level colored + time +
        # per-record-type format (records provide pretty()).
        color = _LEVEL_COLORS.get(level, 37)
        ts = time.strftime("%H:%M:%S", time.localtime(now))
        head = f"[38;5;{color}m{level_name(level):<6}[0m [{ts}] "
        if hasattr(message, "pretty"):
            return head + message.pretty()
        return head + str(message)

    # -- public leveled API — reference: logger.go:19-28 (Logger interface) --
    def Debug(self, *args: Any) -> None:
        self.logf(DEBUG, *self._spread(args))

    def Debugf(self, fmt: str, *args: Any) -> None:
        self.logf(DEBUG, fmt, *args)

    def Info(self, *args: Any) -> None:
        self.logf(INFO, *self._spread(args))

    def Infof(self, fmt: str, *args: Any) -> None:
        self.logf(INFO, fmt, *args)

    def Notice(self, *args: Any) -> None:
        self.logf(NOTICE, *self._spread(args))

    def Noticef(self, fmt: str, *args: Any) -> None:
        self.logf(NOTICE, fmt, *args)

    def Warn(self, *args: Any) -> None:
        self.logf(WARN, *self._spread(args))

    def Warnf(self, fmt: str, *args: Any) -> None:
        self.logf(WARN, fmt, *args)

    def Error(self, *args: Any) -> None:
        self.logf(ERROR, *self._spread(args))

    def Errorf(self, fmt: str, *args: Any) -> None:
        self.logf(ERROR, fmt, *args)

    def Fatal(self, *args: Any) -> None:
        self.logf(FATAL, *self._spread(args))
        raise SystemExit(1)

    def Fatalf(self, fmt: str, *args: Any) -> None:
        self.logf(FATAL, fmt, *args)
        raise SystemExit(1)

    @staticmethod
    def _spread(args):
        if len(args) == 1:
            a = args[0]
            if isinstance(a, str):
                return (a,)
            return ("%s", a) if not hasattr(a, "pretty") else ("", a)
        if not args:
            return ("",)
        return (" ".join(["%s"] * len(args)),) + args

    # typed-record entry points, used by middlewares/datasources
    def debug_record(self, record: Any) -> None:
        self.log_record(DEBUG, record)

    def info_record(self, record: Any) -> None:
        self.log_record(INFO, record)

    # pythonic aliases
    debug = Debugf
    info = Infof
    notice = Noticef
    warn = Warnf
    error = Errorf


def NewLogger(level: int = INFO, **kw) -> Logger:
    """Reference: logging/logger.go:153-164."""
    return Logger(level=level, **kw)


def NewSilentLogger() -> Logger:
    """Reference: logging/logger.go:167-174 — discards everything."""
    sink = io.StringIO()
    return Logger(level=FATAL + 1, out=sink, err=sink, force_json=True)


def new_logger_from_env(config) -> Logger:
    """LOG_LEVEL-driven constructor — reference: container/container.go:42."""
    return NewLogger(level_from_string(config.Get("LOG_LEVEL")))
