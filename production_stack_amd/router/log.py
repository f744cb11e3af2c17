"""Logging: colored text / JSON formatters, TRACE level, secret redaction.

Parity: reference src/vllm_router/log.py (JsonFormatter :81, redaction
filter :133-155, init_logger :194).
"""

from __future__ import annotations

import json
import logging
import re
import sys

TRACE = 5
logging.addLevelName(TRACE, "TRACE")

_REDACT_KEYS = re.compile(
    r"(api[-_]?key|authorization|token|secret|password)", re.IGNORECASE
)
_BEARER = re.compile(r"(Bearer\s+)[A-Za-z0-9._\-]+")


def redact(text: str) -> str:
    text = _BEARER.sub(r"\1[REDACTED]", text)

    def _kv(m: re.Match) -> str:
        return f"{m.group(1)}=[REDACTED]"

    return re.sub(
        r"(" + _REDACT_KEYS.pattern + r")\s*[=:]\s*[^\s,;]+",
        _kv,
        text,
        flags=re.IGNORECASE,
    )


class RedactionFilter(logging.Filter):
    def filter(self, record: logging.LogRecord) -> bool:
        try:
            msg = record.getMessage()
            red = redact(msg)
            if red != msg:
                record.msg = red
                record.args = ()
        except Exception:
            pass
        return True


class JsonFormatter(logging.Formatter):
    def format(self, record: logging.LogRecord) -> str:
        out = {
            "time": self.formatTime(record),
            "level": record.levelname,
            "logger": record.name,
            "message": record.getMessage(),
        }
        if record.exc_info:
            out["exc_info"] = self.formatException(record.exc_info)
        return json.dumps(out)


class ColorFormatter(logging.Formatter):
    COLORS = {
        "DEBUG": "\033[36m",
        "INFO": "\033[32m",
        "WARNING": "\033[33m",
        "ERROR": "\033[31m",
        "CRITICAL": "\033[35m",
    }
    RESET = "\033[0m"

    def format(self, record: logging.LogRecord) -> str:
        color = self.COLORS.get(record.levelname, "")
        base = super().format(record)
        return f"{color}{base}{self.RESET}" if sys.stderr.isatty() else base


def init_logger(
    level: str = "info", fmt: str = "text"
) -> logging.Logger:
    root = logging.getLogger()
    root.handlers.clear()
    handler = logging.StreamHandler()
    if fmt == "json":
        handler.setFormatter(JsonFormatter())
    else:
        handler.setFormatter(
            ColorFormatter("[%(asctime)s] %(levelname)s %(name)s: %(message)s")
        )
    handler.addFilter(RedactionFilter())
    root.addHandler(handler)
    lvl = TRACE if level == "trace" else getattr(
        logging, level.upper(), logging.INFO
    )
    root.setLevel(lvl)
    return root
