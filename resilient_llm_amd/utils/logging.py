"""Shared console-logging idioms.

The reference duplicates these per-script: an ANSI-colored
ms-precision logger (reference src/demo_cris.py:26-39 and four copies) and a
``sanitize_error`` mapping raw exceptions to a small set of safe categories
(src/demo_fallback.py:32-43 and three copies).  Here they live once.
"""

from __future__ import annotations

import datetime
import os
import sys

_COLORS = {
    "grey": "\033[90m",
    "red": "\033[91m",
    "green": "\033[92m",
    "yellow": "\033[93m",
    "blue": "\033[94m",
    "magenta": "\033[95m",
    "cyan": "\033[96m",
    "white": "\033[97m",
    "reset": "\033[0m",
    "bold": "\033[1m",
}


def _use_color() -> bool:
    if os.environ.get("NO_COLOR"):
        return False
    return sys.stdout.isatty() or bool(os.environ.get("FORCE_COLOR"))


def log_with_timestamp(message: str, color: str | None = None) -> None:
    """Print ``[HH:MM:SS.mmm] message``, optionally ANSI-colored."""
    ts = datetime.datetime.now().strftime("%H:%M:%S.%f")[:-3]
    if color and color in _COLORS and _use_color():
        print(f"{_COLORS['grey']}[{ts}]{_COLORS['reset']} "
              f"{_COLORS[color]}{message}{_COLORS['reset']}", flush=True)
    else:
        print(f"[{ts}] {message}", flush=True)


def colorize(message: str, color: str) -> str:
    if color in _COLORS and _use_color():
        return f"{_COLORS[color]}{message}{_COLORS['reset']}"
    return message


def sanitize_error(err: BaseException | str) -> str:
    """Map an exception to one of a few safe, comparable categories.

    Mirrors the reference's taxonomy (src/demo_fallback.py:32-43;
    throttling detection src/demo_cris.py:261-283): rate-limit, timeout,
    connection, authentication, everything else.
    """
    text = str(err).lower()
    name = type(err).__name__.lower() if isinstance(err, BaseException) else ""
    if any(k in text or k in name for k in
           ("ratelimit", "rate limit", "throttl", "quota", "429", "too many requests")):
        return "Rate limit or quota exceeded"
    if any(k in text or k in name for k in ("timeout", "timed out")):
        return "Request timeout"
    if any(k in text or k in name for k in
           ("connection", "connect", "refused", "unreachable", "reset by peer")):
        return "Connection error"
    if any(k in text or k in name for k in
           ("auth", "credential", "forbidden", "401", "403", "api key", "apikey")):
        return "Authentication error"
    if any(k in text or k in name for k in ("worker", "backend", "gpu", "hip", "device")):
        return "Backend error"
    return "Service error"
