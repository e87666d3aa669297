"""Logging + sanitization.

Reference: /root/reference/app_logging.py (LogSanitizingFilter :86 —
log-injection-safe newline stripping) + sanitization.py (:36
sanitize_for_log / DB+JSON sanitizers) + ssrf_guard.py (outbound URL
validation).
"""

from __future__ import annotations

import ipaddress
import logging
import re
import sys
from urllib.parse import urlparse

_CTRL = re.compile(r"[\x00-\x1f\x7f]")


def sanitize_for_log(value) -> str:
    """Strip control chars / newlines so user input can't forge log lines
    (sanitization.py:36)."""
    s = str(value)
    s = _CTRL.sub(" ", s)
    if len(s) > 500:
        s = s[:500] + "..."
    return s


class LogSanitizingFilter(logging.Filter):
    """Applies sanitize_for_log to interpolated args (app_logging.py:86)."""

    def filter(self, record: logging.LogRecord) -> bool:
        if record.args:
            try:
                if isinstance(record.args, tuple):
                    record.args = tuple(
                        sanitize_for_log(a) if isinstance(a, str) else a
                        for a in record.args)
            except Exception:
                pass
        if isinstance(record.msg, str) and ("\n" in record.msg or "\r" in record.msg):
            record.msg = sanitize_for_log(record.msg)
        return True


def configure_logging(level: int = logging.INFO) -> None:
    """reference: app_logging.configure_logging :114"""
    root = logging.getLogger()
    if any(isinstance(f, LogSanitizingFilter)
           for h in root.handlers for f in h.filters):
        return
    handler = logging.StreamHandler(sys.stderr)
    handler.setFormatter(logging.Formatter(
        "%(asctime)s %(levelname)s %(name)s: %(message)s"))
    handler.addFilter(LogSanitizingFilter())
    root.addHandler(handler)
    root.setLevel(level)


# -- SSRF guard (ssrf_guard.py:58) ------------------------------------------

_BLOCKED_SCHEMES = {"file", "ftp", "gopher", "dict"}


def validate_outbound_url(url: str, allow_private: bool = False) -> bool:
    """Reject URLs targeting internal networks or non-HTTP schemes."""
    try:
        parsed = urlparse(url)
    except Exception:
        return False
    if parsed.scheme not in ("http", "https"):
        return False
    host = parsed.hostname or ""
    if not host:
        return False
    if allow_private:
        return True
    if host in ("localhost",) or host.endswith(".local"):
        return False
    try:
        ip = ipaddress.ip_address(host)
        return not (ip.is_private or ip.is_loopback or ip.is_link_local
                    or ip.is_reserved or ip.is_multicast)
    except ValueError:
        return True  # hostname: resolution-time checks are the provider's job
