"""Error subsystem: numeric codes + classifier + AudioMuseError.

Reference: /root/reference/error/ (417 LoC; docs/ERROR_CODES.md) —
a numeric code dictionary, an exception classifier, one-line messages
to the UI, tracebacks only to logs (error_manager.py:112-177).
"""

from __future__ import annotations

import logging
from typing import Tuple

logger = logging.getLogger(__name__)

# numeric code dictionary (categories mirror docs/ERROR_CODES.md)
E_UNKNOWN = 1000
E_DB = 1100
E_DB_LOCKED = 1101
E_QUEUE = 1200
E_QUEUE_CANCELLED = 1201
E_MEDIA_SERVER = 1300
E_MEDIA_UNREACHABLE = 1301
E_MEDIA_AUTH = 1302
E_AUDIO_DECODE = 1400
E_MODEL = 1500
E_MODEL_OOM = 1501
E_HIP = 1502
E_INDEX = 1600
E_INDEX_MISSING = 1601
E_CONFIG = 1700
E_AUTH = 1800

MESSAGES = {
    E_UNKNOWN: "An unexpected error occurred.",
    E_DB: "Database error.",
    E_DB_LOCKED: "Database is busy; retry shortly.",
    E_QUEUE: "Task queue error.",
    E_QUEUE_CANCELLED: "Task was cancelled.",
    E_MEDIA_SERVER: "Media server error.",
    E_MEDIA_UNREACHABLE: "Media server is unreachable.",
    E_MEDIA_AUTH: "Media server rejected the credentials.",
    E_AUDIO_DECODE: "Could not decode the audio file.",
    E_MODEL: "Model inference failed.",
    E_MODEL_OOM: "GPU ran out of memory; the job will retry smaller.",
    E_HIP: "GPU kernel error.",
    E_INDEX: "Similarity index error.",
    E_INDEX_MISSING: "Similarity index is not built yet.",
    E_CONFIG: "Configuration error.",
    E_AUTH: "Authentication failed.",
}


class AudioMuseError(RuntimeError):
    def __init__(self, code: int, detail: str = ""):
        self.code = code
        self.detail = detail
        super().__init__(f"[{code}] {MESSAGES.get(code, '')} {detail}".strip())

    @property
    def user_message(self) -> str:
        return MESSAGES.get(self.code, MESSAGES[E_UNKNOWN])


def classify_exception(exc: BaseException) -> int:
    """Map arbitrary exceptions onto numeric codes (error classifier)."""
    if isinstance(exc, AudioMuseError):
        return exc.code
    name = type(exc).__name__
    text = str(exc).lower()
    import sqlite3

    if isinstance(exc, sqlite3.OperationalError):
        return E_DB_LOCKED if "locked" in text else E_DB
    if isinstance(exc, sqlite3.Error):
        return E_DB
    if "out of memory" in text or "hip_error_out_of_memory" in text:
        return E_MODEL_OOM
    if "hip" in text and "error" in text:
        return E_HIP
    if name in ("ConnectionError", "Timeout", "ConnectTimeout", "ReadTimeout"):
        return E_MEDIA_UNREACHABLE
    if "401" in text or "unauthorized" in text:
        return E_MEDIA_AUTH
    if name in ("EOFError", "wave.Error") or "wav" in text:
        return E_AUDIO_DECODE
    return E_UNKNOWN


def report_error(exc: BaseException, context: str = "") -> Tuple[int, str]:
    """Log the traceback; return (code, one-line user message)
    (error_manager.py behavior: tracebacks never reach the UI)."""
    code = classify_exception(exc)
    logger.exception("[%s] %s (code %d)", context, exc, code)
    return code, MESSAGES.get(code, MESSAGES[E_UNKNOWN])
