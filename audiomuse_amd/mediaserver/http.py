"""Shared HTTP plumbing for media-server adapters.

Reference analog: /root/reference/tasks/mediaserver/http.py plus the
per-provider request helpers (navidrome.py:161-220 `_navidrome_request_ex`
retry/redaction, jellyfin.py:310-359 paged fetches). Centralised here so
every provider gets the same retry, timeout, error-classification and
secret-redaction behavior.
"""

from __future__ import annotations

import logging
import re
import time
from typing import Callable, Dict, Iterator, List, Optional

from audiomuse_amd import config as C
from audiomuse_amd.utils.errors import (E_MEDIA_AUTH, E_MEDIA_SERVER,
                                        E_MEDIA_UNREACHABLE, AudioMuseError)

logger = logging.getLogger(__name__)

_SECRET_PARAMS = re.compile(
    r"((?:token|t|s|p|password|apikey|api_key|X-Plex-Token)=)[^&\s]+",
    re.IGNORECASE)


def redact(text: str) -> str:
    """Strip credential query params before a URL reaches a log line."""
    return _SECRET_PARAMS.sub(r"\1<redacted>", str(text))


class MediaHttp:
    """requests.Session wrapper: bounded retry with backoff on transient
    failures (connection errors, 429, 5xx), typed AudioMuseError on
    terminal ones, redacted logging."""

    def __init__(self, session=None, timeout: Optional[float] = None,
                 download_timeout: Optional[float] = None,
                 retries: Optional[int] = None,
                 backoff: Optional[float] = None):
        if session is None:
            import requests
            session = requests.Session()
        self.http = session
        self.timeout = timeout if timeout is not None else \
            getattr(C, "MEDIASERVER_TIMEOUT_SECONDS", 30.0)
        self.download_timeout = download_timeout if download_timeout is not None \
            else getattr(C, "MEDIASERVER_DOWNLOAD_TIMEOUT_SECONDS", 300.0)
        self.retries = retries if retries is not None else \
            getattr(C, "MEDIASERVER_RETRIES", 3)
        self.backoff = backoff if backoff is not None else \
            getattr(C, "MEDIASERVER_RETRY_BACKOFF_SECONDS", 1.0)

    def request(self, method: str, url: str, *, params: Optional[Dict] = None,
                headers: Optional[Dict] = None, json_body=None,
                data=None, stream: bool = False,
                timeout: Optional[float] = None):
        """One logical request; retries transparently. Raises
        AudioMuseError(E_MEDIA_*) when the server stays unreachable,
        rejects credentials, or keeps failing."""
        last_exc: Optional[Exception] = None
        tmo = timeout if timeout is not None else self.timeout
        for attempt in range(self.retries + 1):
            if attempt:
                time.sleep(self.backoff * (2 ** (attempt - 1)))
            try:
                r = self.http.request(method, url, params=params,
                                      headers=headers, json=json_body,
                                      data=data, stream=stream, timeout=tmo)
            except Exception as exc:  # connection/timeout family
                last_exc = exc
                logger.warning("media request failed (%s, attempt %d/%d): %s",
                               redact(url), attempt + 1, self.retries + 1,
                               redact(exc))
                continue
            if r.status_code in (401, 403):
                raise AudioMuseError(
                    E_MEDIA_AUTH,
                    f"credentials rejected by {redact(url)} "
                    f"(HTTP {r.status_code})")
            if r.status_code == 429 or r.status_code >= 500:
                last_exc = AudioMuseError(
                    E_MEDIA_SERVER,
                    f"{redact(url)} returned HTTP {r.status_code}")
                logger.warning("media server busy/erroring "
                               "(HTTP %d, attempt %d/%d): %s",
                               r.status_code, attempt + 1, self.retries + 1,
                               redact(url))
                continue
            if r.status_code >= 400:
                raise AudioMuseError(
                    E_MEDIA_SERVER,
                    f"{redact(url)} returned HTTP {r.status_code}")
            return r
        if isinstance(last_exc, AudioMuseError):
            raise last_exc
        raise AudioMuseError(
            E_MEDIA_UNREACHABLE,
            f"{redact(url)} unreachable after {self.retries + 1} attempts: "
            f"{redact(last_exc)}")

    def get(self, url: str, **kw):
        return self.request("GET", url, **kw)

    def post(self, url: str, **kw):
        return self.request("POST", url, **kw)

    def delete(self, url: str, **kw):
        return self.request("DELETE", url, **kw)


def paged(fetch_page: Callable[[int, int], List],
          page_size: Optional[int] = None,
          limit: int = 0) -> Iterator:
    """Drive a (start_index, page_size) -> items pager to exhaustion
    (reference: jellyfin.py:158-238 / plex.py:178-204 page loops).
    Stops on a short page, an empty page, or ``limit`` items."""
    size = page_size or getattr(C, "MEDIASERVER_PAGE_SIZE", 500)
    start = 0
    yielded = 0
    while True:
        items = fetch_page(start, size)
        if not items:
            return
        for it in items:
            yield it
            yielded += 1
            if limit and yielded >= limit:
                return
        if len(items) < size:
            return
        start += len(items)
