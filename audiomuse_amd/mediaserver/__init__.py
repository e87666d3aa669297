"""Media-server adapters.

Reference: /root/reference/tasks/mediaserver/ — a uniform 19-call
surface (`_PUBLIC_SERVER_API`, __init__.py:56-62) dispatched to provider
modules (navidrome/jellyfin/emby/lyrion/plex), a registry with
provider<->canonical id translation (registry.py), and thread-local
server binding (context.py).

This build defines the same surface as a Provider ABC with:
- `synthetic`: a deterministic in-process provider generating WAV tracks
  (ops.audio_io.synthetic_track) — the test/bench source in an image
  with no network;
- `subsonic`: a Subsonic-API HTTP provider (Navidrome speaks Subsonic)
  as the real-network reference implementation; the remaining providers
  (jellyfin/emby/lyrion/plex) are HTTP adapters over the same ABC and
  register identically.
"""

from __future__ import annotations

import threading
from typing import Dict, List, Optional

from audiomuse_amd.mediaserver.base import Provider

_PROVIDERS: Dict[str, type] = {}
_LOCAL = threading.local()


def register_provider(name: str):
    def deco(cls):
        _PROVIDERS[name] = cls
        return cls
    return deco


def provider_types() -> List[str]:
    return sorted(_PROVIDERS)


def config_defaults(server_type: str) -> Dict:
    """Connection kwargs for `server_type` from the legacy single-server
    env/config surface (reference PARAMETERS.md "Mediaserver General":
    NAVIDROME_URL/JELLYFIN_TOKEN/... + MUSIC_LIBRARIES). Registry rows
    in music_servers override these per server."""
    from audiomuse_amd import config as C

    libs = [x.strip() for x in (C.MUSIC_LIBRARIES or "").split(",")
            if x.strip()]
    if server_type in ("navidrome", "subsonic"):
        return {"base_url": C.NAVIDROME_URL, "username": C.NAVIDROME_USER,
                "credential": C.NAVIDROME_API_KEY or C.NAVIDROME_PASSWORD,
                "music_folder_ids": libs}
    if server_type == "jellyfin":
        return {"base_url": C.JELLYFIN_URL, "credential": C.JELLYFIN_TOKEN,
                "user_id": C.JELLYFIN_USER_ID, "library_ids": libs}
    if server_type == "emby":
        return {"base_url": C.EMBY_URL, "credential": C.EMBY_TOKEN,
                "user_id": C.EMBY_USER_ID, "library_ids": libs}
    if server_type == "lyrion":
        return {"base_url": C.LYRION_URL, "target_paths": libs}
    if server_type == "plex":
        return {"base_url": C.PLEX_URL, "credential": C.PLEX_TOKEN,
                "section_ids": libs}
    return {}


def make_provider(server_type: str, **kwargs) -> Provider:
    cls = _PROVIDERS.get(server_type)
    if cls is None:
        raise ValueError(f"unsupported media server type {server_type!r} "
                         f"(supported: {', '.join(provider_types())})")
    merged = config_defaults(server_type)
    merged.update(kwargs)
    return cls(**merged)


class BoundServer:
    """Thread-local active-server binding (reference: context.py:35 +
    BoundServer __init__.py:392)."""

    def __init__(self, provider: Provider, server_id: str):
        self.provider = provider
        self.server_id = server_id

    def __enter__(self):
        stack = getattr(_LOCAL, "stack", None)
        if stack is None:
            stack = _LOCAL.stack = []
        stack.append(self)
        return self.provider

    def __exit__(self, *exc):
        _LOCAL.stack.pop()
        return False


def active_server() -> Optional[BoundServer]:
    stack = getattr(_LOCAL, "stack", None)
    return stack[-1] if stack else None


# register built-ins
from audiomuse_amd.mediaserver import synthetic  # noqa: E402,F401
from audiomuse_amd.mediaserver import subsonic  # noqa: E402,F401
from audiomuse_amd.mediaserver import jellyfin  # noqa: E402,F401
from audiomuse_amd.mediaserver import emby  # noqa: E402,F401
from audiomuse_amd.mediaserver import plex  # noqa: E402,F401
from audiomuse_amd.mediaserver import lyrion  # noqa: E402,F401
