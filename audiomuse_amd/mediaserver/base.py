"""Provider ABC: the uniform media-server call surface.

Mirrors the reference's 19-function `_PUBLIC_SERVER_API`
(/root/reference/tasks/mediaserver/__init__.py:40-46): albums, tracks,
download, playlists CRUD, top/last-played, lyrics, connection test.
"""

from __future__ import annotations

from abc import ABC, abstractmethod
from dataclasses import dataclass, field
from typing import Dict, List, Optional


@dataclass
class Track:
    provider_id: str
    title: str = ""
    author: str = ""
    album: str = ""
    duration: float = 0.0
    file_path: str = ""
    year: Optional[int] = None


@dataclass
class Album:
    provider_id: str
    name: str = ""
    author: str = ""
    track_ids: List[str] = field(default_factory=list)


class Provider(ABC):
    """One instance per configured server."""

    # -- connection / libraries ----------------------------------------

    @abstractmethod
    def test_connection(self) -> bool: ...

    def list_libraries(self) -> List[Dict]:
        return []

    # -- catalogue ------------------------------------------------------

    @abstractmethod
    def get_recent_albums(self, limit: int = 0) -> List[Album]: ...

    def get_recent_music_items(self, limit: int = 0) -> List[Album]:
        """Albums plus provider-specific loose items (Emby overrides to
        merge standalone tracks; reference emby.py:435)."""
        return self.get_recent_albums(limit)

    def resolve_user(self, identifier: Optional[str] = None) -> List[Dict]:
        """Username/id -> user records on providers with user accounts
        (reference: resolve_emby_jellyfin_user, __init__.py:62)."""
        return []

    @abstractmethod
    def get_tracks_from_album(self, album_id: str) -> List[Track]: ...

    @abstractmethod
    def get_all_songs(self) -> List[Track]: ...

    def search_albums(self, query: str) -> List[Album]:
        q = (query or "").lower()
        return [a for a in self.get_recent_albums() if q in a.name.lower()]

    # -- audio -----------------------------------------------------------

    @abstractmethod
    def download_track(self, track_id: str) -> Optional[bytes]:
        """Raw audio bytes (WAV for the synthetic provider)."""

    def get_lyrics(self, track_id: str) -> Optional[str]:
        return None

    # -- playlists -------------------------------------------------------

    @abstractmethod
    def get_all_playlists(self) -> List[Dict]: ...

    def get_playlist_by_name(self, name: str) -> Optional[Dict]:
        for p in self.get_all_playlists():
            if p.get("name") == name:
                return p
        return None

    @abstractmethod
    def get_playlist_track_ids(self, playlist_id: str) -> List[str]: ...

    @abstractmethod
    def create_playlist(self, name: str, track_ids: List[str]) -> Optional[str]: ...

    def create_instant_playlist(self, name: str, track_ids: List[str]) -> Optional[str]:
        return self.create_playlist(name, track_ids)

    def create_or_replace_playlist(self, name: str, track_ids: List[str]) -> Optional[str]:
        existing = self.get_playlist_by_name(name)
        if existing is not None:
            self.delete_playlist(existing["id"])
        return self.create_playlist(name, track_ids)

    def delete_playlist(self, playlist_id: str) -> bool:
        return False

    def delete_playlists_by_suffix(self, suffix: str) -> int:
        n = 0
        for p in self.get_all_playlists():
            if p.get("name", "").endswith(suffix) and self.delete_playlist(p["id"]):
                n += 1
        return n

    def delete_automatic_playlists(self) -> int:
        return self.delete_playlists_by_suffix("_automatic")

    # -- listening stats --------------------------------------------------

    def get_top_played_songs(self, limit: int = 100) -> List[Track]:
        return []

    def get_last_played_time(self, track_id: str) -> Optional[float]:
        return None
