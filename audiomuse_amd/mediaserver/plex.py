"""Plex provider (full 19-call surface).

Reference analog: /root/reference/tasks/mediaserver/plex.py (703 LoC) —
the Plex Media Server REST API with X-Plex-Token auth and JSON accepts.
Behavioral parity points carried over: container pagination via
X-Plex-Container-Start/Size (plex.py:178-204), multi-section library
targeting (:125-177), machineIdentifier-scoped item URIs for playlist
mutation (:487-527), batched playlist creation (:528), Media/Part
resolution for downloads (:249-300), lastViewedAt listening stats
(:560-ish) and tolerant field fallbacks (:81-123)."""

from __future__ import annotations

from typing import Dict, List, Optional

from audiomuse_amd.mediaserver import register_provider
from audiomuse_amd.mediaserver.base import Album, Provider, Track
from audiomuse_amd.mediaserver.http import MediaHttp

_PAGE = 500
_PLAYLIST_BATCH = 200


@register_provider("plex")
class PlexProvider(Provider):
    def __init__(self, base_url: str = "", credential: str = "",
                 section_id: str = "",
                 section_ids: Optional[List[str]] = None,
                 session=None, timeout: Optional[float] = None, **_ignored):
        self.base_url = base_url.rstrip("/")
        self.token = credential
        self.section_ids = list(section_ids or
                                ([section_id] if section_id else []))
        self.http = MediaHttp(session=session, timeout=timeout)
        self._machine_id: Optional[str] = None

    def _get(self, path: str, **params):
        params["X-Plex-Token"] = self.token
        r = self.http.get(f"{self.base_url}{path}", params=params,
                          headers={"Accept": "application/json"})
        return r.json().get("MediaContainer", {})

    def _paged(self, path: str, limit: int = 0, **params) -> List[Dict]:
        """Container page loop (plex.py:178-204)."""
        out: List[Dict] = []
        start = 0
        while True:
            body = self._get(path, **params,
                             **{"X-Plex-Container-Start": start,
                                "X-Plex-Container-Size": _PAGE})
            items = body.get("Metadata", []) or []
            out.extend(items)
            total = int(body.get("totalSize", body.get("size", len(items))))
            start += len(items)
            if (not items or len(items) < _PAGE or start >= total
                    or (limit and len(out) >= limit)):
                return out[:limit] if limit else out

    def _music_sections(self) -> List[str]:
        if self.section_ids:
            return self.section_ids
        body = self._get("/library/sections")
        self.section_ids = [str(d["key"]) for d in body.get("Directory", [])
                            if d.get("type") == "artist"]
        return self.section_ids

    def _machine_identifier(self) -> str:
        """Playlist URIs are scoped to the server id (plex.py:487)."""
        if self._machine_id is None:
            body = self._get("/identity")
            self._machine_id = body.get("machineIdentifier", "local")
        return self._machine_id

    def _items_uri(self, item_ids: List[str]) -> str:
        mid = self._machine_identifier()
        return (f"server://{mid}/com.plexapp.plugins.library"
                f"/library/metadata/" + ",".join(item_ids))

    @staticmethod
    def _track(t: Dict) -> Track:
        part = ((t.get("Media") or [{}])[0].get("Part") or [{}])[0]
        return Track(
            provider_id=str(t.get("ratingKey", "")),
            title=t.get("title", ""),
            author=t.get("grandparentTitle") or t.get("originalTitle", ""),
            album=t.get("parentTitle", ""),
            duration=(t.get("duration") or 0) / 1000.0,
            file_path=part.get("file", ""), year=t.get("year"))

    @staticmethod
    def _album(a: Dict) -> Album:
        return Album(provider_id=str(a.get("ratingKey", "")),
                     name=a.get("title", ""),
                     author=a.get("parentTitle", ""))

    # -- surface --------------------------------------------------------

    def test_connection(self) -> bool:
        try:
            self._get("/identity")
            self._music_sections()
            return True
        except Exception:
            return False

    def list_libraries(self) -> List[Dict]:
        body = self._get("/library/sections")
        return [{"id": str(d["key"]), "name": d.get("title", ""),
                 "type": d.get("type", "")}
                for d in body.get("Directory", [])]

    def get_recent_albums(self, limit: int = 0) -> List[Album]:
        out: List[Album] = []
        for sec in self._music_sections():
            out.extend(self._album(a) for a in self._paged(
                f"/library/sections/{sec}/all", limit=limit,
                type=9, sort="addedAt:desc"))
        out = list({a.provider_id: a for a in out}.values())
        return out[:limit] if limit else out

    def get_tracks_from_album(self, album_id: str) -> List[Track]:
        items = self._paged(f"/library/metadata/{album_id}/children")
        return [self._track(t) for t in items]

    def get_all_songs(self) -> List[Track]:
        out: List[Track] = []
        for sec in self._music_sections():
            out.extend(self._track(t) for t in self._paged(
                f"/library/sections/{sec}/all", type=10))
        return list({t.provider_id: t for t in out}.values())

    def search_albums(self, query: str) -> List[Album]:
        out: List[Album] = []
        for sec in self._music_sections():
            body = self._get(f"/library/sections/{sec}/all", type=9,
                             title=query,
                             **{"X-Plex-Container-Size": 10})
            out.extend(self._album(a) for a in body.get("Metadata", []))
        return out[:10]

    # -- audio -----------------------------------------------------------

    def download_track(self, track_id: str) -> Optional[bytes]:
        try:
            body = self._get(f"/library/metadata/{track_id}")
            meta = (body.get("Metadata") or [{}])[0]
            part = ((meta.get("Media") or [{}])[0].get("Part") or [{}])[0]
            key = part.get("key")
            if not key:
                return None
            r = self.http.get(f"{self.base_url}{key}",
                              params={"X-Plex-Token": self.token,
                                      "download": 1},
                              timeout=self.http.download_timeout)
            return r.content
        except Exception:
            return None

    def get_lyrics(self, track_id: str) -> Optional[str]:
        """Lyric streams ride Media/Part/Stream type 4 (plex lyrics
        agent); fetch the stream key when present."""
        try:
            body = self._get(f"/library/metadata/{track_id}")
            meta = (body.get("Metadata") or [{}])[0]
            part = ((meta.get("Media") or [{}])[0].get("Part") or [{}])[0]
            for stream in part.get("Stream", []) or []:
                if stream.get("streamType") == 4 and stream.get("key"):
                    from audiomuse_amd import config as C
                    r = self.http.get(f"{self.base_url}{stream['key']}",
                                      params={"X-Plex-Token": self.token},
                                      timeout=C.MUSICSERVER_LYRICS_TIMEOUT)
                    return r.text or None
            return None
        except Exception:
            return None

    # -- playlists --------------------------------------------------------

    def get_all_playlists(self) -> List[Dict]:
        body = self._get("/playlists", playlistType="audio")
        return [{"id": str(p["ratingKey"]), "name": p.get("title", "")}
                for p in body.get("Metadata", [])]

    def get_playlist_track_ids(self, playlist_id: str) -> List[str]:
        items = self._paged(f"/playlists/{playlist_id}/items")
        return [str(t["ratingKey"]) for t in items]

    def create_playlist(self, name: str,
                        track_ids: List[str]) -> Optional[str]:
        """Create with the first batch, PUT the rest — giant uri lists
        blow the URL limit (plex.py:528 batched creation)."""
        first = track_ids[:_PLAYLIST_BATCH]
        r = self.http.post(f"{self.base_url}/playlists",
                           params={"X-Plex-Token": self.token, "title": name,
                                   "type": "audio", "smart": 0,
                                   "uri": self._items_uri(first)},
                           headers={"Accept": "application/json"})
        meta = r.json().get("MediaContainer", {}).get("Metadata", [{}])
        pid = str(meta[0].get("ratingKey")) if meta else None
        if pid:
            rest = track_ids[_PLAYLIST_BATCH:]
            for i in range(0, len(rest), _PLAYLIST_BATCH):
                self.http.request(
                    "PUT", f"{self.base_url}/playlists/{pid}/items",
                    params={"X-Plex-Token": self.token,
                            "uri": self._items_uri(
                                rest[i:i + _PLAYLIST_BATCH])},
                    headers={"Accept": "application/json"})
        return pid

    def delete_playlist(self, playlist_id: str) -> bool:
        try:
            self.http.delete(f"{self.base_url}/playlists/{playlist_id}",
                             params={"X-Plex-Token": self.token})
            return True
        except Exception:
            return False

    # -- listening stats ---------------------------------------------------

    def get_top_played_songs(self, limit: int = 100) -> List[Track]:
        out: List[Track] = []
        for sec in self._music_sections():
            out.extend(self._track(t) for t in self._paged(
                f"/library/sections/{sec}/all", limit=limit,
                type=10, sort="viewCount:desc"))
        return out[:limit]

    def get_last_played_time(self, track_id: str) -> Optional[float]:
        try:
            body = self._get(f"/library/metadata/{track_id}")
            meta = (body.get("Metadata") or [{}])[0]
            lv = meta.get("lastViewedAt")
            return float(lv) if lv else None
        except Exception:
            return None
