"""Plex provider.

Reference analog: /root/reference/tasks/mediaserver/plex.py (703 LoC) —
the Plex Media Server REST API with X-Plex-Token auth and JSON accepts.
"""

from __future__ import annotations

from typing import Dict, List, Optional

from audiomuse_amd.mediaserver import register_provider
from audiomuse_amd.mediaserver.base import Album, Provider, Track


@register_provider("plex")
class PlexProvider(Provider):
    def __init__(self, base_url: str = "", credential: str = "",
                 section_id: str = "", session=None, timeout: float = 30.0,
                 **_ignored):
        import requests

        self.base_url = base_url.rstrip("/")
        self.token = credential
        self.section_id = section_id
        self.timeout = timeout
        self.http = session or requests.Session()

    def _get(self, path: str, **params):
        params["X-Plex-Token"] = self.token
        r = self.http.get(f"{self.base_url}{path}", params=params,
                          headers={"Accept": "application/json"},
                          timeout=self.timeout)
        r.raise_for_status()
        return r.json().get("MediaContainer", {})

    def _music_section(self) -> Optional[str]:
        if self.section_id:
            return self.section_id
        body = self._get("/library/sections")
        for d in body.get("Directory", []):
            if d.get("type") == "artist":
                self.section_id = str(d["key"])
                return self.section_id
        return None

    # -- surface --------------------------------------------------------

    def test_connection(self) -> bool:
        try:
            self._get("/identity")
            return True
        except Exception:
            return False

    def list_libraries(self) -> List[Dict]:
        body = self._get("/library/sections")
        return [{"id": str(d["key"]), "name": d.get("title", "")}
                for d in body.get("Directory", [])]

    def get_recent_albums(self, limit: int = 0) -> List[Album]:
        sec = self._music_section()
        if sec is None:
            return []
        body = self._get(f"/library/sections/{sec}/albums",
                         sort="addedAt:desc")
        out = [Album(provider_id=str(a["ratingKey"]),
                     name=a.get("title", ""),
                     author=a.get("parentTitle", ""))
               for a in body.get("Metadata", [])]
        return out[:limit] if limit else out

    def get_tracks_from_album(self, album_id: str) -> List[Track]:
        body = self._get(f"/library/metadata/{album_id}/children")
        out = []
        for t in body.get("Metadata", []):
            part = ((t.get("Media") or [{}])[0].get("Part") or [{}])[0]
            out.append(Track(
                provider_id=str(t["ratingKey"]), title=t.get("title", ""),
                author=t.get("grandparentTitle", ""),
                album=t.get("parentTitle", ""),
                duration=(t.get("duration") or 0) / 1000.0,
                file_path=part.get("file", ""), year=t.get("year")))
        return out

    def get_all_songs(self) -> List[Track]:
        out: List[Track] = []
        for a in self.get_recent_albums():
            out.extend(self.get_tracks_from_album(a.provider_id))
        return out

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
                              timeout=max(self.timeout, 300.0))
            r.raise_for_status()
            return r.content
        except Exception:
            return None

    def get_all_playlists(self) -> List[Dict]:
        body = self._get("/playlists", playlistType="audio")
        return [{"id": str(p["ratingKey"]), "name": p.get("title", "")}
                for p in body.get("Metadata", [])]

    def get_playlist_track_ids(self, playlist_id: str) -> List[str]:
        body = self._get(f"/playlists/{playlist_id}/items")
        return [str(t["ratingKey"]) for t in body.get("Metadata", [])]

    def create_playlist(self, name: str, track_ids: List[str]) -> Optional[str]:
        sec = self._music_section()
        uri = (f"server://local/com.plexapp.plugins.library/library/metadata/"
               + ",".join(track_ids))
        r = self.http.post(f"{self.base_url}/playlists",
                           params={"X-Plex-Token": self.token, "title": name,
                                   "type": "audio", "smart": 0, "uri": uri},
                           headers={"Accept": "application/json"},
                           timeout=self.timeout)
        r.raise_for_status()
        meta = r.json().get("MediaContainer", {}).get("Metadata", [{}])
        return str(meta[0].get("ratingKey")) if meta else None

    def delete_playlist(self, playlist_id: str) -> bool:
        r = self.http.delete(f"{self.base_url}/playlists/{playlist_id}",
                             params={"X-Plex-Token": self.token},
                             timeout=self.timeout)
        return r.status_code in (200, 204)

    def get_top_played_songs(self, limit: int = 100) -> List[Track]:
        sec = self._music_section()
        if sec is None:
            return []
        body = self._get(f"/library/sections/{sec}/all", type=10,
                         sort="viewCount:desc")
        out = [Track(provider_id=str(t["ratingKey"]), title=t.get("title", ""),
                     author=t.get("grandparentTitle", ""))
               for t in body.get("Metadata", [])[:limit]]
        return out
