"""Jellyfin + Emby providers (shared REST family).

Reference analogs: /root/reference/tasks/mediaserver/jellyfin.py (756
LoC) and emby.py (1071 LoC) — the two speak the same /Items API with
minor auth differences (Jellyfin: X-Emby-Token header too). Contract
covered by canned-response tests (tests/test_mediaserver.py), matching
the reference's strategy.
"""

from __future__ import annotations

from typing import Dict, List, Optional

from audiomuse_amd.mediaserver import register_provider
from audiomuse_amd.mediaserver.base import Album, Provider, Track


@register_provider("jellyfin")
@register_provider("emby")
class JellyfinProvider(Provider):
    def __init__(self, base_url: str = "", username: str = "",
                 credential: str = "", user_id: str = "", session=None,
                 timeout: float = 30.0, **_ignored):
        import requests

        self.base_url = base_url.rstrip("/")
        self.token = credential
        self.user_id = user_id
        self.timeout = timeout
        self.http = session or requests.Session()

    def _headers(self) -> Dict[str, str]:
        return {"X-Emby-Token": self.token,
                "Authorization": f'MediaBrowser Token="{self.token}", '
                                 'Client="audiomuse-amd", Device="server", '
                                 'DeviceId="audiomuse", Version="1.0"'}

    def _get(self, path: str, **params):
        r = self.http.get(f"{self.base_url}{path}", params=params,
                          headers=self._headers(), timeout=self.timeout)
        r.raise_for_status()
        return r.json()

    def _get_raw(self, path: str, **params) -> bytes:
        r = self.http.get(f"{self.base_url}{path}", params=params,
                          headers=self._headers(),
                          timeout=max(self.timeout, 300.0))
        r.raise_for_status()
        return r.content

    def _resolve_user(self) -> str:
        """reference: resolve_emby_jellyfin_user (__init__.py:62)."""
        if self.user_id:
            return self.user_id
        users = self._get("/Users")
        if users:
            self.user_id = users[0]["Id"]
        return self.user_id

    # -- surface --------------------------------------------------------

    def test_connection(self) -> bool:
        try:
            self._get("/System/Info/Public")
            return True
        except Exception:
            return False

    def list_libraries(self) -> List[Dict]:
        uid = self._resolve_user()
        views = self._get(f"/Users/{uid}/Views")
        return [{"id": v["Id"], "name": v.get("Name", "")}
                for v in views.get("Items", [])]

    def get_recent_albums(self, limit: int = 0) -> List[Album]:
        uid = self._resolve_user()
        out: List[Album] = []
        start = 0
        page = 500
        while True:
            body = self._get(f"/Users/{uid}/Items",
                             IncludeItemTypes="MusicAlbum", Recursive="true",
                             SortBy="DateCreated", SortOrder="Descending",
                             StartIndex=start, Limit=page)
            items = body.get("Items", [])
            for a in items:
                out.append(Album(provider_id=str(a["Id"]),
                                 name=a.get("Name", ""),
                                 author=a.get("AlbumArtist", "")))
            if len(items) < page or (limit and len(out) >= limit):
                break
            start += page
        return out[:limit] if limit else out

    def get_tracks_from_album(self, album_id: str) -> List[Track]:
        uid = self._resolve_user()
        body = self._get(f"/Users/{uid}/Items", ParentId=album_id,
                         IncludeItemTypes="Audio")
        out = []
        for s in body.get("Items", []):
            ticks = s.get("RunTimeTicks") or 0
            out.append(Track(
                provider_id=str(s["Id"]), title=s.get("Name", ""),
                author=(s.get("Artists") or [""])[0],
                album=s.get("Album", ""), duration=ticks / 1e7,
                file_path=s.get("Path", ""),
                year=s.get("ProductionYear")))
        return out

    def get_all_songs(self) -> List[Track]:
        out: List[Track] = []
        for a in self.get_recent_albums():
            out.extend(self.get_tracks_from_album(a.provider_id))
        return out

    def download_track(self, track_id: str) -> Optional[bytes]:
        try:
            return self._get_raw(f"/Items/{track_id}/Download")
        except Exception:
            return None

    def get_lyrics(self, track_id: str) -> Optional[str]:
        try:
            body = self._get(f"/Audio/{track_id}/Lyrics")
            lines = body.get("Lyrics", [])
            return "\n".join(l.get("Text", "") for l in lines) or None
        except Exception:
            return None

    def get_all_playlists(self) -> List[Dict]:
        uid = self._resolve_user()
        body = self._get(f"/Users/{uid}/Items",
                         IncludeItemTypes="Playlist", Recursive="true")
        return [{"id": str(p["Id"]), "name": p.get("Name", "")}
                for p in body.get("Items", [])]

    def get_playlist_track_ids(self, playlist_id: str) -> List[str]:
        body = self._get(f"/Playlists/{playlist_id}/Items",
                         UserId=self._resolve_user())
        return [str(i["Id"]) for i in body.get("Items", [])]

    def create_playlist(self, name: str, track_ids: List[str]) -> Optional[str]:
        r = self.http.post(f"{self.base_url}/Playlists",
                           headers=self._headers(),
                           json={"Name": name, "Ids": track_ids,
                                 "UserId": self._resolve_user(),
                                 "MediaType": "Audio"},
                           timeout=self.timeout)
        r.raise_for_status()
        return str(r.json().get("Id"))

    def delete_playlist(self, playlist_id: str) -> bool:
        r = self.http.delete(f"{self.base_url}/Items/{playlist_id}",
                             headers=self._headers(), timeout=self.timeout)
        return r.status_code in (200, 204)

    def get_top_played_songs(self, limit: int = 100) -> List[Track]:
        uid = self._resolve_user()
        body = self._get(f"/Users/{uid}/Items", IncludeItemTypes="Audio",
                         Recursive="true", SortBy="PlayCount",
                         SortOrder="Descending", Limit=limit)
        return [Track(provider_id=str(s["Id"]), title=s.get("Name", ""),
                      author=(s.get("Artists") or [""])[0])
                for s in body.get("Items", [])]
