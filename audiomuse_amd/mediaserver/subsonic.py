"""Subsonic-API provider (Navidrome and compatible servers).

Reference analog: /root/reference/tasks/mediaserver/navidrome.py (875
LoC; Navidrome speaks the Subsonic REST API). Same uniform surface as
the synthetic provider; HTTP via `requests`. Untestable without a live
server in this image — covered by contract tests against a canned
response fake (tests/test_mediaserver.py), matching the reference's
test strategy (test_mediaserver.py canned HTTP responses).
"""

from __future__ import annotations

import hashlib
import secrets
from typing import Dict, List, Optional

from audiomuse_amd.mediaserver import register_provider
from audiomuse_amd.mediaserver.base import Album, Provider, Track


@register_provider("subsonic")
@register_provider("navidrome")
class SubsonicProvider(Provider):
    def __init__(self, base_url: str = "", username: str = "",
                 credential: str = "", session=None, timeout: float = 30.0,
                 **_ignored):
        import requests

        self.base_url = base_url.rstrip("/")
        self.username = username
        self.credential = credential
        self.timeout = timeout
        self.http = session or requests.Session()

    def _params(self) -> Dict[str, str]:
        # md5(password + salt) is the Subsonic API's own token scheme
        # (protocol-mandated; not a choice of hash for security here)
        salt = secrets.token_hex(8)
        token = hashlib.md5((self.credential + salt).encode()).hexdigest()
        return {"u": self.username, "t": token, "s": salt, "v": "1.16.1",
                "c": "audiomuse-amd", "f": "json"}

    def _get(self, endpoint: str, **params):
        p = self._params()
        p.update(params)
        r = self.http.get(f"{self.base_url}/rest/{endpoint}", params=p,
                          timeout=self.timeout)
        r.raise_for_status()
        body = r.json().get("subsonic-response", {})
        if body.get("status") != "ok":
            raise RuntimeError(f"subsonic error: {body.get('error')}")
        return body

    def _get_raw(self, endpoint: str, **params) -> bytes:
        p = self._params()
        p.update(params)
        r = self.http.get(f"{self.base_url}/rest/{endpoint}", params=p,
                          timeout=max(self.timeout, 300.0))
        r.raise_for_status()
        return r.content

    # -- surface --------------------------------------------------------

    def test_connection(self) -> bool:
        try:
            self._get("ping")
            return True
        except Exception:
            return False

    def get_recent_albums(self, limit: int = 0) -> List[Album]:
        albums: List[Album] = []
        offset = 0
        page = 500
        while True:
            body = self._get("getAlbumList2", type="newest", size=page,
                             offset=offset)
            items = (body.get("albumList2") or {}).get("album", [])
            for a in items:
                albums.append(Album(provider_id=str(a["id"]),
                                    name=a.get("name", ""),
                                    author=a.get("artist", "")))
            if len(items) < page or (limit and len(albums) >= limit):
                break
            offset += page
        return albums[:limit] if limit else albums

    def get_tracks_from_album(self, album_id: str) -> List[Track]:
        body = self._get("getAlbum", id=album_id)
        songs = (body.get("album") or {}).get("song", [])
        return [Track(provider_id=str(s["id"]), title=s.get("title", ""),
                      author=s.get("artist", ""), album=s.get("album", ""),
                      duration=float(s.get("duration", 0)),
                      file_path=s.get("path", ""), year=s.get("year"))
                for s in songs]

    def get_all_songs(self) -> List[Track]:
        out: List[Track] = []
        for a in self.get_recent_albums():
            out.extend(self.get_tracks_from_album(a.provider_id))
        return out

    def download_track(self, track_id: str) -> Optional[bytes]:
        try:
            return self._get_raw("download", id=track_id)
        except Exception:
            return None

    def get_lyrics(self, track_id: str) -> Optional[str]:
        try:
            body = self._get("getLyrics", id=track_id)
            lyr = body.get("lyrics") or {}
            return lyr.get("value")
        except Exception:
            return None

    def get_all_playlists(self) -> List[Dict]:
        body = self._get("getPlaylists")
        pls = (body.get("playlists") or {}).get("playlist", [])
        return [{"id": str(p["id"]), "name": p.get("name", "")} for p in pls]

    def get_playlist_track_ids(self, playlist_id: str) -> List[str]:
        body = self._get("getPlaylist", id=playlist_id)
        entries = (body.get("playlist") or {}).get("entry", [])
        return [str(e["id"]) for e in entries]

    def create_playlist(self, name: str, track_ids: List[str]) -> Optional[str]:
        body = self._get("createPlaylist", name=name, songId=track_ids)
        pl = body.get("playlist") or {}
        return str(pl.get("id")) if pl else None

    def delete_playlist(self, playlist_id: str) -> bool:
        try:
            self._get("deletePlaylist", id=playlist_id)
            return True
        except Exception:
            return False

    def get_top_played_songs(self, limit: int = 100) -> List[Track]:
        body = self._get("getAlbumList2", type="frequent", size=50)
        out: List[Track] = []
        for a in (body.get("albumList2") or {}).get("album", []):
            out.extend(self.get_tracks_from_album(str(a["id"])))
            if len(out) >= limit:
                break
        return out[:limit]
