"""Lyrion Music Server (LMS / Logitech Media Server) provider.

Reference analog: /root/reference/tasks/mediaserver/lyrion.py (1142 LoC)
— the LMS JSON-RPC endpoint (/jsonrpc.js, `slim.request` envelopes).
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

from audiomuse_amd.mediaserver import register_provider
from audiomuse_amd.mediaserver.base import Album, Provider, Track


@register_provider("lyrion")
class LyrionProvider(Provider):
    def __init__(self, base_url: str = "", session=None,
                 timeout: float = 30.0, **_ignored):
        import requests

        self.base_url = base_url.rstrip("/")
        self.timeout = timeout
        self.http = session or requests.Session()

    def _rpc(self, *command) -> Dict[str, Any]:
        r = self.http.post(
            f"{self.base_url}/jsonrpc.js",
            json={"id": 1, "method": "slim.request",
                  "params": ["", list(command)]},
            timeout=self.timeout)
        r.raise_for_status()
        return r.json().get("result", {})

    # -- surface --------------------------------------------------------

    def test_connection(self) -> bool:
        try:
            self._rpc("version", "?")
            return True
        except Exception:
            return False

    def get_recent_albums(self, limit: int = 0) -> List[Album]:
        n = limit or 100000
        body = self._rpc("albums", 0, n, "sort:new", "tags:la")
        return [Album(provider_id=str(a["id"]), name=a.get("album", ""),
                      author=a.get("artist", ""))
                for a in body.get("albums_loop", [])]

    def get_tracks_from_album(self, album_id: str) -> List[Track]:
        body = self._rpc("titles", 0, 1000, f"album_id:{album_id}",
                         "tags:aldu")
        out = []
        for t in body.get("titles_loop", []):
            out.append(Track(
                provider_id=str(t["id"]), title=t.get("title", ""),
                author=t.get("artist", ""), album=t.get("album", ""),
                duration=float(t.get("duration", 0)),
                file_path=(t.get("url", "") or "").replace("file://", "")))
        return out

    def get_all_songs(self) -> List[Track]:
        out: List[Track] = []
        for a in self.get_recent_albums():
            out.extend(self.get_tracks_from_album(a.provider_id))
        return out

    def download_track(self, track_id: str) -> Optional[bytes]:
        try:
            r = self.http.get(f"{self.base_url}/music/{track_id}/download",
                              timeout=max(self.timeout, 300.0))
            r.raise_for_status()
            return r.content
        except Exception:
            return None

    def get_all_playlists(self) -> List[Dict]:
        body = self._rpc("playlists", 0, 10000)
        return [{"id": str(p["id"]), "name": p.get("playlist", "")}
                for p in body.get("playlists_loop", [])]

    def get_playlist_track_ids(self, playlist_id: str) -> List[str]:
        body = self._rpc("playlists", "tracks", 0, 10000,
                         f"playlist_id:{playlist_id}")
        return [str(t["id"]) for t in body.get("playlisttracks_loop", [])]

    def create_playlist(self, name: str, track_ids: List[str]) -> Optional[str]:
        body = self._rpc("playlists", "new", f"name:{name}")
        pid = body.get("overwritten_playlist_id") or body.get("playlist_id")
        if pid is None:
            return None
        for tid in track_ids:
            self._rpc("playlists", "edit", f"playlist_id:{pid}", "cmd:add",
                      f"track_id:{tid}")
        return str(pid)

    def delete_playlist(self, playlist_id: str) -> bool:
        try:
            self._rpc("playlists", "delete", f"playlist_id:{playlist_id}")
            return True
        except Exception:
            return False

    def get_top_played_songs(self, limit: int = 100) -> List[Track]:
        body = self._rpc("titles", 0, limit, "sort:playcount", "tags:aldu")
        return [Track(provider_id=str(t["id"]), title=t.get("title", ""),
                      author=t.get("artist", ""))
                for t in body.get("titles_loop", [])]
