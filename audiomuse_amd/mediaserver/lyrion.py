"""Lyrion Music Server (LMS / Logitech Media Server) provider.

Reference analog: /root/reference/tasks/mediaserver/lyrion.py (1142 LoC)
— the LMS JSON-RPC endpoint (/jsonrpc.js, ``slim.request`` envelopes).
Behavioral parity points carried over: start/count pagination on every
loop query (lyrion.py:246-273, :306-393), remote-stream tracks skipped
(only ``file://`` urls are analyzable, :49-68), tolerant field fallback
when servers omit tags (:69-109), target-path library filtering
(:126-138, :566-594), playlist creation in batches through
``playlistcontrol`` (:729-897), songinfo-based lyrics (:1103) and
playcount/lastplayed listening stats (:1057-1101). Optional HTTP basic
auth for password-protected servers.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional
from urllib.parse import unquote

from audiomuse_amd.mediaserver import register_provider
from audiomuse_amd.mediaserver.base import Album, Provider, Track
from audiomuse_amd.mediaserver.http import MediaHttp

_PAGE = 500
_PLAYLIST_BATCH = 200


def _file_path(url: str) -> Optional[str]:
    """file:// URL -> local path; None for remote streams (lyrion.py:49:
    remote tracks cannot be downloaded for analysis)."""
    if not url:
        return None
    if url.startswith("file://"):
        return unquote(url[len("file://"):])
    return None


def _first(d: Dict, *keys, default=""):
    """LMS responses vary by version/tags; take the first present field
    (reference lyrion.py:83 ``_try``)."""
    for k in keys:
        v = d.get(k)
        if v not in (None, ""):
            return v
    return default


@register_provider("lyrion")
class LyrionProvider(Provider):
    def __init__(self, base_url: str = "", username: str = "",
                 credential: str = "",
                 target_paths: Optional[List[str]] = None,
                 session=None, timeout: Optional[float] = None, **_ignored):
        self.base_url = base_url.rstrip("/")
        self.target_paths = [p.rstrip("/") for p in (target_paths or [])]
        self.http = MediaHttp(session=session, timeout=timeout)
        self._auth = (username, credential) if username else None

    def _rpc(self, *command) -> Dict[str, Any]:
        kw: Dict[str, Any] = {"json_body": {"id": 1, "method": "slim.request",
                                            "params": ["", list(command)]}}
        if self._auth:
            import base64
            cred = base64.b64encode(
                f"{self._auth[0]}:{self._auth[1]}".encode()).decode()
            kw["headers"] = {"Authorization": f"Basic {cred}"}
        r = self.http.post(f"{self.base_url}/jsonrpc.js", **kw)
        return r.json().get("result", {})

    def _paged_loop(self, loop_key: str, *command) -> List[Dict]:
        """start/count page loop over one LMS query (lyrion.py:246)."""
        out: List[Dict] = []
        start = 0
        while True:
            body = self._rpc(command[0], start, _PAGE, *command[1:])
            items = body.get(loop_key, []) or []
            out.extend(items)
            if len(items) < _PAGE:
                return out
            start += len(items)

    def _in_target_paths(self, path: Optional[str]) -> bool:
        if not self.target_paths:
            return True
        if not path:
            return False
        return any(path == t or path.startswith(t + "/")
                   for t in self.target_paths)

    # -- surface --------------------------------------------------------

    def test_connection(self) -> bool:
        try:
            self._rpc("version", "?")
            return True
        except Exception:
            return False

    def list_libraries(self) -> List[Dict]:
        """Media folders (used by the migration wizard's path-format
        probe; lyrion.py:139-172)."""
        body = self._rpc("pref", "mediadirs", "?")
        dirs = body.get("_p2") or body.get("mediadirs") or []
        if isinstance(dirs, str):
            dirs = [dirs]
        return [{"id": d, "name": d} for d in dirs]

    def get_recent_albums(self, limit: int = 0) -> List[Album]:
        items = self._paged_loop("albums_loop", "albums",
                                 "sort:new", "tags:la")
        out = [Album(provider_id=str(a["id"]),
                     name=_first(a, "album", "title"),
                     author=_first(a, "artist", "albumartist"))
               for a in items]
        return out[:limit] if limit else out

    def _title_track(self, t: Dict) -> Optional[Track]:
        path = _file_path(_first(t, "url", default=""))
        if path is None and _first(t, "url", default=""):
            return None  # remote stream: skip (lyrion.py:49)
        return Track(
            provider_id=str(t["id"]), title=_first(t, "title", "name"),
            author=_first(t, "artist", "trackartist", "albumartist"),
            album=_first(t, "album"),
            duration=float(t.get("duration", 0) or 0),
            file_path=path or "",
            year=t.get("year") or None)

    def get_tracks_from_album(self, album_id: str) -> List[Track]:
        items = self._paged_loop("titles_loop", "titles",
                                 f"album_id:{album_id}", "tags:aldyu")
        out = []
        for t in items:
            tr = self._title_track(t)
            if tr is not None and self._in_target_paths(tr.file_path or None):
                out.append(tr)
        return out

    def get_all_songs(self) -> List[Track]:
        items = self._paged_loop("titles_loop", "titles", "tags:aldyu")
        out = []
        for t in items:
            tr = self._title_track(t)
            if tr is not None and self._in_target_paths(tr.file_path or None):
                out.append(tr)
        return out

    def search_albums(self, query: str) -> List[Album]:
        body = self._rpc("albums", 0, 10, f"search:{query}", "tags:la")
        return [Album(provider_id=str(a["id"]),
                      name=_first(a, "album", "title"),
                      author=_first(a, "artist"))
                for a in body.get("albums_loop", []) or []]

    def download_track(self, track_id: str) -> Optional[bytes]:
        try:
            r = self.http.get(f"{self.base_url}/music/{track_id}/download",
                              timeout=self.http.download_timeout)
            return r.content
        except Exception:
            return None

    def get_lyrics(self, track_id: str) -> Optional[str]:
        """songinfo carries lyrics when tagged (lyrion.py:1103)."""
        try:
            from audiomuse_amd import config as C
            old = self.http.timeout
            self.http.timeout = C.MUSICSERVER_LYRICS_TIMEOUT
            try:
                body = self._rpc("songinfo", 0, 100,
                                 f"track_id:{track_id}", "tags:L")
            finally:
                self.http.timeout = old
            for entry in body.get("songinfo_loop", []) or []:
                if "lyrics" in entry:
                    return entry["lyrics"] or None
            return None
        except Exception:
            return None

    # -- playlists -------------------------------------------------------

    def get_all_playlists(self) -> List[Dict]:
        items = self._paged_loop("playlists_loop", "playlists")
        return [{"id": str(p["id"]), "name": _first(p, "playlist", "name")}
                for p in items]

    def get_playlist_track_ids(self, playlist_id: str) -> List[str]:
        items = self._paged_loop("playlisttracks_loop", "playlists",
                                 "tracks", f"playlist_id:{playlist_id}")
        return [str(t["id"]) for t in items]

    def create_playlist(self, name: str,
                        track_ids: List[str]) -> Optional[str]:
        """New playlist + batched track adds (lyrion.py:831-897:
        one-command-per-track round trips are too slow at playlist
        sizes; playlistcontrol takes comma lists)."""
        body = self._rpc("playlists", "new", f"name:{name}")
        pid = body.get("overwritten_playlist_id") or body.get("playlist_id")
        if pid is None:
            return None
        for i in range(0, len(track_ids), _PLAYLIST_BATCH):
            chunk = ",".join(track_ids[i:i + _PLAYLIST_BATCH])
            self._rpc("playlists", "edit", f"playlist_id:{pid}",
                      "cmd:add", f"track_id:{chunk}")
        return str(pid)

    def delete_playlist(self, playlist_id: str) -> bool:
        try:
            self._rpc("playlists", "delete", f"playlist_id:{playlist_id}")
            return True
        except Exception:
            return False

    # -- listening stats ---------------------------------------------------

    def get_top_played_songs(self, limit: int = 100) -> List[Track]:
        body = self._rpc("titles", 0, limit, "sort:playcount", "tags:aldyu")
        out = []
        for t in body.get("titles_loop", []) or []:
            tr = self._title_track(t)
            if tr is not None:
                out.append(tr)
        return out

    def get_last_played_time(self, track_id: str) -> Optional[float]:
        try:
            body = self._rpc("songinfo", 0, 100,
                             f"track_id:{track_id}", "tags:n")
            for entry in body.get("songinfo_loop", []) or []:
                if "lastplayed" in entry:
                    return float(entry["lastplayed"]) or None
            return None
        except Exception:
            return None
