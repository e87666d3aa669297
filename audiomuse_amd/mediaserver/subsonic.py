"""Subsonic-API provider (Navidrome and compatible servers).

Reference analog: /root/reference/tasks/mediaserver/navidrome.py
(875 LoC; Navidrome speaks the Subsonic REST API). Behavioral parity
points carried over: salted-md5 token auth (protocol-mandated scheme,
navidrome.py:106-146), secret redaction in logs (:147), music-folder
targeting (:37-105), paginated search3 full-library walk (:351-467),
batched playlist creation that dodges URL-length limits (:592-666,
create then updatePlaylist songIdToAdd chunks), playlist clear via
update (:817-845), Navidrome's ``played`` timestamp for last-played
(:784) and getLyrics (:794). Subsonic error code 40 = bad credentials
-> typed E_MEDIA_AUTH.
"""

from __future__ import annotations

import datetime as _dt
import hashlib
import secrets
from typing import Dict, List, Optional

from audiomuse_amd import config as C
from audiomuse_amd.mediaserver import register_provider
from audiomuse_amd.mediaserver.base import Album, Provider, Track
from audiomuse_amd.mediaserver.http import MediaHttp
from audiomuse_amd.utils.errors import (E_MEDIA_AUTH, E_MEDIA_SERVER,
                                        AudioMuseError)

_PLAYLIST_BATCH = 200  # ids per request; Subsonic servers cap URL length


def _parse_played(stamp: Optional[str]) -> Optional[float]:
    if not stamp:
        return None
    try:
        return _dt.datetime.fromisoformat(stamp.rstrip("Z")).replace(
            tzinfo=_dt.timezone.utc).timestamp()
    except ValueError:
        return None


@register_provider("subsonic")
@register_provider("navidrome")
class SubsonicProvider(Provider):
    def __init__(self, base_url: str = "", username: str = "",
                 credential: str = "",
                 music_folder_ids: Optional[List[str]] = None,
                 session=None, timeout: Optional[float] = None, **_ignored):
        self.base_url = base_url.rstrip("/")
        self.username = username
        self.credential = credential
        self.music_folder_ids = list(music_folder_ids or [])
        self.http = MediaHttp(session=session, timeout=timeout)

    def _params(self) -> Dict[str, str]:
        # md5(password + salt) is the Subsonic API's own token scheme
        # (protocol-mandated; not a choice of hash for security here)
        salt = secrets.token_hex(8)
        token = hashlib.md5((self.credential + salt).encode()).hexdigest()
        return {"u": self.username, "t": token, "s": salt, "v": "1.16.1",
                "c": "audiomuse-amd", "f": "json"}

    def _get(self, endpoint: str, **params):
        p = self._params()
        for k, v in params.items():
            if v is not None:
                p[k] = v
        r = self.http.get(f"{self.base_url}/rest/{endpoint}", params=p)
        body = r.json().get("subsonic-response", {})
        if body.get("status") != "ok":
            err = body.get("error") or {}
            code = err.get("code")
            if code in (40, 41, 42, 43, 44):  # credential family
                raise AudioMuseError(
                    E_MEDIA_AUTH,
                    f"subsonic auth error {code}: {err.get('message')}")
            raise AudioMuseError(
                E_MEDIA_SERVER,
                f"subsonic error {code}: {err.get('message')}")
        return body

    def _get_raw(self, endpoint: str, **params) -> bytes:
        p = self._params()
        p.update(params)
        r = self.http.get(f"{self.base_url}/rest/{endpoint}", params=p,
                          timeout=self.http.download_timeout)
        return r.content

    def _folders(self) -> List[Optional[str]]:
        return list(self.music_folder_ids) or [None]

    # -- surface --------------------------------------------------------

    def test_connection(self) -> bool:
        try:
            self._get("ping")
            return True
        except Exception:
            return False

    def list_libraries(self) -> List[Dict]:
        body = self._get("getMusicFolders")
        folders = (body.get("musicFolders") or {}).get("musicFolder", [])
        return [{"id": str(f["id"]), "name": f.get("name", "")}
                for f in folders]

    def get_recent_albums(self, limit: int = 0) -> List[Album]:
        albums: List[Album] = []
        page = 500
        for folder in self._folders():
            offset = 0
            while True:
                body = self._get("getAlbumList2", type="newest", size=page,
                                 offset=offset, musicFolderId=folder)
                items = (body.get("albumList2") or {}).get("album", [])
                for a in items:
                    albums.append(Album(provider_id=str(a["id"]),
                                        name=a.get("name", ""),
                                        author=a.get("artist", "")))
                if len(items) < page or (limit and len(albums) >= limit):
                    break
                offset += page
        albums = list({a.provider_id: a for a in albums}.values())
        return albums[:limit] if limit else albums

    @staticmethod
    def _track(s: Dict) -> Track:
        return Track(provider_id=str(s["id"]), title=s.get("title", ""),
                     author=s.get("artist", ""), album=s.get("album", ""),
                     duration=float(s.get("duration", 0)),
                     file_path=s.get("path", ""), year=s.get("year"))

    def get_tracks_from_album(self, album_id: str) -> List[Track]:
        body = self._get("getAlbum", id=album_id)
        songs = (body.get("album") or {}).get("song", [])
        return [self._track(s) for s in songs]

    def get_all_songs(self) -> List[Track]:
        """Paginated search3 full-library walk (navidrome.py:351-467) —
        one request per page instead of one per album."""
        out: List[Track] = []
        page = 500
        for folder in self._folders():
            offset = 0
            while True:
                body = self._get("search3", query='""', songCount=page,
                                 songOffset=offset, artistCount=0,
                                 albumCount=0, musicFolderId=folder)
                songs = (body.get("searchResult3") or {}).get("song", [])
                out.extend(self._track(s) for s in songs)
                if len(songs) < page:
                    break
                offset += page
        return list({t.provider_id: t for t in out}.values())

    def search_albums(self, query: str) -> List[Album]:
        body = self._get("search3", query=query, albumCount=10,
                         songCount=0, artistCount=0)
        items = (body.get("searchResult3") or {}).get("album", [])
        return [Album(provider_id=str(a["id"]), name=a.get("name", ""),
                      author=a.get("artist", "")) for a in items]

    def download_track(self, track_id: str) -> Optional[bytes]:
        try:
            return self._get_raw("download", id=track_id)
        except Exception:
            return None

    def get_lyrics(self, track_id: str) -> Optional[str]:
        try:
            p = self._params()
            p["id"] = track_id
            r = self.http.get(f"{self.base_url}/rest/getLyrics", params=p,
                              timeout=C.MUSICSERVER_LYRICS_TIMEOUT)
            body = r.json().get("subsonic-response", {})
            if body.get("status") != "ok":
                return None
            lyr = body.get("lyrics") or {}
            return lyr.get("value")
        except Exception:
            return None

    # -- playlists -------------------------------------------------------

    def get_all_playlists(self) -> List[Dict]:
        body = self._get("getPlaylists")
        pls = (body.get("playlists") or {}).get("playlist", [])
        return [{"id": str(p["id"]), "name": p.get("name", "")} for p in pls]

    def get_playlist_track_ids(self, playlist_id: str) -> List[str]:
        body = self._get("getPlaylist", id=playlist_id)
        entries = (body.get("playlist") or {}).get("entry", [])
        return [str(e["id"]) for e in entries]

    def create_playlist(self, name: str,
                        track_ids: List[str]) -> Optional[str]:
        """Create with the first batch, append the rest via
        updatePlaylist chunks (navidrome.py:592-666: long URLs 414)."""
        first, rest = track_ids[:_PLAYLIST_BATCH], track_ids[_PLAYLIST_BATCH:]
        body = self._get("createPlaylist", name=name, songId=first)
        pl = body.get("playlist") or {}
        pid = str(pl.get("id")) if pl else None
        if pid:
            for i in range(0, len(rest), _PLAYLIST_BATCH):
                self._get("updatePlaylist", playlistId=pid,
                          songIdToAdd=rest[i:i + _PLAYLIST_BATCH])
        return pid

    def create_or_replace_playlist(self, name: str,
                                   track_ids: List[str]) -> Optional[str]:
        """Clear-then-refill preserving the playlist id
        (navidrome.py:817-878)."""
        existing = self.get_playlist_by_name(name)
        if existing is None:
            return self.create_playlist(name, track_ids)
        pid = existing["id"]
        n = len(self.get_playlist_track_ids(pid))
        if n:
            # remove back-to-front so indexes stay valid
            self._get("updatePlaylist", playlistId=pid,
                      songIndexToRemove=list(range(n - 1, -1, -1)))
        for i in range(0, len(track_ids), _PLAYLIST_BATCH):
            self._get("updatePlaylist", playlistId=pid,
                      songIdToAdd=track_ids[i:i + _PLAYLIST_BATCH])
        return pid

    def delete_playlist(self, playlist_id: str) -> bool:
        try:
            self._get("deletePlaylist", id=playlist_id)
            return True
        except Exception:
            return False

    # -- listening stats --------------------------------------------------

    def get_top_played_songs(self, limit: int = 100) -> List[Track]:
        body = self._get("getAlbumList2", type="frequent", size=50)
        out: List[Track] = []
        for a in (body.get("albumList2") or {}).get("album", []):
            out.extend(self.get_tracks_from_album(str(a["id"])))
            if len(out) >= limit:
                break
        return out[:limit]

    def get_last_played_time(self, track_id: str) -> Optional[float]:
        """Navidrome stamps ``played`` on the song (navidrome.py:784)."""
        try:
            body = self._get("getSong", id=track_id)
            return _parse_played((body.get("song") or {}).get("played"))
        except Exception:
            return None
