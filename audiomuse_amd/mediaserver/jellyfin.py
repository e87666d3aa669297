"""Jellyfin provider (full 19-call surface).

Reference analog: /root/reference/tasks/mediaserver/jellyfin.py (756
LoC). Behavioral parity points carried over: token *or*
username/password (AuthenticateByName) auth, user resolution
(__init__.py:62 resolve_emby_jellyfin_user), target-library filtering
(jellyfin.py:36-94), StartIndex/Limit page loops for albums/songs
(:158-238, :310-359), entry-level playlist replace that preserves the
playlist id (:651-714), PlayCount-sorted top-played (:507),
UserData.LastPlayedDate (:545), and the /Audio/{id}/Lyrics endpoint
(:566). Emby — a different server with overlapping API — lives in its
own module (emby.py) as in the reference.
"""

from __future__ import annotations

import datetime as _dt
from typing import Dict, List, Optional

from audiomuse_amd import config as C
from audiomuse_amd.mediaserver import register_provider
from audiomuse_amd.mediaserver.base import Album, Provider, Track
from audiomuse_amd.mediaserver.http import MediaHttp, paged

_TICKS_PER_SECOND = 10_000_000


def _parse_iso_epoch(stamp: Optional[str]) -> Optional[float]:
    if not stamp:
        return None
    try:
        clean = stamp.rstrip("Z")
        if "." in clean:  # trim sub-microsecond digits
            head, frac = clean.split(".", 1)
            clean = f"{head}.{frac[:6]}"
        return _dt.datetime.fromisoformat(clean).replace(
            tzinfo=_dt.timezone.utc).timestamp()
    except ValueError:
        return None


@register_provider("jellyfin")
class JellyfinProvider(Provider):
    AUTH_CLIENT = ('MediaBrowser Client="audiomuse-amd", Device="server", '
                   'DeviceId="audiomuse", Version="1.0"')

    def __init__(self, base_url: str = "", username: str = "",
                 credential: str = "", user_id: str = "",
                 library_ids: Optional[List[str]] = None, session=None,
                 timeout: Optional[float] = None, **_ignored):
        self.base_url = base_url.rstrip("/")
        self.username = username
        self.token = ""
        self.password = ""
        # credential is an API token unless a username is configured, in
        # which case it is that user's password (AuthenticateByName flow)
        if username:
            self.password = credential
        else:
            self.token = credential
        self.user_id = user_id
        self.library_ids = list(library_ids or [])
        self.http = MediaHttp(session=session, timeout=timeout)

    # -- auth / plumbing -----------------------------------------------

    def _headers(self) -> Dict[str, str]:
        self._ensure_token()
        return {"X-Emby-Token": self.token,
                "Authorization": f'{self.AUTH_CLIENT}, Token="{self.token}"'}

    def _ensure_token(self) -> None:
        if self.token or not self.username:
            return
        r = self.http.post(
            f"{self.base_url}/Users/AuthenticateByName",
            json_body={"Username": self.username, "Pw": self.password},
            headers={"Authorization": self.AUTH_CLIENT})
        body = r.json()
        self.token = body.get("AccessToken", "")
        if not self.user_id:
            self.user_id = (body.get("User") or {}).get("Id", "")

    def _get(self, path: str, **params):
        r = self.http.get(f"{self.base_url}{path}", params=params,
                          headers=self._headers())
        return r.json()

    def _uid(self) -> str:
        if not self.user_id:
            users = self.resolve_user(self.username or None)
            if users:
                self.user_id = users[0]["id"]
        return self.user_id

    def resolve_user(self, identifier: Optional[str] = None) -> List[Dict]:
        """Name/id -> user records (reference: resolve_emby_jellyfin_user,
        jellyfin.py:146)."""
        users = self._get("/Users")
        out = [{"id": u.get("Id", ""), "name": u.get("Name", "")}
               for u in (users or [])]
        if identifier:
            ident = identifier.lower()
            out = [u for u in out
                   if u["id"] == identifier or u["name"].lower() == ident]
        return out

    # -- libraries -------------------------------------------------------

    def list_libraries(self) -> List[Dict]:
        body = self._get(f"/Users/{self._uid()}/Views")
        return [{"id": v.get("Id", ""), "name": v.get("Name", ""),
                 "type": v.get("CollectionType", "")}
                for v in (body.get("Items") or [])
                if v.get("CollectionType") in ("music", None)]

    def _target_parents(self) -> List[Optional[str]]:
        """Configured library scoping; [None] = whole server."""
        return list(self.library_ids) or [None]

    # -- catalogue -------------------------------------------------------

    def _items_page(self, start: int, size: int, *,
                    parent: Optional[str], **extra) -> List[Dict]:
        params = {"Recursive": "true", "StartIndex": start, "Limit": size,
                  **extra}
        if parent:
            params["ParentId"] = parent
        body = self._get(f"/Users/{self._uid()}/Items", **params)
        return body.get("Items") or []

    @staticmethod
    def _album(a: Dict) -> Album:
        return Album(provider_id=str(a.get("Id", "")),
                     name=a.get("Name", ""),
                     author=a.get("AlbumArtist", "")
                     or ", ".join(a.get("Artists") or []))

    @staticmethod
    def _track(t: Dict) -> Track:
        artists = t.get("Artists") or []
        return Track(provider_id=str(t.get("Id", "")),
                     title=t.get("Name", ""),
                     author=(artists[0] if artists
                             else t.get("AlbumArtist", "")),
                     album=t.get("Album", ""),
                     duration=float(t.get("RunTimeTicks", 0))
                     / _TICKS_PER_SECOND,
                     file_path=t.get("Path", ""),
                     year=t.get("ProductionYear"))

    def get_recent_albums(self, limit: int = 0) -> List[Album]:
        out: List[Album] = []
        for parent in self._target_parents():
            for a in paged(lambda s, n, p=parent: self._items_page(
                    s, n, parent=p, IncludeItemTypes="MusicAlbum",
                    SortBy="DateCreated", SortOrder="Descending"),
                    limit=limit):
                out.append(self._album(a))
        out = list({a.provider_id: a for a in out}.values())
        return out[:limit] if limit else out

    def get_tracks_from_album(self, album_id: str) -> List[Track]:
        items = self._items_page(0, 10000, parent=album_id,
                                 IncludeItemTypes="Audio",
                                 SortBy="SortName")
        return [self._track(t) for t in items]

    def get_all_songs(self) -> List[Track]:
        out: List[Track] = []
        for parent in self._target_parents():
            out.extend(self._track(t) for t in paged(
                lambda s, n, p=parent: self._items_page(
                    s, n, parent=p, IncludeItemTypes="Audio")))
        return list({t.provider_id: t for t in out}.values())

    def search_albums(self, query: str) -> List[Album]:
        items = self._items_page(0, 10, parent=None,
                                 IncludeItemTypes="MusicAlbum",
                                 SearchTerm=query)
        return [self._album(a) for a in items]

    # -- audio -----------------------------------------------------------

    def download_track(self, track_id: str) -> Optional[bytes]:
        try:
            r = self.http.get(f"{self.base_url}/Items/{track_id}/Download",
                              headers=self._headers(),
                              timeout=self.http.download_timeout)
            return r.content
        except Exception:
            return None

    def get_lyrics(self, track_id: str) -> Optional[str]:
        try:
            r = self.http.get(f"{self.base_url}/Audio/{track_id}/Lyrics",
                              headers=self._headers(),
                              timeout=C.MUSICSERVER_LYRICS_TIMEOUT)
            body = r.json()
            lines = [l.get("Text", "") for l in (body.get("Lyrics") or [])]
            text = "\n".join(x for x in lines if x)
            return text or None
        except Exception:
            return None

    # -- connection -------------------------------------------------------

    def test_connection(self) -> bool:
        try:
            self._get("/System/Info/Public")
            self._items_page(0, 1, parent=None, IncludeItemTypes="Audio")
            return True
        except Exception:
            return False

    # -- playlists --------------------------------------------------------

    def get_all_playlists(self) -> List[Dict]:
        items = self._items_page(0, 10000, parent=None,
                                 IncludeItemTypes="Playlist")
        return [{"id": str(p.get("Id", "")), "name": p.get("Name", "")}
                for p in items]

    def get_playlist_track_ids(self, playlist_id: str) -> List[str]:
        body = self._get(f"/Playlists/{playlist_id}/Items",
                         UserId=self._uid())
        return [str(i.get("Id", "")) for i in (body.get("Items") or [])]

    def _playlist_entry_ids(self, playlist_id: str) -> List[str]:
        body = self._get(f"/Playlists/{playlist_id}/Items",
                         UserId=self._uid())
        return [str(i.get("PlaylistItemId", ""))
                for i in (body.get("Items") or [])]

    def create_playlist(self, name: str,
                        track_ids: List[str]) -> Optional[str]:
        r = self.http.post(f"{self.base_url}/Playlists",
                           headers=self._headers(),
                           json_body={"Name": name, "Ids": track_ids,
                                      "UserId": self._uid(),
                                      "MediaType": "Audio"})
        return str(r.json().get("Id")) if r.content else None

    def create_or_replace_playlist(self, name: str,
                                   track_ids: List[str]) -> Optional[str]:
        """Entry-level replace preserving the playlist id (reference:
        jellyfin.py:651-714) — clients keep their favorites pinned."""
        existing = self.get_playlist_by_name(name)
        if existing is None:
            return self.create_playlist(name, track_ids)
        pid = existing["id"]
        entries = self._playlist_entry_ids(pid)
        for i in range(0, len(entries), 100):
            chunk = ",".join(entries[i:i + 100])
            self.http.delete(
                f"{self.base_url}/Playlists/{pid}/Items",
                params={"EntryIds": chunk}, headers=self._headers())
        for i in range(0, len(track_ids), 100):
            chunk = ",".join(track_ids[i:i + 100])
            self.http.post(
                f"{self.base_url}/Playlists/{pid}/Items",
                params={"Ids": chunk, "UserId": self._uid()},
                headers=self._headers())
        return pid

    def delete_playlist(self, playlist_id: str) -> bool:
        try:
            self.http.delete(f"{self.base_url}/Items/{playlist_id}",
                             headers=self._headers())
            return True
        except Exception:
            return False

    # -- listening stats ---------------------------------------------------

    def get_top_played_songs(self, limit: int = 100) -> List[Track]:
        items = self._items_page(0, limit, parent=None,
                                 IncludeItemTypes="Audio",
                                 SortBy="PlayCount",
                                 SortOrder="Descending",
                                 Filters="IsPlayed")
        return [self._track(t) for t in items]

    def get_last_played_time(self, track_id: str) -> Optional[float]:
        body = self._get(f"/Users/{self._uid()}/Items/{track_id}")
        return _parse_iso_epoch(
            (body.get("UserData") or {}).get("LastPlayedDate"))
