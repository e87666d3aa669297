"""Emby provider (full 19-call surface).

Reference analog: /root/reference/tasks/mediaserver/emby.py (1071 LoC).
Emby and Jellyfin share an API ancestry but have real differences this
module captures (the reference keeps them split for the same reason):

- user listing is ``/Users/Query`` returning ``{Items: []}``
  (emby.py:150) vs Jellyfin's bare ``/Users`` array
- playlist delete is ``POST /Items/{id}/Delete`` (emby.py:791) —
  Emby rejects the DELETE verb Jellyfin accepts
- playlist create passes ``Ids`` as a comma string (emby.py:729)
- "recent music" merges recent albums with recent *standalone* tracks
  that belong to no album (emby.py:181-470 get_recent_music_items);
  each standalone batch surfaces as a single-track pseudo-album so the
  analysis work map still sees them
- lyrics live at ``/Items/{id}/Lyrics`` (emby.py:861)
"""

from __future__ import annotations

from typing import Dict, List, Optional

from audiomuse_amd import config as C
from audiomuse_amd.mediaserver import register_provider
from audiomuse_amd.mediaserver.base import Album, Provider, Track
from audiomuse_amd.mediaserver.http import paged
from audiomuse_amd.mediaserver.jellyfin import JellyfinProvider


@register_provider("emby")
class EmbyProvider(JellyfinProvider):
    """Shares the MediaBrowser transport with Jellyfin; overrides every
    point where Emby's API genuinely diverges."""

    AUTH_CLIENT = ('Emby Client="audiomuse-amd", Device="server", '
                   'DeviceId="audiomuse", Version="1.0"')

    def resolve_user(self, identifier: Optional[str] = None) -> List[Dict]:
        """Emby: /Users/Query -> {Items} (reference emby.py:150-173)."""
        body = self._get("/Users/Query")
        out = [{"id": u.get("Id", ""), "name": u.get("Name", "")}
               for u in (body.get("Items") or [])]
        if identifier:
            ident = identifier.lower()
            out = [u for u in out
                   if u["id"] == identifier or u["name"].lower() == ident]
        return out

    def delete_playlist(self, playlist_id: str) -> bool:
        """Emby deletes via POST /Items/{id}/Delete (emby.py:791)."""
        try:
            self.http.post(f"{self.base_url}/Items/{playlist_id}/Delete",
                           headers=self._headers())
            return True
        except Exception:
            return False

    def create_playlist(self, name: str,
                        track_ids: List[str]) -> Optional[str]:
        """Emby takes Ids as one comma string (emby.py:729-776)."""
        r = self.http.post(f"{self.base_url}/Playlists",
                           headers=self._headers(),
                           params={"Name": name,
                                   "Ids": ",".join(track_ids),
                                   "MediaType": "Audio"})
        return str(r.json().get("Id")) if r.content else None

    def get_lyrics(self, track_id: str) -> Optional[str]:
        try:
            r = self.http.get(f"{self.base_url}/Items/{track_id}/Lyrics",
                              headers=self._headers(),
                              timeout=C.MUSICSERVER_LYRICS_TIMEOUT)
            body = r.json()
            lines = [l.get("Text", "") for l in (body.get("Lyrics") or [])]
            text = "\n".join(x for x in lines if x)
            return text or None
        except Exception:
            return None

    # -- recent music incl. standalone tracks ---------------------------

    def _recent_standalone_tracks(self, limit: int) -> List[Track]:
        """Audio items with no AlbumId — Emby libraries commonly carry
        loose files the album scan would miss (emby.py:181-335)."""
        out: List[Track] = []
        for parent in self._target_parents():
            for t in paged(lambda s, n, p=parent: self._items_page(
                    s, n, parent=p, IncludeItemTypes="Audio",
                    SortBy="DateCreated", SortOrder="Descending"),
                    limit=max(limit * 4, 200) if limit else 0):
                if not t.get("AlbumId"):
                    out.append(self._track(t))
                if limit and len(out) >= limit:
                    break
        return out[:limit] if limit else out

    def get_recent_music_items(self, limit: int = 0) -> List[Album]:
        """Albums + one single-track pseudo-album per standalone track,
        so the analysis work map covers loose files (emby.py:435-470)."""
        albums = self.get_recent_albums(limit)
        singles = self._recent_standalone_tracks(limit)
        pseudo = [Album(provider_id=f"standalone:{t.provider_id}",
                        name=t.title, author=t.author,
                        track_ids=[t.provider_id])
                  for t in singles]
        merged = albums + pseudo
        return merged[:limit] if limit else merged

    def get_tracks_from_album(self, album_id: str) -> List[Track]:
        if album_id.startswith("standalone:"):
            tid = album_id.split(":", 1)[1]
            body = self._get(f"/Users/{self._uid()}/Items/{tid}")
            return [self._track(body)] if body else []
        return super().get_tracks_from_album(album_id)
