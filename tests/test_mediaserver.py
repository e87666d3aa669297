"""Provider adapter contract tests against canned HTTP responses
(reference strategy: test_mediaserver.py, 3423 LoC of canned fakes)."""

import json
from typing import Dict

import pytest

from audiomuse_amd.mediaserver import make_provider, provider_types


class FakeResponse:
    def __init__(self, payload, content: bytes = b""):
        self._payload = payload
        self.content = content or json.dumps(payload).encode()
        self.status_code = 200

    def json(self):
        return self._payload

    def raise_for_status(self):
        pass


class FakeSession:
    """Canned request->response map keyed by path substring."""

    def __init__(self, routes: Dict[str, object]):
        self.routes = routes
        self.calls = []

    def _match(self, url):
        for key, payload in self.routes.items():
            if key in url:
                return FakeResponse(payload() if callable(payload) else payload)
        raise AssertionError(f"no canned route for {url}")

    def get(self, url, **kw):
        self.calls.append(("GET", url, kw.get("params")))
        return self._match(url)

    def post(self, url, **kw):
        self.calls.append(("POST", url, kw.get("json") or kw.get("params")))
        return self._match(url)

    def delete(self, url, **kw):
        self.calls.append(("DELETE", url, None))
        return self._match(url)


def test_all_provider_types_registered():
    assert {"synthetic", "subsonic", "navidrome", "jellyfin", "emby",
            "plex", "lyrion"} <= set(provider_types())


def test_synthetic_full_contract():
    p = make_provider("synthetic", n_albums=2, tracks_per_album=3,
                      seconds=2.0, sr=8000)
    assert p.test_connection()
    albums = p.get_recent_albums()
    assert len(albums) == 2
    tracks = p.get_tracks_from_album(albums[0].provider_id)
    assert len(tracks) == 3 and tracks[0].author
    blob = p.download_track(tracks[0].provider_id)
    assert blob.startswith(b"RIFF")          # WAV bytes
    assert len(p.get_all_songs()) == 6
    pid = p.create_playlist("mine", [t.provider_id for t in tracks])
    assert p.get_playlist_track_ids(pid) == [t.provider_id for t in tracks]
    assert p.get_playlist_by_name("mine")["id"] == pid
    p.create_or_replace_playlist("mine", [tracks[0].provider_id])
    assert len(p.get_playlist_track_ids(p.get_playlist_by_name("mine")["id"])) == 1
    assert len(p.get_top_played_songs(4)) == 4
    assert p.get_last_played_time(tracks[0].provider_id) is not None
    p.create_playlist("x_automatic", [])
    assert p.delete_automatic_playlists() == 1



# The per-provider HTTP contract tests (pagination, auth flows, retry,
# typed errors — one class per provider) live in
# tests/test_mediaserver_contract.py, superseding the early canned-route
# versions that used to live here.
