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


def test_subsonic_contract():
    routes = {
        "/rest/ping": {"subsonic-response": {"status": "ok"}},
        "/rest/getAlbumList2": {"subsonic-response": {"status": "ok",
            "albumList2": {"album": [{"id": "al1", "name": "First",
                                      "artist": "Art"}]}}},
        "/rest/getAlbum": {"subsonic-response": {"status": "ok",
            "album": {"song": [{"id": "s1", "title": "T1", "artist": "Art",
                                "album": "First", "duration": 200,
                                "path": "/m/a/t1.flac"}]}}},
        "/rest/getPlaylists": {"subsonic-response": {"status": "ok",
            "playlists": {"playlist": [{"id": "p1", "name": "Faves"}]}}},
        "/rest/getPlaylist?": {"subsonic-response": {"status": "ok",
            "playlist": {"entry": [{"id": "s1"}]}}},
        "/rest/getPlaylist": {"subsonic-response": {"status": "ok",
            "playlist": {"entry": [{"id": "s1"}]}}},
        "/rest/createPlaylist": {"subsonic-response": {"status": "ok",
            "playlist": {"id": "p9"}}},
        "/rest/getLyrics": {"subsonic-response": {"status": "ok",
            "lyrics": {"value": "la la"}}},
    }
    sess = FakeSession(routes)
    p = make_provider("navidrome", base_url="http://x", username="u",
                      credential="pw", session=sess)
    assert p.test_connection()
    albums = p.get_recent_albums()
    assert albums[0].name == "First"
    tracks = p.get_tracks_from_album("al1")
    assert tracks[0].duration == 200.0
    assert p.get_all_playlists()[0]["name"] == "Faves"
    assert p.get_playlist_track_ids("p1") == ["s1"]
    assert p.create_playlist("new", ["s1"]) == "p9"
    assert p.get_lyrics("s1") == "la la"
    # auth params present on every call
    for _m, _url, params in sess.calls:
        assert params and "t" in params and "s" in params


def test_jellyfin_contract():
    routes = {
        "/System/Info/Public": {"Version": "10"},
        "/Users/u1/Items": {"Items": [
            {"Id": "alb1", "Name": "Album", "AlbumArtist": "Z"}]},
        "/Users": [{"Id": "u1"}],
        "/Playlists": {"Id": "pl1"},
    }
    sess = FakeSession(routes)
    p = make_provider("jellyfin", base_url="http://j", credential="tok",
                      session=sess)
    assert p.test_connection()
    albums = p.get_recent_albums(limit=5)
    assert albums[0].provider_id == "alb1"
    # token header attached
    assert p.create_playlist("n", ["alb1"]) == "pl1"


def test_lyrion_contract():
    routes = {
        "/jsonrpc.js": {"result": {
            "albums_loop": [{"id": 7, "album": "LMS Album", "artist": "Q"}],
            "titles_loop": [{"id": 9, "title": "T", "artist": "Q",
                             "album": "LMS Album", "duration": 100,
                             "url": "file:///m/t.flac"}],
            "playlists_loop": [], "_version": "9"}},
    }
    sess = FakeSession(routes)
    p = make_provider("lyrion", base_url="http://lms", session=sess)
    assert p.test_connection()
    albums = p.get_recent_albums()
    assert albums[0].name == "LMS Album"
    tracks = p.get_tracks_from_album("7")
    assert tracks[0].file_path == "/m/t.flac"


def test_plex_contract():
    routes = {
        "/identity": {"MediaContainer": {"machineIdentifier": "m"}},
        "/library/sections/5/albums": {"MediaContainer": {"Metadata": [
            {"ratingKey": "301", "title": "P Album", "parentTitle": "W"}]}},
        "/library/sections": {"MediaContainer": {"Directory": [
            {"key": "5", "type": "artist", "title": "Music"}]}},
        "/library/metadata/301/children": {"MediaContainer": {"Metadata": [
            {"ratingKey": "401", "title": "PT", "grandparentTitle": "W",
             "parentTitle": "P Album", "duration": 180000,
             "Media": [{"Part": [{"file": "/m/p.mp3", "key": "/parts/1"}]}]}]}},
    }
    sess = FakeSession(routes)
    p = make_provider("plex", base_url="http://plex", credential="tok",
                      session=sess)
    assert p.test_connection()
    albums = p.get_recent_albums()
    assert albums[0].author == "W"
    tracks = p.get_tracks_from_album("301")
    assert tracks[0].duration == 180.0
    assert p.list_libraries()[0]["name"] == "Music"
