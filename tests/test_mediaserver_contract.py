"""Per-provider media-server contract tests against canned HTTP
responses — the strategy of the reference's test_mediaserver.py
(3 423 LoC of canned payloads): every public API call per provider,
pagination joins, auth flows, retry/error behavior.

No network: a FakeSession records every request and answers from a
routing table keyed on (METHOD, path)."""

import json as _json
import hashlib

import pytest

from audiomuse_amd import config as C
from audiomuse_amd.mediaserver import make_provider, provider_types
from audiomuse_amd.mediaserver.http import MediaHttp, paged, redact
from audiomuse_amd.utils.errors import (E_MEDIA_AUTH, E_MEDIA_UNREACHABLE,
                                        AudioMuseError)


class FakeResponse:
    def __init__(self, status=200, payload=None, content=b"", text=""):
        self.status_code = status
        self._payload = payload
        self.content = content or (_json.dumps(payload).encode()
                                   if payload is not None else b"")
        self.text = text or self.content.decode("utf-8", "replace")

    def json(self):
        if self._payload is None:
            raise ValueError("no json")
        return self._payload


class FakeSession:
    """Routes (METHOD, path) -> payload | callable(params, headers,
    json_body) -> FakeResponse/payload. Records calls for assertions."""

    def __init__(self, routes):
        self.routes = routes
        self.calls = []

    def request(self, method, url, params=None, headers=None, json=None,
                data=None, stream=False, timeout=None):
        path = url.split("://", 1)[-1]
        path = "/" + path.split("/", 1)[1] if "/" in path else "/"
        path = path.split("?")[0]
        self.calls.append((method.upper(), path, dict(params or {}),
                           dict(headers or {}), json))
        handler = self.routes.get((method.upper(), path))
        if handler is None:
            return FakeResponse(status=404)
        if callable(handler):
            out = handler(params or {}, headers or {}, json)
        else:
            out = handler
        if isinstance(out, FakeResponse):
            return out
        return FakeResponse(payload=out)


@pytest.fixture(autouse=True)
def _fast_retries(monkeypatch):
    monkeypatch.setattr(C, "MEDIASERVER_RETRY_BACKOFF_SECONDS", 0.001)
    monkeypatch.setattr(C, "MEDIASERVER_RETRIES", 2)


# ---------------------------------------------------------------------------
# shared HTTP layer
# ---------------------------------------------------------------------------

def test_registry_has_all_providers():
    for name in ("synthetic", "subsonic", "navidrome", "jellyfin", "emby",
                 "plex", "lyrion"):
        assert name in provider_types()


def test_http_retries_transient_then_succeeds():
    attempts = []

    def flaky(params, headers, body):
        attempts.append(1)
        if len(attempts) < 3:
            return FakeResponse(status=503)
        return {"ok": True}

    sess = FakeSession({("GET", "/x"): flaky})
    http = MediaHttp(session=sess)
    assert http.get("http://s/x").json() == {"ok": True}
    assert len(attempts) == 3


def test_http_429_retries_and_gives_typed_error():
    sess = FakeSession({("GET", "/x"): FakeResponse(status=429)})
    http = MediaHttp(session=sess)
    with pytest.raises(AudioMuseError) as ei:
        http.get("http://s/x")
    assert ei.value.code != E_MEDIA_AUTH
    assert len(sess.calls) == 3  # retried to exhaustion


def test_http_auth_errors_do_not_retry():
    sess = FakeSession({("GET", "/x"): FakeResponse(status=401)})
    http = MediaHttp(session=sess)
    with pytest.raises(AudioMuseError) as ei:
        http.get("http://s/x")
    assert ei.value.code == E_MEDIA_AUTH
    assert len(sess.calls) == 1  # terminal, no retry


def test_http_connection_failure_is_unreachable():
    class Boom:
        def request(self, *a, **k):
            raise OSError("connection refused")

    http = MediaHttp(session=Boom())
    with pytest.raises(AudioMuseError) as ei:
        http.get("http://gone/x")
    assert ei.value.code == E_MEDIA_UNREACHABLE


def test_redact_strips_secrets():
    out = redact("http://s/rest/ping?u=bob&t=deadbeef&s=salt&v=1.16.1"
                 "&X-Plex-Token=tok123")
    assert "deadbeef" not in out and "tok123" not in out and "bob" in out


def test_paged_drives_to_exhaustion_and_limit():
    pages = {0: list(range(3)), 3: list(range(3, 5))}
    got = list(paged(lambda s, n: pages.get(s, []), page_size=3))
    assert got == [0, 1, 2, 3, 4]
    got = list(paged(lambda s, n: pages.get(s, []), page_size=3, limit=2))
    assert got == [0, 1]


# ---------------------------------------------------------------------------
# Jellyfin
# ---------------------------------------------------------------------------

def _jf_items(items, total=None):
    return {"Items": items, "TotalRecordCount": total or len(items)}


def _jf_track(i, album="Alb", artists=("Art",)):
    return {"Id": f"t{i}", "Name": f"Song {i}", "Artists": list(artists),
            "Album": album, "RunTimeTicks": 1800000000 + i,
            "Path": f"/m/{album}/{i}.flac", "ProductionYear": 2020}


@pytest.fixture
def jellyfin():
    albums_page1 = [{"Id": f"a{i}", "Name": f"Album {i}",
                     "AlbumArtist": "AA"} for i in range(500)]
    albums_page2 = [{"Id": "a500", "Name": "Album 500",
                     "AlbumArtist": "AA"}]

    def items(params, headers, body):
        assert headers.get("X-Emby-Token") == "tok", "auth header required"
        t = params.get("IncludeItemTypes")
        start = int(params.get("StartIndex", 0))
        if t == "MusicAlbum" and params.get("SearchTerm"):
            return _jf_items([{"Id": "as", "Name": "Found",
                               "AlbumArtist": "X"}])
        if t == "MusicAlbum":
            return _jf_items(albums_page1 if start == 0 else albums_page2,
                             total=501)
        if t == "Audio" and params.get("ParentId") == "a1":
            return _jf_items([_jf_track(1), _jf_track(2)])
        if t == "Audio" and params.get("SortBy") == "PlayCount":
            return _jf_items([_jf_track(9)])
        if t == "Audio":
            return _jf_items([_jf_track(5)] if start == 0 else [])
        if t == "Playlist":
            return _jf_items([{"Id": "p1", "Name": "Mix_automatic"}])
        return _jf_items([])

    routes = {
        ("GET", "/Users"): [{"Id": "u1", "Name": "Admin"},
                            {"Id": "u2", "Name": "Kid"}],
        ("GET", "/Users/u1/Items"): items,
        ("GET", "/Users/u1/Views"): _jf_items(
            [{"Id": "lib1", "Name": "Music", "CollectionType": "music"},
             {"Id": "lib2", "Name": "Movies", "CollectionType": "movies"}]),
        ("GET", "/System/Info/Public"): {"Version": "10.9"},
        ("GET", "/Items/t1/Download"): FakeResponse(content=b"RIFFaudio"),
        ("GET", "/Audio/t1/Lyrics"): {"Lyrics": [{"Text": "la"},
                                                 {"Text": "laa"}]},
        ("GET", "/Playlists/p1/Items"): _jf_items(
            [{"Id": "t1", "PlaylistItemId": "e1"},
             {"Id": "t2", "PlaylistItemId": "e2"}]),
        ("POST", "/Playlists"): {"Id": "pNew"},
        ("POST", "/Playlists/p1/Items"): FakeResponse(status=204),
        ("DELETE", "/Playlists/p1/Items"): FakeResponse(status=204),
        ("DELETE", "/Items/p1"): FakeResponse(status=204),
        ("GET", "/Users/u1/Items/t1"): {
            **_jf_track(1),
            "UserData": {"LastPlayedDate": "2026-01-02T03:04:05.1234567Z"}},
    }
    sess = FakeSession(routes)
    p = make_provider("jellyfin", base_url="http://jf", credential="tok",
                      user_id="u1", session=sess)
    return p, sess


class TestJellyfin:
    def test_connection(self, jellyfin):
        p, _ = jellyfin
        assert p.test_connection() is True

    def test_list_libraries_filters_music(self, jellyfin):
        p, _ = jellyfin
        libs = p.list_libraries()
        assert [l["id"] for l in libs] == ["lib1"]

    def test_recent_albums_joins_pages(self, jellyfin):
        p, sess = jellyfin
        albums = p.get_recent_albums()
        assert len(albums) == 501
        starts = [c[2].get("StartIndex") for c in sess.calls
                  if c[2].get("IncludeItemTypes") == "MusicAlbum"]
        assert starts == [0, 500]  # the page loop advanced

    def test_album_tracks_mapping(self, jellyfin):
        p, _ = jellyfin
        ts = p.get_tracks_from_album("a1")
        assert [t.provider_id for t in ts] == ["t1", "t2"]
        assert ts[0].author == "Art" and ts[0].album == "Alb"
        assert abs(ts[0].duration - 180.0) < 0.01  # ticks -> seconds
        assert ts[0].year == 2020

    def test_search_albums(self, jellyfin):
        p, _ = jellyfin
        assert p.search_albums("found")[0].name == "Found"

    def test_download(self, jellyfin):
        p, _ = jellyfin
        assert p.download_track("t1") == b"RIFFaudio"

    def test_lyrics_joined(self, jellyfin):
        p, _ = jellyfin
        assert p.get_lyrics("t1") == "la\nlaa"

    def test_playlists_and_entry_level_replace(self, jellyfin):
        p, sess = jellyfin
        pls = p.get_all_playlists()
        assert pls == [{"id": "p1", "name": "Mix_automatic"}]
        assert p.get_playlist_track_ids("p1") == ["t1", "t2"]
        # replace must keep the id: delete entries then add new ids
        pid = p.create_or_replace_playlist("Mix_automatic", ["t5", "t6"])
        assert pid == "p1"
        deletes = [c for c in sess.calls
                   if c[0] == "DELETE" and c[1] == "/Playlists/p1/Items"]
        adds = [c for c in sess.calls
                if c[0] == "POST" and c[1] == "/Playlists/p1/Items"]
        assert deletes[0][2]["EntryIds"] == "e1,e2"
        assert adds[0][2]["Ids"] == "t5,t6"

    def test_create_fresh_playlist(self, jellyfin):
        p, sess = jellyfin
        assert p.create_playlist("New", ["t1"]) == "pNew"
        post = [c for c in sess.calls if c[1] == "/Playlists"][0]
        assert post[4]["MediaType"] == "Audio"

    def test_delete_automatic_playlists(self, jellyfin):
        p, _ = jellyfin
        assert p.delete_automatic_playlists() == 1

    def test_top_and_last_played(self, jellyfin):
        p, _ = jellyfin
        top = p.get_top_played_songs(5)
        assert top and top[0].provider_id == "t9"
        epoch = p.get_last_played_time("t1")
        assert epoch is not None and 1767000000 < epoch < 1800000000

    def test_resolve_user(self, jellyfin):
        p, _ = jellyfin
        assert p.resolve_user("admin") == [{"id": "u1", "name": "Admin"}]
        assert p.resolve_user("u2")[0]["name"] == "Kid"

    def test_auth_by_name_flow(self):
        def auth(params, headers, body):
            assert body == {"Username": "bob", "Pw": "pw"}
            assert "MediaBrowser" in headers.get("Authorization", "")
            return {"AccessToken": "fresh", "User": {"Id": "u9"}}

        sess = FakeSession({
            ("POST", "/Users/AuthenticateByName"): auth,
            ("GET", "/System/Info/Public"): {},
            ("GET", "/Users/u9/Items"): _jf_items([]),
        })
        p = make_provider("jellyfin", base_url="http://jf", username="bob",
                          credential="pw", session=sess)
        assert p.test_connection() is True
        assert p.token == "fresh" and p.user_id == "u9"

    def test_bad_token_is_typed_auth_error(self):
        sess = FakeSession({("GET", "/Users/u1/Items"):
                            FakeResponse(status=401)})
        p = make_provider("jellyfin", base_url="http://jf",
                          credential="bad", user_id="u1", session=sess)
        with pytest.raises(AudioMuseError) as ei:
            p.get_recent_albums()
        assert ei.value.code == E_MEDIA_AUTH

    def test_library_scoping_passes_parent(self):
        sess = FakeSession({("GET", "/Users/u1/Items"):
                            lambda p, h, b: _jf_items([])})
        p = make_provider("jellyfin", base_url="http://jf", credential="t",
                          user_id="u1", library_ids=["libA", "libB"],
                          session=sess)
        p.get_recent_albums()
        parents = [c[2].get("ParentId") for c in sess.calls]
        assert parents == ["libA", "libB"]


# ---------------------------------------------------------------------------
# Emby
# ---------------------------------------------------------------------------

@pytest.fixture
def emby():
    def items(params, headers, body):
        t = params.get("IncludeItemTypes")
        if t == "MusicAlbum":
            return _jf_items([{"Id": "a1", "Name": "Album", "AlbumArtist": "AA"}])
        if t == "Audio" and params.get("SortBy") == "DateCreated":
            return _jf_items([
                {**_jf_track(7), "AlbumId": ""},        # standalone
                {**_jf_track(8), "AlbumId": "a1"},      # in album
            ])
        if t == "Playlist":
            return _jf_items([{"Id": "p1", "Name": "Mix"}])
        return _jf_items([])

    routes = {
        ("GET", "/Users/Query"): {"Items": [{"Id": "u1", "Name": "Admin"}]},
        ("GET", "/Users/u1/Items"): items,
        ("GET", "/Users/u1/Items/t7"): _jf_track(7),
        ("POST", "/Items/p1/Delete"): FakeResponse(status=204),
        ("POST", "/Playlists"): {"Id": "pE"},
        ("GET", "/Items/t1/Lyrics"): {"Lyrics": [{"Text": "emby line"}]},
    }
    sess = FakeSession(routes)
    p = make_provider("emby", base_url="http://emby", credential="tok",
                      user_id="u1", session=sess)
    return p, sess


class TestEmby:
    def test_users_query_shape(self, emby):
        p, _ = emby
        assert p.resolve_user("admin")[0]["id"] == "u1"

    def test_delete_playlist_uses_post(self, emby):
        p, sess = emby
        assert p.delete_playlist("p1") is True
        assert ("POST", "/Items/p1/Delete") in [(c[0], c[1])
                                                for c in sess.calls]

    def test_create_playlist_comma_ids(self, emby):
        p, sess = emby
        assert p.create_playlist("M", ["t1", "t2"]) == "pE"
        post = [c for c in sess.calls if c[1] == "/Playlists"][0]
        assert post[2]["Ids"] == "t1,t2"

    def test_recent_music_items_include_standalone(self, emby):
        p, _ = emby
        items = p.get_recent_music_items()
        names = [a.provider_id for a in items]
        assert "a1" in names and "standalone:t7" in names
        # standalone pseudo-album resolves to its single track
        ts = p.get_tracks_from_album("standalone:t7")
        assert len(ts) == 1 and ts[0].provider_id == "t7"

    def test_lyrics_items_endpoint(self, emby):
        p, _ = emby
        assert p.get_lyrics("t1") == "emby line"


# ---------------------------------------------------------------------------
# Subsonic / Navidrome
# ---------------------------------------------------------------------------

def _sub_ok(extra):
    return {"subsonic-response": {"status": "ok", "version": "1.16.1",
                                  **extra}}


def _sub_song(i):
    return {"id": f"s{i}", "title": f"Song {i}", "artist": "Art",
            "album": "Alb", "duration": 180 + i, "path": f"m/{i}.flac",
            "year": 2021}


@pytest.fixture
def subsonic():
    state = {"updates": [], "created": []}

    def albumlist(params, headers, body):
        assert params["f"] == "json"
        # token = md5(password + salt) must verify
        tok = hashlib.md5(("pw" + params["s"]).encode()).hexdigest()
        assert params["t"] == tok and params["u"] == "bob"
        off = int(params.get("offset", 0))
        if params.get("type") == "frequent":
            return _sub_ok({"albumList2": {"album": [{"id": "a1"}]}})
        if off == 0:
            return _sub_ok({"albumList2": {"album": [
                {"id": f"a{i}", "name": f"Al {i}", "artist": "AA"}
                for i in range(500)]}})
        return _sub_ok({"albumList2": {"album": [
            {"id": "a500", "name": "Al 500", "artist": "AA"}]}})

    def search3(params, headers, body):
        off = int(params.get("songOffset", 0))
        if params.get("albumCount") == "10" or params.get("albumCount") == 10:
            return _sub_ok({"searchResult3": {"album": [
                {"id": "aX", "name": "Hit", "artist": "AA"}]}})
        if off == 0:
            return _sub_ok({"searchResult3": {"song": [
                _sub_song(i) for i in range(500)]}})
        return _sub_ok({"searchResult3": {"song": [_sub_song(500)]}})

    def create_playlist(params, headers, body):
        ids = params.get("songId", [])
        if isinstance(ids, str):
            ids = [ids]
        state["created"].append(list(ids))
        return _sub_ok({"playlist": {"id": "pl9", "name": params["name"]}})

    def update_playlist(params, headers, body):
        state["updates"].append(dict(params))
        return _sub_ok({})

    routes = {
        ("GET", "/rest/ping"): _sub_ok({}),
        ("GET", "/rest/getMusicFolders"): _sub_ok(
            {"musicFolders": {"musicFolder": [{"id": 1, "name": "Music"}]}}),
        ("GET", "/rest/getAlbumList2"): albumlist,
        ("GET", "/rest/search3"): search3,
        ("GET", "/rest/getAlbum"): _sub_ok(
            {"album": {"song": [_sub_song(1), _sub_song(2)]}}),
        ("GET", "/rest/download"): FakeResponse(content=b"FLACbytes"),
        ("GET", "/rest/getLyrics"): _sub_ok(
            {"lyrics": {"value": "sub lyrics"}}),
        ("GET", "/rest/getPlaylists"): _sub_ok(
            {"playlists": {"playlist": [{"id": "pl1", "name": "Mix"}]}}),
        ("GET", "/rest/getPlaylist"): _sub_ok(
            {"playlist": {"entry": [{"id": "s1"}, {"id": "s2"}]}}),
        ("GET", "/rest/createPlaylist"): create_playlist,
        ("GET", "/rest/updatePlaylist"): update_playlist,
        ("GET", "/rest/deletePlaylist"): _sub_ok({}),
        ("GET", "/rest/getSong"): _sub_ok(
            {"song": {**_sub_song(1), "played": "2026-02-03T04:05:06Z"}}),
    }
    sess = FakeSession(routes)
    p = make_provider("navidrome", base_url="http://nd", username="bob",
                      credential="pw", session=sess)
    return p, sess, state


class TestSubsonic:
    def test_connection_and_salted_token(self, subsonic):
        p, sess, _ = subsonic
        assert p.test_connection() is True
        params = sess.calls[0][2]
        assert params["t"] != "pw" and len(params["s"]) == 16

    def test_libraries(self, subsonic):
        p, _, _ = subsonic
        assert p.list_libraries() == [{"id": "1", "name": "Music"}]

    def test_recent_albums_pagination(self, subsonic):
        p, _, _ = subsonic
        assert len(p.get_recent_albums()) == 501

    def test_all_songs_via_search3_pages(self, subsonic):
        p, sess, _ = subsonic
        songs = p.get_all_songs()
        assert len(songs) == 501
        offs = [c[2].get("songOffset") for c in sess.calls
                if c[1] == "/rest/search3"]
        assert offs == [0, 500]

    def test_search_albums(self, subsonic):
        p, _, _ = subsonic
        assert p.search_albums("hit")[0].name == "Hit"

    def test_album_tracks_and_download_and_lyrics(self, subsonic):
        p, _, _ = subsonic
        ts = p.get_tracks_from_album("a1")
        assert [t.provider_id for t in ts] == ["s1", "s2"]
        assert ts[0].duration == 181.0
        assert p.download_track("s1") == b"FLACbytes"
        assert p.get_lyrics("s1") == "sub lyrics"

    def test_playlist_batched_create(self, subsonic):
        p, _, state = subsonic
        ids = [f"s{i}" for i in range(450)]
        pid = p.create_playlist("Big", ids)
        assert pid == "pl9"
        assert len(state["created"][0]) == 200       # first batch in create
        adds = [u for u in state["updates"] if "songIdToAdd" in u]
        assert len(adds) == 2                        # 200 + 50 appended

    def test_replace_clears_then_refills(self, subsonic):
        p, _, state = subsonic
        pid = p.create_or_replace_playlist("Mix", ["s7"])
        assert pid == "pl1"
        removes = [u for u in state["updates"] if "songIndexToRemove" in u]
        assert removes and removes[0]["songIndexToRemove"] == [1, 0]
        adds = [u for u in state["updates"] if "songIdToAdd" in u]
        assert adds[-1]["songIdToAdd"] == ["s7"]

    def test_auth_error_code_is_typed(self):
        routes = {("GET", "/rest/ping"): {
            "subsonic-response": {"status": "failed",
                                  "error": {"code": 40,
                                            "message": "Wrong credentials"}}}}
        p = make_provider("subsonic", base_url="http://nd", username="b",
                          credential="x", session=FakeSession(routes))
        with pytest.raises(AudioMuseError) as ei:
            p._get("ping")
        assert ei.value.code == E_MEDIA_AUTH

    def test_last_played(self, subsonic):
        p, _, _ = subsonic
        epoch = p.get_last_played_time("s1")
        assert epoch is not None and epoch > 1767000000


# ---------------------------------------------------------------------------
# Lyrion (JSON-RPC)
# ---------------------------------------------------------------------------

@pytest.fixture
def lyrion():
    state = {"added": []}

    def rpc(params, headers, body):
        cmd = body["params"][1]
        if cmd[0] == "version":
            return {"result": {"_version": "9.0"}}
        if cmd[0] == "albums":
            start = cmd[1]
            items = ([{"id": i, "album": f"Al {i}", "artist": "AA"}
                      for i in range(500)] if start == 0 else
                     [{"id": 500, "album": "Al 500", "artist": "AA"}])
            if any(str(c).startswith("search:") for c in cmd):
                items = [{"id": 9, "album": "Hit", "artist": "AA"}]
            return {"result": {"albums_loop": items}}
        if cmd[0] == "titles":
            return {"result": {"titles_loop": [
                {"id": 1, "title": "Local", "artist": "Art",
                 "album": "Alb", "duration": 100,
                 "url": "file:///m/a%20b.flac", "year": 1999},
                {"id": 2, "title": "Stream", "artist": "Art",
                 "url": "http://radio/stream"},          # remote: skip
            ]}}
        if cmd[0] == "playlists" and len(cmd) > 1 and cmd[1] == "new":
            return {"result": {"playlist_id": 77}}
        if cmd[0] == "playlists" and len(cmd) > 1 and cmd[1] == "edit":
            state["added"].append([c for c in cmd
                                   if str(c).startswith("track_id:")][0])
            return {"result": {}}
        if cmd[0] == "playlists" and len(cmd) > 1 and cmd[1] == "delete":
            return {"result": {}}
        if cmd[0] == "playlists" and "tracks" in cmd:
            return {"result": {"playlisttracks_loop": [{"id": 1}, {"id": 3}]}}
        if cmd[0] == "playlists":
            return {"result": {"playlists_loop": [
                {"id": 5, "playlist": "Mix"}]}}
        if cmd[0] == "songinfo":
            return {"result": {"songinfo_loop": [
                {"id": 1}, {"lyrics": "lms lyrics"},
                {"lastplayed": 1767000001}]}}
        if cmd[0] == "pref":
            return {"result": {"_p2": ["/music"]}}
        return {"result": {}}

    sess = FakeSession({("POST", "/jsonrpc.js"): rpc})
    p = make_provider("lyrion", base_url="http://lms", session=sess)
    return p, sess, state


class TestLyrion:
    def test_connection_and_envelope(self, lyrion):
        p, sess, _ = lyrion
        assert p.test_connection() is True
        body = sess.calls[0][4]
        assert body["method"] == "slim.request"
        assert body["params"][1][0] == "version"

    def test_albums_paginated(self, lyrion):
        p, _, _ = lyrion
        assert len(p.get_recent_albums()) == 501

    def test_remote_tracks_skipped_and_url_decoded(self, lyrion):
        p, _, _ = lyrion
        ts = p.get_tracks_from_album("1")
        assert len(ts) == 1                      # stream skipped
        assert ts[0].file_path == "/m/a b.flac"  # percent-decoded
        assert ts[0].year == 1999

    def test_target_path_filter(self, lyrion):
        _, sess, _ = lyrion
        p2 = make_provider("lyrion", base_url="http://lms",
                           target_paths=["/elsewhere"], session=sess)
        assert p2.get_tracks_from_album("1") == []

    def test_search(self, lyrion):
        p, _, _ = lyrion
        assert p.search_albums("hit")[0].name == "Hit"

    def test_playlists_batched_add(self, lyrion):
        p, _, state = lyrion
        pid = p.create_playlist("Big", [str(i) for i in range(450)])
        assert pid == "77"
        assert len(state["added"]) == 3          # 200+200+50 chunks
        assert state["added"][0].count(",") == 199

    def test_playlist_tracks_and_delete(self, lyrion):
        p, _, _ = lyrion
        assert p.get_all_playlists() == [{"id": "5", "name": "Mix"}]
        assert p.get_playlist_track_ids("5") == ["1", "3"]
        assert p.delete_playlist("5") is True

    def test_lyrics_and_lastplayed_from_songinfo(self, lyrion):
        p, _, _ = lyrion
        assert p.get_lyrics("1") == "lms lyrics"
        assert p.get_last_played_time("1") == 1767000001.0

    def test_libraries(self, lyrion):
        p, _, _ = lyrion
        assert p.list_libraries() == [{"id": "/music", "name": "/music"}]

    def test_basic_auth_header(self):
        seen = {}

        def rpc(params, headers, body):
            seen.update(headers)
            return {"result": {"_version": "9"}}

        sess = FakeSession({("POST", "/jsonrpc.js"): rpc})
        p = make_provider("lyrion", base_url="http://lms", username="u",
                          credential="p", session=sess)
        p.test_connection()
        assert seen.get("Authorization", "").startswith("Basic ")


# ---------------------------------------------------------------------------
# Plex
# ---------------------------------------------------------------------------

def _px_container(meta, total=None):
    return {"MediaContainer": {"Metadata": meta, "size": len(meta),
                               "totalSize": total or len(meta)}}


def _px_track(i):
    return {"ratingKey": f"{i}", "title": f"Song {i}",
            "grandparentTitle": "Art", "parentTitle": "Alb",
            "duration": 181000,
            "Media": [{"Part": [{"file": f"/m/{i}.mp3",
                                 "key": f"/library/parts/{i}/file.mp3"}]}],
            "year": 2022, "lastViewedAt": 1767000500}


@pytest.fixture
def plex():
    def section_all(params, headers, body):
        t = int(params.get("type", 0))
        start = int(params.get("X-Plex-Container-Start", 0))
        if t == 9:  # albums
            meta = ([{"ratingKey": f"a{i}", "title": f"Al {i}",
                      "parentTitle": "AA"} for i in range(500)]
                    if start == 0 else
                    [{"ratingKey": "a500", "title": "Al 500",
                      "parentTitle": "AA"}])
            if params.get("title"):
                meta = [{"ratingKey": "aH", "title": "Hit",
                         "parentTitle": "AA"}]
                return _px_container(meta, total=1)
            return _px_container(meta, total=501)
        if t == 10:
            if params.get("sort") == "viewCount:desc":
                return _px_container([_px_track(3)], total=1)
            return _px_container([_px_track(1)], total=1)
        return _px_container([])

    routes = {
        ("GET", "/identity"): {"MediaContainer":
                               {"machineIdentifier": "mach1"}},
        ("GET", "/library/sections"): {"MediaContainer": {"Directory": [
            {"key": "3", "type": "artist", "title": "Music"},
            {"key": "4", "type": "movie", "title": "Films"}]}},
        ("GET", "/library/sections/3/all"): section_all,
        ("GET", "/library/metadata/a1/children"): _px_container(
            [_px_track(1), _px_track(2)]),
        ("GET", "/library/metadata/1"): _px_container([_px_track(1)]),
        ("GET", "/library/parts/1/file.mp3"): FakeResponse(content=b"MP3!"),
        ("GET", "/playlists"): _px_container(
            [{"ratingKey": "p1", "title": "Mix"}]),
        ("GET", "/playlists/p1/items"): _px_container([_px_track(1)]),
        ("POST", "/playlists"): _px_container([{"ratingKey": "p9"}]),
        ("PUT", "/playlists/p9/items"): _px_container([]),
        ("DELETE", "/playlists/p1"): FakeResponse(status=204),
    }
    sess = FakeSession(routes)
    p = make_provider("plex", base_url="http://px", credential="tok",
                      session=sess)
    return p, sess


class TestPlex:
    def test_connection_discovers_sections(self, plex):
        p, _ = plex
        assert p.test_connection() is True
        assert p.section_ids == ["3"]

    def test_token_param_everywhere(self, plex):
        p, sess = plex
        p.test_connection()
        assert all(c[2].get("X-Plex-Token") == "tok" for c in sess.calls)

    def test_albums_container_paging(self, plex):
        p, sess = plex
        albums = p.get_recent_albums()
        assert len(albums) == 501
        starts = [c[2].get("X-Plex-Container-Start") for c in sess.calls
                  if c[1] == "/library/sections/3/all"]
        assert 500 in starts

    def test_album_tracks_mapping(self, plex):
        p, _ = plex
        ts = p.get_tracks_from_album("a1")
        assert ts[0].duration == 181.0 and ts[0].file_path == "/m/1.mp3"

    def test_search(self, plex):
        p, _ = plex
        assert p.search_albums("hit")[0].name == "Hit"

    def test_download_resolves_part(self, plex):
        p, _ = plex
        assert p.download_track("1") == b"MP3!"

    def test_playlist_create_uses_machine_uri_and_batches(self, plex):
        p, sess = plex
        ids = [str(i) for i in range(350)]
        pid = p.create_playlist("Big", ids)
        assert pid == "p9"
        post = [c for c in sess.calls if c[1] == "/playlists"
                and c[0] == "POST"][0]
        assert post[2]["uri"].startswith(
            "server://mach1/com.plexapp.plugins.library/library/metadata/")
        puts = [c for c in sess.calls if c[0] == "PUT"]
        assert len(puts) == 1       # 200 in create + 150 appended

    def test_playlists_and_delete(self, plex):
        p, _ = plex
        assert p.get_all_playlists() == [{"id": "p1", "name": "Mix"}]
        assert p.get_playlist_track_ids("p1") == ["1"]
        assert p.delete_playlist("p1") is True

    def test_top_and_last_played(self, plex):
        p, _ = plex
        assert p.get_top_played_songs(5)[0].provider_id == "3"
        assert p.get_last_played_time("1") == 1767000500.0
