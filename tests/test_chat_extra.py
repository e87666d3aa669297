"""Chat refinements: filler push-down, hard artist cut, one replan."""

import numpy as np
import pytest

from audiomuse_amd.db import connect
from audiomuse_amd.db.schema import init_db
from audiomuse_amd.db.store import save_track_analysis_and_embedding
from audiomuse_amd.analysis.index import run_all_index_builds
from audiomuse_amd.web.app import create_app


@pytest.fixture(scope="module")
def chat_client(tmp_path_factory):
    url = "sqlite:///" + str(tmp_path_factory.mktemp("chat") / "c.db")
    conn = connect(url)
    init_db(conn)
    rng = np.random.default_rng(3)
    titles = ["Great Song", "Intro", "Skit Time", "Another Tune",
              "Banger", "Interlude", "Closer"]
    for i, t in enumerate(titles + [f"Track {j}" for j in range(10)]):
        save_track_analysis_and_embedding(
            conn, f"c{i}", title=t,
            author="Bad Artist" if i % 3 == 0 else f"Artist {i}",
            tempo=100 + i, energy=0.5,
            mood_vector={"rock": 0.8},
            embedding=rng.standard_normal(200).astype(np.float32))
    run_all_index_builds(conn)
    app = create_app(url, auth_disabled=True)
    app.testing = True
    with app.test_client() as client:
        yield client
    conn.close()


def test_filler_titles_pushed_down(chat_client):
    body = chat_client.post("/chat/api/chatPlaylist",
                            json={"prompt": "17 rock songs"}).json
    titles = [t["title"] for t in body["tracks"]]
    filler_pos = [i for i, t in enumerate(titles)
                  if t in ("Intro", "Skit Time", "Interlude")]
    real_pos = [i for i, t in enumerate(titles) if t not in
                ("Intro", "Skit Time", "Interlude")]
    assert filler_pos and real_pos
    assert min(filler_pos) > max(real_pos)  # all filler after all real


def test_exclude_artists_hard_cut(chat_client):
    body = chat_client.post(
        "/chat/api/chatPlaylist",
        json={"prompt": "17 rock songs",
              "exclude_artists": ["Bad Artist"]}).json
    assert body["tracks"]
    assert all(t["author"] != "Bad Artist" for t in body["tracks"])


def test_ordered_playlist_option(chat_client):
    body = chat_client.post("/chat/api/chatPlaylist",
                            json={"prompt": "12 rock songs",
                                  "order": True}).json
    assert len(body["tracks"]) >= 10


def test_ai_readonly_connection_cannot_write(tmp_sqlite_url):
    """The AI tool connection is a real privilege boundary (reference
    mcp_helper low-privilege role): SELECT works, writes raise."""
    import sqlite3

    import pytest as _pytest

    from audiomuse_amd.ai.dbrole import readonly_connection, reset_cache
    from audiomuse_amd.db import connect
    from audiomuse_amd.db.schema import init_db

    admin = connect(tmp_sqlite_url)
    init_db(admin)
    from audiomuse_amd.db import write_txn
    with write_txn(admin):
        admin.execute("INSERT INTO score (item_id, title) VALUES "
                      "('fp_1', 'Song')")
    reset_cache()
    ro = readonly_connection(tmp_sqlite_url)
    assert ro.execute("SELECT title FROM score").fetchone()["title"] == "Song"
    with _pytest.raises(sqlite3.OperationalError):
        ro.execute("INSERT INTO score (item_id) VALUES ('evil')")
    with _pytest.raises(sqlite3.OperationalError):
        ro.execute("DELETE FROM score")
    reset_cache()
    admin.close()


def test_search_database_tool_runs_readonly(chat_client):
    """The chat search_database tool path goes through the read-only
    connection and still answers filters."""
    r = chat_client.post("/chat/api/chatPlaylist",
                         json={"prompt": "10 rock songs faster than 90 bpm"})
    assert r.status_code == 200
    body = r.json
    # the plan includes a database filter call and the read-only path
    # produced candidates
    assert any(c["tool"] == "search_database" for c in body["plan"])
    assert body["tracks"]
