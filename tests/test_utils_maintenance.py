"""Aux subsystems: errors, logging sanitization, SSRF guard, resources,
maintenance tasks (sweep/cleaning/backup/dashboard)."""

import logging
import sqlite3

import numpy as np
import pytest

from audiomuse_amd.analysis.maintenance import (align_server_tracks,
                                                backup_database,
                                                normalize_title,
                                                refresh_dashboard_stats)
from audiomuse_amd.db import connect
from audiomuse_amd.db.schema import init_db
from audiomuse_amd.db.store import save_track_analysis_and_embedding
from audiomuse_amd.mediaserver.base import Track
from audiomuse_amd.utils.errors import (E_DB, E_MEDIA_UNREACHABLE,
                                        E_MODEL_OOM, E_UNKNOWN,
                                        AudioMuseError, classify_exception)
from audiomuse_amd.utils.logging_utils import (LogSanitizingFilter,
                                               sanitize_for_log,
                                               validate_outbound_url)
from audiomuse_amd.utils.resources import ModelLifecycle, usable_cpu_count


def test_error_classifier():
    assert classify_exception(sqlite3.OperationalError("db locked")) == 1101
    assert classify_exception(sqlite3.IntegrityError("x")) == E_DB
    assert classify_exception(RuntimeError("HIP out of memory")) == E_MODEL_OOM
    assert classify_exception(ValueError("whatever")) == E_UNKNOWN
    err = AudioMuseError(E_MEDIA_UNREACHABLE, "navidrome at 10.0.0.2")
    assert classify_exception(err) == E_MEDIA_UNREACHABLE
    assert "unreachable" in err.user_message.lower()


def test_log_sanitization():
    assert "\n" not in sanitize_for_log("evil\ninjected line")
    assert len(sanitize_for_log("x" * 2000)) < 600
    rec = logging.LogRecord("t", logging.INFO, "f", 1,
                            "user said %s", ("a\r\nFAKE",), None)
    LogSanitizingFilter().filter(rec)
    assert "\n" not in rec.getMessage()


def test_ssrf_guard():
    assert validate_outbound_url("https://api.openai.com/v1")
    assert not validate_outbound_url("http://127.0.0.1/admin")
    assert not validate_outbound_url("http://10.0.0.5/x")
    assert not validate_outbound_url("file:///etc/passwd")
    assert not validate_outbound_url("http://localhost:8080")
    assert validate_outbound_url("http://192.168.1.1", allow_private=False) is False


def test_usable_cpu_count_positive():
    n = usable_cpu_count()
    assert 1 <= n <= 1024


def test_model_lifecycle_warm_unload():
    calls = []
    lc = ModelLifecycle(lambda: calls.append(1) or object(), idle_seconds=0.0)
    a = lc.get()
    assert lc.loaded and len(calls) == 1
    assert lc.get() is a and len(calls) == 1
    assert lc.maybe_unload()
    assert not lc.loaded
    lc.get()
    assert len(calls) == 2


def test_normalize_title():
    assert normalize_title("Song (Remastered 2011)") == "song"
    assert normalize_title("Track feat. Someone") == "track"
    assert normalize_title("It's All Good!") == "it s all good"


@pytest.fixture
def db(tmp_db_url):
    conn = connect(tmp_db_url)
    init_db(conn)
    yield conn
    conn.close()


def _seed_catalogue(conn):
    from audiomuse_amd.db import write_txn

    rng = np.random.default_rng(0)
    for i in range(5):
        save_track_analysis_and_embedding(
            conn, f"fp_4{'%050x' % i}", title=f"Song {i}", author="A",
            embedding=rng.standard_normal(200).astype(np.float32))
        with write_txn(conn):
            conn.execute(
                """INSERT INTO track_server_map (provider_id, server_id,
                       item_id, title, author, file_path)
                   VALUES (?,?,?,?,?,?)""",
                (f"p{i}", "srv1", f"fp_4{'%050x' % i}", f"Song {i}", "A",
                 f"/m/A/Song {i}.flac"))


def test_align_server_tracks_tiers(db):
    _seed_catalogue(db)
    new_tracks = [
        Track(provider_id="x0", title="zzz", author="zzz",
              file_path="/m/A/Song 0.flac"),                       # path
        Track(provider_id="x1", title="zzz", author="zzz",
              file_path="/other/Song 1.flac"),                     # tail
        Track(provider_id="x2", title="Song 2", author="A"),       # exact
        Track(provider_id="x3", title="Song 3 (Live)", author="A"),  # norm
        Track(provider_id="x4", title="Brand New", author="B"),    # none
    ]
    tiers = align_server_tracks(db, "srv2", new_tracks)
    assert tiers == {"path": 1, "tail": 1, "exact": 1, "normalized": 1,
                     "unmatched": 1}
    n = db.execute("SELECT COUNT(*) AS n FROM track_server_map "
                   "WHERE server_id='srv2'").fetchone()["n"]
    assert n == 4


def test_backup_roundtrip(db, tmp_path):
    _seed_catalogue(db)
    dest = str(tmp_path / "backup.db")
    backup_database(db, dest)
    b = sqlite3.connect(dest)
    assert b.execute("SELECT COUNT(*) FROM score").fetchone()[0] == 5
    b.close()


def test_dashboard_stats(db):
    _seed_catalogue(db)
    stats = refresh_dashboard_stats(db)
    assert stats["tracks"] == 5 and stats["embeddings"] == 5
    row = db.execute("SELECT value FROM dashboard_stats WHERE key='tracks'"
                     ).fetchone()
    assert row["value"] == "5"


def test_catalogue_never_deleted_by_cleaning(db, tmp_db_url):
    """The reference's guarded invariant (test_catalogue_is_never_deleted):
    cleaning removes mappings only."""
    import json

    from audiomuse_amd.taskqueue import SUCCESS, enqueue, task_row
    from audiomuse_amd.taskqueue.worker import Worker

    _seed_catalogue(db)
    from audiomuse_amd.db import write_txn

    with write_txn(db):
        db.execute(
            """INSERT INTO track_server_map (provider_id, server_id, item_id)
               VALUES ('orphan', 'srv1', 'fp_4dead')""")
    url = tmp_db_url
    tid = enqueue(db, "clean_orphans", {"delete": True})
    Worker(db_url=url, max_jobs=1).run_forever(idle_timeout=3.0)
    row = task_row(db, tid)
    assert row["status"] == SUCCESS
    result = json.loads(row["result"])
    assert result["orphans"] == 1 and result["deleted"] == 1
    # catalogue untouched
    assert db.execute("SELECT COUNT(*) FROM score").fetchone()[0] == 5
    assert db.execute("SELECT COUNT(*) FROM embedding").fetchone()[0] == 5


def test_oom_retry_halves_batch():
    import torch

    from audiomuse_amd.utils.resources import oom_retry

    calls = []

    def fn(b):
        calls.append(b.shape[0])
        if b.shape[0] > 3:
            raise torch.cuda.OutOfMemoryError("fake OOM")
        return b * 2

    x = torch.arange(10).reshape(10, 1).float()
    assert torch.equal(oom_retry(fn, x), x * 2)
    assert calls[0] == 10 and max(calls[2:]) <= 3  # halved until it fit

    def always(b):
        raise torch.cuda.OutOfMemoryError("cannot fit even one row")

    import pytest as _pytest
    with _pytest.raises(torch.cuda.OutOfMemoryError):
        oom_retry(always, x)
