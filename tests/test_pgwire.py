"""First-party PostgreSQL wire driver (db/pgwire.py) against the
in-process protocol stub (tests/pgstub.py) — startup + SCRAM, codecs,
placeholders, transactions, LISTEN/NOTIFY, advisory locks, and the
storage layer's backend dispatch."""

import threading
import time

import pytest

from audiomuse_amd.db import (CHAN_JOB, backend_kind, connect,
                              insert_returning_id, listen, notify,
                              wait_notify, write_txn)
from audiomuse_amd.db.pgwire import (PGError, ProtocolError, Row,
                                     connect_url, qmark_to_dollar,
                                     sqlite_dialect_to_pg)
from audiomuse_amd.db.schema import DDL, DDL_PG, init_db, to_postgres
from tests.pgstub import StubServer, pg_to_sqlite


@pytest.fixture
def stub(tmp_path):
    srv = StubServer(tmp_path / "stub.db").start()
    yield srv
    srv.stop()


@pytest.fixture
def pg(stub):
    conn = connect_url(stub.url)
    yield conn
    conn.close()


# -- SQL munging -------------------------------------------------------------

def test_qmark_translation():
    assert qmark_to_dollar("SELECT ?, ?") == "SELECT $1, $2"
    # literals survive
    assert qmark_to_dollar("SELECT '?', ?") == "SELECT '?', $1"
    assert qmark_to_dollar("SELECT 'it''s ?', ?") == "SELECT 'it''s ?', $1"


def test_dialect_translation():
    assert sqlite_dialect_to_pg("BEGIN IMMEDIATE") == "BEGIN"
    out = sqlite_dialect_to_pg(
        "UPDATE t SET x=(julianday('now') - 2440587.5) * 86400.0")
    assert "EXTRACT(EPOCH FROM now())" in out and "julianday" not in out


def test_ddl_translation_is_mechanical():
    assert "BIGSERIAL PRIMARY KEY" in DDL_PG
    assert "BLOB" not in DDL_PG and "BYTEA" in DDL_PG
    assert "julianday" not in DDL_PG
    # round trip through the stub's down-translation reproduces the
    # canonical sqlite DDL (modulo whitespace) — one table catalogue
    assert pg_to_sqlite(to_postgres(DDL)).split() == DDL.split()


# -- protocol ----------------------------------------------------------------

def test_scram_auth_and_bad_password(stub):
    conn = connect_url(stub.url)
    assert conn.execute("SELECT 1 AS one").fetchone()["one"] == 1
    conn.close()
    bad = stub.url.replace("audiomuse-test", "wrong")
    with pytest.raises((PGError, ProtocolError)):
        connect_url(bad)


@pytest.mark.parametrize("mode", ["md5", "cleartext"])
def test_legacy_auth_modes(tmp_path, mode):
    """Servers configured for md5 or password auth (pgwire.py:343-352)
    still connect; a wrong password is refused the same way."""
    srv = StubServer(tmp_path / f"{mode}.db", auth_mode=mode).start()
    try:
        conn = connect_url(srv.url)
        assert conn.execute("SELECT 7 AS x").fetchone()["x"] == 7
        conn.close()
        with pytest.raises((PGError, ProtocolError, OSError)):
            connect_url(srv.url.replace("audiomuse-test", "wrong"))
    finally:
        srv.stop()


def test_executemany_accumulates_rowcount(pg):
    pg.execute("CREATE TABLE em (v INTEGER)")
    cur = pg.executemany("INSERT INTO em VALUES (?)",
                         [(i,) for i in range(5)])
    assert cur.rowcount == 5
    rows = pg.execute("SELECT v FROM em ORDER BY v").fetchall()
    assert [r["v"] for r in rows] == list(range(5))


def test_type_round_trip(pg):
    pg.execute("CREATE TABLE t (i INTEGER, f DOUBLE PRECISION, s TEXT, "
               "b BYTEA, n TEXT)")
    blob = bytes(range(256))
    pg.execute("INSERT INTO t VALUES (?,?,?,?,?)",
               (42, 1.5, "héllo 'quoted'", blob, None))
    row = pg.execute("SELECT * FROM t").fetchone()
    assert row["i"] == 42 and row["f"] == 1.5
    assert row["s"] == "héllo 'quoted'"
    assert row["b"] == blob and row["n"] is None
    # index access like sqlite3.Row
    assert row[0] == 42 and len(row) == 5 and "i" in row.keys()


def test_rowcount_and_returning(pg):
    pg.execute("CREATE TABLE r (id INTEGER PRIMARY KEY AUTOINCREMENT, "
               "v TEXT)")
    cur = pg.execute("INSERT INTO r (v) VALUES (?)", ("a",))
    assert cur.rowcount == 1
    rid = insert_returning_id(pg, "INSERT INTO r (v) VALUES (?)", ("b",))
    assert rid == 2
    cur = pg.execute("UPDATE r SET v='z'")
    assert cur.rowcount == 2


def test_transactions_and_write_txn(pg):
    pg.execute("CREATE TABLE tx (v INTEGER)")
    assert not pg.in_transaction
    with write_txn(pg):
        assert pg.in_transaction
        pg.execute("INSERT INTO tx VALUES (1)")
    assert not pg.in_transaction
    with pytest.raises(RuntimeError):
        with write_txn(pg):
            pg.execute("INSERT INTO tx VALUES (2)")
            raise RuntimeError("boom")
    rows = pg.execute("SELECT v FROM tx").fetchall()
    assert [r["v"] for r in rows] == [1]  # second insert rolled back


def test_error_resync(pg):
    with pytest.raises(PGError):
        pg.execute("SELECT * FROM missing_table")
    # connection still usable after the error
    assert pg.execute("SELECT 2 AS two").fetchone()["two"] == 2


def test_listen_notify(stub):
    a = connect_url(stub.url)
    b = connect_url(stub.url)
    a.listen("audiomuse_job")
    b.notify("audiomuse_job", "t123")
    got = a.wait_notify(timeout=5.0)
    assert got and got[0][1] == "audiomuse_job" and got[0][2] == "t123"
    # no queued notifications -> empty after timeout
    assert a.wait_notify(timeout=0.05) == []
    a.close()
    b.close()


def test_advisory_locks_die_with_connection(stub):
    from audiomuse_amd.taskqueue import sql as qsql

    a = connect_url(stub.url)
    b = connect_url(stub.url)
    assert qsql.try_advisory_lock(a, qsql.LOCK_CLASS_TASK, "t1")
    assert not qsql.try_advisory_lock(b, qsql.LOCK_CLASS_TASK, "t1")
    a.close()
    time.sleep(0.2)  # stub releases on socket close
    assert qsql.try_advisory_lock(b, qsql.LOCK_CLASS_TASK, "t1")
    qsql.advisory_unlock(b, qsql.LOCK_CLASS_TASK, "t1")
    b.close()


# -- storage layer on the PG backend ----------------------------------------

def test_backend_dispatch_and_schema(stub):
    assert backend_kind(stub.url) == "postgres"
    conn = connect(stub.url)
    assert backend_kind(conn) == "postgres"
    init_db(conn)
    init_db(conn)  # idempotent
    conn.execute("INSERT INTO score (item_id, title) VALUES (?,?)",
                 ("fp_1", "Song"))
    row = conn.execute("SELECT * FROM score WHERE item_id=?",
                       ("fp_1",)).fetchone()
    assert row["title"] == "Song"
    conn.close()


def test_db_notify_helpers(stub):
    a = connect(stub.url)
    b = connect(stub.url)
    listen(a, CHAN_JOB)
    notify(b, CHAN_JOB, "wake")
    got = wait_notify(a, timeout=5.0)
    assert got and got[0][1] == CHAN_JOB
    a.close()
    b.close()


def test_concurrent_claims_are_exclusive_on_pg(stub):
    """FOR UPDATE SKIP LOCKED path: N threads, no double-claims."""
    from audiomuse_amd.taskqueue import enqueue
    from audiomuse_amd.taskqueue import sql as qsql

    seed = connect(stub.url)
    init_db(seed)
    ids = [enqueue(seed, "noop") for _ in range(20)]
    claimed = []
    lock = threading.Lock()

    def worker(wid):
        conn = connect(stub.url)
        try:
            while True:
                row = qsql.claim(conn, wid)
                if row is None:
                    return
                with lock:
                    claimed.append(row["task_id"])
        finally:
            conn.close()

    threads = [threading.Thread(target=worker, args=(f"w{i}",))
               for i in range(4)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=30)
    assert sorted(claimed) == sorted(ids)
    assert len(set(claimed)) == len(ids)
    seed.close()
