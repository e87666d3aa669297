"""In-process PostgreSQL wire-protocol test double ("pgstub").

No PostgreSQL server, psycopg wheel, or pgserver exists in this image
(docs/POSTGRES.md records the attempted installs), so backend tests
speak the real v3 wire protocol — startup, SCRAM-SHA-256, extended
query, LISTEN/NOTIFY, advisory locks — against this stub, which executes
the SQL on a shared SQLite file. It is the moral equivalent of the
reference's ephemeral embedded Postgres fixture
(/root/reference/test/integration/conftest.py:74-96) scaled down to
what the image allows.

Fidelity notes (deliberate simplifications, all test-only):
- SQL runs on SQLite after a mechanical PG->SQLite down-translation
  (``$n`` params, ``FOR UPDATE SKIP LOCKED`` serialized via the
  single-writer BEGIN IMMEDIATE lock, EXTRACT(EPOCH FROM now())).
- Parameter types are inferred from text form (int -> float -> hex
  bytea -> str); fine for this schema, not a general PG.
- Advisory locks and LISTEN/NOTIFY are implemented in the stub process
  and release on connection close — exactly the semantics the queue's
  orphan-reclaim logic depends on (reference maintenance.py:177).

Set AUDIOMUSE_TEST_DATABASE_URL to run the same tests against a real
PostgreSQL instead.
"""

from __future__ import annotations

import base64
import hashlib
import hmac
import re
import secrets
import socket
import sqlite3
import struct
import threading
from typing import Dict, List, Optional, Set, Tuple

_SCRAM_ITERS = 4096
STUB_USER = "audiomuse"
STUB_PASSWORD = "audiomuse-test"
STUB_DB = "audiomuse"


def _msg(tag: bytes, payload: bytes) -> bytes:
    return tag + struct.pack("!I", len(payload) + 4) + payload


def pg_to_sqlite(sql: str) -> str:
    sql = re.sub(r"\$(\d+)", r"?\1", sql)
    sql = sql.replace("FOR UPDATE SKIP LOCKED", "")
    sql = sql.replace("EXTRACT(EPOCH FROM now())",
                      "(julianday('now') - 2440587.5) * 86400.0")
    sql = sql.replace("BIGSERIAL PRIMARY KEY",
                      "INTEGER PRIMARY KEY AUTOINCREMENT")
    sql = sql.replace("BYTEA", "BLOB")
    sql = sql.replace("DOUBLE PRECISION", "REAL")
    head = sql.lstrip().upper()
    if head.startswith("BEGIN"):
        # preserve PG's writers-block-writers claim semantics through
        # SQLite's single-writer lock
        return "BEGIN IMMEDIATE"
    return sql


def _infer_param(raw: Optional[bytes]):
    if raw is None:
        return None
    text = raw.decode("utf-8")
    try:
        return int(text)
    except ValueError:
        pass
    try:
        return float(text)
    except ValueError:
        pass
    if text.startswith("\\x"):
        try:
            return bytes.fromhex(text[2:])
        except ValueError:
            pass
    return text


def _oid_for(values) -> int:
    for v in values:
        if v is None:
            continue
        if isinstance(v, bool):
            return 16
        if isinstance(v, int):
            return 20
        if isinstance(v, float):
            return 701
        if isinstance(v, (bytes, memoryview)):
            return 17
        return 25
    return 25


def _encode_cell(v) -> Optional[bytes]:
    if v is None:
        return None
    if isinstance(v, bool):
        return b"t" if v else b"f"
    if isinstance(v, (bytes, memoryview)):
        return b"\\x" + bytes(v).hex().encode()
    if isinstance(v, (int, float)):
        return repr(v).encode()
    return str(v).encode("utf-8")


class StubServer:
    """Threaded PG-wire server over one shared SQLite file."""

    def __init__(self, db_path: str, require_auth: bool = True,
                 auth_mode: str = "scram"):
        self.db_path = str(db_path)
        self.require_auth = require_auth
        self.auth_mode = auth_mode  # scram | md5 | cleartext
        self._srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._srv.bind(("127.0.0.1", 0))
        self._srv.listen(32)
        self.port = self._srv.getsockname()[1]
        self._stop = threading.Event()
        self._threads: List[threading.Thread] = []
        self._state_lock = threading.Lock()
        self._advisory: Dict[Tuple[int, str], int] = {}
        self._listeners: Dict[int, Set[str]] = {}
        self._conn_socks: Dict[int, Tuple[socket.socket, threading.Lock]] = {}
        self._next_id = 0

    @property
    def url(self) -> str:
        return (f"postgresql://{STUB_USER}:{STUB_PASSWORD}"
                f"@127.0.0.1:{self.port}/{STUB_DB}")

    def start(self) -> "StubServer":
        t = threading.Thread(target=self._accept_loop, daemon=True)
        t.start()
        self._threads.append(t)
        return self

    def stop(self) -> None:
        self._stop.set()
        try:
            self._srv.close()
        except OSError:
            pass

    def _accept_loop(self) -> None:
        while not self._stop.is_set():
            try:
                cli, _ = self._srv.accept()
            except OSError:
                return
            with self._state_lock:
                self._next_id += 1
                cid = self._next_id
            t = threading.Thread(target=self._serve_client,
                                 args=(cli, cid), daemon=True)
            t.start()
            self._threads.append(t)

    # ---- per-connection ----

    def _serve_client(self, sock: socket.socket, cid: int) -> None:
        wlock = threading.Lock()
        with self._state_lock:
            self._conn_socks[cid] = (sock, wlock)
        db = sqlite3.connect(self.db_path, timeout=30.0, isolation_level=None)
        db.row_factory = sqlite3.Row
        db.execute("PRAGMA journal_mode=WAL")
        db.execute("PRAGMA busy_timeout=30000")
        try:
            if not self._handshake(sock, wlock):
                return
            self._message_loop(sock, wlock, db, cid)
        except (ConnectionError, OSError, _ClientGone):
            pass
        finally:
            db.close()
            with self._state_lock:
                self._conn_socks.pop(cid, None)
                self._listeners.pop(cid, None)
                # advisory locks die with the connection — the liveness
                # semantics the orphan reclaim depends on
                for key in [k for k, v in self._advisory.items() if v == cid]:
                    del self._advisory[key]
            try:
                sock.close()
            except OSError:
                pass

    @staticmethod
    def _read_exact(sock: socket.socket, n: int) -> bytes:
        buf = b""
        while len(buf) < n:
            chunk = sock.recv(n - len(buf))
            if not chunk:
                raise _ClientGone()
            buf += chunk
        return buf

    def _handshake(self, sock: socket.socket, wlock: threading.Lock) -> bool:
        while True:
            (length,) = struct.unpack("!I", self._read_exact(sock, 4))
            payload = self._read_exact(sock, length - 4)
            (code,) = struct.unpack("!I", payload[:4])
            if code == 80877103:  # SSLRequest
                sock.sendall(b"N")
                continue
            if code != 196608:
                return False
            break
        out = b""
        if self.require_auth:
            handler = {"scram": self._scram, "md5": self._md5,
                       "cleartext": self._cleartext}[self.auth_mode]
            if not handler(sock):
                return False
        out += _msg(b"R", struct.pack("!I", 0))
        out += _msg(b"S", b"server_version\x00pgstub 15.0\x00")
        out += _msg(b"K", struct.pack("!II", 4242, 4242))
        out += _msg(b"Z", b"I")
        with wlock:
            sock.sendall(out)
        return True

    def _auth_failed(self, sock: socket.socket) -> bool:
        sock.sendall(_msg(b"E", b"SFATAL\x00C28P01\x00"
                          b"Mpassword authentication failed\x00\x00"))
        return False

    def _read_password_msg(self, sock: socket.socket) -> bytes:
        tag = self._read_exact(sock, 1)
        (length,) = struct.unpack("!I", self._read_exact(sock, 4))
        body = self._read_exact(sock, length - 4)
        if tag != b"p":
            raise _ClientGone()
        return body.rstrip(b"\x00")

    def _md5(self, sock: socket.socket) -> bool:
        salt = secrets.token_bytes(4)
        sock.sendall(_msg(b"R", struct.pack("!I", 5) + salt))
        body = self._read_password_msg(sock)
        inner = hashlib.md5(
            STUB_PASSWORD.encode() + STUB_USER.encode()).hexdigest()
        expect = b"md5" + hashlib.md5(
            inner.encode() + salt).hexdigest().encode()
        if not hmac.compare_digest(body, expect):
            return self._auth_failed(sock)
        return True

    def _cleartext(self, sock: socket.socket) -> bool:
        sock.sendall(_msg(b"R", struct.pack("!I", 3)))
        if self._read_password_msg(sock) != STUB_PASSWORD.encode():
            return self._auth_failed(sock)
        return True

    def _scram(self, sock: socket.socket) -> bool:
        sock.sendall(_msg(b"R", struct.pack("!I", 10) + b"SCRAM-SHA-256\x00\x00"))
        tag = self._read_exact(sock, 1)
        (length,) = struct.unpack("!I", self._read_exact(sock, 4))
        body = self._read_exact(sock, length - 4)
        if tag != b"p":
            return False
        mech, rest = body.split(b"\x00", 1)
        if mech != b"SCRAM-SHA-256":
            return False
        (dlen,) = struct.unpack("!I", rest[:4])
        client_first = rest[4:4 + dlen].decode()
        bare = client_first.split(",", 2)[2]
        client_nonce = dict(kv.split("=", 1)
                            for kv in bare.split(","))["r"]
        nonce = client_nonce + base64.b64encode(
            secrets.token_bytes(12)).decode()
        salt = secrets.token_bytes(16)
        server_first = (f"r={nonce},s={base64.b64encode(salt).decode()},"
                        f"i={_SCRAM_ITERS}")
        sock.sendall(_msg(b"R", struct.pack("!I", 11) + server_first.encode()))
        tag = self._read_exact(sock, 1)
        (length,) = struct.unpack("!I", self._read_exact(sock, 4))
        final = self._read_exact(sock, length - 4).decode()
        fields = dict(kv.split("=", 1) for kv in final.split(","))
        final_no_proof = final[:final.rindex(",p=")]
        auth_message = ",".join([bare, server_first, final_no_proof]).encode()
        salted = hashlib.pbkdf2_hmac("sha256", STUB_PASSWORD.encode(),
                                     salt, _SCRAM_ITERS)
        client_key = hmac.new(salted, b"Client Key", hashlib.sha256).digest()
        stored_key = hashlib.sha256(client_key).digest()
        signature = hmac.new(stored_key, auth_message,
                             hashlib.sha256).digest()
        expect = bytes(a ^ b for a, b in zip(client_key, signature))
        if not hmac.compare_digest(base64.b64decode(fields["p"]), expect):
            sock.sendall(_msg(b"E", b"SFATAL\x00C28P01\x00"
                              b"Mpassword authentication failed\x00\x00"))
            return False
        server_key = hmac.new(salted, b"Server Key", hashlib.sha256).digest()
        server_sig = hmac.new(server_key, auth_message,
                              hashlib.sha256).digest()
        sock.sendall(_msg(b"R", struct.pack("!I", 12) + b"v="
                          + base64.b64encode(server_sig)))
        return True

    # ---- query handling ----

    def _message_loop(self, sock, wlock, db, cid) -> None:
        pending_sql = ""
        pending_params: List = []
        out = bytearray()
        while True:
            tag = self._read_exact(sock, 1)
            (length,) = struct.unpack("!I", self._read_exact(sock, 4))
            body = self._read_exact(sock, length - 4)
            if tag == b"X":
                return
            if tag == b"Q":
                sql = body.rstrip(b"\x00").decode("utf-8")
                resp = self._run_statement(db, cid, sql, [], simple=True)
                resp += _msg(b"Z", b"T" if db.in_transaction else b"I")
                with wlock:
                    sock.sendall(resp)
            elif tag == b"P":
                _name, rest = body.split(b"\x00", 1)
                sql, _rest = rest.split(b"\x00", 1)
                pending_sql = sql.decode("utf-8")
                out += _msg(b"1", b"")
            elif tag == b"B":
                off = body.index(b"\x00") + 1
                off = body.index(b"\x00", off) + 1
                (nfmt,) = struct.unpack("!H", body[off:off + 2])
                off += 2 + 2 * nfmt
                (nparams,) = struct.unpack("!H", body[off:off + 2])
                off += 2
                pending_params = []
                for _ in range(nparams):
                    (ln,) = struct.unpack("!i", body[off:off + 4])
                    off += 4
                    if ln < 0:
                        pending_params.append(None)
                    else:
                        pending_params.append(_infer_param(body[off:off + ln]))
                        off += ln
                out += _msg(b"2", b"")
            elif tag == b"D":
                out += _msg(b"n", b"")
            elif tag == b"E":
                out += self._run_statement(db, cid, pending_sql,
                                           pending_params, simple=False)
            elif tag == b"S":
                out += _msg(b"Z", b"T" if db.in_transaction else b"I")
                with wlock:
                    sock.sendall(bytes(out))
                out = bytearray()
            # ignore anything else

    def _run_statement(self, db, cid: int, sql: str, params: List,
                       simple: bool) -> bytes:
        try:
            return self._dispatch(db, cid, sql, params)
        except sqlite3.Error as e:
            state = "23505" if isinstance(e, sqlite3.IntegrityError) else "XX000"
            fields = (b"SERROR\x00C" + state.encode() + b"\x00M"
                      + str(e).encode("utf-8", "replace") + b"\x00\x00")
            if db.in_transaction and not simple:
                pass  # driver raises after Z; txn left for ROLLBACK
            return _msg(b"E", fields)

    def _dispatch(self, db, cid: int, sql: str, params: List) -> bytes:
        compact = " ".join(sql.split())
        lowered = compact.lower()
        if "pg_try_advisory_lock" in lowered:
            klass, key = int(params[0]), str(params[1])
            with self._state_lock:
                owner = self._advisory.get((klass, key))
                got = owner is None or owner == cid
                if got:
                    self._advisory[(klass, key)] = cid
            return self._rows_response([("got", [got])])
        if "pg_advisory_unlock" in lowered:
            klass, key = int(params[0]), str(params[1])
            with self._state_lock:
                if self._advisory.get((klass, key)) == cid:
                    del self._advisory[(klass, key)]
                    ok = True
                else:
                    ok = False
            return self._rows_response([("pg_advisory_unlock", [ok])])
        if "pg_notify" in lowered:
            self._deliver_notify(str(params[0]), str(params[1] or ""))
            return self._rows_response([("pg_notify", [None])])
        if lowered.startswith("listen"):
            chan = compact.split(None, 1)[1].strip().strip(';').strip('"')
            with self._state_lock:
                self._listeners.setdefault(cid, set()).add(chan)
            return _msg(b"C", b"LISTEN\x00")
        if lowered.startswith("notify"):
            rest = compact.split(None, 1)[1]
            chan = rest.split(",", 1)[0].strip().strip('"')
            payload = ""
            if "," in rest:
                payload = rest.split(",", 1)[1].strip().strip("'")
            self._deliver_notify(chan, payload)
            return _msg(b"C", b"NOTIFY\x00")
        sqlite_sql = pg_to_sqlite(sql)
        # multi-statement scripts (DDL) run via executescript
        stripped = sqlite_sql.strip().rstrip(";")
        if ";" in stripped:
            db.executescript(sqlite_sql)
            return _msg(b"C", b"OK\x00")
        cur = db.execute(sqlite_sql, params)
        rows = cur.fetchall() if cur.description else []
        verb = stripped.split(None, 1)[0].upper() if stripped else "OK"
        if cur.description:
            cols = [d[0] for d in cur.description]
            data = [tuple(r) for r in rows]
            resp = self._table_response(cols, data)
            resp += _msg(b"C", b"SELECT %d\x00" % len(data))
            return resp
        n = max(cur.rowcount, 0)
        if verb == "INSERT":
            tag = b"INSERT 0 %d" % n
        elif verb in ("UPDATE", "DELETE"):
            tag = b"%s %d" % (verb.encode(), n)
        else:
            tag = verb.encode()
        return _msg(b"C", tag + b"\x00")

    def _deliver_notify(self, chan: str, payload: str) -> None:
        with self._state_lock:
            targets = [(self._conn_socks[c])
                       for c, chans in self._listeners.items()
                       if chan in chans and c in self._conn_socks]
        body = (struct.pack("!I", 4242) + chan.encode() + b"\x00"
                + payload.encode() + b"\x00")
        for sock, wlock in targets:
            try:
                with wlock:
                    sock.sendall(_msg(b"A", body))
            except OSError:
                pass

    @staticmethod
    def _row_description(cols: List[str], oids: List[int]) -> bytes:
        body = struct.pack("!H", len(cols))
        for name, oid in zip(cols, oids):
            body += name.encode() + b"\x00"
            body += struct.pack("!IHIhih", 0, 0, oid, -1, -1, 0)
        return _msg(b"T", body)

    def _table_response(self, cols: List[str], data: List[tuple]) -> bytes:
        oids = [_oid_for([row[i] for row in data])
                for i in range(len(cols))]
        resp = self._row_description(cols, oids)
        for row in data:
            body = struct.pack("!H", len(row))
            for v in row:
                enc = _encode_cell(v)
                if enc is None:
                    body += struct.pack("!i", -1)
                else:
                    body += struct.pack("!i", len(enc)) + enc
            resp += _msg(b"D", body)
        return resp

    def _rows_response(self, cols_vals: List[Tuple[str, List]]) -> bytes:
        cols = [c for c, _ in cols_vals]
        data = [tuple(vs[0] for _, vs in cols_vals)]
        resp = self._table_response(cols, data)
        resp += _msg(b"C", b"SELECT 1\x00")
        return resp


class _ClientGone(Exception):
    pass
