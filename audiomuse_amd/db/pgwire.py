"""First-party PostgreSQL frontend/backend (v3) wire-protocol driver.

The reference talks to PostgreSQL through psycopg2
(/root/reference/database.py:1, taskqueue/sql.py:415-462 FOR UPDATE SKIP
LOCKED claim, sql.py:48-52 LISTEN/NOTIFY channels, advisory locks). This
image has no psycopg/psycopg2/pg8000 wheel and no package index (see
docs/POSTGRES.md for the attempted-install log), so the PG backend ships
its own driver: a single-file, dependency-free implementation of the
protocol v3 with

- startup + auth: trust, cleartext, md5, SCRAM-SHA-256 (RFC 7677 via
  hashlib/hmac — no channel binding)
- extended query protocol (Parse/Bind/Execute/Sync) with ``?``
  placeholder translation to ``$n`` so callers reuse the exact SQL the
  SQLite backend runs
- text-format parameter/result codecs for the types the schema uses
  (bool, ints, floats, numeric, text, bytea)
- async NotificationResponse collection + ``wait_notify`` (LISTEN/NOTIFY
  — the queue's wake channels)
- an sqlite3-shaped surface (``execute`` -> cursor with
  fetchone/fetchall/rowcount/lastrowid/iteration, ``executescript``,
  ``in_transaction``) so the storage layer runs unchanged on either
  backend

Deliberately NOT implemented (unneeded here): binary row format, COPY,
SSL/GSS negotiation, multiple result sets per Parse, portals with
partial fetches.
"""

from __future__ import annotations

import base64
import hashlib
import hmac
import os
import re
import secrets
import select
import socket
import struct
from collections import deque
from typing import Any, Deque, Dict, Iterator, List, Optional, Sequence, Tuple
from urllib.parse import unquote, urlparse

PROTOCOL_V3 = 196608


class PGError(Exception):
    """Server-reported error (ErrorResponse)."""

    def __init__(self, fields: Dict[str, str]):
        self.fields = fields
        self.sqlstate = fields.get("C", "")
        super().__init__(
            f"{fields.get('S', 'ERROR')}: {fields.get('M', '?')} "
            f"(sqlstate {self.sqlstate})")


class ProtocolError(Exception):
    pass


# -- SQL text munging --------------------------------------------------------

_QMARK_RE = re.compile(r"\?")
_STR_OR_QMARK = re.compile(r"'(?:[^']|'')*'|\?")


def qmark_to_dollar(sql: str) -> str:
    """``?`` positional placeholders -> ``$1..$n`` (skips string literals)."""
    n = 0

    def repl(m: re.Match) -> str:
        nonlocal n
        tok = m.group(0)
        if tok != "?":
            return tok
        n += 1
        return f"${n}"

    return _STR_OR_QMARK.sub(repl, sql)


_JULIAN_NOW = "(julianday('now') - 2440587.5) * 86400.0"


def sqlite_dialect_to_pg(sql: str) -> str:
    """Translate the few SQLite-isms the storage layer emits so one SQL
    corpus serves both backends (kept deliberately small + literal)."""
    if _JULIAN_NOW in sql:
        sql = sql.replace(_JULIAN_NOW, "EXTRACT(EPOCH FROM now())")
    s = sql.lstrip()[:16].upper()
    if s.startswith("BEGIN IMMEDIATE"):
        sql = "BEGIN"
    return sql


# -- rows / cursors ----------------------------------------------------------

class Row:
    """Mapping+sequence row, same access patterns as sqlite3.Row."""

    __slots__ = ("_cols", "_vals")

    def __init__(self, cols: Dict[str, int], vals: tuple):
        self._cols = cols
        self._vals = vals

    def __getitem__(self, key):
        if isinstance(key, str):
            return self._vals[self._cols[key]]
        return self._vals[key]

    def __iter__(self):
        return iter(self._vals)

    def __len__(self):
        return len(self._vals)

    def keys(self) -> List[str]:
        return list(self._cols)

    def __contains__(self, key):
        return key in self._cols if isinstance(key, str) else key in self._vals

    def __eq__(self, other):
        if isinstance(other, Row):
            return self._vals == other._vals and self._cols == other._cols
        return NotImplemented

    def __repr__(self):
        return "Row(%s)" % ", ".join(
            f"{k}={self._vals[i]!r}" for k, i in self._cols.items())


def _dict_row_compat(row: Row) -> dict:
    return {k: row[k] for k in row.keys()}


class Cursor:
    """Materialized result of one statement (text protocol)."""

    def __init__(self, rows: List[Row], rowcount: int,
                 lastrowid: Optional[int] = None):
        self._rows = rows
        self._pos = 0
        self.rowcount = rowcount
        self.lastrowid = lastrowid

    def fetchone(self) -> Optional[Row]:
        if self._pos >= len(self._rows):
            return None
        row = self._rows[self._pos]
        self._pos += 1
        return row

    def fetchall(self) -> List[Row]:
        out = self._rows[self._pos:]
        self._pos = len(self._rows)
        return out

    def fetchmany(self, size: int = 1) -> List[Row]:
        out = self._rows[self._pos:self._pos + size]
        self._pos += len(out)
        return out

    def __iter__(self) -> Iterator[Row]:
        while True:
            row = self.fetchone()
            if row is None:
                return
            yield row


# -- type codecs (text format) -----------------------------------------------

_OID_BOOL = 16
_OID_BYTEA = 17
_OID_INT8, _OID_INT2, _OID_INT4 = 20, 21, 23
_OID_FLOAT4, _OID_FLOAT8 = 700, 701
_OID_NUMERIC = 1700

_INT_OIDS = {_OID_INT2, _OID_INT4, _OID_INT8}
_FLOAT_OIDS = {_OID_FLOAT4, _OID_FLOAT8, _OID_NUMERIC}


def decode_value(oid: int, raw: Optional[bytes]) -> Any:
    if raw is None:
        return None
    if oid in _INT_OIDS:
        return int(raw)
    if oid in _FLOAT_OIDS:
        return float(raw)
    if oid == _OID_BOOL:
        return raw == b"t"
    if oid == _OID_BYTEA:
        if raw.startswith(b"\\x"):
            return bytes.fromhex(raw[2:].decode())
        # legacy escape format (server setting bytea_output='escape')
        return _decode_bytea_escape(raw)
    return raw.decode("utf-8")


def _decode_bytea_escape(raw: bytes) -> bytes:
    out = bytearray()
    i = 0
    while i < len(raw):
        b = raw[i]
        if b == 0x5C:  # backslash
            if raw[i + 1:i + 2] == b"\\":
                out.append(0x5C)
                i += 2
            else:
                out.append(int(raw[i + 1:i + 4], 8))
                i += 4
        else:
            out.append(b)
            i += 1
    return bytes(out)


def encode_param(value: Any) -> Optional[bytes]:
    """Python value -> text-format wire bytes (None stays NULL)."""
    if value is None:
        return None
    if isinstance(value, bool):
        return b"t" if value else b"f"
    if isinstance(value, (bytes, bytearray, memoryview)):
        return b"\\x" + bytes(value).hex().encode()
    if isinstance(value, (int, float)):
        return repr(value).encode()
    if isinstance(value, str):
        return value.encode("utf-8")
    raise TypeError(f"unsupported parameter type {type(value)!r}")


# -- connection --------------------------------------------------------------

_COMPLETE_ROWS = re.compile(rb"^[A-Z ]+?(?: \d+)? (\d+)$")


class PGConnection:
    """One socket, one session. Not thread-safe (use one per thread, as
    the storage layer already does for sqlite3)."""

    kind = "postgres"

    def __init__(self, host: str, port: int = 5432, user: str = "postgres",
                 password: str = "", dbname: str = "postgres",
                 connect_timeout: float = 10.0,
                 application_name: str = "audiomuse-amd"):
        self.notifications: Deque[Tuple[int, str, str]] = deque()
        self.parameters: Dict[str, str] = {}
        self._txn_status = b"I"
        self._backend_pid = 0
        self._backend_key = 0
        self._buf = b""
        self._stmt_counter = 0
        self._closed = False
        if host.startswith("/"):
            self._sock = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
            self._sock.settimeout(connect_timeout)
            self._sock.connect(os.path.join(host, f".s.PGSQL.{port}"))
        else:
            self._sock = socket.create_connection((host, port),
                                                  timeout=connect_timeout)
            self._sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        self._sock.settimeout(None)
        self._startup(user, password, dbname, application_name)

    # ---- low-level framing ----

    def _send(self, data: bytes) -> None:
        self._sock.sendall(data)

    def _msg(self, tag: bytes, payload: bytes) -> bytes:
        return tag + struct.pack("!I", len(payload) + 4) + payload

    def _read_exact(self, n: int) -> bytes:
        while len(self._buf) < n:
            chunk = self._sock.recv(65536)
            if not chunk:
                raise ProtocolError("server closed connection")
            self._buf += chunk
        out, self._buf = self._buf[:n], self._buf[n:]
        return out

    def _recv_message(self) -> Tuple[bytes, bytes]:
        head = self._read_exact(5)
        tag = head[:1]
        (length,) = struct.unpack("!I", head[1:5])
        payload = self._read_exact(length - 4)
        return tag, payload

    @staticmethod
    def _cstrings(payload: bytes) -> List[bytes]:
        return payload.split(b"\x00")

    def _handle_async(self, tag: bytes, payload: bytes) -> bool:
        """NoticeResponse / ParameterStatus / NotificationResponse can
        arrive interleaved anywhere; returns True if consumed."""
        if tag == b"A":
            (pid,) = struct.unpack("!I", payload[:4])
            chan, msg, _ = payload[4:].split(b"\x00", 2)
            self.notifications.append((pid, chan.decode(), msg.decode()))
            return True
        if tag == b"N":
            return True  # notices are non-fatal; ignore
        if tag == b"S":
            k, v, _ = payload.split(b"\x00", 2)
            self.parameters[k.decode()] = v.decode()
            return True
        return False

    @staticmethod
    def _error_fields(payload: bytes) -> Dict[str, str]:
        fields: Dict[str, str] = {}
        for part in payload.split(b"\x00"):
            if part:
                fields[chr(part[0])] = part[1:].decode("utf-8", "replace")
        return fields

    # ---- startup / auth ----

    def _startup(self, user: str, password: str, dbname: str,
                 app_name: str) -> None:
        kv = (f"user\x00{user}\x00database\x00{dbname}\x00"
              f"application_name\x00{app_name}\x00"
              f"client_encoding\x00UTF8\x00\x00").encode()
        payload = struct.pack("!I", PROTOCOL_V3) + kv
        self._send(struct.pack("!I", len(payload) + 4) + payload)
        scram = None
        while True:
            tag, body = self._recv_message()
            if self._handle_async(tag, body):
                continue
            if tag == b"E":
                raise PGError(self._error_fields(body))
            if tag == b"R":
                (code,) = struct.unpack("!I", body[:4])
                if code == 0:
                    continue  # AuthenticationOk
                if code == 3:  # cleartext
                    self._send(self._msg(b"p", password.encode() + b"\x00"))
                elif code == 5:  # md5
                    salt = body[4:8]
                    inner = hashlib.md5(
                        password.encode() + user.encode()).hexdigest()
                    digest = hashlib.md5(
                        inner.encode() + salt).hexdigest()
                    self._send(self._msg(b"p", b"md5" + digest.encode()
                                         + b"\x00"))
                elif code == 10:  # SASL mechanisms
                    mechs = [m for m in body[4:].split(b"\x00") if m]
                    if b"SCRAM-SHA-256" not in mechs:
                        raise ProtocolError(f"unsupported SASL mechs {mechs}")
                    scram = _ScramClient(user, password)
                    first = scram.client_first()
                    self._send(self._msg(
                        b"p", b"SCRAM-SHA-256\x00"
                        + struct.pack("!I", len(first)) + first))
                elif code == 11:  # SASL continue
                    assert scram is not None
                    self._send(self._msg(b"p", scram.client_final(body[4:])))
                elif code == 12:  # SASL final
                    assert scram is not None
                    scram.verify_server_final(body[4:])
                else:
                    raise ProtocolError(f"unsupported auth code {code}")
            elif tag == b"K":
                self._backend_pid, self._backend_key = struct.unpack(
                    "!II", body)
            elif tag == b"Z":
                self._txn_status = body
                return
            else:
                raise ProtocolError(f"unexpected startup message {tag!r}")

    # ---- public sqlite3-shaped surface ----

    @property
    def in_transaction(self) -> bool:
        return self._txn_status in (b"T", b"E")

    def close(self) -> None:
        if not self._closed:
            self._closed = True
            try:
                self._send(self._msg(b"X", b""))
            except OSError:
                pass
            self._sock.close()

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass

    def execute(self, sql: str, params: Sequence[Any] = ()) -> Cursor:
        sql = sqlite_dialect_to_pg(sql)
        stripped = sql.lstrip()[:6].upper()
        if stripped.startswith("PRAGMA"):
            return Cursor([], -1)  # sqlite-only; harmless no-op on PG
        if not params:
            return self._simple_query(sql)
        return self._extended_query(qmark_to_dollar(sql), list(params))

    def executemany(self, sql: str, seq_of_params) -> Cursor:
        total = 0
        for p in seq_of_params:
            cur = self.execute(sql, p)
            if cur.rowcount > 0:
                total += cur.rowcount
        return Cursor([], total)

    def executescript(self, script: str) -> Cursor:
        return self._simple_query(script)

    def commit(self) -> None:
        if self.in_transaction:
            self._simple_query("COMMIT")

    def rollback(self) -> None:
        if self.in_transaction:
            self._simple_query("ROLLBACK")

    # ---- queue/notify surface ----

    def listen(self, channel: str) -> None:
        self._simple_query(f'LISTEN "{channel}"')

    def notify(self, channel: str, payload: str = "") -> None:
        self.execute("SELECT pg_notify(?, ?)", (channel, payload))

    def wait_notify(self, timeout: float = 5.0) -> List[Tuple[int, str, str]]:
        """Drain queued notifications; if none, block up to ``timeout``
        for socket traffic and parse whatever arrives."""
        if not self.notifications:
            if self._buf or select.select([self._sock], [], [], timeout)[0]:
                self._drain_async()
        out = list(self.notifications)
        self.notifications.clear()
        return out

    def _drain_async(self) -> None:
        """Consume complete messages already queued on the socket
        (outside a query cycle only async messages are expected)."""
        self._sock.setblocking(False)
        try:
            while True:
                try:
                    chunk = self._sock.recv(65536)
                except (BlockingIOError, InterruptedError):
                    break
                if not chunk:
                    raise ProtocolError("server closed connection")
                self._buf += chunk
        finally:
            self._sock.setblocking(True)
        while len(self._buf) >= 5:
            (length,) = struct.unpack("!I", self._buf[1:5])
            if len(self._buf) < 1 + length:
                break
            tag, payload = self._recv_message()
            if not self._handle_async(tag, payload):
                if tag == b"E":
                    raise PGError(self._error_fields(payload))
                raise ProtocolError(f"unexpected async message {tag!r}")

    # ---- query cycles ----

    def _collect_results(self, *, until_sync: bool) -> Cursor:
        rows: List[Row] = []
        cols: Dict[str, int] = {}
        oids: List[int] = []
        rowcount = -1
        lastrowid = None
        error: Optional[PGError] = None
        while True:
            tag, body = self._recv_message()
            if self._handle_async(tag, body):
                continue
            if tag == b"E":
                error = PGError(self._error_fields(body))
            elif tag == b"T":
                cols, oids = self._parse_row_description(body)
                rows = []
            elif tag == b"D":
                rows.append(self._parse_data_row(body, cols, oids))
            elif tag == b"C":
                rowcount = self._parse_complete_tag(body.rstrip(b"\x00"))
            elif tag in (b"1", b"2", b"3", b"n", b"t", b"s", b"I"):
                pass  # ParseComplete/BindComplete/... — nothing to do
            elif tag == b"Z":
                self._txn_status = body
                if error is not None:
                    raise error
                return Cursor(rows, rowcount, lastrowid)
            else:
                raise ProtocolError(f"unexpected message {tag!r}")

    @staticmethod
    def _parse_row_description(body: bytes) -> Tuple[Dict[str, int], List[int]]:
        (nfields,) = struct.unpack("!H", body[:2])
        cols: Dict[str, int] = {}
        oids: List[int] = []
        off = 2
        for i in range(nfields):
            end = body.index(b"\x00", off)
            name = body[off:end].decode()
            off = end + 1
            _table, _attnum, oid, _size, _mod, _fmt = struct.unpack(
                "!IHIhih", body[off:off + 18])
            off += 18
            if name not in cols:  # first wins, like sqlite3.Row
                cols[name] = i
            oids.append(oid)
        return cols, oids

    @staticmethod
    def _parse_data_row(body: bytes, cols: Dict[str, int],
                        oids: List[int]) -> Row:
        (nfields,) = struct.unpack("!H", body[:2])
        off = 2
        vals = []
        for i in range(nfields):
            (ln,) = struct.unpack("!i", body[off:off + 4])
            off += 4
            if ln < 0:
                vals.append(None)
            else:
                vals.append(decode_value(oids[i], body[off:off + ln]))
                off += ln
        return Row(cols, tuple(vals))

    @staticmethod
    def _parse_complete_tag(tag: bytes) -> int:
        # "INSERT 0 5" / "UPDATE 3" / "SELECT 7" / "DELETE 0" / "BEGIN"
        parts = tag.split(b" ")
        try:
            return int(parts[-1])
        except ValueError:
            return -1

    def _simple_query(self, sql: str) -> Cursor:
        self._send(self._msg(b"Q", sql.encode("utf-8") + b"\x00"))
        return self._collect_results(until_sync=False)

    def _extended_query(self, sql: str, params: List[Any]) -> Cursor:
        parse = b"\x00" + sql.encode("utf-8") + b"\x00" + struct.pack("!H", 0)
        encoded = [encode_param(p) for p in params]
        bind = bytearray()
        bind += b"\x00\x00"                      # unnamed portal + stmt
        bind += struct.pack("!H", 0)             # all params text format
        bind += struct.pack("!H", len(encoded))
        for e in encoded:
            if e is None:
                bind += struct.pack("!i", -1)
            else:
                bind += struct.pack("!i", len(e)) + e
        bind += struct.pack("!H", 0)             # all results text format
        execute = b"\x00" + struct.pack("!I", 0)
        self._send(self._msg(b"P", parse)
                   + self._msg(b"B", bytes(bind))
                   + self._msg(b"D", b"P\x00")
                   + self._msg(b"E", execute)
                   + self._msg(b"S", b""))
        return self._collect_results(until_sync=True)


class _ScramClient:
    """SCRAM-SHA-256 client side (RFC 5802/7677), no channel binding."""

    def __init__(self, user: str, password: str):
        self._password = password.encode("utf-8")
        self._nonce = base64.b64encode(secrets.token_bytes(18)).decode()
        self._client_first_bare = f"n=,r={self._nonce}"
        self._auth_message = b""
        self._salted = b""

    def client_first(self) -> bytes:
        return ("n,," + self._client_first_bare).encode()

    def client_final(self, server_first: bytes) -> bytes:
        fields = dict(kv.split("=", 1)
                      for kv in server_first.decode().split(","))
        combined_nonce = fields["r"]
        if not combined_nonce.startswith(self._nonce):
            raise ProtocolError("SCRAM nonce mismatch")
        salt = base64.b64decode(fields["s"])
        iterations = int(fields["i"])
        self._salted = hashlib.pbkdf2_hmac(
            "sha256", self._password, salt, iterations)
        client_key = hmac.new(self._salted, b"Client Key",
                              hashlib.sha256).digest()
        stored_key = hashlib.sha256(client_key).digest()
        final_no_proof = f"c=biws,r={combined_nonce}"
        self._auth_message = ",".join([
            self._client_first_bare, server_first.decode(),
            final_no_proof]).encode()
        signature = hmac.new(stored_key, self._auth_message,
                             hashlib.sha256).digest()
        proof = bytes(a ^ b for a, b in zip(client_key, signature))
        return (final_no_proof
                + ",p=" + base64.b64encode(proof).decode()).encode()

    def verify_server_final(self, server_final: bytes) -> None:
        fields = dict(kv.split("=", 1)
                      for kv in server_final.decode().split(","))
        server_key = hmac.new(self._salted, b"Server Key",
                              hashlib.sha256).digest()
        expect = hmac.new(server_key, self._auth_message,
                          hashlib.sha256).digest()
        if not hmac.compare_digest(
                base64.b64decode(fields["v"]), expect):
            raise ProtocolError("SCRAM server signature mismatch")


def connect_url(url: str, **kw) -> PGConnection:
    """postgresql://user:pass@host:port/dbname (host may be a unix
    socket dir via percent-encoding, e.g. %2Fvar%2Frun%2Fpostgresql)."""
    p = urlparse(url)
    host = unquote(p.hostname or "127.0.0.1")
    return PGConnection(
        host=host,
        port=p.port or 5432,
        user=unquote(p.username or "postgres"),
        password=unquote(p.password or ""),
        dbname=(p.path or "/postgres").lstrip("/") or "postgres",
        **kw)
