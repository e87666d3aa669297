"""Schema DDL — mirrors the reference layout (database.py:1286-1651 +
taskqueue/sql.py:79-260; catalogued in SURVEY.md §2.5)."""

from __future__ import annotations

import sqlite3

DDL = """
CREATE TABLE IF NOT EXISTS score (
    item_id TEXT PRIMARY KEY,
    title TEXT, author TEXT, album TEXT,
    tempo REAL, key TEXT, scale TEXT,
    mood_vector TEXT, other_features TEXT,
    energy REAL, year INTEGER, rating REAL,
    file_path TEXT, duration REAL
);
CREATE TABLE IF NOT EXISTS embedding (
    item_id TEXT PRIMARY KEY REFERENCES score(item_id),
    embedding BLOB NOT NULL
);
CREATE TABLE IF NOT EXISTS clap_embedding (
    item_id TEXT PRIMARY KEY,
    embedding BLOB NOT NULL
);
CREATE TABLE IF NOT EXISTS lyrics_embedding (
    item_id TEXT PRIMARY KEY,
    embedding BLOB,
    axis_scores TEXT,
    lyrics_text TEXT,
    language TEXT,
    instrumental INTEGER DEFAULT 0
);
CREATE TABLE IF NOT EXISTS chromaprint (
    item_id TEXT PRIMARY KEY,
    fingerprint BLOB,
    duration REAL
);
CREATE TABLE IF NOT EXISTS track_server_map (
    provider_id TEXT NOT NULL,
    server_id TEXT NOT NULL,
    item_id TEXT NOT NULL,
    title TEXT, author TEXT, album TEXT, file_path TEXT,
    PRIMARY KEY (provider_id, server_id)
);
CREATE INDEX IF NOT EXISTS idx_tsm_item ON track_server_map(item_id);
CREATE TABLE IF NOT EXISTS artist_server_map (
    provider_artist_id TEXT NOT NULL,
    server_id TEXT NOT NULL,
    artist_name TEXT NOT NULL,
    PRIMARY KEY (provider_artist_id, server_id)
);
CREATE TABLE IF NOT EXISTS music_servers (
    server_id TEXT PRIMARY KEY,
    server_type TEXT NOT NULL,
    base_url TEXT, username TEXT, credential TEXT,
    enabled INTEGER DEFAULT 1,
    config TEXT
);
CREATE TABLE IF NOT EXISTS task_status (
    task_id TEXT PRIMARY KEY,
    task_type TEXT NOT NULL,
    parent_task_id TEXT,
    queue TEXT NOT NULL DEFAULT 'default',
    status TEXT NOT NULL DEFAULT 'PENDING',
    priority INTEGER DEFAULT 0,
    attempts INTEGER DEFAULT 0,
    max_attempts INTEGER DEFAULT 3,
    worker_id TEXT,
    lease_expires REAL,
    progress REAL DEFAULT 0,
    details TEXT,
    payload TEXT,
    shared_token TEXT,
    result TEXT,
    error_code INTEGER,
    created_at REAL DEFAULT ((julianday('now') - 2440587.5) * 86400.0),
    started_at REAL,
    finished_at REAL
);
CREATE INDEX IF NOT EXISTS idx_task_claim
    ON task_status(queue, status, priority DESC, created_at);
CREATE INDEX IF NOT EXISTS idx_task_parent ON task_status(parent_task_id);
CREATE TABLE IF NOT EXISTS task_history (
    id INTEGER PRIMARY KEY AUTOINCREMENT,
    task_id TEXT, task_type TEXT, status TEXT, note TEXT,
    created_at REAL DEFAULT ((julianday('now') - 2440587.5) * 86400.0)
);
CREATE TABLE IF NOT EXISTS shared_payload (
    token TEXT PRIMARY KEY,
    payload BLOB NOT NULL,
    refcount INTEGER DEFAULT 1,
    created_at REAL DEFAULT ((julianday('now') - 2440587.5) * 86400.0)
);
CREATE TABLE IF NOT EXISTS ivf_dir (
    index_name TEXT PRIMARY KEY,
    meta TEXT NOT NULL,
    n_parts INTEGER NOT NULL,
    updated_at REAL DEFAULT ((julianday('now') - 2440587.5) * 86400.0)
);
CREATE TABLE IF NOT EXISTS ivf_cell (
    index_name TEXT NOT NULL,
    part INTEGER NOT NULL,
    blob BLOB NOT NULL,
    PRIMARY KEY (index_name, part)
);
CREATE TABLE IF NOT EXISTS map_projection_data (
    name TEXT PRIMARY KEY,
    blob BLOB NOT NULL,
    updated_at REAL DEFAULT ((julianday('now') - 2440587.5) * 86400.0)
);
CREATE TABLE IF NOT EXISTS artist_metadata_data (
    name TEXT PRIMARY KEY,
    blob BLOB NOT NULL
);
CREATE TABLE IF NOT EXISTS playlist (
    id INTEGER PRIMARY KEY AUTOINCREMENT,
    name TEXT NOT NULL,
    server_id TEXT,
    item_ids TEXT NOT NULL,
    kind TEXT DEFAULT 'manual',
    created_at REAL DEFAULT ((julianday('now') - 2440587.5) * 86400.0)
);
CREATE TABLE IF NOT EXISTS playlist_name_history (
    id INTEGER PRIMARY KEY AUTOINCREMENT,
    name TEXT NOT NULL,
    created_at REAL DEFAULT ((julianday('now') - 2440587.5) * 86400.0)
);
CREATE TABLE IF NOT EXISTS migration_session (
    id INTEGER PRIMARY KEY AUTOINCREMENT,
    status TEXT DEFAULT 'open',
    server_type TEXT NOT NULL,
    server_config TEXT DEFAULT '{}',
    source_server_id TEXT DEFAULT 'default',
    decisions TEXT DEFAULT '{}',
    report BLOB,
    target_meta BLOB,
    created_at REAL DEFAULT ((julianday('now') - 2440587.5) * 86400.0)
);
CREATE TABLE IF NOT EXISTS cron (
    id INTEGER PRIMARY KEY AUTOINCREMENT,
    name TEXT, schedule TEXT NOT NULL, task_type TEXT NOT NULL,
    payload TEXT, enabled INTEGER DEFAULT 1,
    last_claimed_minute TEXT
);
CREATE TABLE IF NOT EXISTS cron_retry (
    id INTEGER PRIMARY KEY AUTOINCREMENT,
    cron_id INTEGER, due_at REAL, attempts INTEGER DEFAULT 0
);
CREATE TABLE IF NOT EXISTS audiomuse_users (
    username TEXT PRIMARY KEY,
    password_hash TEXT NOT NULL,
    role TEXT DEFAULT 'admin',
    created_at REAL DEFAULT ((julianday('now') - 2440587.5) * 86400.0)
);
CREATE TABLE IF NOT EXISTS app_config (
    key TEXT PRIMARY KEY,
    value TEXT
);
CREATE TABLE IF NOT EXISTS plugin (
    name TEXT PRIMARY KEY,
    blob BLOB NOT NULL,
    enabled INTEGER DEFAULT 1,
    uploaded_at REAL DEFAULT ((julianday('now') - 2440587.5) * 86400.0)
);
CREATE TABLE IF NOT EXISTS dashboard_stats (
    key TEXT PRIMARY KEY,
    value TEXT,
    updated_at REAL DEFAULT ((julianday('now') - 2440587.5) * 86400.0)
);
CREATE TABLE IF NOT EXISTS alchemy_anchors (
    name TEXT PRIMARY KEY,
    vector BLOB NOT NULL,
    created_at REAL DEFAULT ((julianday('now') - 2440587.5) * 86400.0)
);
CREATE TABLE IF NOT EXISTS alchemy_radios (
    name TEXT PRIMARY KEY,
    definition TEXT NOT NULL
);
CREATE TABLE IF NOT EXISTS control_request (
    id INTEGER PRIMARY KEY AUTOINCREMENT,
    action TEXT NOT NULL,
    payload TEXT,
    created_at REAL DEFAULT ((julianday('now') - 2440587.5) * 86400.0),
    expires_at REAL
);
CREATE TABLE IF NOT EXISTS control_ack (
    request_id INTEGER NOT NULL,
    listener TEXT NOT NULL,
    acked_at REAL DEFAULT ((julianday('now') - 2440587.5) * 86400.0),
    PRIMARY KEY (request_id, listener)
);
"""


def to_postgres(ddl: str) -> str:
    """Derive the PostgreSQL DDL from the canonical (SQLite-typed) DDL.

    The reference's schema is native PG (database.py:1286-1651); here the
    translation is mechanical so both backends share one table catalogue
    and the conventions test can diff them column by column.
    """
    out = ddl.replace("INTEGER PRIMARY KEY AUTOINCREMENT",
                      "BIGSERIAL PRIMARY KEY")
    out = out.replace("BLOB", "BYTEA")
    out = out.replace("REAL", "DOUBLE PRECISION")
    out = out.replace("(julianday('now') - 2440587.5) * 86400.0",
                      "EXTRACT(EPOCH FROM now())")
    return out


DDL_PG = to_postgres(DDL)


def init_db(conn) -> None:
    if getattr(conn, "kind", "sqlite") == "postgres":
        conn.executescript(DDL_PG)
    else:
        conn.executescript(DDL)
