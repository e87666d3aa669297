import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an AMD GPU (MI355X)")
    config.addinivalue_line("markers", "slow: long-running test")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU on this host")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture(params=["sqlite", "pg"])
def tmp_db_url(tmp_path, request):
    """Every DB-touching test runs on BOTH backends: SQLite directly,
    and PostgreSQL through the first-party wire driver. The PG side uses
    a real server when AUDIOMUSE_TEST_DATABASE_URL is set, else the
    in-process wire-protocol stub (tests/pgstub.py) — the image has no
    PostgreSQL server or client wheel (docs/POSTGRES.md)."""
    if request.param == "sqlite":
        yield "sqlite:///" + str(tmp_path / "test.db")
        return
    real = os.environ.get("AUDIOMUSE_TEST_DATABASE_URL")
    if real:
        from audiomuse_amd.db import connect as _connect
        conn = _connect(real)
        for row in conn.execute(
                "SELECT tablename FROM pg_tables WHERE schemaname='public'"
        ).fetchall():
            conn.execute(f'DROP TABLE IF EXISTS "{row[0]}" CASCADE')
        conn.close()
        yield real
        return
    from tests.pgstub import StubServer
    srv = StubServer(tmp_path / "pgstub.db").start()
    yield srv.url
    srv.stop()


@pytest.fixture
def tmp_sqlite_url(tmp_path):
    """SQLite-only fixture for tests exercising sqlite-specific paths."""
    return "sqlite:///" + str(tmp_path / "test.db")
