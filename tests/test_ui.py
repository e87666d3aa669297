"""Browser-UI coverage: the SPA's JS modules run in a stub DOM (node)
against a live server and drive every view's data path — setup wizard,
dashboard, library search/similar/path, WebGL-map data+grid, hyperbolic
explorer browse (incl. lazy warm), alchemy, chat planner, admin config
editor and the migration-wizard probe. Reference analog: the
screenshot/E2E driver (screenshot/example/tools/driver.py)."""

import shutil
import socket
import subprocess
import threading

import numpy as np
import pytest

from audiomuse_amd.db import connect
from audiomuse_amd.db.schema import init_db
from audiomuse_amd.db.store import save_track_analysis_and_embedding

node = shutil.which("node")
pytestmark = pytest.mark.skipif(node is None, reason="node unavailable")


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.fixture(scope="module")
def live_server(tmp_path_factory):
    url = "sqlite:///" + str(tmp_path_factory.mktemp("ui") / "ui.db")
    conn = connect(url)
    init_db(conn)
    rng = np.random.default_rng(0)
    for i in range(80):
        save_track_analysis_and_embedding(
            conn, f"fp_4{'%050x' % i}", title=f"Song {i}",
            author=f"Artist {i % 7}", album=f"Album {i % 9}",
            tempo=90 + i, energy=(i % 10) / 10, key="C", scale="major",
            duration=180.0,
            mood_vector={"rock": (i % 3) / 2, "jazz": ((i + 1) % 3) / 2,
                         "pop": ((i + 2) % 3) / 2},
            other_features={"happy": 0.5},
            embedding=rng.standard_normal(200).astype(np.float32))
    from audiomuse_amd.analysis.index import run_all_index_builds
    run_all_index_builds(conn)

    from audiomuse_amd.web.app import create_app
    app = create_app(url, auth_disabled=True)
    port = _free_port()
    from werkzeug.serving import make_server
    srv = make_server("127.0.0.1", port, app, threaded=True)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    yield f"http://127.0.0.1:{port}"
    srv.shutdown()
    conn.close()


def test_index_shell_and_assets(live_server):
    import urllib.request
    body = urllib.request.urlopen(live_server + "/").read().decode()
    assert "AudioMuse-AMD" in body
    for asset in ["app.js", "map.js", "explorer.js", "chat.js", "admin.js",
                  "setup.js", "tasks.js", "library.js", "alchemy.js",
                  "style.css"]:
        assert asset in body
        r = urllib.request.urlopen(f"{live_server}/static/{asset}")
        assert r.status == 200 and len(r.read()) > 500


def test_spa_drives_every_view(live_server):
    out = subprocess.run(
        [node, "tests/ui_smoke.js", live_server],
        capture_output=True, text=True, timeout=180)
    assert out.returncode == 0, f"UI smoke failed:\n{out.stdout}\n{out.stderr}"
    assert "UI_SMOKE_OK views=7" in out.stdout
