"""Plugin system: zip-packaged extensions with hooks.

Reference: /root/reference/plugin/ (2146 LoC) — third-party plugins as
zips stored in the DB: safe extraction, role-scoped load, hooks
(`song_analyzed`), cron task and analysis-provider extension points
(PluginManager.load :537). Here: the same zip format (a `plugin.py`
module exporting `register(api)`), safe extraction (path traversal
guarded), an API object exposing hook registration + cron task
registration, and a `song_analyzed` hook invoked by the analysis
pipeline. pip-requirement install is intentionally absent (no network
in target deployments is supported; document requirements instead).
"""

from __future__ import annotations

import importlib.util
import io
import logging
import os
import tempfile
import zipfile
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional

from audiomuse_amd import config as C

logger = logging.getLogger(__name__)


class HookRegistry:
    def __init__(self):
        self._hooks: Dict[str, List[Callable]] = {}

    def register(self, name: str, fn: Callable) -> None:
        self._hooks.setdefault(name, []).append(fn)

    def fire(self, name: str, *args, **kwargs) -> None:
        for fn in self._hooks.get(name, []):
            try:
                fn(*args, **kwargs)
            except Exception:  # noqa: BLE001 — plugin errors never break analysis
                logger.exception("plugin hook %s failed", name)

    def clear(self) -> None:
        self._hooks.clear()


hook_registry = HookRegistry()


@dataclass
class PluginAPI:
    """What a plugin's register(api) receives."""

    name: str
    hooks: HookRegistry
    cron_tasks: List[dict] = field(default_factory=list)

    def on_song_analyzed(self, fn: Callable) -> None:
        self.hooks.register("song_analyzed", fn)

    def add_cron_task(self, schedule: str, task_type: str,
                      payload: Optional[dict] = None) -> None:
        self.cron_tasks.append({"schedule": schedule, "task_type": task_type,
                                "payload": payload or {}})

    def add_task_handler(self, task_type: str, fn: Callable) -> None:
        """Analysis-provider extension point (reference: plugin cron task
        + analysis-provider extension points, PluginManager.load :537):
        the plugin contributes a queue task type, runnable by workers and
        schedulable via add_cron_task. Namespaced to the plugin so a
        plugin cannot shadow built-in handlers."""
        from audiomuse_amd.taskqueue.worker import _REGISTRY

        full = f"plugin.{self.name}.{task_type}"
        _REGISTRY[full] = fn
        self.task_types = getattr(self, "task_types", [])
        self.task_types.append(full)


def _safe_extract(zf: zipfile.ZipFile, dest: str) -> None:
    for member in zf.namelist():
        target = os.path.realpath(os.path.join(dest, member))
        if not target.startswith(os.path.realpath(dest) + os.sep):
            raise ValueError(f"unsafe path in plugin zip: {member!r}")
    zf.extractall(dest)


class PluginManager:
    def __init__(self, hooks: Optional[HookRegistry] = None):
        self.hooks = hooks or hook_registry
        self.loaded: Dict[str, PluginAPI] = {}

    def load_zip(self, name: str, blob: bytes) -> PluginAPI:
        """Extract + import plugin.py + call register(api)
        (reference: PluginManager.load :537)."""
        base = os.path.join(C.DATA_DIR, "plugins")
        os.makedirs(base, exist_ok=True)
        dest = tempfile.mkdtemp(prefix=f"{name}-", dir=base)
        with zipfile.ZipFile(io.BytesIO(blob)) as zf:
            _safe_extract(zf, dest)
        mod_path = os.path.join(dest, "plugin.py")
        if not os.path.exists(mod_path):
            raise FileNotFoundError("plugin zip must contain plugin.py")
        # pip-install of plugin requirements is intentionally
        # unsupported in this offline build; PLUGIN_ALLOW_PIP=1 fails
        # loudly instead of silently skipping the dependency step
        if os.path.exists(os.path.join(dest, "requirements.txt")):
            if C.PLUGIN_ALLOW_PIP:
                raise RuntimeError(
                    "PLUGIN_ALLOW_PIP is set but pip installs are "
                    "unsupported in this offline build; vendor the "
                    "dependencies inside the plugin zip instead")
            logger.warning("plugin %s ships requirements.txt; pip install "
                           "is disabled (vendor dependencies in the zip)",
                           name)
        spec = importlib.util.spec_from_file_location(
            f"audiomuse_plugin_{name}", mod_path)
        module = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(module)  # type: ignore[union-attr]
        if not hasattr(module, "register"):
            raise AttributeError("plugin.py must export register(api)")
        api = PluginAPI(name=name, hooks=self.hooks)
        module.register(api)
        self.loaded[name] = api
        return api

    def fire_song_analyzed(self, item_id: str, analysis: dict) -> None:
        self.hooks.fire("song_analyzed", item_id, analysis)

    def load_from_db(self, conn) -> int:
        """Load every enabled plugin stored in the `plugin` table (the
        reference keeps plugin zips in the DB and loads them at web and
        worker boot), then sync their cron tasks. Individual plugin
        failures are logged and skipped. Returns plugins loaded.
        PLUGINS_ENABLED=0 disables the whole subsystem."""
        from audiomuse_amd import config as C
        if not C.PLUGINS_ENABLED:
            return 0
        rows = conn.execute(
            "SELECT name, blob FROM plugin WHERE enabled = 1").fetchall()
        n = 0
        for r in rows:
            try:
                self.load_zip(r["name"], r["blob"])
                n += 1
            except Exception:  # noqa: BLE001 — one bad plugin never blocks boot
                logger.exception("plugin %s failed to load", r["name"])
        self.sync_cron(conn)
        return n

    def sync_cron(self, conn) -> int:
        """Upsert every loaded plugin's cron tasks into the cron table
        (name 'plugin:<plugin>:<i>'); prunes rows of plugins no longer
        loaded. Returns the number of active plugin cron rows."""
        import json as _json

        from audiomuse_amd.db import write_txn

        want = {}
        for api in self.loaded.values():
            for i, ct in enumerate(api.cron_tasks):
                want[f"plugin:{api.name}:{i}"] = ct
        with write_txn(conn):
            rows = conn.execute(
                "SELECT id, name FROM cron WHERE name LIKE 'plugin:%'"
            ).fetchall()
            for r in rows:
                if r["name"] not in want:
                    conn.execute("DELETE FROM cron WHERE id = ?", (r["id"],))
            have = {r["name"] for r in rows}
            for name, ct in want.items():
                if name in have:
                    conn.execute(
                        "UPDATE cron SET schedule=?, task_type=?, payload=? "
                        "WHERE name=?",
                        (ct["schedule"], ct["task_type"],
                         _json.dumps(ct["payload"]), name))
                else:
                    conn.execute(
                        "INSERT INTO cron (name, schedule, task_type, "
                        "payload, enabled) VALUES (?,?,?,?,1)",
                        (name, ct["schedule"], ct["task_type"],
                         _json.dumps(ct["payload"])))
        return len(want)
