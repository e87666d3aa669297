"""Plugin package: zip-packaged extensions (see manager.py)."""

from audiomuse_amd.plugin.manager import (PluginManager,  # noqa: F401
                                          hook_registry)

# process-wide manager: web app and workers load DB-stored plugins into
# this instance at boot (manager.load_from_db)
plugin_manager = PluginManager()
