from audiomuse_amd.plugin.manager import PluginManager, hook_registry  # noqa: F401
