from .framework import (  # noqa: F401
    HookType,
    Plugin,
    PluginContext,
    PluginManager,
    PluginMode,
    PluginResult,
    PluginViolationError,
)
from .loader import build_plugin, default_chain_specs, load_plugin_manager  # noqa: F401
