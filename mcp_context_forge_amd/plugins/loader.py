"""Plugin config loading (reference: plugins/config.yaml format).

YAML shape (reference-compatible):

    plugins:
      - name: deny_filter
        kind: builtin                # or dotted path "pkg.mod.ClassName"
        hooks: [tool_pre_invoke]     # optional override
        mode: enforce
        priority: 10
        conditions: {tools: ["secure-*"]}
        config: {words: [foo, bar]}
"""

from __future__ import annotations

import importlib
from pathlib import Path
from typing import Any, Dict, List, Optional

import yaml

from .builtin import BUILTIN_PLUGINS
from .framework import HookType, Plugin, PluginManager


def _all_builtin():
    from .content import EXTRA_PLUGINS
    from .integrations import INTEGRATION_PLUGINS

    out = dict(BUILTIN_PLUGINS)
    out.update(EXTRA_PLUGINS)
    out.update(INTEGRATION_PLUGINS)
    return out


def _resolve_class(name: str, kind: str):
    if kind in ("builtin", "native", ""):
        cls = _all_builtin().get(name)
        if cls is None:
            raise KeyError(f"unknown builtin plugin {name!r}")
        return cls
    if kind == "external":
        # external-process plugin service (reference: plugins/external/*)
        from .external import ExternalServicePlugin

        return ExternalServicePlugin
    mod_name, _, cls_name = kind.rpartition(".")
    mod = importlib.import_module(mod_name)
    return getattr(mod, cls_name)


def build_plugin(spec: Dict[str, Any]) -> Plugin:
    name = spec["name"]
    cls = _resolve_class(name, spec.get("kind", "builtin"))
    cfg = dict(spec.get("config") or {})
    if "mode" in spec:
        cfg["mode"] = spec["mode"]
    if "priority" in spec:
        cfg["priority"] = spec["priority"]
    if "conditions" in spec:
        cfg["conditions"] = spec["conditions"]
    plugin = cls(cfg)
    if spec.get("kind") == "external":
        plugin.name = name  # external instances are named by their spec
    if spec.get("hooks"):
        plugin.hooks = tuple(HookType(h) for h in spec["hooks"])
    return plugin


def load_plugin_manager(config_file: Optional[str] = None, enabled: bool = True,
                        specs: Optional[List[Dict[str, Any]]] = None) -> PluginManager:
    if specs is None:
        specs = []
        if config_file and Path(config_file).exists():
            raw = yaml.safe_load(Path(config_file).read_text()) or {}
            specs = raw.get("plugins", [])
    plugins = [build_plugin(s) for s in specs]
    return PluginManager(plugins, enabled=enabled)


def default_chain_specs() -> List[Dict[str, Any]]:
    """The full BASELINE.json plugin chain, default configs."""
    return [
        {"name": "response_cache_by_prompt"},
        {"name": "deny_filter"},
        {"name": "regex_filter"},
        {"name": "argument_normalizer"},
        {"name": "pii_filter"},
        {"name": "schema_guard"},
        {"name": "content_moderation"},
        {"name": "harmful_content_detector"},
        {"name": "toon_encoder"},
        {"name": "output_length_guard"},
    ]
