"""Loader for the forge_pybridge CPython extension (ops/csrc/pybridge.c):
C-loop scatter of response arena spans into the responses list — replaces
the per-row Python slicing loop that dominated the finalize stage."""

from __future__ import annotations

import importlib.util

_mod = None


def get():
    global _mod
    if _mod is None:
        from .build import PYBRIDGE, build_pybridge

        build_pybridge(verbose=False)
        spec = importlib.util.spec_from_file_location("forge_pybridge", PYBRIDGE)
        _mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(_mod)
    return _mod
