"""In-tree native-extension loader.

Extensions are built by ``build.py`` (driven by ``__graft_entry__.build()``)
straight into ``dist_tuto_pth_amd/_native/*.so`` so the compiled objects
travel with the repo snapshot to GPU machines (no JIT cache involved).

Loading is lazy and failures are LOUD on GPU machines: a GPU run must
never fall back to an eager/PyTorch path silently.
"""

from __future__ import annotations

import importlib.util
import os
import sys

_NATIVE_DIR = os.path.join(os.path.dirname(os.path.dirname(__file__)),
                           "_native")
_cache = {}


def native_path(name: str) -> str:
    return os.path.join(_NATIVE_DIR, name + ".so")


def native_available(name: str) -> bool:
    return os.path.exists(native_path(name))


def load_native(name: str):
    """Import ``_native/<name>.so`` as a module (cached)."""
    if name in _cache:
        return _cache[name]
    path = native_path(name)
    if not os.path.exists(path):
        raise ImportError(
            f"native extension {name!r} not built (expected {path}). "
            f"Run `python build.py` (or __graft_entry__.build()) first — "
            f"GPU paths do not fall back to eager.")
    spec = importlib.util.spec_from_file_location(name, path)
    mod = importlib.util.module_from_spec(spec)
    sys.modules[name] = mod
    spec.loader.exec_module(mod)
    _cache[name] = mod
    return mod
