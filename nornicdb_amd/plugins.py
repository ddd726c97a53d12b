"""Runtime plugin loading.

Parity: reference pkg/nornicdb/plugins.go (Go plugin .so loading; two
plugin types: APOC function plugins and Heimdall subsystem plugins,
plugins/README.md). Python equivalent: load *.py files from a plugins
directory; each module may define:

    def register_functions(register):   # register(name, fn) -> Cypher fns
    def register_procedures(db):        # returns {name: proc}
    HEIMDALL_PLUGINS = [HeimdallPlugin subclasses or instances]
"""

from __future__ import annotations

import importlib.util
import os
import sys
from typing import Dict, List


def load_plugins_from_dir(db, plugin_dir: str) -> Dict[str, list]:
    """Load plugin modules and wire them into the database instance."""
    loaded = {"modules": [], "functions": [], "procedures": [], "heimdall": []}
    if not plugin_dir or not os.path.isdir(plugin_dir):
        return loaded
    from .cypher.functions import FUNCTIONS

    for fname in sorted(os.listdir(plugin_dir)):
        if not fname.endswith(".py") or fname.startswith("_"):
            continue
        path = os.path.join(plugin_dir, fname)
        mod_name = f"nornicdb_plugin_{fname[:-3]}"
        spec = importlib.util.spec_from_file_location(mod_name, path)
        mod = importlib.util.module_from_spec(spec)
        try:
            sys.modules[mod_name] = mod
            spec.loader.exec_module(mod)
        except Exception as e:
            loaded.setdefault("errors", []).append(f"{fname}: {e}")
            continue
        loaded["modules"].append(fname)

        if hasattr(mod, "register_functions"):
            before = set(FUNCTIONS)
            def _reg(name, fn):
                FUNCTIONS[name.lower()] = fn
            mod.register_functions(_reg)
            loaded["functions"].extend(sorted(set(FUNCTIONS) - before))

        if hasattr(mod, "register_procedures"):
            procs = mod.register_procedures(db) or {}
            for name, fn in procs.items():
                db.executor.procedures[name.lower()] = fn
                loaded["procedures"].append(name)

        for p in getattr(mod, "HEIMDALL_PLUGINS", []):
            loaded["heimdall"].append(getattr(p, "name", str(p)))
    return loaded
