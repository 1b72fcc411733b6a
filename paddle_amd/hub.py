"""paddle.hub (reference: python/paddle/hapi/hub.py -- list/help/load from
a hubconf.py).  Local-source only: this image has no network egress."""
from __future__ import annotations

import importlib.util
import os


def _load_hubconf(repo_dir):
    path = os.path.join(repo_dir, "hubconf.py")
    if not os.path.exists(path):
        raise FileNotFoundError(f"no hubconf.py in {repo_dir}")
    spec = importlib.util.spec_from_file_location("hubconf", path)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod

def _check_source(source):
    if source != "local":
        raise RuntimeError("paddle.hub: only source='local' is available "
                           "(no network egress in this environment)")

def list(repo_dir, source="local", force_reload=False):
    _check_source(source)
    mod = _load_hubconf(repo_dir)
    return [n for n in dir(mod) if callable(getattr(mod, n)) and not n.startswith("_")]


def help(repo_dir, model, source="local", force_reload=False):
    _check_source(source)
    return getattr(_load_hubconf(repo_dir), model).__doc__


def load(repo_dir, model, source="local", force_reload=False, **kwargs):
    _check_source(source)
    return getattr(_load_hubconf(repo_dir), model)(**kwargs)
