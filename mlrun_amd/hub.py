# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Function hub: import functions from indexed local sources.

Parity target: reference hub:// handling (run.py:330 dispatch +
server/api/crud/hub.py).  Network-less: sources are local directories
of function yaml files registered via add_hub_source; the "default"
source maps to {base_dir}/hub.
"""

import os

import yaml

from .config import config
from .errors import MLRunNotFoundError

_sources: dict = {}


def add_hub_source(name: str, path: str, order: int = -1):
    _sources[name] = {"name": name, "path": path, "order": order}


def list_hub_sources() -> list:
    _ensure_default()
    return sorted(_sources.values(), key=lambda s: s["order"])


BUILTIN_DIR = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                           "hub_functions")


def _ensure_default():
    if "builtin" not in _sources:
        _sources["builtin"] = {"name": "builtin", "path": BUILTIN_DIR,
                               "order": 0}
    if "default" not in _sources:
        _sources["default"] = {
            "name": "default",
            "path": os.path.join(config.base_dir, "hub"),
            "order": 999,
        }


def get_hub_catalog(source: str = "default") -> list:
    _ensure_default()
    src = _sources.get(source)
    if not src or not os.path.isdir(src["path"]):
        return []
    items = []
    for entry in sorted(os.listdir(src["path"])):
        item_dir = os.path.join(src["path"], entry)
        yaml_path = os.path.join(item_dir, "function.yaml")
        if os.path.isfile(yaml_path):
            items.append({"name": entry, "source": source, "path": yaml_path})
    return items


def get_hub_function(url_body: str):
    """Resolve hub://[source/]name[:tag] to a function object."""
    from .run import new_function

    _ensure_default()
    body = url_body.strip("/")
    source = None
    if "/" in body:
        source, body = body.split("/", 1)
    name = body.split(":")[0]
    candidates = [source] if source else \
        [s["name"] for s in sorted(_sources.values(),
                                   key=lambda s: s["order"])]
    yaml_path = None
    for cand in candidates:
        src = _sources.get(cand)
        if not src:
            continue
        path = os.path.join(src["path"], name, "function.yaml")
        if os.path.isfile(path):
            yaml_path = path
            break
    if yaml_path is None:
        raise MLRunNotFoundError(
            f"hub function {name} not found in sources {candidates}")
    with open(yaml_path) as fp:
        struct = yaml.safe_load(fp)
    return new_function(runtime=struct)
