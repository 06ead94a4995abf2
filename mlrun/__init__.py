# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Drop-in import alias: ``import mlrun`` resolves to mlrun_amd so
reference user code runs unchanged on the MI355X-native framework.

Any ``mlrun.X.Y`` import — including paths not yet loaded at alias-import
time — is redirected to the *same* ``mlrun_amd.X.Y`` module object via a
meta-path finder, so module-level state (locks, registries, caches) and
``isinstance`` checks are shared across both names.
"""

import importlib as _importlib
import importlib.abc as _abc
import importlib.machinery as _machinery
import sys as _sys

import mlrun_amd as _impl
from mlrun_amd import *  # noqa: F401,F403
from mlrun_amd import (  # noqa: F401
    __all__,
    __version__,
    mlconf,
    feature_store,
    serving,
    artifacts,
    datastore,
    projects,
    runtimes,
    frameworks,
)


class _AliasLoader(_abc.Loader):
    """Loads ``mlrun.X`` by importing ``mlrun_amd.X`` and registering the
    SAME module object under both names (no second instance)."""

    def create_module(self, spec):
        target = "mlrun_amd" + spec.name[len("mlrun"):]
        module = _importlib.import_module(target)
        _sys.modules[spec.name] = module
        return module

    def exec_module(self, module):
        # already executed by the mlrun_amd import in create_module
        pass


class _AliasFinder(_abc.MetaPathFinder):
    def find_spec(self, fullname, path=None, target=None):
        if not fullname.startswith("mlrun."):
            return None
        impl_name = "mlrun_amd" + fullname[len("mlrun"):]
        # only alias paths that exist under mlrun_amd
        try:
            impl_spec = _importlib.util.find_spec(impl_name)
        except (ImportError, ValueError):
            return None
        if impl_spec is None:
            return None
        spec = _machinery.ModuleSpec(fullname, _AliasLoader(), is_package=impl_spec.submodule_search_locations is not None)
        return spec


if not any(isinstance(f, _AliasFinder) for f in _sys.meta_path):
    _sys.meta_path.insert(0, _AliasFinder())

# alias submodules already imported (fast path + keeps attribute access
# like ``mlrun.feature_store`` consistent before any fresh import)
for _name, _mod in list(_sys.modules.items()):
    if _name.startswith("mlrun_amd.") or _name == "mlrun_amd":
        _sys.modules["mlrun" + _name[len("mlrun_amd"):]] = _mod
