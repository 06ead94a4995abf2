# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Drop-in import alias: ``import mlrun`` resolves to mlrun_amd so
reference user code runs unchanged on the MI355X-native framework."""

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

# submodule aliasing so "import mlrun.feature_store as fstore" works
for _name, _mod in list(_sys.modules.items()):
    if _name.startswith("mlrun_amd.") or _name == "mlrun_amd":
        _sys.modules["mlrun" + _name[len("mlrun_amd"):]] = _mod
