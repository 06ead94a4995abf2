# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""mlrun_amd — an MI355X-native MLOps orchestration & serving framework.

A from-scratch rebuild of the mlrun/mlrun capability surface for a
single 8xMI355X node: projects/functions/runs/artifacts + a serving
graph engine whose hot path runs hand-written CDNA4 HIP kernels, a
GPU-resident feature store, and RCCL-over-xGMI distributed runtimes.

Public API parity: reference mlrun/__init__.py (__all__ :17,
set_environment :107).
"""

__version__ = "0.1.0"

__all__ = [
    "get_version",
    "set_environment",
    "code_to_function",
    "import_function",
    "new_function",
    "run_local",
    "get_or_create_ctx",
    "new_task",
    "new_project",
    "load_project",
    "get_current_project",
    "get_or_create_project",
    "run_function",
    "build_function",
    "deploy_function",
    "get_run_db",
    "mlconf",
    "handler",
    "get_secret_or_env",
    "mount_v3io",
    "auto_mount",
    "v3io_cred",
    "VolumeMount",
]

from .config import config as mlconf  # noqa: E402
from .errors import (  # noqa: F401,E402
    MLRunBaseError,
    MLRunInvalidArgumentError,
    MLRunNotFoundError,
    MLRunRuntimeError,
)
from .model import RunObject, RunTemplate, new_task  # noqa: F401,E402
from .execution import MLClientCtx  # noqa: F401,E402
from .db import get_run_db  # noqa: F401,E402
from .datastore import DataItem, store_manager, get_dataitem, get_object  # noqa: F401,E402
from .run import (  # noqa: F401,E402
    code_to_function,
    function_to_module,
    get_or_create_ctx,
    get_pipeline,
    import_function,
    new_function,
    new_model_server,
    run_local,
    wait_for_pipeline_completion,
)
from .projects import pipeline_context  # noqa: F401,E402
from .projects import (  # noqa: F401,E402
    MlrunProject,
    ProjectMetadata,
    build_function,
    deploy_function,
    get_or_create_project,
    load_project,
    new_project,
    run_function,
)
from .package import (  # noqa: F401,E402
    ArtifactType,
    DefaultPackager,
    Packager,
    handler,
)
from .secrets import get_secret_or_env  # noqa: F401,E402
from .platforms import (  # noqa: F401,E402
    VolumeMount,
    auto_mount,
    mount_v3io,
    v3io_cred,
)


def get_version() -> str:
    return __version__


def set_environment(api_path: str = None, artifact_path: str = "",
                    env_file: str = None, mock_functions: str = None):
    """Set global configuration: run-DB target + default artifact path.

    Returns (default_project_name, artifact_path) — parity with the
    reference set_environment.
    """
    if env_file:
        _load_env_file(env_file)
    if api_path:
        mlconf.dbpath = api_path
        from .db import get_run_db as _get

        _get(api_path, force_reconnect=True)
    if artifact_path:
        import os

        if not artifact_path.startswith("/") and "://" not in artifact_path:
            artifact_path = os.path.abspath(artifact_path)
        mlconf.artifact_path = artifact_path
    return mlconf.default_project, mlconf.artifact_path


def _load_env_file(path: str):
    set_env_from_file(path)


def set_env_from_file(env_file: str, return_dict: bool = False):
    """Read KEY=VALUE lines from a .env file into the process env and
    reload mlrun config (reference mlrun/__init__.py:187)."""
    import os

    from .errors import MLRunInvalidArgumentError, MLRunNotFoundError

    env_file = os.path.expanduser(env_file)
    if not os.path.isfile(env_file):
        raise MLRunNotFoundError(f"env file {env_file} does not exist")
    env_vars = {}
    with open(env_file) as fp:
        for line in fp:
            line = line.strip()
            if not line or line.startswith("#"):
                continue
            if "=" not in line:
                raise MLRunInvalidArgumentError(
                    "env file lines must be in the form key=value")
            key, _, value = line.partition("=")
            env_vars[key.strip()] = value.strip()
    for key, value in env_vars.items():
        os.environ[key] = value
    from .config import _populate

    _populate()
    return env_vars if return_dict else None


def get_sample_path(subpath: str = "") -> str:
    """Url of a sample dataset or model (reference
    mlrun/__init__.py:175)."""
    import os

    samples_path = os.environ.get("SAMPLE_DATA_SOURCE_URL_PREFIX",
                                  mlconf.default_samples_path)
    if subpath:
        samples_path = os.path.join(samples_path, subpath.lstrip("/"))
    return samples_path


def get_current_project(silent: bool = False):
    """The project active in the current pipeline/workflow context
    (reference mlrun/__init__.py:167)."""
    if pipeline_context.project is None and not silent:
        from .errors import MLRunInvalidArgumentError

        raise MLRunInvalidArgumentError(
            "no project is active; load/create one first")
    return pipeline_context.project

# expose common submodules as attributes (reference: mlrun.feature_store
# etc. are importable directly off the package)
from . import feature_store, serving  # noqa: F401,E402
from .artifacts import get_model  # noqa: F401,E402
