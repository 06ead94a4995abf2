# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Runtime kinds registry.

The reference exposes ~11 runtime kinds (job/mpijob/dask/spark/nuclio/
serving/...).  The MI355X-native framework maps them to node-local
engines:

- local / handler  -> in-process execution
- job              -> managed local process w/ GPU lease
- mpijob           -> N local ranks (1/GPU) over RCCL/xGMI
- serving          -> serving-graph engine (HIP kernels + hipGraph)
- application      -> long-lived local HTTP app
- remote (nuclio)  -> local HTTP function host
- dask / spark / databricks kinds from the reference have no MI355X
  analog (cluster-external engines) — requesting them raises with a
  pointer to the node-local equivalents (job with parallel_runs /
  mpijob).
"""

from ..errors import MLRunInvalidArgumentError
from .base import BaseRuntime, FunctionSpec, FunctionMetadata  # noqa: F401
from .local import HandlerRuntime, LocalRuntime  # noqa: F401
from .job import KubejobRuntime  # noqa: F401


class RuntimeKinds:
    local = "local"
    handler = "handler"
    job = "job"
    mpijob = "mpijob"
    serving = "serving"
    remote = "remote"
    nuclio = "nuclio"
    application = "application"
    dask = "dask"
    spark = "spark"
    databricks = "databricks"

    @staticmethod
    def all():
        return [RuntimeKinds.local, RuntimeKinds.handler, RuntimeKinds.job,
                RuntimeKinds.mpijob, RuntimeKinds.serving,
                RuntimeKinds.remote, RuntimeKinds.nuclio,
                RuntimeKinds.application]

    @staticmethod
    def runtime_with_handlers():
        return [RuntimeKinds.local, RuntimeKinds.handler, RuntimeKinds.job,
                RuntimeKinds.mpijob]

    @staticmethod
    def local_runtimes():
        return [RuntimeKinds.local, RuntimeKinds.handler]


def get_runtime_class(kind: str):
    if kind in (None, "", "local"):
        return LocalRuntime
    if kind == RuntimeKinds.handler:
        return HandlerRuntime
    if kind == RuntimeKinds.job:
        return KubejobRuntime
    if kind == RuntimeKinds.mpijob:
        from .mpijob import MpiRuntime

        return MpiRuntime
    if kind in (RuntimeKinds.serving,):
        from .serving import ServingRuntime

        return ServingRuntime
    if kind in (RuntimeKinds.remote, RuntimeKinds.nuclio,
                RuntimeKinds.application):
        from .remote import RemoteRuntime, ApplicationRuntime

        return ApplicationRuntime if kind == RuntimeKinds.application \
            else RemoteRuntime
    if kind in (RuntimeKinds.dask, RuntimeKinds.spark,
                RuntimeKinds.databricks):
        raise MLRunInvalidArgumentError(
            f"runtime kind {kind!r} has no MI355X node-local engine; use "
            f"kind='job' with hyper_param_options.parallel_runs for task "
            f"fan-out or kind='mpijob' for distributed (RCCL/xGMI) runs")
    raise MLRunInvalidArgumentError(f"unsupported runtime kind {kind!r}")


class RunError(Exception):
    """A run failed inside a runtime (reference runtimes/utils.py
    RunError)."""


def is_local_runtime(kind: str) -> bool:
    return kind in RuntimeKinds.local_runtimes() or not kind


from .mpijob import MpiRuntime  # noqa: F401,E402
from .remote import ApplicationRuntime, RemoteRuntime  # noqa: F401,E402
from .serving import ServingRuntime  # noqa: F401,E402
from ..serving.v1_serving import (  # noqa: F401,E402
    MLModelServer,
    new_v1_model_server,
)


def new_model_server(*args, **kwargs):
    """Create a serving function pre-loaded with model routes
    (delegates to run.new_model_server — import deferred to avoid a
    cycle)."""
    from ..run import new_model_server as _factory

    return _factory(*args, **kwargs)


# reference runtimes/__init__ aliases & constants
MpiRuntimeV1 = MpiRuntime
serving_subkind = "v2"


class MPIJobCRDVersions:
    v1 = "v1"
    v1alpha1 = "v1alpha1"

    @staticmethod
    def all():
        return [MPIJobCRDVersions.v1, MPIJobCRDVersions.v1alpha1]

    @staticmethod
    def default():
        return MPIJobCRDVersions.v1


class RuntimeClassMode:
    """How a runtime class is used (reference runtimes/__init__
    RuntimeClassMode): building a run object vs monitoring one."""

    run = "run"
    build = "build"


def new_v2_model_server(name: str, model_class: str, models: dict = None,
                        filename: str = "", protocol: str = "",
                        image: str = "", workers: int = 8, **kwargs):
    """Create a V2 (KFServing-v2 protocol) model-server function
    (reference runtimes/__init__.py new_v2_model_server)."""
    return new_model_server(name, model_class=model_class, models=models,
                            filename=filename, protocol=protocol or "v2",
                            image=image, **kwargs)
from .api_gateway import (  # noqa: F401,E402
    APIGateway,
    APIGatewayMetadata,
    APIGatewaySpec,
)
