# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Project-level operations: run/build/deploy a function.

Parity target: reference mlrun/projects/operations.py (run_function
:60, build_function :256, deploy_function :372).
"""


from ..errors import MLRunInvalidArgumentError
from .project import pipeline_context


def _resolve_function(function, project_object=None):
    from ..runtimes import BaseRuntime

    if isinstance(function, BaseRuntime):
        return function
    project = project_object or pipeline_context.project
    if project is None:
        raise MLRunInvalidArgumentError(
            "function name given but no active project; pass a function "
            "object or use project.run_function")
    return project.get_function(function)


def run_function(function, handler=None, name="", params=None, inputs=None,
                 hyperparams=None, hyper_param_options=None,
                 artifact_path=None, workdir="", watch=True, local=None,
                 schedule=None, returns=None, notifications=None,
                 project_object=None, **kwargs):
    fn = _resolve_function(function, project_object)
    run = fn.run(handler=handler, name=name, params=params, inputs=inputs,
                 hyperparams=hyperparams,
                 hyper_param_options=hyper_param_options,
                 artifact_path=artifact_path, workdir=workdir, watch=watch,
                 local=local, schedule=schedule, returns=returns,
                 notifications=notifications, **kwargs)
    if pipeline_context.project is not None:
        # inside an active workflow — track for the pipeline status
        pipeline_context.runs.append(run)
    return run


def build_function(function, with_mlrun=None, skip_deployed=False,
                   image=None, base_image=None, commands=None,
                   requirements=None, project_object=None, **kwargs):
    fn = _resolve_function(function, project_object)
    if image:
        fn.spec.image = image
    if hasattr(fn, "build_config"):
        fn.build_config(image=image or "", base_image=base_image or "",
                        commands=commands, requirements=requirements)
    if hasattr(fn, "deploy"):
        fn.deploy(watch=False)
    return fn


def deploy_function(function, models=None, env=None, tag=None, verbose=None,
                    builder_env=None, mock=None, project_object=None,
                    **kwargs):
    fn = _resolve_function(function, project_object)
    if env:
        fn.set_envs(env)
    if models:
        for model in models:
            if isinstance(model, dict):
                fn.add_model(**model)
            else:
                fn.add_model(model)
    if mock:
        return fn.to_mock_server()
    address = fn.deploy(tag=tag) if "tag" in _sig_params(fn.deploy) \
        else fn.deploy()
    return DeployStatus(state="ready", address=address
                        if isinstance(address, str) else None,
                        function=fn)


def _sig_params(fn) -> list:
    import inspect

    try:
        return list(inspect.signature(fn).parameters)
    except (ValueError, TypeError):
        return []


class DeployStatus:
    def __init__(self, state=None, address=None, function=None):
        self.state = state
        self.address = address
        self.function = function

    def __repr__(self):
        return f"DeployStatus(state={self.state}, address={self.address})"
