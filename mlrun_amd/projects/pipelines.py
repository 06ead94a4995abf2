# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Workflow execution: the local pipeline runner.

The reference compiles workflows to Kubeflow Pipelines
(projects/pipelines.py:542 _KFPRunner) or runs them locally
(:673 _LocalRunner).  The MI355X-native build keeps the local runner
as the only engine: a workflow is a python function using
mlrun_amd.run_function / project.run_function, executed in-process
with pipeline_context set.
"""

import os
import typing

from ..errors import MLRunInvalidArgumentError, MLRunRuntimeError
from ..model import RunObject, generate_uid
from ..utils import logger
from .project import pipeline_context


_pipeline_runs: dict = {}


def list_pipeline_runs(project: str = None) -> list:
    """All recorded workflow runs (newest last), optionally filtered
    by project name (reference: pipelines list endpoint)."""
    runs = list(_pipeline_runs.values())
    if project:
        runs = [r for r in runs
                if getattr(r._project, "metadata", None) is not None
                and r._project.metadata.name == project]
    return runs


def get_pipeline(run_id: str) -> "_PipelineRunStatus":
    """Look up a completed workflow run by id (reference run.py
    get_pipeline over KFP)."""
    from ..errors import MLRunNotFoundError

    if run_id not in _pipeline_runs:
        raise MLRunNotFoundError(f"pipeline run {run_id} not found")
    return _pipeline_runs[run_id]


def wait_for_pipeline_completion(run_id: str, timeout: int = 3600,
                                 expected_statuses=None):
    """Local workflows complete synchronously; this resolves the
    stored status (kept for reference-code compatibility)."""
    status = get_pipeline(run_id)
    if expected_statuses and status.state not in expected_statuses:
        from ..errors import MLRunRuntimeError

        raise MLRunRuntimeError(
            f"pipeline {run_id} in state {status.state}, expected "
            f"{expected_statuses}")
    return status


class _PipelineRunStatus:
    def __init__(self, run_id, project, workflow_name, state="completed",
                 runs=None, error=None):
        self.run_id = run_id
        self._project = project
        self.workflow_name = workflow_name
        self.state = state
        self.runs = runs or []
        self.error = error

    def to_dict(self) -> dict:
        return {"run_id": self.run_id,
                "workflow_name": self.workflow_name,
                "state": self.state,
                "project": self._project.metadata.name
                if getattr(self._project, "metadata", None) else "",
                "runs": [r.metadata.uid for r in (self.runs or [])]}

    def wait_for_completion(self, timeout=None, expected_statuses=None):
        return self.state

    def __str__(self):
        return str(self.run_id)


class FunctionStep:
    """A lazily-executed function invocation inside a workflow
    (returned by function.as_step — parity: reference runtimes/base.py:666)."""

    def __init__(self, function, runspec=None, handler=None, name="",
                 params=None, inputs=None, outputs=None, artifact_path=""):
        self.function = function
        self.runspec = runspec
        self.handler = handler
        self.name = name
        self.params = params
        self.inputs = inputs
        self.outputs = outputs or []
        self.artifact_path = artifact_path
        self._run: typing.Optional[RunObject] = None

    def run(self) -> RunObject:
        self._run = self.function.run(
            self.runspec, handler=self.handler, name=self.name,
            params=self.params, inputs=self.inputs,
            artifact_path=self.artifact_path, watch=False)
        return self._run

    @property
    def outputs_map(self) -> dict:
        if self._run is None:
            self.run()
        return self._run.outputs

    def output(self, key):
        if self._run is None:
            self.run()
        return self._run.output(key)

    # KFP-compat sugar: step.after(other) -> self (execution is already
    # sequential in the local runner)
    def after(self, *steps):
        return self


def load_workflow_module(path: str):
    from ..runtimes.local import load_module

    return load_module(path)


def run_workflow(project, path=None, handler=None, arguments=None,
                 artifact_path=None, watch=True) -> _PipelineRunStatus:
    """Execute a workflow python file's handler with pipeline_context
    bound to the project."""
    arguments = arguments or {}
    run_id = generate_uid()[:8]
    pipeline_context.set(project)
    error = None
    try:
        if callable(handler):
            workflow_fn = handler
        else:
            if not path:
                raise MLRunInvalidArgumentError(
                    "workflow needs a path or a callable handler")
            if not os.path.isabs(path) and project.context:
                candidate = os.path.join(project.context, path)
                if os.path.isfile(candidate):
                    path = candidate
            module = load_workflow_module(path)
            fn_name = handler or "pipeline"
            if not hasattr(module, fn_name):
                # fall back: first public function
                candidates = [n for n in dir(module)
                              if callable(getattr(module, n))
                              and not n.startswith("_")]
                if not candidates:
                    raise MLRunInvalidArgumentError(
                        f"no workflow handler found in {path}")
                fn_name = candidates[0]
            workflow_fn = getattr(module, fn_name)
        import inspect

        sig = inspect.signature(workflow_fn)
        kwargs = {}
        for pname, param in sig.parameters.items():
            if pname in arguments:
                kwargs[pname] = arguments[pname]
            elif pname in ("project",):
                kwargs[pname] = project
        workflow_fn(**kwargs)
        state = "completed"
    except Exception as exc:
        logger.error("workflow failed", error=str(exc))
        state = "error"
        error = str(exc)
    runs = list(pipeline_context.runs)
    pipeline_context.clear()
    status = _PipelineRunStatus(run_id, project,
                                getattr(workflow_fn, "__name__", "workflow")
                                if "workflow_fn" in dir() else "workflow",
                                state=state, runs=runs, error=error)
    _pipeline_runs[run_id] = status
    if state == "error":
        raise MLRunRuntimeError(f"workflow failed: {error}")
    return status


def load_and_run(context, url: str = None, project_name: str = "",
                 init_git: bool = None, subpath: str = None,
                 clone: bool = False, save: bool = True,
                 workflow_name: str = None, workflow_path: str = None,
                 workflow_arguments: dict = None,
                 artifact_path: str = None, workflow_handler=None,
                 namespace: str = None, sync: bool = False,
                 dirty: bool = False, engine: str = None,
                 local: bool = None, schedule=None,
                 cleanup_ttl: int = None, load_only: bool = False,
                 wait_for_completion: bool = False,
                 project_context: str = None):
    """Load a project from a source and run one of its workflows —
    the handler the remote/scheduled workflow runner executes
    (reference pipelines.py:987)."""
    from .project import load_project

    project = load_project(
        context=project_context or f"./{project_name}", url=url,
        name=project_name, init_git=init_git, subpath=subpath,
        clone=clone, save=save)
    if context is not None and hasattr(context, "logger"):
        context.logger.info(f"Loaded project {project.name} successfully")
    if load_only:
        return
    status = project.run(
        name=workflow_name or "", workflow_path=workflow_path or "",
        arguments=workflow_arguments, artifact_path=artifact_path or "",
        workflow_handler=workflow_handler, namespace=namespace,
        sync=sync)
    if context is not None and hasattr(context, "log_result"):
        context.log_result("workflow_id", getattr(status, "run_id", ""))
        context.log_result("workflow_state",
                           getattr(status, "state", "completed"))
    return status
