# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Local execution runtimes: in-process handler + subprocess command.

Parity target: reference mlrun/runtimes/local.py (LocalRuntime :199,
HandlerRuntime :172, run_exec :423, exec_from_params :481) and the
packager-lite return handling of mlrun/package.
"""

import importlib.util
import io
import json
import os
import socket
import subprocess
import sys
import tempfile
import traceback
import typing

from ..errors import MLRunInvalidArgumentError
from ..execution import MLClientCtx
from ..model import RunObject, RunStates
from ..utils import logger
from .base import BaseRuntime

META_TMPFILE_ENV = "MLRUN_META_TMPFILE"


def load_module(file_path: str, module_name: str = None):
    """Import a python file as a module."""
    module_name = module_name or \
        os.path.splitext(os.path.basename(file_path))[0]
    spec = importlib.util.spec_from_file_location(module_name, file_path)
    if spec is None:
        raise MLRunInvalidArgumentError(f"cannot load module from {file_path}")
    module = importlib.util.module_from_spec(spec)
    sys.modules[module_name] = module
    spec.loader.exec_module(module)
    return module


class _DupStdout(io.TextIOBase):
    """Duplicate stdout into a buffer for run-log capture
    (parity: reference runtimes/local.py:468)."""

    def __init__(self, stream):
        self._stream = stream
        self.buf = io.StringIO()

    def write(self, text):
        self._stream.write(text)
        self.buf.write(text)
        return len(text)

    def flush(self):
        self._stream.flush()


class LocalRuntime(BaseRuntime):
    kind = "local"

    def to_job(self, image=""):
        from .job import KubejobRuntime

        job = KubejobRuntime.from_dict(self.to_dict())
        if image:
            job.spec.image = image
        return job

    def _run(self, run: RunObject, execution: MLClientCtx) -> dict:
        handler = run.spec.handler
        handler_obj = getattr(run.spec, "handler_obj", None)
        command = self.spec.command
        workdir = self.spec.workdir
        old_dir = os.getcwd()
        if workdir:
            os.chdir(workdir)
        try:
            if handler_obj is not None or handler:
                if handler_obj is None:
                    if not command:
                        raise MLRunInvalidArgumentError(
                            "handler requested but function has no command "
                            "(python file) to load it from")
                    module = load_module(command)
                    fn_name = handler.split("::")[-1]
                    if not hasattr(module, fn_name):
                        raise MLRunInvalidArgumentError(
                            f"handler {fn_name} not found in {command}")
                    handler_obj = getattr(module, fn_name)
                return exec_from_params(handler_obj, run, execution)
            if not command:
                raise MLRunInvalidArgumentError(
                    "function has neither handler nor command")
            return run_exec_command(command, self.spec.args, run, execution)
        finally:
            if workdir:
                os.chdir(old_dir)


class HandlerRuntime(BaseRuntime):
    """Run a python callable directly (parity: reference HandlerRuntime)."""

    kind = "handler"

    def __init__(self, metadata=None, spec=None, handler=None):
        super().__init__(metadata, spec)
        self.handler = handler

    def _run(self, run: RunObject, execution: MLClientCtx) -> dict:
        handler = getattr(run.spec, "handler_obj", None) or self.handler
        if handler is None:
            raise MLRunInvalidArgumentError("no handler callable set")
        return exec_from_params(handler, run, execution)


def exec_from_params(handler: typing.Callable, run: RunObject,
                     execution: MLClientCtx) -> dict:
    """Call a user handler: inject context/params/inputs, log returns.

    Parity: reference runtimes/local.py:481 + package context handler.
    """
    import inspect

    context = execution
    sig = inspect.signature(handler)
    kwargs = {}
    params = run.spec.parameters or {}
    inputs = run.spec.inputs or {}
    for name, param in sig.parameters.items():
        if name in ("context", "ctx") or (
                param.annotation is MLClientCtx):
            kwargs[name] = context
        elif name in params:
            kwargs[name] = params[name]
        elif name in inputs:
            kwargs[name] = context.get_input(name)
        elif param.default is inspect.Parameter.empty and \
                param.kind not in (inspect.Parameter.VAR_POSITIONAL,
                                   inspect.Parameter.VAR_KEYWORD):
            kwargs[name] = None

    dup = _DupStdout(sys.stdout)
    old_stdout = sys.stdout
    sys.stdout = dup
    host = socket.gethostname()
    context.set_hostname(host)
    error = None
    try:
        returns = handler(**kwargs)
        _log_returns(context, run, returns)
        context.set_state(RunStates.completed, commit=False)
    except Exception as exc:
        error = exc
        logger.error("run failed", error=str(exc))
        traceback.print_exc()
        context.set_state(error=str(exc), commit=False)
    finally:
        sys.stdout = old_stdout
    _store_log(context, dup.buf.getvalue())
    context.commit_db()
    if error and run.spec.verbose:
        raise error
    return context.to_dict()


def _log_returns(context: MLClientCtx, run: RunObject, returns):
    """Pack handler return value(s) into results/artifacts.

    A packager-lite: dicts -> results; DataFrames -> dataset artifacts;
    bytes/str with a configured return key -> artifact; scalars ->
    'return' result.  The reference does this via mlrun.package type-hint
    packagers (package/packagers_manager.py).
    """
    if returns is None:
        return
    keys = run.spec.returns or []
    values = returns if isinstance(returns, tuple) else (returns,)
    for i, value in enumerate(values):
        key_spec = keys[i] if i < len(keys) else None
        key, kind = _parse_return_key(key_spec, i, len(values))
        _log_single(context, key, kind, value)


def _parse_return_key(key_spec, index, total):
    if key_spec is None:
        key = "return" if total == 1 else f"return_{index}"
        return key, None
    if isinstance(key_spec, dict):
        return key_spec.get("key", f"return_{index}"), \
            key_spec.get("artifact_type")
    if ":" in str(key_spec):
        key, kind = str(key_spec).split(":", 1)
        return key, kind.strip()
    return str(key_spec), None


def _log_single(context: MLClientCtx, key: str, kind, value):
    import numpy as np

    try:
        import pandas as pd

        is_df = isinstance(value, pd.DataFrame)
    except ImportError:
        is_df = False
    if kind == "result" or (kind is None and isinstance(
            value, (int, float, str, bool, np.generic))):
        context.log_result(key, value)
    elif isinstance(value, dict) and kind in (None, "result"):
        for rkey, rval in value.items():
            context.log_result(f"{rkey}" if key.startswith("return")
                               else f"{key}_{rkey}", rval)
    elif is_df or kind == "dataset":
        context.log_dataset(key, df=value)
    elif isinstance(value, np.ndarray) or kind == "file":
        if isinstance(value, np.ndarray):
            buf = io.BytesIO()
            np.save(buf, value)
            context.log_artifact(key, body=buf.getvalue(), format="npy")
        else:
            context.log_artifact(key, body=value)
    elif kind == "model":
        context.log_model(key, body=value)
    else:
        context.log_artifact(key, body=str(value))


def _store_log(context: MLClientCtx, text: str):
    if not text:
        return
    try:
        db = context._db
        if db is not None:
            db.store_log(context._uid, context.project, text.encode(),
                         append=True)
    except Exception:
        pass


def run_exec_command(command: str, args: list, run: RunObject,
                     execution: MLClientCtx) -> dict:
    """Run `python -u <command> [args]` as a subprocess, passing the run
    spec via env and reading final state back from MLRUN_META_TMPFILE
    (parity: reference run_exec :423)."""
    tmpfile = tempfile.NamedTemporaryFile(suffix=".json", delete=False)
    tmpfile.close()
    env = os.environ.copy()
    env[META_TMPFILE_ENV] = tmpfile.name
    env["MLRUN_EXEC_CONFIG"] = json.dumps(run.to_dict(), default=str)
    # make the framework importable in the child regardless of install mode
    pkg_root = os.path.dirname(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    env["PYTHONPATH"] = pkg_root + os.pathsep + env.get("PYTHONPATH", "")
    cmd = [sys.executable, "-u", command] + [str(a) for a in (args or [])]
    process = subprocess.run(cmd, env=env, capture_output=True, text=True)
    out = process.stdout or ""
    if process.stderr:
        out += "\n" + process.stderr
    _store_log(execution, out)
    print(out)
    if process.returncode != 0:
        execution.set_state(error=f"exit code {process.returncode}")
        return execution.to_dict()
    # read back state written by the child via get_or_create_ctx commit
    try:
        with open(tmpfile.name) as fp:
            text = fp.read()
        if text.strip():
            child = json.loads(text)
            results = child.get("status", {}).get("results", {}) or {}
            for key, value in results.items():
                execution.log_result(key, value)
            for artifact in child.get("status", {}).get("artifacts", []) or []:
                # already stored in the DB by the child; attach to status
                execution._artifacts_manager.artifacts[
                    artifact["metadata"]["key"]] = \
                    _make_artifact_obj(artifact)
    except (OSError, ValueError):
        pass
    finally:
        try:
            os.remove(tmpfile.name)
        except OSError:
            pass
    execution.set_state(RunStates.completed, commit=False)
    execution.commit_db()
    return execution.to_dict()


def _make_artifact_obj(struct: dict):
    from ..artifacts import Artifact

    return Artifact.from_dict(struct)
