# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Function factories and run entry points.

Parity target: reference mlrun/run.py (new_function :425,
code_to_function :581, get_or_create_ctx :198, import_function :330,
function_to_module :77, run_local wrapper).
"""

import json
import os
import typing

from .config import config
from .errors import MLRunInvalidArgumentError
from .execution import MLClientCtx
from .model import RunObject
from .runtimes import get_runtime_class, RuntimeKinds
from .runtimes.local import META_TMPFILE_ENV, load_module
from .utils import normalize_name


def new_function(name: str = "", project: str = "", tag: str = "",
                 kind: str = "", command: str = "", image: str = "",
                 args: list = None, mode=None, handler=None, source=None,
                 requirements=None, kfp=None, runtime=None,
                 requirements_file: str = "") -> "BaseRuntime":
    """Create a function object of the given runtime kind."""
    if runtime:
        if hasattr(runtime, "to_dict"):
            runtime = runtime.to_dict()
        kind = kind or runtime.get("kind", "")
        cls = get_runtime_class(kind)
        fn = cls.from_dict(runtime)
    else:
        # split "file.py#handler" command form
        if command and "#" in command:
            command, inline_handler = command.split("#", 1)
            handler = handler or inline_handler
        cls = get_runtime_class(kind)
        fn = cls()
        fn.spec.command = command
        fn.spec.args = args or []
        fn.spec.image = image
        if mode:
            fn.spec.mode = mode
    fn.kind = kind or fn.kind
    fn.metadata.name = normalize_name(
        name or fn.metadata.name
        or (os.path.splitext(os.path.basename(command))[0] if command else "")
        or "fn")
    fn.metadata.project = project or fn.metadata.project or \
        config.default_project
    fn.metadata.tag = tag or fn.metadata.tag
    if handler:
        if callable(handler):
            if hasattr(fn, "handler"):
                fn.handler = handler
            fn.spec.default_handler = getattr(handler, "__name__", "handler")
        else:
            fn.spec.default_handler = handler
    if source:
        fn.spec.build["source"] = source
    if requirements_file:
        with open(requirements_file) as fp:
            requirements = (requirements or []) + [
                line.strip() for line in fp
                if line.strip() and not line.startswith("#")]
    if requirements:
        fn.spec.build["requirements"] = requirements
    return fn


def _notebook_to_code(path: str) -> str:
    """Extract the code cells of a .ipynb (cells starting with
    `# mlrun: ignore` and ipython magics are skipped)."""
    import json as _json

    with open(path) as fp:
        nb = _json.load(fp)
    chunks = []
    for cell in nb.get("cells", []):
        if cell.get("cell_type") != "code":
            continue
        lines = cell.get("source") or []
        if isinstance(lines, str):
            lines = lines.splitlines(keepends=True)
        if lines and lines[0].strip().lower().startswith("# mlrun: ignore"):
            continue
        kept = [ln for ln in lines
                if not ln.lstrip().startswith(("%", "!"))]
        if kept:
            chunks.append("".join(kept).rstrip() + "\n")
    return "\n".join(chunks)


def code_to_function(name: str = "", project: str = "", tag: str = "",
                     filename: str = "", handler: str = "", kind: str = "",
                     image: str = "", code_output: str = "", embed_code=True,
                     description: str = "", requirements=None,
                     categories=None, labels=None,
                     with_doc=True, ignored_tags=None,
                     code_: str = None,
                     requirements_file: str = "") -> "BaseRuntime":
    """Package a python file (or the calling notebook's code) into a
    function object.  ``code_`` embeds source directly (reference
    code_to_function code_ param) instead of reading a file."""
    if code_:
        fn = new_function(name=name or "fn", project=project, tag=tag,
                          kind=kind or "job", image=image,
                          handler=handler)
        fn.spec.description = description
        fn.spec.build["functionSourceCode"] = code_
        if requirements:
            fn.spec.build["requirements"] = requirements
        if labels:
            fn.metadata.labels.update(labels)
        return fn
    if not filename:
        raise MLRunInvalidArgumentError(
            "filename is required (notebook capture is not supported in "
            "the node-local build)")
    if not os.path.isfile(filename):
        raise MLRunInvalidArgumentError(f"file {filename} not found")
    source = None
    command = filename
    if filename.endswith(".ipynb"):
        # notebook -> code (reference: nuclio-jupyter export); code
        # cells concatenate, `# mlrun: ignore` cells are dropped
        source = _notebook_to_code(filename)
        command = ""
    fn = new_function(name=name or os.path.splitext(
        os.path.basename(filename))[0], project=project, tag=tag,
        kind=kind or "job", command=command, image=image, handler=handler)
    fn.spec.description = description
    if source is not None:
        fn.spec.build["functionSourceCode"] = source
    elif embed_code:
        with open(filename) as fp:
            fn.spec.build["functionSourceCode"] = fp.read()
    if requirements_file:
        with open(requirements_file) as fp:
            requirements = (requirements or []) + [
                line.strip() for line in fp
                if line.strip() and not line.startswith("#")]
    if requirements:
        fn.spec.build["requirements"] = requirements
    if categories:
        fn.metadata.categories = categories
    if labels:
        fn.metadata.labels.update(labels)
    if with_doc:
        _extract_entry_points(fn, filename)
    return fn


def _extract_entry_points(fn, filename):
    import ast

    try:
        with open(filename) as fp:
            tree = ast.parse(fp.read())
        for node in ast.walk(tree):
            if isinstance(node, ast.FunctionDef) and not \
                    node.name.startswith("_"):
                fn.spec.entry_points[node.name] = {
                    "name": node.name,
                    "doc": ast.get_docstring(node) or "",
                    "parameters": [a.arg for a in node.args.args],
                }
    except SyntaxError:
        pass


def import_function(url: str = "", project: str = "", name: str = "",
                    new_name: str = "", secrets=None,
                    db="") -> "BaseRuntime":
    """Load a function object from a yaml file, db:// reference, or
    hub://  (parity: reference run.py:330)."""
    import yaml

    if url.startswith("db://"):
        body = url[len("db://"):]
        proj, _, rest = body.partition("/")
        if not rest:
            proj, rest = project or "default", proj
        fn_name, _, tag = rest.partition(":")
        from .db import get_run_db

        struct = get_run_db(db or None).get_function(
            fn_name, proj, tag=tag or "latest")
        fn = new_function(runtime=struct)
    elif url.startswith("hub://"):
        from .hub import get_hub_function

        fn = get_hub_function(url[len("hub://"):])
    else:
        from .datastore import get_object

        body = get_object(url)
        struct = yaml.safe_load(body)
        fn = new_function(runtime=struct)
    if new_name or name:
        fn.metadata.name = normalize_name(new_name or name)
    if project:
        fn.metadata.project = project
    return fn


def function_to_module(code: str, workdir=None, secrets=None, silent=False):
    """Import a function file (or db:// function) as a python module."""
    if code.startswith("db://"):
        fn = import_function(code)
        source = fn.spec.build.get("functionSourceCode")
        if not source:
            raise MLRunInvalidArgumentError(
                "function has no embedded source code")
        import tempfile

        tmp = tempfile.NamedTemporaryFile(suffix=".py", delete=False,
                                          mode="w")
        tmp.write(source)
        tmp.close()
        code = tmp.name
    path = os.path.join(workdir or "", code)
    return load_module(path)


def get_or_create_ctx(name: str, event=None, spec=None, with_env: bool = True,
                      rundb: str = "", project: str = "",
                      upload_artifacts=False,
                      labels: dict = None) -> MLClientCtx:
    """Entry point for user scripts: returns the active run context.

    Inside a framework-launched run (local subprocess, job process, or
    mpijob rank), the run spec arrives via MLRUN_EXEC_CONFIG; otherwise
    a fresh context is created (parity: reference run.py:198).
    """
    spec_struct = None
    if spec:
        spec_struct = spec if isinstance(spec, dict) else json.loads(spec)
    elif with_env and os.environ.get("MLRUN_EXEC_CONFIG"):
        spec_struct = json.loads(os.environ["MLRUN_EXEC_CONFIG"])
    if spec_struct is None:
        spec_struct = {"metadata": {"name": name, "project":
                       project or config.default_project}}
    else:
        spec_struct.setdefault("metadata", {}).setdefault("name", name)
        if project:
            spec_struct["metadata"]["project"] = project

    tmpfile = os.environ.get(META_TMPFILE_ENV, "")
    db = None
    if rundb:
        from .db import create_run_db

        db = create_run_db(rundb)
    ctx = MLClientCtx.from_dict(spec_struct, rundb=db, autocommit=True,
                                tmp=tmpfile)
    for key, value in (labels or {}).items():
        ctx.set_label(key, value)
    return ctx


def run_local(task=None, command: str = "", name: str = "", args: list = None,
              workdir=None, project: str = "", tag: str = "", secrets=None,
              handler=None, params: dict = None, inputs: dict = None,
              artifact_path: str = "", mode=None, allow_empty_resources=None,
              notifications=None, returns=None) -> RunObject:
    """Run a task locally (handler callable or python file).

    Parity: reference run_local — baseline config 1's entry point.
    """
    if callable(command) and not handler:
        handler = command
        command = ""
    fn = new_function(name=name, project=project, tag=tag,
                     kind=RuntimeKinds.handler if callable(handler)
                     else RuntimeKinds.local,
                     command=command, args=args, mode=mode)
    if callable(handler):
        fn.handler = handler
    if workdir:
        fn.spec.workdir = workdir
    return fn.run(task, handler=handler, name=name, params=params,
                  inputs=inputs, artifact_path=artifact_path,
                  notifications=notifications, returns=returns)


def get_object(url, secrets=None, size=None, offset=0):
    from .datastore import get_object as _get

    return _get(url, secrets, size, offset)


def get_dataitem(url, secrets=None):
    from .datastore import get_dataitem as _get

    return _get(url, secrets)


def get_pipeline(run_id: str):
    from .projects.pipelines import get_pipeline as _get

    return _get(run_id)


def wait_for_pipeline_completion(run_id: str, timeout: int = 3600,
                                 expected_statuses=None):
    from .projects.pipelines import wait_for_pipeline_completion as _wait

    return _wait(run_id, timeout, expected_statuses)


def new_model_server(name: str, model_class: typing.Union[str, type] = None,
                     models: dict = None, filename: str = "",
                     protocol: str = "v2", image: str = "", **class_args):
    """Create a serving function pre-loaded with model routes
    (reference runtimes new_model_server helper)."""
    fn = new_function(name=name, kind="serving", command=filename,
                      image=image)
    for key, model_path in (models or {}).items():
        fn.add_model(key, model_path=model_path, class_name=model_class,
                     **class_args)
    return fn
