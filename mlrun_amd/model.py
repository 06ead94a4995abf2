# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Serializable object model: runs, tasks, notifications.

The core spine of the framework — every run/function/artifact object
derives from :class:`ModelObj`, a dict-round-trippable base.  API
parity target: reference mlrun/model.py (ModelObj :46, RunSpec :904,
RunStatus :1262, RunTemplate :1312, RunObject :1454, Notification :681,
HyperParamOptions :856), re-designed without k8s-specific fields.
"""

import inspect
import os
import time
import typing
import uuid
from copy import deepcopy

from .errors import MLRunInvalidArgumentError


class RunStates:
    created = "created"
    pending = "pending"
    running = "running"
    completed = "completed"
    error = "error"
    aborted = "aborted"
    aborting = "aborting"
    skipped = "skipped"
    unknown = "unknown"

    @staticmethod
    def all():
        return [
            RunStates.created,
            RunStates.pending,
            RunStates.running,
            RunStates.completed,
            RunStates.error,
            RunStates.aborted,
            RunStates.aborting,
            RunStates.skipped,
            RunStates.unknown,
        ]

    @staticmethod
    def terminal_states():
        return [RunStates.completed, RunStates.error, RunStates.aborted,
                RunStates.skipped]

    @staticmethod
    def is_terminal(state: str) -> bool:
        return state in RunStates.terminal_states()


class ModelObj:
    """Base class for serializable spec objects (to_dict/from_dict/copy)."""

    _dict_fields: typing.Optional[list] = None
    # fields serialized via their own to_dict
    _default_fields_to_strip: list = []

    @staticmethod
    def _verify_list(param, name):
        if param is not None and not isinstance(param, list):
            raise MLRunInvalidArgumentError(f"parameter {name} must be a list")

    @staticmethod
    def _verify_dict(param, name, new_type=None):
        if param is not None and not isinstance(param, (dict, ModelObj)):
            raise MLRunInvalidArgumentError(f"parameter {name} must be a dict")
        if new_type and isinstance(param, dict):
            return new_type.from_dict(param)
        if new_type and param is None:
            return new_type()
        return param

    def to_dict(self, fields: list = None, exclude: list = None, strip: bool = False) -> dict:
        struct = {}
        fields = fields or self._dict_fields
        if not fields:
            fields = [k for k in inspect.signature(self.__init__).parameters]
        for field in fields:
            if exclude and field in exclude:
                continue
            val = getattr(self, field, None)
            if val is None:
                continue
            if hasattr(val, "to_dict"):
                val = val.to_dict(strip=strip) if _accepts_strip(val) else val.to_dict()
                if val:
                    struct[field] = val
            else:
                struct[field] = val
        return struct

    @classmethod
    def from_dict(cls, struct: dict = None, fields: list = None, deprecated_fields: dict = None):
        struct = struct or {}
        deprecated_fields = deprecated_fields or {}
        fields = fields or cls._dict_fields
        if not fields:
            fields = [k for k in inspect.signature(cls.__init__).parameters
                      if k != "self"]
        new_obj = cls()
        for key, val in struct.items():
            if key in fields and key not in deprecated_fields:
                setattr(new_obj, key, deepcopy(val))
        return new_obj

    def to_json(self, exclude: list = None) -> str:
        import json

        return json.dumps(self.to_dict(exclude=exclude))

    def to_yaml(self, exclude: list = None) -> str:
        import yaml

        return yaml.safe_dump(self.to_dict(exclude=exclude), default_flow_style=False)

    def copy(self):
        return deepcopy(self)

    def __repr__(self):
        return f"{self.__class__.__name__}({self.to_dict()!r})"


def _accepts_strip(obj) -> bool:
    try:
        return "strip" in inspect.signature(obj.to_dict).parameters
    except (ValueError, TypeError):
        return False


class Notification(ModelObj):
    """A notification spec attached to a run (console/webhook/...)."""

    def __init__(self, kind=None, name=None, message=None, severity=None,
                 when=None, condition=None, params=None, status=None,
                 sent_time=None):
        self.kind = kind or "console"
        self.name = name or ""
        self.message = message or ""
        self.severity = severity or "info"
        self.when = when or ["completed"]
        self.condition = condition or ""
        self.params = params or {}
        self.status = status
        self.sent_time = sent_time


class HyperParamStrategies:
    grid = "grid"
    random = "random"
    list = "list"
    custom = "custom"

    @staticmethod
    def all():
        return [HyperParamStrategies.grid, HyperParamStrategies.random,
                HyperParamStrategies.list, HyperParamStrategies.custom]


class HyperParamOptions(ModelObj):
    """Hyperparameter run options (strategy, selector, stop condition)."""

    def __init__(self, param_file=None, strategy=None, selector=None,
                 stop_condition=None, parallel_runs=None, max_iterations=None,
                 max_errors=None, teardown_dask=None):
        self.param_file = param_file
        self.strategy = strategy
        self.selector = selector
        self.stop_condition = stop_condition
        self.parallel_runs = parallel_runs
        self.max_iterations = max_iterations
        self.max_errors = max_errors
        self.teardown_dask = teardown_dask


class RunMetadata(ModelObj):
    def __init__(self, uid=None, name=None, project=None, labels=None,
                 annotations=None, iteration=None):
        self.uid = uid
        self.name = name
        self.project = project or ""
        self.labels = labels or {}
        self.annotations = annotations or {}
        self.iteration = iteration


class RunSpec(ModelObj):
    """What to execute: handler/function ref, parameters, inputs, outputs."""

    def __init__(self, parameters=None, hyperparams=None, param_file=None,
                 selector=None, handler=None, inputs=None, outputs=None,
                 input_path=None, output_path=None, function=None,
                 secret_sources=None, data_stores=None, strategy=None,
                 verbose=None, scrape_metrics=None,
                 hyper_param_options=None, allow_empty_resources=None,
                 inputs_type_hints=None, returns=None, notifications=None,
                 state_thresholds=None, node_selector=None):
        self.parameters = parameters or {}
        self.hyperparams = hyperparams or {}
        self.param_file = param_file
        self.selector = selector
        self.handler = handler
        self.inputs = inputs or {}
        self.outputs = outputs or []
        self.input_path = input_path
        self.output_path = output_path
        self.function = function
        self.secret_sources = secret_sources or []
        self.data_stores = data_stores or []
        self.strategy = strategy
        self.verbose = verbose
        self.scrape_metrics = scrape_metrics
        self._hyper_param_options = None
        self.hyper_param_options = hyper_param_options
        self.allow_empty_resources = allow_empty_resources
        self.inputs_type_hints = inputs_type_hints or {}
        self.returns = returns or []
        self.notifications = notifications or []
        self.state_thresholds = state_thresholds or {}
        self.node_selector = node_selector or {}

    @property
    def hyper_param_options(self) -> HyperParamOptions:
        return self._hyper_param_options

    @hyper_param_options.setter
    def hyper_param_options(self, value):
        self._hyper_param_options = self._verify_dict(
            value, "hyper_param_options", HyperParamOptions)

    def to_dict(self, fields=None, exclude=None, strip=False):
        struct = super().to_dict(fields, exclude=["hyper_param_options"])
        if self._hyper_param_options:
            hpo = self._hyper_param_options.to_dict()
            if hpo:
                struct["hyper_param_options"] = hpo
        return struct


class RunStatus(ModelObj):
    """Run state + results/artifacts as reported by the executor."""

    def __init__(self, state=None, error=None, host=None, commit=None,
                 status_text=None, results=None, artifacts=None,
                 start_time=None, last_update=None, end_time=None,
                 iterations=None, ui_url=None, reason=None,
                 notifications=None, artifact_uris=None):
        self.state = state or RunStates.created
        self.error = error
        self.host = host
        self.commit = commit
        self.status_text = status_text
        self.results = results
        self.artifacts = artifacts
        self.start_time = start_time
        self.last_update = last_update
        self.end_time = end_time
        self.iterations = iterations
        self.ui_url = ui_url
        self.reason = reason
        self.notifications = notifications or {}
        self.artifact_uris = artifact_uris or {}

    def is_failed(self) -> typing.Optional[bool]:
        if self.state in [RunStates.error, RunStates.aborted]:
            return True
        if self.state in [RunStates.completed]:
            return False
        return None


class RunTemplate(ModelObj):
    """A task template: metadata + spec, no status yet."""

    def __init__(self, spec: RunSpec = None, metadata: RunMetadata = None):
        self._spec = None
        self._metadata = None
        self.spec = spec
        self.metadata = metadata

    @property
    def spec(self) -> RunSpec:
        return self._spec

    @spec.setter
    def spec(self, spec):
        self._spec = self._verify_dict(spec, "spec", RunSpec)

    @property
    def metadata(self) -> RunMetadata:
        return self._metadata

    @metadata.setter
    def metadata(self, metadata):
        self._metadata = self._verify_dict(metadata, "metadata", RunMetadata)

    def with_params(self, **kwargs):
        self.spec.parameters = kwargs
        return self

    def with_input(self, name, path):
        self.spec.inputs[name] = path
        return self

    def with_param_file(self, param_file, selector=None, strategy=None,
                        **options):
        """Hyper-param values from a csv/json file url (reference
        model.py:1379)."""
        opts = self.spec.hyper_param_options or HyperParamOptions()
        for key, value in options.items():
            setattr(opts, key, value)
        opts.param_file = param_file
        if selector:
            opts.selector = selector
        if strategy:
            opts.strategy = strategy
        self.spec.hyper_param_options = opts
        self.spec.selector = selector or self.spec.selector
        return self

    def with_hyper_params(self, hyperparams, selector=None, strategy=None,
                          **options):
        self.spec.hyperparams = hyperparams
        self.spec.selector = selector
        opts = self.spec.hyper_param_options or HyperParamOptions()
        if strategy:
            opts.strategy = strategy
        if selector:
            opts.selector = selector
        for key, val in options.items():
            setattr(opts, key, val)
        self.spec.hyper_param_options = opts
        return self

    def with_secrets(self, kind, source):
        self.spec.secret_sources.append({"kind": kind, "source": source})
        return self

    def set_label(self, key, value):
        self.metadata.labels[key] = str(value)
        return self

    def to_dict(self, fields=None, exclude=None, strip=False):
        return {
            "kind": "run",
            "metadata": self.metadata.to_dict(),
            "spec": self.spec.to_dict(),
        }

    @classmethod
    def from_dict(cls, struct=None, fields=None, deprecated_fields=None):
        struct = struct or {}
        obj = cls()
        obj.metadata = struct.get("metadata")
        obj.spec = struct.get("spec")
        return obj


class RunObject(RunTemplate):
    """A tracked run: template + status; knows how to refresh from the DB."""

    def __init__(self, spec: RunSpec = None, metadata: RunMetadata = None,
                 status: RunStatus = None):
        super().__init__(spec, metadata)
        self._status = None
        self.status = status
        self.outputs_wait_for_completion = True

    @property
    def status(self) -> RunStatus:
        return self._status

    @status.setter
    def status(self, status):
        self._status = self._verify_dict(status, "status", RunStatus)

    @classmethod
    def from_template(cls, template: RunTemplate) -> "RunObject":
        return cls(spec=template.spec.copy(), metadata=template.metadata.copy())

    @property
    def uid(self):
        return self.metadata.uid

    def output(self, key):
        """Return a result value or artifact uri by key."""
        if self.status.results and key in self.status.results:
            return self.status.results.get(key)
        artifact = self._artifact(key)
        if artifact:
            return get_artifact_target(artifact, self.metadata.project)
        return None

    @property
    def outputs(self) -> dict:
        outputs = {}
        if self.status.results:
            outputs = dict(self.status.results)
        if self.status.artifacts:
            for artifact in self.status.artifacts:
                key = artifact["metadata"]["key"] if "metadata" in artifact \
                    else artifact.get("key")
                outputs[key] = get_artifact_target(artifact, self.metadata.project)
        return outputs

    def artifact(self, key):
        """Return a DataItem for a named output artifact."""
        artifact = self._artifact(key)
        if artifact is None:
            return None
        from .datastore import get_store_resource_uri_item

        return get_store_resource_uri_item(
            get_artifact_target(artifact, self.metadata.project))

    def _artifact(self, key):
        if self.status.artifacts:
            for artifact in self.status.artifacts:
                akey = artifact["metadata"]["key"] if "metadata" in artifact \
                    else artifact.get("key")
                if akey == key:
                    return artifact
        return None

    def state(self) -> str:
        """Refresh from DB if possible and return current state."""
        self.refresh()
        return self.status.state or RunStates.unknown

    def abort(self):
        """Abort this run (reference model.py RunObject.abort)."""
        from .db import get_run_db

        get_run_db().abort_run(self.metadata.uid,
                               project=self.metadata.project,
                               iter=self.metadata.iteration or 0)
        return self.refresh()

    def refresh(self):
        from .db import get_run_db

        db = get_run_db()
        if db is None:
            return self
        try:
            updated = db.read_run(
                uid=self.metadata.uid, project=self.metadata.project,
                iter=self.metadata.iteration)
        except Exception:
            return self
        if updated:
            self.status = updated.get("status", {})
        return self

    @property
    def error(self) -> str:
        """Error string of a failed run (reference RunObject.error)."""
        if self.status and self.status.state in ("error", "failed"):
            return (getattr(self.status, "error", "") or
                    getattr(self.status, "status_text", "") or
                    "unknown error")
        return ""

    @property
    def ui_url(self) -> str:
        """UI link (node-local: the run's REST resource URL)."""
        from .config import config

        base = str(config.ui_url or
                   f"http://127.0.0.1:{config.httpdb.port}")
        project = self.metadata.project or "default"
        return f"{base}/api/v1/run/{project}/{self.metadata.uid}"

    def is_failed(self) -> bool:
        return self.status is not None and \
            self.status.state in ("error", "failed", "aborted")

    def show(self):
        """Render the run as an HTML table in notebooks (reference
        RunObject.show -> render)."""
        from .render import run_to_html

        return run_to_html(self.to_dict())

    def wait_for_completion(self, timeout: int = 600, sleep: float = 0.5,
                            raise_on_failure: bool = True) -> str:
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            state = self.state()
            if RunStates.is_terminal(state):
                if raise_on_failure and state != RunStates.completed:
                    from .errors import MLRunRuntimeError

                    raise MLRunRuntimeError(
                        f"run {self.metadata.name} did not complete "
                        f"(state={state}): {self.status.error}")
                return state
            time.sleep(sleep)
        from .errors import MLRunTimeoutError

        raise MLRunTimeoutError(
            f"run {self.metadata.name} did not reach terminal state "
            f"within {timeout}s")

    def to_dict(self, fields=None, exclude=None, strip=False):
        struct = super().to_dict()
        struct["status"] = self.status.to_dict()
        return struct

    @classmethod
    def from_dict(cls, struct=None, fields=None, deprecated_fields=None):
        struct = struct or {}
        obj = cls()
        obj.metadata = struct.get("metadata")
        obj.spec = struct.get("spec")
        obj.status = struct.get("status")
        return obj

    def logs(self, watch=False, db=None):
        db = db or _get_db()
        if db is None:
            return ""
        state, text = db.get_log(self.metadata.uid, self.metadata.project)
        if isinstance(text, bytes):
            text = text.decode(errors="replace")
        return text


def _get_db():
    from .db import get_run_db

    return get_run_db()


def get_artifact_target(item: dict, project: str = None) -> str:
    """Build a store:// uri for an artifact dict (or return its target
    path).  Reference helpers.py:109: the store uri is used only when
    the artifact was registered in the DB (``spec.db_key`` set);
    otherwise the raw target path is the address."""
    metadata = item.get("metadata", item)
    spec = item.get("spec", item)
    kind = item.get("kind", "artifact")
    db_key = spec.get("db_key")
    project = metadata.get("project") or project or "default"
    tree = metadata.get("tree")
    tag = metadata.get("tag")
    if kind in ["dataset", "model", "artifact"] and db_key:
        uri = f"store://artifacts/{project}/{db_key}"
        uri += f":{tag}" if tag else ":latest"
        if tree:
            uri += f"@{tree}"
        return uri
    return spec.get("target_path", "")


def new_task(name=None, project=None, handler=None, params=None, hyper_params=None,
             param_file=None, selector=None, hyper_param_options=None, inputs=None,
             outputs=None, in_path=None, out_path=None, artifact_path=None,
             secrets=None, base=None, returns=None) -> RunTemplate:
    """Create a run task template (API parity: reference mlrun/model.py new_task)."""
    if base:
        run = deepcopy(base)
    else:
        run = RunTemplate()
    run.metadata.name = name or run.metadata.name or "task"
    run.metadata.project = project or run.metadata.project
    run.spec.handler = handler or run.spec.handler
    run.spec.parameters = params or run.spec.parameters
    run.spec.hyperparams = hyper_params or run.spec.hyperparams
    run.spec.param_file = param_file or run.spec.param_file
    run.spec.selector = selector or run.spec.selector
    if hyper_param_options:
        run.spec.hyper_param_options = hyper_param_options
    run.spec.inputs = inputs or run.spec.inputs
    run.spec.outputs = outputs or run.spec.outputs
    run.spec.returns = returns or run.spec.returns
    run.spec.input_path = in_path or run.spec.input_path
    run.spec.output_path = artifact_path or out_path or run.spec.output_path
    if secrets:
        run.spec.secret_sources = secrets
    return run


def generate_uid() -> str:
    return uuid.uuid4().hex


class ObjectDict:
    """Dict of typed child objects keyed by name, kind-dispatched
    (reference model.py:272) — used by router routes & graph steps."""

    kind = "object_dict"

    def __init__(self, classes_map: dict, default_kind: str = ""):
        self._children: dict = {}
        self._default_kind = default_kind
        self._classes_map = classes_map

    def values(self):
        return self._children.values()

    def keys(self):
        return self._children.keys()

    def items(self):
        return self._children.items()

    def __len__(self):
        return len(self._children)

    def __iter__(self):
        yield from self._children.keys()

    def __getitem__(self, name):
        return self._children[name]

    def __setitem__(self, name, item):
        self._children[name] = self._resolve(name, item)

    def __delitem__(self, name):
        del self._children[name]

    def update(self, key, item):
        self._children[key] = self._resolve(key, item)
        return self._children[key]

    def to_dict(self) -> dict:
        return {name: child.to_dict()
                for name, child in self._children.items()}

    @classmethod
    def from_dict(cls, classes_map: dict, children: dict,
                  default_kind: str = ""):
        new_obj = cls(classes_map, default_kind)
        for name, child in (children or {}).items():
            new_obj[name] = child
        return new_obj

    def _resolve(self, name, item):
        if hasattr(item, "to_dict") and not isinstance(item, dict):
            return item
        if isinstance(item, dict):
            kind = item.get("kind", self._default_kind)
            cls = self._classes_map.get(kind)
            if cls is None:
                raise MLRunInvalidArgumentError(
                    f"illegal object kind {kind!r} for child {name}")
            if hasattr(cls, "from_dict"):
                child = cls.from_dict(item)
            else:
                from .serving.states import step_from_dict

                child = step_from_dict(item)
            if hasattr(child, "name"):
                child.name = name
            return child
        raise MLRunInvalidArgumentError(
            f"child {name} must be an object or dict")


class ObjectList:
    """Ordered list of typed child objects addressable by name
    (reference model.py:361) — e.g. feature/entity lists."""

    def __init__(self, child_class):
        self._children: dict = {}
        self._child_class = child_class

    def values(self):
        return self._children.values()

    def keys(self):
        return self._children.keys()

    def items(self):
        return self._children.items()

    def __len__(self):
        return len(self._children)

    def __iter__(self):
        yield from self._children.values()

    def __getitem__(self, name):
        if isinstance(name, int):
            return list(self._children.values())[name]
        return self._children[name]

    def __setitem__(self, key, item):
        self.update(item, key)

    def __delitem__(self, key):
        del self._children[key]

    def __contains__(self, key):
        return key in self._children

    def update(self, item, key=None):
        if isinstance(item, dict):
            item = self._child_class.from_dict(item)
        key = key or getattr(item, "name", None)
        if key is None:
            raise MLRunInvalidArgumentError("child item has no name")
        self._children[key] = item
        return item

    def to_dict(self) -> list:
        return [child.to_dict() for child in self._children.values()]

    @classmethod
    def from_list(cls, child_class, children: list = None):
        new_obj = cls(child_class)
        for child in children or []:
            new_obj.update(child)
        return new_obj


class Credentials(ModelObj):
    """Function/run credentials spec (reference model.py:427)."""

    generate_access_key = "$generate"
    secret_reference_prefix = "$ref:"

    def __init__(self, access_key: str = None):
        self.access_key = access_key


class ImageBuilder(ModelObj):
    """Function build spec (reference model.py:485): source, base
    image, commands, requirements."""

    def __init__(self, functionSourceCode=None, source=None, image=None,  # noqa: N803
                 base_image=None, commands=None, extra=None, secret=None,
                 code_origin=None, registry=None, load_source_on_run=None,
                 origin_filename=None, with_mlrun=None, auto_build=None,
                 build_pod=None, requirements: list = None,
                 extra_args=None, source_code_target_dir=None):
        self.functionSourceCode = functionSourceCode  # noqa: N803
        self.source = source
        self.image = image
        self.base_image = base_image
        self.commands = commands or []
        self.extra = extra
        self.secret = secret
        self.code_origin = code_origin
        self.registry = registry
        self.load_source_on_run = load_source_on_run
        self.origin_filename = origin_filename
        self.with_mlrun = with_mlrun
        self.auto_build = auto_build
        self.requirements = requirements or []
        self.extra_args = extra_args
        self.source_code_target_dir = source_code_target_dir


class EntrypointParam(ModelObj):
    """One handler parameter's doc/type/default (reference
    model.py:1865)."""

    def __init__(self, name="", type=None, default=None, doc="",
                 required=None, choices: list = None):
        self.name = name
        self.type = type
        self.default = default
        self.doc = doc
        self.required = required
        self.choices = choices


class FunctionEntrypoint(ModelObj):
    """One handler's signature summary (reference model.py:1883)."""

    def __init__(self, name="", doc="", parameters=None, outputs=None,
                 lineno=-1, has_varargs=None, has_kwargs=None):
        self.name = name
        self.doc = doc
        self.parameters = parameters or []
        self.outputs = outputs or []
        self.lineno = lineno
        self.has_varargs = has_varargs
        self.has_kwargs = has_kwargs


class TargetPathObject:
    """Target path with optional {run_id} templating (reference
    model.py:1983)."""

    def __init__(self, base_path=None, run_id=None,
                 is_single_file=False):
        self.run_id = run_id
        self.full_path_template = base_path
        self.is_single_file = is_single_file
        if run_id is not None and "{run_id}" not in (base_path or ""):
            if not is_single_file:
                self.full_path_template = os.path.join(
                    base_path or "", "{run_id}") + "/"
            else:
                directory, fname = os.path.split(base_path or "")
                self.full_path_template = os.path.join(
                    directory, "{run_id}", fname)

    def get_templated_path(self) -> str:
        return self.full_path_template

    def get_absolute_path(self, project_name: str = None) -> str:
        path = self.full_path_template or ""
        if self.run_id is not None:
            path = path.replace("{run_id}", str(self.run_id))
        if project_name:
            path = path.replace("{project}", project_name)
        return path
