# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Function/runtime object model.

BaseRuntime is the "function" object users build via new_function /
code_to_function; run() routes through a launcher.  Parity target:
reference mlrun/runtimes/base.py (BaseRuntime :171, FunctionSpec :96,
run :314).  The k8s pod-spec machinery of the reference
(runtimes/pod.py) is replaced by a node-local resource spec:
gpus/cpus/memory requested from the local GPU scheduler.
"""

import getpass
import os
import typing

from ..config import config
from ..model import ModelObj, RunObject, RunTemplate
from ..utils import logger, normalize_name


class FunctionMetadata(ModelObj):
    def __init__(self, name=None, tag=None, hash=None, project=None,
                 labels=None, annotations=None, categories=None, updated=None,
                 credentials=None):
        self.name = name or ""
        self.tag = tag or ""
        self.hash = hash or ""
        self.project = project or ""
        self.labels = labels or {}
        self.annotations = annotations or {}
        self.categories = categories or []
        self.updated = updated
        self.credentials = credentials or {}


class ResourceSpec(ModelObj):
    """Node-local resource request: GPUs (MI355X devices), cpus, mem."""

    def __init__(self, gpus=0, cpus=0, memory="", gpu_memory_gb=0):
        self.gpus = gpus
        self.cpus = cpus
        self.memory = memory
        self.gpu_memory_gb = gpu_memory_gb


class FunctionSpec(ModelObj):
    def __init__(self, command=None, args=None, image=None, mode=None,
                 build=None, entry_points=None, description=None,
                 workdir=None, default_handler=None, pythonpath=None,
                 disable_auto_mount=None, allow_empty_resources=None,
                 resources=None):
        self.command = command or ""
        self.args = args or []
        self.image = image or ""
        self.mode = mode or ""
        self.build = build or {}
        self.entry_points = entry_points or {}
        self.description = description or ""
        self.workdir = workdir or ""
        self.default_handler = default_handler or ""
        self.pythonpath = pythonpath or ""
        self.disable_auto_mount = disable_auto_mount
        self.allow_empty_resources = allow_empty_resources
        self._resources = None
        self.resources = resources

    @property
    def resources(self) -> ResourceSpec:
        return self._resources

    @resources.setter
    def resources(self, value):
        self._resources = self._verify_dict(value, "resources", ResourceSpec)

    def to_dict(self, fields=None, exclude=None, strip=False):
        struct = super().to_dict(fields, exclude=["resources"])
        if self._resources:
            res = self._resources.to_dict()
            if res:
                struct["resources"] = res
        return struct


class FunctionStatus(ModelObj):
    def __init__(self, state=None, build_pod=None, address=None,
                 internal_invocation_urls=None, external_invocation_urls=None):
        self.state = state
        self.build_pod = build_pod
        self.address = address
        self.internal_invocation_urls = internal_invocation_urls or []
        self.external_invocation_urls = external_invocation_urls or []


class BaseRuntime(ModelObj):
    kind = "base"
    _is_nested = False
    _is_remote = False

    def __init__(self, metadata=None, spec=None):
        self._metadata = None
        self.metadata = metadata
        self._spec = None
        self.spec = spec
        self._status = None
        self.status = FunctionStatus()
        self._db = None
        self.verbose = False
        self._enriched = False

    @property
    def metadata(self) -> FunctionMetadata:
        return self._metadata

    @metadata.setter
    def metadata(self, value):
        self._metadata = self._verify_dict(value, "metadata", FunctionMetadata)

    @property
    def spec(self) -> FunctionSpec:
        return self._spec

    @spec.setter
    def spec(self, value):
        self._spec = self._verify_dict(value, "spec", FunctionSpec)

    @property
    def status(self) -> FunctionStatus:
        return self._status

    @status.setter
    def status(self, value):
        self._status = self._verify_dict(value, "status", FunctionStatus)

    def apply(self, modifier):
        """Apply a function modifier (reference KubeResource.apply —
        e.g. ``fn.apply(mlrun.platforms.auto_mount())``); modifiers
        are callables taking the runtime object."""
        modifier(self)
        return self

    def is_deployed(self) -> bool:
        return True

    @property
    def uri(self) -> str:
        project = self.metadata.project or config.default_project
        uri = f"{project}/{self.metadata.name}"
        if self.metadata.hash:
            uri += f"@{self.metadata.hash}"
        elif self.metadata.tag:
            uri += f":{self.metadata.tag}"
        return uri

    def _get_db(self):
        if self._db is None:
            from ..db import get_run_db

            self._db = get_run_db()
        return self._db

    def set_db_connection(self, db):
        self._db = db

    def to_dict(self, fields=None, exclude=None, strip=False):
        return {
            "kind": self.kind,
            "metadata": self.metadata.to_dict(),
            "spec": self.spec.to_dict(),
            "status": self.status.to_dict() if not strip else {},
        }

    @classmethod
    def from_dict(cls, struct=None, fields=None, deprecated_fields=None):
        struct = struct or {}
        obj = cls()
        obj.metadata = struct.get("metadata")
        obj.spec = struct.get("spec")
        obj.status = struct.get("status")
        return obj

    # -------------------------------------------------------- resources
    def with_limits(self, mem=None, cpu=None, gpus=None, gpu_type=None):
        """Request node-local resources; gpus = number of MI355X devices."""
        if mem:
            self.spec.resources.memory = mem
        if cpu:
            self.spec.resources.cpus = cpu
        if gpus is not None:
            self.spec.resources.gpus = gpus
        return self

    def with_requests(self, mem=None, cpu=None):
        return self.with_limits(mem=mem, cpu=cpu)

    def with_node_selection(self, node_name=None, node_selector=None,
                            affinity=None, tolerations=None):
        """Node placement (reference pod.py:1156).  Single-node
        deployment: recorded as labels for parity; device placement
        happens via the GPU allocator instead of k8s scheduling."""
        if node_name:
            self.set_label("node-name", node_name)
        for key, value in (node_selector or {}).items():
            self.set_label(f"node-selector/{key}", value)
        return self

    def with_preemption_mode(self, mode):
        """Preemptible-node policy (reference pod.py:1207): recorded
        only — there are no spot nodes in the node-local deployment."""
        self.set_label("preemption-mode", str(mode))
        return self

    def with_priority_class(self, name=""):
        self.set_label("priority-class", name)
        return self

    def set_env(self, name, value):
        self.spec.build.setdefault("env", {})[name] = str(value)
        return self

    def set_envs(self, env_vars: dict):
        for name, value in (env_vars or {}).items():
            self.set_env(name, value)
        return self

    def set_label(self, key, value):
        self.metadata.labels[key] = str(value)
        return self

    # -------------------------------------------------------------- run
    def run(self, runspec: typing.Union[RunTemplate, RunObject, dict] = None,
            handler=None, name: str = "", project: str = "", params: dict = None,
            inputs: dict = None, out_path: str = "", workdir: str = "",
            artifact_path: str = "", watch: bool = True, schedule=None,
            hyperparams: dict = None, hyper_param_options=None, verbose=None,
            scrape_metrics=None, local: bool = None, local_code_path=None,
            auto_build=None, param_file_secrets=None, notifications=None,
            returns=None, state_thresholds=None, selector: str = None,
            reset_on_run: bool = None, labels: dict = None,
            **launcher_kwargs) -> RunObject:
        """Run this function (locally or submitted to the service)."""
        from ..launcher import LauncherFactory

        if selector:  # reference API: run(selector="max.accuracy")
            hyper_param_options = dict(hyper_param_options or {})
            hyper_param_options.setdefault("selector", selector)
        run = self._enrich_run_template(
            runspec, handler=handler, name=name, project=project,
            params=params, inputs=inputs, out_path=out_path,
            artifact_path=artifact_path, workdir=workdir,
            hyperparams=hyperparams, hyper_param_options=hyper_param_options,
            verbose=verbose, scrape_metrics=scrape_metrics,
            notifications=notifications, returns=returns,
            state_thresholds=state_thresholds)
        if labels:
            run.metadata.labels.update(labels)
        if launcher_kwargs:
            from ..utils import logger

            logger.warning("unknown run() arguments ignored",
                           arguments=sorted(launcher_kwargs))
        launcher = LauncherFactory.create_launcher(
            is_remote=self._is_remote and not local, local=local)
        return launcher.launch(self, run, schedule=schedule, watch=watch)

    def _enrich_run_template(self, runspec=None, handler=None, name="",
                             project="", params=None, inputs=None,
                             out_path="", artifact_path="", workdir="",
                             hyperparams=None, hyper_param_options=None,
                             verbose=None, scrape_metrics=None,
                             notifications=None, returns=None,
                             state_thresholds=None) -> RunObject:
        if runspec is None:
            runspec = RunTemplate()
        if isinstance(runspec, dict):
            runspec = RunTemplate.from_dict(runspec)
        if isinstance(runspec, RunTemplate) and not isinstance(runspec,
                                                               RunObject):
            run = RunObject.from_template(runspec)
        else:
            run = runspec.copy()
        spec = run.spec
        spec.handler = handler or spec.handler or self.spec.default_handler
        if callable(spec.handler):
            spec.handler_obj = spec.handler
            spec.handler = getattr(spec.handler, "__name__", "handler")
        run.metadata.name = normalize_name(
            name or run.metadata.name or spec.handler or self.metadata.name
            or "run")
        run.metadata.project = (project or run.metadata.project
                                or self.metadata.project
                                or config.default_project)
        spec.parameters = params or spec.parameters
        spec.inputs = inputs or spec.inputs
        spec.returns = returns or spec.returns
        spec.hyperparams = hyperparams or spec.hyperparams
        if hyper_param_options:
            spec.hyper_param_options = hyper_param_options
        spec.verbose = verbose if verbose is not None else spec.verbose
        spec.scrape_metrics = scrape_metrics if scrape_metrics is not None \
            else spec.scrape_metrics
        spec.output_path = out_path or artifact_path or spec.output_path
        if workdir:
            self.spec.workdir = workdir
        if notifications:
            spec.notifications = notifications
        if state_thresholds:
            spec.state_thresholds = state_thresholds
        spec.function = self.uri
        run.metadata.labels.setdefault("kind", self.kind)
        run.metadata.labels.setdefault("owner",
                                       os.environ.get("USER",
                                                      _safe_username()))
        return run

    def _run(self, run: RunObject, execution) -> dict:
        raise NotImplementedError

    def _store_function(self, run: RunObject, db):
        meta = self.metadata
        meta.tag = meta.tag or "latest"
        meta.project = run.metadata.project or meta.project
        try:
            hash_key = db.store_function(self.to_dict(), meta.name,
                                         meta.project, tag=meta.tag,
                                         versioned=True)
            meta.hash = hash_key
            run.spec.function = self.uri
        except Exception as exc:
            logger.warning("failed to store function", error=str(exc))

    def save(self, tag="", versioned=False, refresh=False) -> str:
        db = self._get_db()
        tag = tag or self.metadata.tag or "latest"
        hash_key = db.store_function(self.to_dict(), self.metadata.name,
                                     self.metadata.project, tag=tag,
                                     versioned=versioned)
        self.metadata.hash = hash_key
        return self.uri

    def export(self, target="", format="yaml") -> str:
        target = target or f"function_{self.metadata.name}.yaml"
        with open(target, "w") as fp:
            fp.write(self.to_yaml())
        return target

    def doc(self):
        print(f"function: {self.metadata.name} ({self.kind})")
        for name, entry in (self.spec.entry_points or {}).items():
            print(f"  handler {name}: {entry.get('doc', '')}")

    def as_step(self, runspec=None, handler=None, name="", project="",
                params=None, inputs=None, outputs=None, workdir="",
                artifact_path="", image="", **kwargs):
        """Return a pipeline step wrapper for this function
        (node-local workflow runner; parity: reference as_step :666)."""
        from ..projects.pipelines import FunctionStep

        return FunctionStep(self, runspec=runspec, handler=handler, name=name,
                            params=params, inputs=inputs, outputs=outputs,
                            artifact_path=artifact_path)


def _safe_username() -> str:
    try:
        return getpass.getuser()
    except Exception:
        return "unknown"
