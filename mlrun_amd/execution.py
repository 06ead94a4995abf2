# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""MLClientCtx — the run execution context.

Handed to user handlers; tracks params/inputs/results/artifacts and
commits state to the run DB.  Parity target: reference
mlrun/execution.py:51 (log_result :541, log_artifact :599, log_model
:749, commit :861, set_state :888, is_logging_worker :1040 — the
rank-0-only logging gate now reads RANK from the RCCL/xGMI local rank
launcher instead of OpenMPI env).
"""

import os
import threading

from .artifacts import ArtifactManager, ArtifactProducer
from .config import config
from .errors import MLRunInvalidArgumentError
from .model import RunStates, generate_uid
from .utils import logger, now_date, to_date_str


class MLClientCtx:
    kind = "run"

    def __init__(self, autocommit=False, tmp="", log_stream=None):
        self._uid = ""
        self.name = ""
        self._iteration = 0
        self._project = ""
        self._tag = ""
        self._labels = {}
        self._annotations = {}
        self._function = ""
        self._parameters = {}
        self._inputs = {}
        self._outputs = []
        self._results = {}
        self._state = RunStates.created
        self._error = None
        self._commit_text = ""
        self._host = None
        self._start_time = now_date()
        self._last_update = now_date()
        self._iteration_results = None
        self._child_iterations = []
        self._autocommit = autocommit
        self._tmpfile = tmp
        self._logger = log_stream or logger
        self._db = None
        self._artifacts_manager = None
        self._out_path = ""
        self._in_path = ""
        self._secrets = {}
        self._state_lock = threading.Lock()
        self._updates_lock = threading.Lock()
        self._is_api = False

    # ------------------------------------------------------ construction
    @classmethod
    def from_dict(cls, attrs: dict, rundb=None, autocommit=False, tmp="",
                  host=None, log_stream=None, is_api=False,
                  store_run=True) -> "MLClientCtx":
        self = cls(autocommit=autocommit, tmp=tmp, log_stream=log_stream)
        meta = attrs.get("metadata", {})
        spec = attrs.get("spec", {})
        self._uid = meta.get("uid") or generate_uid()
        self.name = meta.get("name") or "run"
        self._iteration = meta.get("iteration") or 0
        self._project = meta.get("project") or config.default_project
        self._labels = meta.get("labels", {}) or {}
        self._annotations = meta.get("annotations", {}) or {}
        self._function = spec.get("function", "")
        self._parameters = spec.get("parameters", {}) or {}
        self._inputs = spec.get("inputs", {}) or {}
        self._outputs = spec.get("outputs", []) or []
        self._in_path = spec.get("input_path") or ""
        self._out_path = spec.get("output_path") or ""
        self._host = host
        self._is_api = is_api
        for source in spec.get("secret_sources", []) or []:
            self._add_secret_source(source)
        if rundb is not None:
            self._db = rundb
        if store_run:
            self._init_db()
            self.commit_db(state=RunStates.running)
        return self

    def _init_db(self):
        if self._db is None:
            from .db import get_run_db

            self._db = get_run_db()
        self._artifacts_manager = ArtifactManager(db=self._db)

    def _add_secret_source(self, source: dict):
        kind = source.get("kind")
        src = source.get("source")
        if kind == "inline" and isinstance(src, dict):
            self._secrets.update(src)
        elif kind == "env" and isinstance(src, str):
            for name in src.split(","):
                name = name.strip()
                if name in os.environ:
                    self._secrets[name] = os.environ[name]
        elif kind == "file" and src and os.path.isfile(src):
            from .utils import list_to_dict

            with open(src) as fp:
                self._secrets.update(list_to_dict(fp.readlines()))

    # -------------------------------------------------------- properties
    @property
    def uid(self):
        if self._iteration:
            return f"{self._uid}-{self._iteration}"
        return self._uid

    @property
    def tag(self):
        return self._tag or self._uid

    @property
    def iteration(self):
        return self._iteration

    @property
    def project(self):
        return self._project

    @property
    def parameters(self):
        return dict(self._parameters)

    @property
    def inputs(self) -> dict:
        """Input DataItems by name."""
        return {k: self.get_input(k) for k in self._inputs}

    @property
    def results(self):
        return dict(self._results)

    @property
    def state(self):
        return self._state

    @property
    def out_path(self):
        return self.artifact_path

    @property
    def artifact_path(self):
        path = self._out_path or config.artifact_path or os.path.join(
            config.base_dir, "artifacts")
        return path.replace("{{project}}", self._project or "default")

    @property
    def in_path(self):
        return self._in_path

    @property
    def labels(self):
        return self._labels

    @property
    def annotations(self):
        return self._annotations

    @property
    def logger(self):
        return self._logger

    def artifact_subpath(self, *subpaths) -> str:
        return os.path.join(self.artifact_path, *subpaths)

    # ------------------------------------------------------- gets & sets
    def get_param(self, key: str, default=None):
        if key not in self._parameters:
            self._parameters[key] = default
            self._update_db()
            return default
        return self._parameters[key]

    def get_input(self, key: str, url: str = ""):
        from .datastore import store_manager

        if key not in self._inputs:
            if not url:
                raise MLRunInvalidArgumentError(f"input {key} not found")
            self._inputs[key] = url
        url = self._inputs[key]
        if self._in_path and not ("://" in url or url.startswith("/")):
            url = os.path.join(self._in_path, url)
        return store_manager.object(url, key=key, project=self._project)

    def get_secret(self, key: str, default=None):
        if key in self._secrets:
            return self._secrets[key]
        return os.environ.get(key, default)

    def set_label(self, key: str, value):
        self._labels[key] = str(value)

    def set_annotation(self, key: str, value):
        self._annotations[key] = str(value)

    def get_meta(self) -> dict:
        return {"name": self.name, "kind": "run", "uri": self._function,
                "owner": self._labels.get("owner")}

    # -------------------------------------------------------- dist gates
    def get_rank(self) -> int:
        return int(os.environ.get("RANK", os.environ.get("OMPI_COMM_WORLD_RANK",
                                                         "0")))

    def is_logging_worker(self) -> bool:
        """Only rank 0 logs results/artifacts in distributed runs
        (parity: reference execution.py:1040)."""
        return self.get_rank() == 0

    # ----------------------------------------------------------- logging
    def log_result(self, key: str, value, commit=False):
        if not self.is_logging_worker():
            return
        self._results[str(key)] = _cast_result(value)
        self._update_db(commit=commit)

    def log_results(self, results: dict, commit=False):
        if not self.is_logging_worker():
            return
        if not isinstance(results, dict):
            raise MLRunInvalidArgumentError("results must be a dict")
        for key, value in results.items():
            self._results[str(key)] = _cast_result(value)
        self._update_db(commit=commit)

    def log_metric(self, key: str, value, timestamp=None, labels=None):
        self.log_result(key, value)

    def log_metrics(self, keyvals: dict, timestamp=None, labels=None):
        self.log_results(keyvals)

    def log_iteration_results(self, best, summary: list, task: dict,
                              commit=False):
        """Record child (hyperparam) iteration summary on the parent run."""
        self._iteration_results = summary
        if best:
            self._results["best_iteration"] = best
            for key, value in get_in_results(task):
                self._results[key] = value
        self._update_db(commit=True)

    def log_artifact(self, item, body=None, local_path="", artifact_path=None,
                     tag="", viewer=None, target_path="", format=None,
                     upload=None, labels=None, db_key=None,
                     src_path: str = None, **kwargs):
        # src_path: deprecated reference alias of local_path
        local_path = local_path or src_path or ""
        if not self.is_logging_worker():
            return None
        self._init_db()
        producer = self._producer()
        item = self._artifacts_manager.log_artifact(
            producer, item, body=body, local_path=local_path,
            artifact_path=artifact_path or self.artifact_path, tag=tag,
            viewer=viewer, target_path=target_path, format=format,
            upload=upload, labels=labels, db_key=db_key, **kwargs)
        self._update_db()
        return item

    def log_dataset(self, key, df, tag="", local_path=None, artifact_path=None,
                    upload=None, labels=None, format="parquet", preview=None,
                    stats=None, db_key=None, target_path="", **kwargs):
        from .artifacts import DatasetArtifact

        ds = DatasetArtifact(key, df=df, format=format, preview=preview,
                             stats=stats, target_path=target_path)
        return self.log_artifact(ds, local_path=local_path,
                                 artifact_path=artifact_path, tag=tag,
                                 upload=upload, labels=labels, db_key=db_key,
                                 **kwargs)

    def log_model(self, key, body=None, framework="", tag="", model_dir=None,
                  model_file=None, algorithm=None, metrics=None,
                  parameters=None, artifact_path=None, upload=None,
                  labels=None, inputs=None, outputs=None, extra_data=None,
                  db_key=None, training_set=None, label_column=None,
                  feature_vector: str = None, feature_weights: list = None,
                  **kwargs):
        """training_set/label_column derive the model's input/output
        feature lists (reference log_model behavior); feature_vector/
        feature_weights record the serving enrichment source."""
        if not self.is_logging_worker():
            return None
        if training_set is not None and hasattr(training_set, "columns"):
            label_columns = [label_column] if isinstance(
                label_column, str) else list(label_column or [])
            inputs = inputs or [
                {"name": col, "value_type": str(dtype)}
                for col, dtype in zip(training_set.columns,
                                      training_set.dtypes)
                if col not in label_columns]
            outputs = outputs or [
                {"name": col,
                 "value_type": str(training_set[col].dtype)}
                for col in label_columns if col in training_set.columns]
        if feature_vector:
            kwargs.setdefault("feature_vector", feature_vector)
        if feature_weights:
            kwargs.setdefault("feature_weights", feature_weights)
        self._init_db()
        producer = self._producer()
        model = self._artifacts_manager.log_model(
            producer, key, body=body, model_file=model_file,
            model_dir=model_dir,
            artifact_path=artifact_path or self.artifact_path,
            framework=framework, algorithm=algorithm, metrics=metrics,
            parameters=parameters, inputs=inputs, outputs=outputs, tag=tag,
            extra_data=extra_data, labels=labels, upload=upload, **kwargs)
        self._update_db()
        return model

    def _producer(self) -> ArtifactProducer:
        producer = ArtifactProducer("run", self._project, self.name,
                                    tag=self._tag,
                                    owner=self._labels.get("owner"),
                                    uid=self._uid)
        producer.iteration = self._iteration
        return producer

    # ------------------------------------------------------ state + sync
    def set_state(self, state: str = None, error: str = None, commit=True):
        with self._state_lock:
            if error:
                self._state = RunStates.error
                self._error = str(error)
            elif state and state != self._state and \
                    not RunStates.is_terminal(self._state):
                self._state = state
            self._last_update = now_date()
        if commit:
            self.commit_db()
        return self._state

    def set_hostname(self, host: str):
        self._host = host

    def get_dataitem(self, url: str, secrets: dict = None):
        """Resolve a data URL to a DataItem handle (reference
        execution.py get_dataitem — the common handler entry for
        reading inputs by URL)."""
        from .datastore import store_manager

        return store_manager.object(url, project=self._project)

    def get_store_resource(self, url: str):
        """Resolve a store:// URI to its artifact/feature object."""
        from .datastore import store_manager

        return store_manager.object(url, project=self._project)

    def to_yaml(self) -> str:
        import yaml

        return yaml.safe_dump(self.to_dict(), default_flow_style=False)

    @property
    def artifacts(self) -> list:
        """Artifact documents logged by this run (reference
        ctx.artifacts — run status artifact list)."""
        if self._artifacts_manager is None:
            return []
        return self._artifacts_manager.artifact_list()

    def update_artifact(self, artifact):
        """Re-log an updated artifact object (reference
        update_artifact)."""
        return self.log_artifact(artifact)

    def store_run(self):
        """Force-store the run document to the DB now."""
        self.commit()

    def commit(self, message: str = "", completed=False):
        self._commit_text = message
        if completed:
            self.set_state(RunStates.completed)
        else:
            self.commit_db()

    # context-manager: `with ctx.get_child_context(..) as child` —
    # exit commits the child (errors mark it failed)
    def __enter__(self):
        return self

    def __exit__(self, exc_type, exc_value, exc_tb):
        if exc_type is not None:
            self.set_state(RunStates.error, str(exc_value))
        elif getattr(self, "_parent", None) is not None:
            self.commit(completed=True)
        return False

    def get_cached_artifact(self, key: str):
        """Logged-artifact object by key (reference execution.py
        get_cached_artifact)."""
        if self._artifacts_manager is None:
            return None
        return self._artifacts_manager.artifacts.get(key)

    def get_notifications(self, unmask_secret_params: bool = False):
        return getattr(self, "_notifications", []) or []

    def get_project_object(self):
        """Load this run's project object from the DB (reference
        execution.py get_project_object)."""
        from .projects import get_or_create_project

        return get_or_create_project(self._project or "default")

    def get_project_param(self, key: str, default=None):
        project = self.get_project_object()
        if project is None:
            return default
        return project.spec.params.get(key, default)

    @property
    def log_level(self) -> str:
        import logging as _logging

        return _logging.getLevelName(
            getattr(self._logger, "level", _logging.INFO)).lower()

    def set_logger_stream(self, stream):
        handler = getattr(self._logger, "_handler", None)
        if handler is not None:
            handler.stream = stream

    def get_child_context(self, with_parent_params: bool = False,
                          **params) -> "MLClientCtx":
        """Child context for a sub-experiment iteration (reference
        execution.py:223): log_xx on the child updates that iteration
        only; ``update_child_iterations`` folds them into the parent;
        ``child.mark_as_best()`` marks its iteration.

        Example::

            for param in param_list:
                with context.get_child_context(p=param) as child:
                    accuracy = child_handler(child, **child.parameters)
                    child.log_result("accuracy", accuracy)
        """
        from .errors import MLRunInvalidArgumentError

        if self._iteration != 0:
            raise MLRunInvalidArgumentError(
                "cannot create child from a child iteration!")
        struct = self.to_dict()
        spec = struct.setdefault("spec", {})
        spec["parameters"] = dict(self._parameters) \
            if with_parent_params else {}
        spec["parameters"].update(params)
        struct["status"] = {}
        struct.setdefault("metadata", {})["iteration"] = \
            len(self._children) + 1
        child = MLClientCtx.from_dict(
            struct, rundb=self._db, autocommit=self._autocommit,
            log_stream=self._logger, store_run=False)
        child._artifacts_manager = self._artifacts_manager
        child._parent = self
        self._children.append(child)
        return child

    def update_child_iterations(self, best_run: int = 0,
                                commit_children: bool = False,
                                completed: bool = True):
        """Fold child-iteration results into this (parent) run
        (reference execution.py:271): builds the iteration table,
        records the best child's results."""
        if not self._children:
            return
        if commit_children:
            for child in self._children:
                child.commit(completed=completed)
        # iteration table: header + one row per child
        result_keys: list = []
        param_keys: list = []
        for child in self._children:
            for key in child._parameters:
                if key not in param_keys:
                    param_keys.append(key)
            for key in child._results:
                if key not in result_keys:
                    result_keys.append(key)
        header = ["state", "iter"] + [f"param.{k}" for k in param_keys] \
            + [f"output.{k}" for k in result_keys]
        rows = [header]
        for child in self._children:
            rows.append([child._state, child._iteration]
                        + [child._parameters.get(k) for k in param_keys]
                        + [child._results.get(k) for k in result_keys])
        task = self._children[best_run - 1].to_dict() if best_run else None
        self.log_iteration_results(best_run, rows, task)

    @property
    def _children(self) -> list:
        return self._child_iterations

    def mark_as_best(self):
        """Mark a child run as the best iteration (reference
        execution.py:291)."""
        parent = getattr(self, "_parent", None)
        if parent is not None and self._iteration:
            parent.log_iteration_results(self._iteration, None,
                                         self.to_dict())
        self.set_label("best_iteration", self._iteration)

    # ------------------------------------------------------------ dicts
    def to_dict(self) -> dict:
        struct = {
            "kind": "run",
            "metadata": {
                "name": self.name,
                "uid": self._uid,
                "iteration": self._iteration,
                "project": self._project,
                "labels": self._labels,
                "annotations": self._annotations,
            },
            "spec": {
                "function": self._function,
                "parameters": self._parameters,
                "inputs": self._inputs,
                "outputs": self._outputs,
                "output_path": self._out_path,
                "input_path": self._in_path,
            },
            "status": {
                "state": self._state,
                "results": self._results,
                "start_time": to_date_str(self._start_time),
                "last_update": to_date_str(self._last_update),
            },
        }
        if self._error is not None:
            struct["status"]["error"] = self._error
        if self._host:
            struct["status"]["host"] = self._host
        if self._iteration_results:
            struct["status"]["iterations"] = self._iteration_results
        if self._artifacts_manager:
            artifacts = self._artifacts_manager.artifact_list()
            if artifacts:
                struct["status"]["artifacts"] = artifacts
                struct["status"]["artifact_uris"] = {
                    a["metadata"]["key"]:
                        f"store://artifacts/{self._project}/"
                        f"{a['metadata']['key']}@{a['metadata'].get('tree', '')}"
                    for a in artifacts}
        return struct

    def to_json(self):
        import json

        return json.dumps(self.to_dict())

    def _update_db(self, commit=False):
        self._last_update = now_date()
        if commit or self._autocommit:
            self.commit_db()

    def commit_db(self, state: str = None):
        if state:
            with self._state_lock:
                if not RunStates.is_terminal(self._state):
                    self._state = state
        if self._db is None:
            return
        with self._updates_lock:
            self._db.store_run(self.to_dict(), self._uid, self._project,
                               iter=self._iteration)
        if self._tmpfile:
            try:
                with open(self._tmpfile, "w") as fp:
                    fp.write(self.to_json())
            except OSError:
                pass


def _cast_result(value):
    try:
        import numpy as np

        if isinstance(value, np.generic):
            return value.item()
    except ImportError:
        pass
    return value


def get_in_results(task: dict):
    results = task.get("status", {}).get("results", {}) or {}
    return results.items()
