# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Projects: the top-level container of functions/artifacts/workflows.

Parity target: reference mlrun/projects/project.py (new_project :122,
load_project :290, get_or_create_project :435, MlrunProject :1136 with
set_function :2325, run :3055, log_model :1735).  Workflows run with
the local runner (no KFP).
"""

import os

from ..artifacts import ArtifactManager, ArtifactProducer
from ..config import config
from ..errors import MLRunNotFoundError
from ..model import ModelObj, generate_uid
from ..utils import logger, normalize_name, now_iso


class ProjectMetadata(ModelObj):
    def __init__(self, name=None, created=None, labels=None, annotations=None):
        self.name = name
        self.created = created
        self.labels = labels or {}
        self.annotations = annotations or {}


class ProjectSpec(ModelObj):
    def __init__(self, description=None, params=None, functions=None,
                 workflows=None, artifacts=None, artifact_path=None,
                 conda=None, source=None, subpath=None, origin_url=None,
                 goals=None, load_source_on_run=None, default_requirements=None,
                 owner=None, disable_auto_mount=None, workdir=None,
                 default_image=None, build=None):
        self.description = description
        self.params = params or {}
        self.functions = functions or []
        self.workflows = workflows or []
        self.artifacts = artifacts or []
        self.artifact_path = artifact_path
        self.source = source or ""
        self.subpath = subpath
        self.origin_url = origin_url
        self.goals = goals
        self.owner = owner
        self.workdir = workdir
        self.default_image = default_image
        self.build = build or {}


class ProjectStatus(ModelObj):
    def __init__(self, state=None):
        self.state = state


class MlrunProject(ModelObj):
    kind = "project"

    def __init__(self, metadata=None, spec=None, context="./"):
        self._metadata = None
        self.metadata = metadata
        self._spec = None
        self.spec = spec
        self.status = ProjectStatus()
        self._function_objects: dict = {}
        self._artifact_manager = None
        self.context = context
        self._db = None

    @property
    def metadata(self) -> ProjectMetadata:
        return self._metadata

    @metadata.setter
    def metadata(self, value):
        self._metadata = self._verify_dict(value, "metadata", ProjectMetadata)

    @property
    def spec(self) -> ProjectSpec:
        return self._spec

    @spec.setter
    def spec(self, value):
        self._spec = self._verify_dict(value, "spec", ProjectSpec)

    @property
    def name(self):
        return self.metadata.name

    @property
    def artifact_path(self):
        path = self.spec.artifact_path or config.artifact_path or \
            os.path.join(config.base_dir, "artifacts")
        return path.replace("{{project}}", self.name or "default")

    def _get_db(self):
        if self._db is None:
            from ..db import get_run_db

            self._db = get_run_db()
        return self._db

    def to_dict(self, fields=None, exclude=None, strip=False):
        return {
            "kind": self.kind,
            "metadata": self.metadata.to_dict(),
            "spec": self.spec.to_dict(),
            "status": self.status.to_dict(),
        }

    @classmethod
    def from_dict(cls, struct=None, fields=None, deprecated_fields=None):
        struct = struct or {}
        proj = cls()
        proj.metadata = struct.get("metadata")
        proj.spec = struct.get("spec")
        return proj

    # --------------------------------------------------------- functions
    def set_function(self, func=None, name: str = "", kind: str = "",
                     image: str = "", handler: str = "", with_repo=None,
                     tag: str = "", requirements=None,
                     function_object=None,
                     function_dict: dict = None) -> "BaseRuntime":
        # reference aliases: function_object (a live runtime),
        # function_dict (a serialized runtime)
        if function_object is not None and func is None:
            func = function_object
        if function_dict is not None and func is None:
            from ..run import new_function

            func = new_function(runtime=function_dict)
        from ..run import code_to_function, import_function, new_function
        from ..runtimes import BaseRuntime

        if isinstance(func, BaseRuntime):
            fn = func
            name = name or fn.metadata.name
        elif callable(func):
            fn = new_function(name=name or func.__name__, kind=kind or
                              "handler", handler=func)
        elif func and (func.endswith(".yaml") or func.startswith("db://")
                       or func.startswith("hub://")):
            fn = import_function(func, project=self.name)
            if image:
                fn.spec.image = image
        elif func is None and handler:
            fn = new_function(name=name or handler, kind=kind or "local",
                              handler=handler)
        else:
            path = func or ""
            if path and not os.path.isabs(path) and self.context:
                full = os.path.join(self.context, path)
                path = full if os.path.isfile(full) else path
            fn = code_to_function(name=name, project=self.name,
                                  filename=path, handler=handler,
                                  kind=kind or "job", image=image)
        if not name:
            name = fn.metadata.name
        name = normalize_name(name)
        fn.metadata.name = name
        fn.metadata.project = self.name
        if tag:
            fn.metadata.tag = tag
        if image:
            fn.spec.image = image
        self._function_objects[name] = fn
        entry = {"name": name, "kind": fn.kind}
        self.spec.functions = [f for f in self.spec.functions
                               if f.get("name") != name] + [entry]
        try:
            fn.save()
        except Exception as exc:
            logger.debug("function save skipped", error=str(exc))
        return fn

    def get_function(self, key: str, sync=False, enrich=False,
                     ignore_cache=False) -> "BaseRuntime":
        if key in self._function_objects and not ignore_cache:
            return self._function_objects[key]
        from ..run import new_function

        struct = self._get_db().get_function(key, self.name)
        fn = new_function(runtime=struct)
        self._function_objects[key] = fn
        return fn

    def get_function_names(self) -> list:
        return [f.get("name") for f in self.spec.functions]

    def remove_function(self, name):
        self._function_objects.pop(name, None)
        self.spec.functions = [f for f in self.spec.functions
                               if f.get("name") != name]

    # --------------------------------------------------------- execution
    def run_function(self, function, handler=None, name="", params=None,
                     inputs=None, hyperparams=None, hyper_param_options=None,
                     artifact_path=None, workdir="", watch=True, local=None,
                     schedule=None, returns=None, notifications=None,
                     labels=None, outputs=None, selector=None,
                     verbose=None, auto_build=None, base_task=None,
                     builder_env=None, reset_on_run=None):
        from .operations import run_function as _run_function

        run = _run_function(
            function, handler=handler, name=name, params=params,
            inputs=inputs, hyperparams=hyperparams,
            hyper_param_options=hyper_param_options,
            artifact_path=artifact_path or self.artifact_path,
            workdir=workdir, watch=watch, local=local, schedule=schedule,
            returns=returns or outputs, notifications=notifications,
            selector=selector, labels=labels, project_object=self)
        return run

    def build_function(self, function, with_mlrun=None, skip_deployed=False,
                       image=None, base_image=None, commands=None,
                       requirements=None):
        fn = function if not isinstance(function, str) else \
            self.get_function(function)
        if hasattr(fn, "deploy"):
            fn.deploy(watch=False)
        return fn

    def deploy_function(self, function, models=None, env=None, tag=None,
                        verbose=None, builder_env=None, mock=None):
        from .operations import deploy_function as _deploy

        return _deploy(function, models=models, env=env, tag=tag,
                       verbose=verbose, mock=mock, project_object=self)

    # --------------------------------------------------------- artifacts
    def _get_artifact_manager(self) -> ArtifactManager:
        if self._artifact_manager is None:
            self._artifact_manager = ArtifactManager(db=self._get_db())
        return self._artifact_manager

    def _producer(self) -> ArtifactProducer:
        return ArtifactProducer("project", self.name, self.name,
                                uid=generate_uid())

    def log_artifact(self, item, body=None, tag="", local_path="",
                     artifact_path=None, format=None, upload=None,
                     labels=None, target_path="", **kwargs):
        return self._get_artifact_manager().log_artifact(
            self._producer(), item, body=body, tag=tag,
            local_path=local_path,
            artifact_path=artifact_path or self.artifact_path,
            format=format, upload=upload, labels=labels,
            target_path=target_path, **kwargs)

    def log_dataset(self, key, df, tag="", local_path=None, format="parquet",
                    preview=None, stats=None, target_path="",
                    artifact_path=None, upload=None, labels=None,
                    extra_data=None, label_column: str = None, **kwargs):
        from ..artifacts import DatasetArtifact

        ds = DatasetArtifact(key, df=df, format=format, preview=preview,
                             stats=stats, target_path=target_path)
        if label_column:
            ds.label_column = label_column
        if extra_data:
            kwargs.setdefault("extra_data", extra_data)
        return self.log_artifact(ds, local_path=local_path,
                                 artifact_path=artifact_path, tag=tag,
                                 upload=upload, labels=labels, **kwargs)

    def log_model(self, key, body=None, framework="", model_dir=None,
                  model_file=None, metrics=None, parameters=None,
                  artifact_path=None, upload=None, labels=None, inputs=None,
                  outputs=None, tag="", extra_data=None, algorithm=None,
                  training_set=None, label_column=None,
                  feature_vector: str = None, feature_weights: list = None,
                  **kwargs):
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
                for col in label_columns
                if col in training_set.columns]
        if feature_vector:
            kwargs.setdefault("feature_vector", feature_vector)
        if feature_weights:
            kwargs.setdefault("feature_weights", feature_weights)
        return self._get_artifact_manager().log_model(
            self._producer(), key, body=body, model_file=model_file,
            model_dir=model_dir,
            artifact_path=artifact_path or self.artifact_path,
            framework=framework, algorithm=algorithm, metrics=metrics,
            parameters=parameters, inputs=inputs, outputs=outputs, tag=tag,
            extra_data=extra_data, labels=labels, upload=upload, **kwargs)

    def get_artifact(self, key, tag=None, iter=None, tree=None):
        return self._get_db().read_artifact(key, tag=tag, iter=iter,
                                            project=self.name, tree=tree)

    def list_artifacts(self, name=None, tag=None, labels=None, kind=None):
        return self._get_db().list_artifacts(name=name or "",
                                             project=self.name,
                                             tag=tag or "", labels=labels,
                                             kind=kind)

    def get_artifact_uri(self, key, category="artifact", tag=None) -> str:
        uri = f"store://{category}s/{self.name}/{key}"
        if tag:
            uri += f":{tag}"
        return uri

    def list_runs(self, name=None, uid=None, labels=None, state=None,
                  last=0, iter=False):
        return self._get_db().list_runs(
            name=name or "", uid=uid, project=self.name, labels=labels,
            state=state, last=last, iter=iter)

    def list_functions(self, name=None, tag=None, labels=None):
        return self._get_db().list_functions(name=name, project=self.name,
                                             tag=tag or "", labels=labels)

    def list_model_monitoring_functions(self):
        return []

    # ----------------------------------------------------------- alerts
    def store_alert_config(self, alert_data, alert_name: str = None):
        """Create/modify an alert (reference project.py:4205); accepts
        an ``mlrun.alerts.AlertConfig`` or a dict."""
        from ..errors import MLRunInvalidArgumentError
        from ..utils import logger

        if not alert_data:
            raise MLRunInvalidArgumentError("Alert data must be provided")
        name = alert_name or (alert_data.get("name") if isinstance(
            alert_data, dict) else alert_data.name)
        alert_project = (alert_data.get("project") if isinstance(
            alert_data, dict) else alert_data.project)
        if alert_project is not None and alert_project != self.name:
            logger.warning("Project in alert does not match project in "
                           "operation", project=alert_project)
        if isinstance(alert_data, dict):
            alert_data["project"] = self.name
        else:
            alert_data.project = self.name
        return self._get_db().store_alert_config(self.name, name,
                                                 alert_data)

    def get_alert_config(self, alert_name: str):
        from ..alerts import AlertConfig

        struct = self._get_db().get_alert_config(self.name, alert_name)
        return AlertConfig.from_dict(struct) if isinstance(
            struct, dict) else struct

    def list_alerts_configs(self) -> list:
        return self._get_db().list_alert_configs(self.name)

    def delete_alert_config(self, alert_data=None, alert_name: str = None):
        alert_name = self._resolve_alert_name(alert_data, alert_name)
        self._get_db().delete_alert_config(self.name, alert_name)

    def reset_alert_config(self, alert_data=None, alert_name: str = None):
        alert_name = self._resolve_alert_name(alert_data, alert_name)
        self._get_db().reset_alert_config(self.name, alert_name)

    @staticmethod
    def _resolve_alert_name(alert_data, alert_name):
        if alert_data is None and alert_name is None:
            raise ValueError(
                "At least one of alert_data or alert_name must be "
                "provided")
        data_name = None if alert_data is None else (
            alert_data.get("name") if isinstance(alert_data, dict)
            else alert_data.name)
        if data_name and alert_name and data_name != alert_name:
            raise ValueError(
                "Alert_data name does not match the provided alert_name")
        return alert_name or data_name

    def get_alert_template(self, template_name: str):
        return self._get_db().get_alert_template(template_name)

    def list_alert_templates(self) -> list:
        return self._get_db().list_alert_templates()

    # --------------------------------------------------------- workflows
    def set_workflow(self, name, workflow_path: str, embed=False,
                     engine=None, args_schema=None, handler=None, **args):
        entry = {"name": name, "path": workflow_path, "engine":
                 engine or "local", "handler": handler, "args": args}
        self.spec.workflows = [w for w in self.spec.workflows
                               if w.get("name") != name] + [entry]

    def run(self, name="", workflow_path="", arguments=None, artifact_path="",
            workflow_handler=None, namespace=None, sync=False, watch=True,
            dirty=False, engine=None, local=None, schedule=None,
            timeout=None, source=None, cleanup_ttl=None,
            notifications=None) -> "_PipelineRunStatus":
        """Run a named (or ad-hoc) workflow with the local runner."""
        from .pipelines import run_workflow

        workflow = None
        if name:
            for entry in self.spec.workflows:
                if entry.get("name") == name:
                    workflow = entry
                    break
            if workflow is None:
                raise MLRunNotFoundError(f"workflow {name} not found")
        path = workflow_path or (workflow or {}).get("path")
        handler = workflow_handler or (workflow or {}).get("handler")
        return run_workflow(self, path=path, handler=handler,
                            arguments=arguments or {},
                            artifact_path=artifact_path or
                            self.artifact_path, watch=watch)

    # ------------------------------------------------------- persistence
    def save(self, filepath=None, store=True) -> str:
        filepath = filepath or os.path.join(self.context or ".",
                                            "project.yaml")
        os.makedirs(os.path.dirname(os.path.abspath(filepath)), exist_ok=True)
        with open(filepath, "w") as fp:
            fp.write(self.to_yaml())
        if store:
            try:
                self._get_db().store_project(self.name, self.to_dict())
            except Exception as exc:
                logger.warning("failed storing project", error=str(exc))
        return filepath

    def export(self, filepath=None):
        """Export the project: yaml spec, or a .zip/.tar.gz archive of
        the whole context directory (reference project.export)."""
        if filepath and str(filepath).endswith((".zip", ".tar.gz",
                                                ".tgz")):
            import shutil
            import tempfile

            context = self.context or "."
            self.save(os.path.join(context, "project.yaml"),
                      store=False)
            base, fmt = (filepath[:-4], "zip") \
                if filepath.endswith(".zip") else \
                (filepath[:-7] if filepath.endswith(".tar.gz")
                 else filepath[:-4], "gztar")
            archive = shutil.make_archive(base, fmt, root_dir=context)
            if archive != filepath:
                shutil.move(archive, filepath)
            return filepath
        return self.save(filepath, store=False)

    def set_secrets(self, secrets: dict = None, file_path=None, provider=None):
        import os as _os

        if file_path:
            from ..utils import list_to_dict

            with open(file_path) as fp:
                secrets = {**(secrets or {}),
                           **list_to_dict(fp.readlines())}
        for key, value in (secrets or {}).items():
            _os.environ[key] = str(value)

    def get_secret(self, key, default=None):
        import os as _os

        return _os.environ.get(key, default)

    def set_model_monitoring_credentials(self, *args, **kwargs):
        pass

    def enable_model_monitoring(self, default_controller_image="",
                                base_period=10, **kwargs):
        from ..model_monitoring import enable_model_monitoring

        return enable_model_monitoring(self, base_period=base_period)


# ------------------------------------------------------------- factories


def new_project(name, context="./", init_git=False, user_project=False,
                remote=None, from_template=None, secrets=None,
                description=None, subpath=None, save=True,
                overwrite=False, parameters=None,
                default_function_node_selector: dict = None
                ) -> MlrunProject:
    name = normalize_name(name)
    project = MlrunProject(context=context)
    project.metadata.name = name
    project.metadata.created = now_iso()
    project.spec.description = description
    project.spec.params = parameters or {}
    project.spec.subpath = subpath
    if save:
        from ..db import get_run_db

        db = get_run_db()
        try:
            if overwrite:
                try:
                    db.delete_project(name)
                except Exception:
                    pass
            db.create_project(project.to_dict())
        except Exception:
            db.store_project(name, project.to_dict())
    return project


def load_project(context="./", url=None, name=None, secrets=None,
                 init_git=False, subpath=None, clone=False, user_project=False,
                 save=True, sync_functions=False, parameters=None
                 ) -> MlrunProject:
    import yaml

    path = url or os.path.join(context, "project.yaml")
    if url and str(url).endswith((".zip", ".tar.gz", ".tgz")) and \
            os.path.isfile(url):
        # archive export -> unpack into the context dir and load
        import shutil

        os.makedirs(context, exist_ok=True)
        shutil.unpack_archive(url, context)
        path = os.path.join(context, "project.yaml")
    if os.path.isfile(path):
        with open(path) as fp:
            struct = yaml.safe_load(fp)
        project = MlrunProject.from_dict(struct)
        project.context = context
    else:
        from ..db import get_run_db

        struct = get_run_db().get_project(name or os.path.basename(
            os.path.abspath(context)))
        project = MlrunProject.from_dict(struct)
        project.context = context
    if name:
        project.metadata.name = normalize_name(name)
    if parameters:
        project.spec.params.update(parameters)
    if save:
        project.save()
    return project


def get_or_create_project(name, context="./", url=None, secrets=None,
                          init_git=False, subpath=None, clone=False,
                          user_project=False, from_template=None, save=True,
                          parameters=None,
                          allow_cross_project: bool = None
                          ) -> MlrunProject:
    from ..db import get_run_db

    name = normalize_name(name)
    try:
        struct = get_run_db().get_project(name)
        project = MlrunProject.from_dict(struct)
        project.context = context
        if parameters:
            project.spec.params.update(parameters)
        return project
    except Exception:
        pass
    try:
        return load_project(context, url=url, name=name, save=save,
                            parameters=parameters)
    except Exception:
        return new_project(name, context=context, save=save,
                           description=None, parameters=parameters)


class _PipelineContext:
    """Holds the active project/workflow during a pipeline run."""

    def __init__(self):
        self.project = None
        self.workflow = None
        self.functions = {}
        self.runs = []

    def set(self, project, workflow=None):
        self.project = project
        self.workflow = workflow

    def clear(self):
        self.project = None
        self.workflow = None
        self.runs = []

    def is_run_local(self):
        return True


pipeline_context = _PipelineContext()

