# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""RunDBExtras: the convenience/alias method families of the client
surface (reference httpdb.py), implemented against the primitive
RunDBInterface methods so BOTH the HTTP client and the node-local
SQLRunDB expose the same 124-method surface (local mode is a drop-in
for client code that calls e.g. ``db.patch_feature_set``)."""

import typing


class RunDBExtras:
    """Mixin over RunDBInterface primitives."""

    # ---------------------------------------------------------- create
    def create_feature_set(self, feature_set, project="", versioned=True):
        body = feature_set.to_dict() if hasattr(feature_set, "to_dict") \
            else feature_set
        name = body.get("metadata", {}).get("name", "")
        return self.store_feature_set(body, name=name, project=project)

    def create_feature_vector(self, feature_vector, project="",
                              versioned=True):
        body = feature_vector.to_dict() if hasattr(
            feature_vector, "to_dict") else feature_vector
        name = body.get("metadata", {}).get("name", "")
        return self.store_feature_vector(body, name=name, project=project)

    def create_model_endpoint(self, project, endpoint_id, model_endpoint):
        body = model_endpoint.to_dict() if hasattr(
            model_endpoint, "to_dict") else model_endpoint
        return self.store_model_endpoint(project, endpoint_id, body)

    def create_project_secrets(self, project, provider="kubernetes",
                               secrets=None):
        return self.store_project_secrets(project, secrets or {})

    def create_user_secrets(self, user, provider="vault", secrets=None):
        raise NotImplementedError(
            "user (vault) secrets are not supported; use project "
            "secrets (store_project_secrets)")

    # ----------------------------------------------------------- patch
    @staticmethod
    def _deep_update(target: dict, patch: dict) -> dict:
        for key, value in (patch or {}).items():
            if isinstance(value, dict) and isinstance(
                    target.get(key), dict):
                RunDBExtras._deep_update(target[key], value)
            else:
                target[key] = value
        return target

    def patch_feature_set(self, name, feature_set_update: dict,
                          project="", tag="latest",
                          patch_mode="additive"):
        current = self.get_feature_set(name, project, tag)
        self._deep_update(current, feature_set_update)
        return self.store_feature_set(current, name=name, project=project,
                                      tag=tag)

    def patch_feature_vector(self, name, feature_vector_update: dict,
                             project="", tag="latest",
                             patch_mode="additive"):
        current = self.get_feature_vector(name, project, tag)
        self._deep_update(current, feature_vector_update)
        return self.store_feature_vector(current, name=name,
                                         project=project, tag=tag)

    def patch_model_endpoint(self, project, endpoint_id,
                             attributes: dict):
        current = self.get_model_endpoint(project, endpoint_id)
        self._deep_update(current, attributes)
        self.store_model_endpoint(project, endpoint_id, current)
        return current

    def patch_project(self, name, project: dict, patch_mode="replace"):
        current = self.get_project(name)
        self._deep_update(current, project)
        self.store_project(name, current)
        return current

    # --------------------------------------------------------- aliases
    def list_alerts_configs(self, project=""):
        return self.list_alert_configs(project)

    def list_features_v2(self, project, name=None, tag=None,
                         entities=None, labels=None):
        return self.list_features(project, name=name, tag=tag,
                                  entities=entities, labels=labels)

    def list_entities_v2(self, project, name=None, tag=None, labels=None):
        return self.list_entities(project, name=name, tag=tag,
                                  labels=labels)

    def get_project_background_task(self, project, name):
        return self.get_background_task(project, name)

    def list_project_background_tasks(self, project):
        return self.list_background_tasks(project)

    def list_project_secrets(self, project, token=None,
                             provider="kubernetes", secrets=None):
        return {"secrets": {k: None for k in
                            self.list_project_secret_keys(project)}}

    # ------------------------------------------------------------ tags
    def tag_artifacts(self, artifacts, project, tag_name, replace=False):
        for artifact in artifacts if isinstance(artifacts, list) \
                else [artifacts]:
            meta = artifact.get("metadata", artifact) if isinstance(
                artifact, dict) else getattr(artifact, "metadata", {})
            get = meta.get if isinstance(meta, dict) else \
                lambda k, d=None: getattr(meta, k, d)
            self.tag_artifact(project, get("key"),
                              get("tree") or get("uid"), tag_name,
                              get("iter", 0) or 0)

    tag_objects = tag_artifacts

    def delete_artifacts_tags(self, artifacts, project, tag_name):
        for artifact in artifacts if isinstance(artifacts, list) \
                else [artifacts]:
            meta = artifact.get("metadata", artifact) if isinstance(
                artifact, dict) else {}
            self.delete_artifact_tag(project, meta.get("key"), tag_name)

    delete_objects_tag = delete_artifacts_tags

    # ------------------------------------------------------------ logs
    def get_log_size(self, uid, project=""):
        _, log = self.get_log(uid, project)
        return len(log or b"")

    def watch_log(self, uid, project="", watch=True, offset=0):
        import sys
        import time as _time

        from ..model import RunStates

        state = ""
        while True:
            state, log = self.get_log(uid, project, offset=offset)
            if log:
                text = log.decode() if isinstance(log, bytes) else log
                sys.stdout.write(text)
                offset += len(log)
            if not watch or RunStates.is_terminal(state):
                return state
            _time.sleep(2)

    # --------------------------------------------------- notifications
    def set_run_notifications(self, project, run_uid,
                              notifications=None):
        run = self.read_run(run_uid, project)
        run.setdefault("spec", {})["notifications"] = [
            n.to_dict() if hasattr(n, "to_dict") else n
            for n in notifications or []]
        self.store_run(run, run_uid, project)

    store_run_notifications = set_run_notifications

    def set_schedule_notifications(self, project, schedule_name,
                                   notifications=None):
        sched = self.get_schedule(project, schedule_name)
        task = sched.get("task") or {}
        task.setdefault("spec", {})["notifications"] = [
            n.to_dict() if hasattr(n, "to_dict") else n
            for n in notifications or []]
        self.update_schedule(project, schedule_name, {"task": task})

    def store_alert_notifications(self, session, notification_objects,
                                  alert_id, project):
        config_ = self.get_alert_config(project, alert_id)
        config_["notifications"] = [
            n.to_dict() if hasattr(n, "to_dict") else n
            for n in notification_objects or []]
        self.store_alert_config(project, alert_id, config_)

    def reset_alert_config(self, project, name):
        if hasattr(self, "reset_alert_state"):
            self.reset_alert_state(project, name)

    # -------------------------------------------------------------- hub
    def list_hub_sources(self, item_name=None, tag=None, version=None):
        from ..hub import list_hub_sources

        return list_hub_sources()

    def get_hub_source(self, source_name):
        for source in self.list_hub_sources():
            if source.get("name") == source_name:
                return source
        from ..errors import MLRunNotFoundError

        raise MLRunNotFoundError(f"hub source {source_name} not found")

    def store_hub_source(self, source_name, source: dict):
        from .. import hub

        body = source if isinstance(source, dict) else source.to_dict()
        spec = body.get("spec", body)
        hub.add_hub_source(source_name, spec.get("path", ""),
                           int(spec.get("order", -1)))
        return body

    create_hub_source = store_hub_source

    def delete_hub_source(self, source_name):
        from .. import hub

        hub._sources.pop(source_name, None)

    def get_hub_catalog(self, source="builtin"):
        from ..hub import get_hub_catalog

        return get_hub_catalog(source)

    def get_hub_item(self, source_name, item_name, version=None,
                     tag="latest", force_refresh=False):
        for item in self.get_hub_catalog(source_name):
            if item.get("name") == item_name or \
                    (item.get("metadata") or {}).get("name") == item_name:
                return item
        from ..errors import MLRunNotFoundError

        raise MLRunNotFoundError(
            f"hub item {item_name} not found in {source_name}")

    def get_hub_asset(self, source_name, item_name, asset_name,
                      version=None, tag="latest"):
        item = self.get_hub_item(source_name, item_name)
        spec = item.get("spec", item)
        path = spec.get("item_uri") or spec.get("filename") or ""
        with open(path, "rb") as fp:
            return fp.read()

    # ------------------------------------------------------------ files
    def get_file(self, path, size=0, offset=0):
        from ..datastore import store_manager

        return store_manager.object(path).get(size=size or None,
                                              offset=offset)

    # ------------------------------------------- workflows / pipelines
    def list_workflows(self, project):
        body = self.get_project(project)
        return ((body or {}).get("spec", {}) or {}).get("workflows", [])

    def submit_workflow(self, project, name, arguments=None):
        from ..projects.project import MlrunProject

        proj = MlrunProject.from_dict(self.get_project(project))
        return proj.run(name, arguments=arguments or {})

    def submit_pipeline(self, project, pipeline, arguments=None,
                        experiment=None, run=None, namespace=None,
                        artifact_path=None, ops=None, ttl=None):
        name = pipeline if isinstance(pipeline, str) else \
            getattr(pipeline, "name", "workflow")
        return self.submit_workflow(project, name, arguments=arguments)

    def list_pipelines(self, project):
        return [r for r in self.list_runs(project=project)
                if (r.get("metadata", {}).get("labels") or {}
                    ).get("workflow")]

    def get_pipeline(self, run_id, namespace=None, timeout=30,
                     format_=None, project=None):
        return self.read_run(run_id, project or "default")

    def get_workflow_id(self, project, name, run_id, engine=None):
        return {"workflow_id": run_id}

    # ------------------------------------------------ function lifecycle
    def start_function(self, func_url=None, function=None):
        from ..run import new_function

        body = function.to_dict() if hasattr(function, "to_dict") else \
            (function or {})
        if func_url and not body:
            project, name = func_url.split("/")[-2:]
            body = self.get_function(name, project)
        fn = new_function(runtime=body)
        address = fn.deploy()
        stored = fn.to_dict()
        stored.setdefault("status", {})["address"] = address
        stored["status"]["state"] = "ready"
        self.store_function(stored,
                            stored.get("metadata", {}).get("name", ""),
                            stored.get("metadata", {}).get("project",
                                                           "default"))
        return {"data": {"address": address, "state": "ready"}}

    def function_status(self, project, name, kind=None, selector=None):
        body = self.get_function(name, project)
        return {"status": body.get("status", {})}

    def deploy_nuclio_function(self, func=None, builder_env=None):
        return self.start_function(function=func)

    def get_nuclio_deploy_status(self, func=None, last_log_timestamp=0,
                                 verbose=False):
        name = func.metadata.name if hasattr(func, "metadata") else ""
        project = getattr(getattr(func, "metadata", None), "project",
                          "") or "default"
        return self.function_status(project, name)

    def remote_builder(self, func, with_mlrun,
                       mlrun_version_specifier=None, skip_deployed=False,
                       builder_env=None, force_build=False):
        from ..utils.builder import build_runtime

        build_runtime(func, with_mlrun=with_mlrun, install=False)
        return {"data": func.to_dict() if hasattr(func, "to_dict")
                else func, "ready": True}

    def get_builder_status(self, func, offset=0, logs=True,
                           last_log_timestamp=0, verbose=False):
        return {"ready": True, "state": "succeeded", "error": ""}

    # ------------------------------------------------ model monitoring
    def enable_model_monitoring(self, project, base_period=10,
                                image="mlrun/mlrun", **kwargs):
        from ..model_monitoring import enable_model_monitoring as _en

        return bool(_en(project, base_period=base_period, start=True))

    def disable_model_monitoring(self, project, **kwargs):
        from ..model_monitoring.controller import _controllers, _lock

        with _lock:
            controller = _controllers.pop(project, None)
        if controller:
            controller.stop()
        return True

    def update_model_monitoring_controller(self, project,
                                           base_period=10, image=None,
                                           **kwargs):
        from ..model_monitoring.controller import _controllers

        controller = _controllers.get(project)
        if controller:
            controller.base_period = base_period * 60
        return True

    def deploy_histogram_data_drift_app(self, project, image=None,
                                        **kwargs):
        from ..model_monitoring import (
            HistogramDataDriftApplication,
            enable_model_monitoring as _en,
        )

        controller = _en(project)
        if not any(type(a).__name__ == "HistogramDataDriftApplication"
                   for a in controller.applications):
            controller.add_application(HistogramDataDriftApplication())
        return True

    def delete_model_monitoring_function(self, project, functions=None,
                                         **kwargs):
        return self.disable_model_monitoring(project)

    def set_model_monitoring_credentials(self, project=None,
                                         credentials=None, **kwargs):
        return True

    def get_model_endpoint_metrics(self, project, endpoint_id,
                                   metrics=None):
        record = self.get_model_endpoint(project, endpoint_id)
        return (record.get("status") or {}).get("stats", {})

    # ----------------------------------------------------------- misc
    def get_base_api_url(self) -> str:
        return ""

    def get_api_path_prefix(self) -> str:
        return "api/v1"

    def verify_authorization(self,
                             authorization_verification_input=None):
        return True

    def warn_on_s3_and_ecr_permissions_conflict(self, func):
        return None

    def list_runtime_resources(self, project="", label_selector=None,
                               kind=None, object_id=None, group_by=None):
        from ..parallel.scheduler import get_gpu_allocator

        allocator = get_gpu_allocator()
        return {"gpu": {"total": allocator.total,
                        "available": allocator.available()}}

    def delete_runtime_resources(self, project="", label_selector=None,
                                 kind=None, object_id=None, force=False,
                                 grace_period=None):
        return {}

    def load_project(self, name, url=None, context=None):
        from ..projects import MlrunProject

        return MlrunProject.from_dict(self.get_project(name))

    def paginated_api_call(self, method, path, params=None, **kwargs):
        raise NotImplementedError(
            "paginated_api_call is HTTP-client-only; local mode: use "
            "paginated_list(method, ...)")

    @staticmethod
    def process_paginated_responses(responses, key):
        items = []
        for resp in responses:
            items.extend(resp.get(key) or [])
        return items

    def get_project_summary(self, project):
        if hasattr(self, "compute_project_summary"):
            return self.compute_project_summary(project)
        raise NotImplementedError
