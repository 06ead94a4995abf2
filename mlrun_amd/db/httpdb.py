# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""HTTP run-DB client speaking to the mlrun_amd API service.

Parity target: reference mlrun/db/httpdb.py:78 HTTPRunDB (~120 REST
methods).  Routes live under /api/v1/ and are served by
mlrun_amd/api/main.py (FastAPI over the same SQLRunDB).
"""

import typing

import requests

from ..errors import err_for_status
from .base import RunDBInterface


class HTTPRunDB(RunDBInterface):
    kind = "http"

    def __init__(self, url: str, token: str = ""):
        self.base_url = url.rstrip("/")
        self.token = token
        self.session = requests.Session()
        self.server_version = ""

    def __repr__(self):
        return f"HTTPRunDB({self.base_url})"

    def api_call(self, method: str, path: str, params: dict = None,
                 body: typing.Any = None, json_body: dict = None,
                 raw: bool = False, timeout: int = 45,
                 retries: int = None):
        """One REST call; idempotent methods retry on transport errors
        with exponential backoff when
        ``httpdb.retry_api_call_on_exception`` is enabled (reference
        httpdb.api_call retry semantics)."""
        import time as _time

        from ..config import config

        url = f"{self.base_url}/api/v1/{path.lstrip('/')}"
        headers = {}
        if self.token:
            headers["Authorization"] = f"Bearer {self.token}"
        if retries is None:
            retry_on = str(config.httpdb.retry_api_call_on_exception
                           ) == "enabled"
            retries = 2 if retry_on and method in ("GET", "HEAD") else 0
        attempt = 0
        while True:
            try:
                resp = self.session.request(
                    method, url, params=params, data=body,
                    json=json_body, headers=headers, timeout=timeout)
                break
            except (requests.ConnectionError, requests.Timeout):
                if attempt >= retries:
                    raise
                _time.sleep(0.2 * (2 ** attempt))
                attempt += 1
        if resp.status_code >= 400:
            try:
                detail = resp.json().get("detail", resp.text)
            except ValueError:
                detail = resp.text
            raise err_for_status(resp.status_code, f"{method} {path}: {detail}")
        if raw:
            return resp
        if resp.content:
            return resp.json()
        return {}

    def connect(self, secrets=None):
        try:
            info = self.api_call("GET", "healthz")
            self.server_version = info.get("version", "")
        except Exception:
            pass
        return self

    # ------------------------------------------------------------- runs
    def store_run(self, struct, uid, project="", iter=0):
        if hasattr(struct, "to_dict"):
            struct = struct.to_dict()
        project = project or "default"
        self.api_call("POST", f"run/{project}/{uid}", params={"iter": iter},
                      json_body=struct)
        return struct

    def update_run(self, updates, uid, project="", iter=0):
        project = project or "default"
        return self.api_call("PATCH", f"run/{project}/{uid}",
                             params={"iter": iter}, json_body=updates)

    def read_run(self, uid, project="", iter=0):
        project = project or "default"
        return self.api_call("GET", f"run/{project}/{uid}",
                             params={"iter": iter}).get("data")

    def list_runs(self, name="", uid=None, project="", labels=None, state=None,
                  sort=True, last=0, iter=False, start_time_from=None,
                  start_time_to=None):
        params = {"name": name, "project": project or "default",
                  "state": state or "", "sort": int(sort), "last": last,
                  "iter": int(iter)}
        if uid:
            params["uid"] = uid
        if labels:
            params["label"] = labels if isinstance(labels, list) else [
                f"{k}={v}" for k, v in labels.items()]
        from ..lists import RunList

        return RunList(self.api_call("GET", "runs",
                                     params=params).get("runs", []))

    def del_run(self, uid, project="", iter=0):
        self.api_call("DELETE", f"run/{project or 'default'}/{uid}",
                      params={"iter": iter})

    def abort_run(self, uid, project="", iter=0, status_text=""):
        self.api_call("POST", f"run/{project or 'default'}/{uid}/abort",
                      json_body={"status_text": status_text})

    # ------------------------------------------------------------- logs
    def store_log(self, uid, project="", body=None, append=False):
        if isinstance(body, str):
            body = body.encode()
        self.api_call("POST", f"log/{project or 'default'}/{uid}",
                      params={"append": int(append)}, body=body)

    def get_log(self, uid, project="", offset=0, size=0):
        resp = self.api_call("GET", f"log/{project or 'default'}/{uid}",
                             params={"offset": offset, "size": size}, raw=True)
        state = resp.headers.get("x-mlrun-run-state", "")
        return state, resp.content

    # -------------------------------------------------------- artifacts
    def store_artifact(self, key, artifact, uid=None, iter=None, tag="",
                       project="", tree=None):
        if hasattr(artifact, "to_dict"):
            artifact = artifact.to_dict()
        return self.api_call(
            "POST", f"artifact/{project or 'default'}/{key}",
            params={"tag": tag, "iter": iter or 0, "tree": tree or "",
                    "uid": uid or ""},
            json_body=artifact)

    def read_artifact(self, key, tag="", iter=None, project="", tree=None,
                      uid=None):
        return self.api_call(
            "GET", f"artifact/{project or 'default'}/{key}",
            params={"tag": tag, "iter": iter or 0, "tree": tree or "",
                    "uid": uid or ""}).get("data")

    def list_artifacts(self, name="", project="", tag="", labels=None,
                       since=None, until=None, kind=None, category=None,
                       iter=None, tree=None):
        params = {"name": name, "project": project or "default", "tag": tag,
                  "kind": kind or "", "tree": tree or ""}
        if labels:
            params["label"] = labels if isinstance(labels, list) else [
                f"{k}={v}" for k, v in labels.items()]
        from ..lists import ArtifactList

        result = ArtifactList(self.api_call(
            "GET", "artifacts", params=params).get("artifacts", []))
        result.tag = tag
        return result

    def del_artifact(self, key, tag="", project="", uid=None, tree=None):
        self.api_call("DELETE", f"artifact/{project or 'default'}/{key}",
                      params={"tag": tag})

    # -------------------------------------------------------- functions
    def store_function(self, function, name, project="", tag="",
                       versioned=False):
        if hasattr(function, "to_dict"):
            function = function.to_dict()
        resp = self.api_call(
            "POST", f"func/{project or 'default'}/{name}",
            params={"tag": tag, "versioned": int(versioned)},
            json_body=function)
        return resp.get("hash_key", "")

    def get_function(self, name, project="", tag="", hash_key=""):
        return self.api_call(
            "GET", f"func/{project or 'default'}/{name}",
            params={"tag": tag, "hash_key": hash_key}).get("func")

    def list_functions(self, name=None, project="", tag="", labels=None):
        params = {"project": project or "default", "tag": tag}
        if name:
            params["name"] = name
        if labels:
            params["label"] = labels if isinstance(labels, list) else [
                f"{k}={v}" for k, v in labels.items()]
        return self.api_call("GET", "funcs", params=params).get("funcs", [])

    def delete_function(self, name, project=""):
        self.api_call("DELETE", f"func/{project or 'default'}/{name}")

    # --------------------------------------------------------- projects
    def create_project(self, project):
        if hasattr(project, "to_dict"):
            project = project.to_dict()
        if isinstance(project, str):
            project = {"metadata": {"name": project}}
        return self.api_call("POST", "projects", json_body=project)

    def get_project(self, name):
        return self.api_call("GET", f"projects/{name}")

    def list_projects(self, owner=None, format_=None, labels=None, state=None):
        return self.api_call("GET", "projects").get("projects", [])

    def store_project(self, name, project):
        if hasattr(project, "to_dict"):
            project = project.to_dict()
        return self.api_call("PUT", f"projects/{name}", json_body=project)

    def delete_project(self, name, deletion_strategy=None):
        self.api_call("DELETE", f"projects/{name}",
                      params={"deletion_strategy": deletion_strategy or ""})

    # -------------------------------------------------------- schedules
    def create_schedule(self, project, schedule):
        self.api_call("POST", f"projects/{project}/schedules",
                      json_body=schedule)

    def update_schedule(self, project, name, schedule):
        self.api_call("PUT", f"projects/{project}/schedules/{name}",
                      json_body=schedule)

    def get_schedule(self, project, name):
        return self.api_call("GET", f"projects/{project}/schedules/{name}")

    def list_schedules(self, project, name=""):
        return self.api_call("GET", f"projects/{project}/schedules",
                             params={"name": name}).get("schedules", [])

    def delete_schedule(self, project, name):
        self.api_call("DELETE", f"projects/{project}/schedules/{name}")

    def invoke_schedule(self, project, name):
        self.api_call("POST", f"projects/{project}/schedules/{name}/invoke")

    # ----------------------------------------------------- feature store
    def store_feature_set(self, feature_set, name=None, project="", tag=None,
                          versioned=False):
        if hasattr(feature_set, "to_dict"):
            feature_set = feature_set.to_dict()
        name = name or feature_set.get("metadata", {}).get("name")
        return self.api_call(
            "PUT", f"projects/{project or 'default'}/feature-sets/{name}",
            params={"tag": tag or ""}, json_body=feature_set)

    def get_feature_set(self, name, project="", tag=None):
        return self.api_call(
            "GET", f"projects/{project or 'default'}/feature-sets/{name}",
            params={"tag": tag or ""})

    def list_feature_sets(self, project="", name=None, tag=None, labels=None):
        return self.api_call(
            "GET", f"projects/{project or 'default'}/feature-sets",
            params={"name": name or ""}).get("feature_sets", [])

    def delete_feature_set(self, name, project="", tag=None):
        self.api_call(
            "DELETE", f"projects/{project or 'default'}/feature-sets/{name}")

    def store_feature_vector(self, feature_vector, name=None, project="",
                             tag=None, versioned=False):
        if hasattr(feature_vector, "to_dict"):
            feature_vector = feature_vector.to_dict()
        name = name or feature_vector.get("metadata", {}).get("name")
        return self.api_call(
            "PUT", f"projects/{project or 'default'}/feature-vectors/{name}",
            params={"tag": tag or ""}, json_body=feature_vector)

    def get_feature_vector(self, name, project="", tag=None):
        return self.api_call(
            "GET", f"projects/{project or 'default'}/feature-vectors/{name}",
            params={"tag": tag or ""})

    def list_feature_vectors(self, project="", name=None, tag=None,
                             labels=None):
        return self.api_call(
            "GET", f"projects/{project or 'default'}/feature-vectors",
            params={"name": name or ""}).get("feature_vectors", [])

    def delete_feature_vector(self, name, project="", tag=None):
        self.api_call(
            "DELETE",
            f"projects/{project or 'default'}/feature-vectors/{name}")

    # --------------------------------------------------- model endpoints
    def store_model_endpoint(self, project, endpoint_id, endpoint):
        if hasattr(endpoint, "to_dict"):
            endpoint = endpoint.to_dict()
        self.api_call(
            "PUT", f"projects/{project}/model-endpoints/{endpoint_id}",
            json_body=endpoint)

    def get_model_endpoint(self, project, endpoint_id):
        return self.api_call(
            "GET", f"projects/{project}/model-endpoints/{endpoint_id}")

    def list_model_endpoints(self, project, model=None, function=None,
                             labels=None):
        return self.api_call(
            "GET", f"projects/{project}/model-endpoints",
            params={"model": model or "", "function": function or ""}).get(
            "endpoints", [])

    def delete_model_endpoint(self, project, endpoint_id):
        self.api_call(
            "DELETE", f"projects/{project}/model-endpoints/{endpoint_id}")

    def get_model_endpoint_metrics(self, project, endpoint_id):
        return self.api_call(
            "GET",
            f"projects/{project}/model-endpoints/{endpoint_id}/metrics")

    # ------------------------------------------------------------ alerts
    def store_alert_config(self, project, name, alert):
        if hasattr(alert, "to_dict"):
            alert = alert.to_dict()
        self.api_call("PUT", f"projects/{project}/alerts/{name}",
                      json_body=alert)

    def get_alert_config(self, project, name):
        return self.api_call("GET", f"projects/{project}/alerts/{name}")

    def list_alert_configs(self, project):
        return self.api_call("GET", f"projects/{project}/alerts").get(
            "alerts", [])

    def delete_alert_config(self, project, name):
        self.api_call("DELETE", f"projects/{project}/alerts/{name}")

    def generate_event(self, project, name, event):
        self.api_call("POST", f"projects/{project}/events/{name}",
                      json_body=event)

    # ----------------------------------------------------------- submit
    def submit_job(self, runspec, schedule=None):
        body = {"task": runspec.to_dict() if hasattr(runspec, "to_dict")
                else runspec}
        if schedule:
            body["schedule"] = schedule
        resp = self.api_call("POST", "submit_job", json_body=body, timeout=120)
        return resp.get("data", resp)

    # -------------------------------------------------- background tasks
    # --------------------------------------------------- new surfaces
    def submit_workflow(self, project, name, arguments=None):
        return self.api_call(
            "POST", f"projects/{project}/workflows/{name}/submit",
            json_body={"arguments": arguments or {}})

    def list_workflows(self, project):
        return self.api_call(
            "GET", f"projects/{project}/workflows").get("workflows", [])

    def store_project_secrets(self, project, secrets: dict,
                              provider="kubernetes"):
        self.api_call("POST", f"projects/{project}/secrets",
                      json_body={"secrets": secrets})

    def list_project_secret_keys(self, project, provider="kubernetes"):
        return self.api_call(
            "GET", f"projects/{project}/secret-keys"
        ).get("secret_keys", [])

    def delete_project_secrets(self, project, keys=None,
                               provider="kubernetes"):
        self.api_call("DELETE", f"projects/{project}/secrets",
                      params={"secrets": ",".join(keys or [])})

    def get_hub_catalog(self, source="builtin"):
        return self.api_call(
            "GET", f"hub/sources/{source}/items").get("catalog", [])

    def list_pipelines(self, project):
        return self.api_call(
            "GET", f"projects/{project}/pipelines").get("runs", [])

    def tag_artifact(self, project, key, tree, tag, iteration=0):
        self.api_call("PUT", f"projects/{project}/tags/{tag}",
                      json_body={"identifiers": [
                          {"key": key, "tree": tree,
                           "iter": iteration}]})

    def list_artifact_tags(self, project, key=""):
        return self.api_call(
            "GET", f"projects/{project}/tags",
            params={"key": key}).get("tags", [])

    def get_file(self, path, size=0, offset=0):
        return self.api_call("GET", "files",
                             params={"path": path, "size": size,
                                     "offset": offset}, raw=True)

    def store_datastore_profile(self, project, profile: dict):
        self.api_call("PUT", f"projects/{project}/datastore-profiles",
                      json_body=profile)

    def list_datastore_profiles(self, project):
        return self.api_call(
            "GET", f"projects/{project}/datastore-profiles"
        ).get("profiles", [])

    def store_alert_template(self, name, template: dict):
        self.api_call("PUT", f"alert-templates/{name}",
                      json_body=template)

    def list_alert_templates(self):
        return self.api_call("GET", "alert-templates"
                             ).get("templates", [])

    def get_background_task(self, project, name):
        return self.api_call(
            "GET", f"projects/{project}/background-tasks/{name}")

    def list_background_tasks(self, project):
        return self.api_call(
            "GET", f"projects/{project}/background-tasks").get(
            "background_tasks", [])

    # ------------------------------------------------------------------
    # Round-2 surface: the remaining reference HTTPRunDB method families
    # (reference mlrun/db/httpdb.py:78; per-method status in
    # docs/PARITY.md §run-DB).  Methods replacing k8s/nuclio behavior
    # carry the node-local semantics in their docstrings.
    # ------------------------------------------------------------------

    # ------------------------------------------------------- operations
    def trigger_migrations(self):
        """Apply pending schema migrations (reference
        trigger_migrations -> /operations/migrations)."""
        return self.api_call("POST", "operations/migrations")

    def get_base_api_url(self) -> str:
        return self.base_url

    def get_api_path_prefix(self) -> str:
        return "api/v1"

    def verify_authorization(self, authorization_verification_input=None):
        """Single-user node-local service: authorization always
        passes (reference checks OPA policies)."""
        return True

    # ------------------------------------------------------------- runs
    def del_runs(self, name="", project="", labels=None, state=None,
                 days_ago=0):
        self.api_call("DELETE", "runs",
                      params={"project": project, "name": name,
                              "state": state or ""})

    def set_run_notifications(self, project, run_uid, notifications=None):
        self.api_call(
            "PUT", f"projects/{project}/runs/{run_uid}/notifications",
            json_body={"notifications": [
                n.to_dict() if hasattr(n, "to_dict") else n
                for n in notifications or []]})

    store_run_notifications = set_run_notifications

    def set_schedule_notifications(self, project, schedule_name,
                                   notifications=None):
        self.api_call(
            "PUT",
            f"projects/{project}/schedules/{schedule_name}/notifications",
            json_body={"notifications": [
                n.to_dict() if hasattr(n, "to_dict") else n
                for n in notifications or []]})

    def watch_log(self, uid, project="", watch=True, offset=0):
        """Poll a run's log until the run reaches a terminal state,
        printing increments (reference watch_log)."""
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

    def get_log_size(self, uid, project=""):
        _, log = self.get_log(uid, project)
        return len(log or b"")

    # -------------------------------------------------------- artifacts
    def del_artifacts(self, name="", project="", tag="", labels=None,
                      days_ago=0):
        self.api_call("DELETE", "artifacts",
                      params={"project": project, "name": name,
                              "tag": tag})

    def tag_artifacts(self, artifacts, project, tag_name, replace=False):
        identifiers = []
        for artifact in artifacts if isinstance(artifacts, list) \
                else [artifacts]:
            meta = artifact.get("metadata", artifact) if isinstance(
                artifact, dict) else getattr(artifact, "metadata", {})
            get = meta.get if isinstance(meta, dict) else \
                lambda k, d=None: getattr(meta, k, d)
            identifiers.append({"key": get("key"),
                                "tree": get("tree") or get("uid"),
                                "iter": get("iter", 0) or 0})
        self.api_call("PUT", f"projects/{project}/tags/{tag_name}",
                      json_body={"identifiers": identifiers})

    tag_objects = tag_artifacts

    def delete_artifacts_tags(self, artifacts, project, tag_name):
        for artifact in artifacts if isinstance(artifacts, list) \
                else [artifacts]:
            meta = artifact.get("metadata", artifact) if isinstance(
                artifact, dict) else {}
            self.api_call("DELETE",
                          f"projects/{project}/tags/{tag_name}",
                          params={"key": meta.get("key")})

    delete_objects_tag = delete_artifacts_tags

    # -------------------------------------------------------- projects
    def patch_project(self, name, project: dict,
                      patch_mode="replace"):
        return self.api_call("PATCH", f"projects/{name}",
                             json_body=project)

    def load_project(self, name, url=None, context=None):
        """Fetch a stored project and return a live MlrunProject."""
        from ..projects import MlrunProject

        body = self.get_project(name)
        return MlrunProject.from_dict(body)

    # --------------------------------------------------- feature store
    def create_feature_set(self, feature_set, project="", versioned=True):
        body = feature_set.to_dict() if hasattr(feature_set, "to_dict") \
            else feature_set
        name = body.get("metadata", {}).get("name", "")
        return self.store_feature_set(body, name=name, project=project)

    def patch_feature_set(self, name, feature_set_update: dict,
                          project="", tag="latest",
                          patch_mode="additive"):
        return self.api_call(
            "PATCH",
            f"projects/{project or 'default'}/feature-sets/{name}",
            params={"tag": tag}, json_body=feature_set_update)

    def create_feature_vector(self, feature_vector, project="",
                              versioned=True):
        body = feature_vector.to_dict() if hasattr(
            feature_vector, "to_dict") else feature_vector
        name = body.get("metadata", {}).get("name", "")
        return self.store_feature_vector(body, name=name, project=project)

    def patch_feature_vector(self, name, feature_vector_update: dict,
                             project="", tag="latest",
                             patch_mode="additive"):
        return self.api_call(
            "PATCH",
            f"projects/{project or 'default'}/feature-vectors/{name}",
            params={"tag": tag}, json_body=feature_vector_update)

    def list_features(self, project, name=None, tag=None, entities=None,
                      labels=None):
        params = {"name": name or ""}
        if entities:
            params["entity"] = entities
        if labels:
            params["label"] = labels
        return self.api_call(
            "GET", f"projects/{project}/features",
            params=params).get("features", [])

    list_features_v2 = list_features

    def list_entities(self, project, name=None, tag=None, labels=None):
        params = {"name": name or ""}
        if labels:
            params["label"] = labels
        return self.api_call(
            "GET", f"projects/{project}/entities",
            params=params).get("entities", [])

    list_entities_v2 = list_entities

    # -------------------------------------------------- model endpoints
    def create_model_endpoint(self, project, endpoint_id, model_endpoint):
        body = model_endpoint.to_dict() if hasattr(
            model_endpoint, "to_dict") else model_endpoint
        return self.store_model_endpoint(project, endpoint_id, body)

    def patch_model_endpoint(self, project, endpoint_id,
                             attributes: dict):
        return self.api_call(
            "PATCH",
            f"projects/{project}/model-endpoints/{endpoint_id}",
            json_body=attributes)

    # ------------------------------------------------------------ alerts
    def list_alerts_configs(self, project=""):
        return self.list_alert_configs(project)

    def reset_alert_config(self, project, name):
        self.api_call("POST", f"projects/{project}/alerts/{name}/reset")

    def get_alert_template(self, name):
        return self.api_call("GET", f"alert-templates/{name}")

    def store_alert_notifications(self, session, notification_objects,
                                  alert_id, project):
        """Notifications ride on the alert config body node-locally."""
        config_ = self.get_alert_config(project, alert_id)
        config_["notifications"] = [
            n.to_dict() if hasattr(n, "to_dict") else n
            for n in notification_objects or []]
        self.store_alert_config(project, alert_id, config_)

    # -------------------------------------------------------------- hub
    def list_hub_sources(self, item_name=None, tag=None, version=None):
        return self.api_call("GET", "hub/sources").get("sources", [])

    def get_hub_source(self, source_name):
        return self.api_call("GET", f"hub/sources/{source_name}")

    def store_hub_source(self, source_name, source: dict):
        return self.api_call("PUT", f"hub/sources/{source_name}",
                             json_body=source if isinstance(source, dict)
                             else source.to_dict())

    create_hub_source = store_hub_source

    def delete_hub_source(self, source_name):
        self.api_call("DELETE", f"hub/sources/{source_name}")

    def get_hub_item(self, source_name, item_name, version=None,
                     tag="latest", force_refresh=False):
        return self.api_call(
            "GET", f"hub/sources/{source_name}/items/{item_name}")

    def get_hub_asset(self, source_name, item_name, asset_name,
                      version=None, tag="latest"):
        """Fetch an item asset (source file) through /files."""
        item = self.get_hub_item(source_name, item_name)
        spec = item.get("spec", item)
        path = spec.get("item_uri") or spec.get("filename") or ""
        return self.get_file(path).content

    # ---------------------------------------------------------- secrets
    def create_project_secrets(self, project, provider="kubernetes",
                               secrets=None):
        return self.store_project_secrets(project, secrets or {},
                                          provider)

    def list_project_secrets(self, project, token=None,
                             provider="kubernetes", secrets=None):
        """Node-local single-user service: returns key list only (the
        reference returns values under session auth; values stay
        server-side here)."""
        return {"secrets": {k: None for k in
                            self.list_project_secret_keys(project)}}

    def create_user_secrets(self, user, provider="vault", secrets=None):
        """Vault user secrets are not part of the node-local design
        (reference vault integration); use project secrets."""
        raise NotImplementedError(
            "user (vault) secrets are not supported; use project "
            "secrets (store_project_secrets)")

    # ------------------------------------------- background tasks (alias)
    def get_project_background_task(self, project, name):
        return self.get_background_task(project, name)

    def list_project_background_tasks(self, project):
        return self.list_background_tasks(project)

    # ------------------------------------------------ runtime resources
    def list_runtime_resources(self, project="", label_selector=None,
                               kind=None, object_id=None,
                               group_by=None):
        return self.api_call(
            "GET", f"projects/{project or 'default'}/runtime-resources")

    def delete_runtime_resources(self, project="", label_selector=None,
                                 kind=None, object_id=None, force=False,
                                 grace_period=None):
        """Node-local runtimes are OS processes reaped by the runs
        monitor; nothing to delete beyond aborting runs."""
        return {}

    # -------------------------------------------- functions lifecycle
    def remote_builder(self, func, with_mlrun, mlrun_version_specifier=None,
                       skip_deployed=False, builder_env=None,
                       force_build=False):
        """Server-side venv/wheel build (reference: kaniko image build;
        node-local: utils/builder in a background task)."""
        return self.api_call(
            "POST", "build/function",
            json_body={"function": func.to_dict() if hasattr(
                func, "to_dict") else func,
                "with_mlrun": bool(with_mlrun)})

    def get_builder_status(self, func, offset=0, logs=True,
                           last_log_timestamp=0, verbose=False):
        name = func.metadata.name if hasattr(func, "metadata") else \
            (func.get("metadata", {}) or {}).get("name", "")
        project = func.metadata.project if hasattr(func, "metadata") \
            else (func.get("metadata", {}) or {}).get("project",
                                                      "default")
        return self.api_call("GET", "build/status",
                             params={"project": project or "default",
                                     "name": name})

    def start_function(self, func_url=None, function=None):
        """Deploy a stored function as a live local host (node-local
        analog of the reference's start_function for dask — here it
        starts serving/remote hosts)."""
        body = function.to_dict() if hasattr(function, "to_dict") else \
            (function or {})
        if func_url and not body:
            project, name = func_url.split("/")[-2:]
            body = self.get_function(name, project)
        return self.api_call("POST", "start/function",
                             json_body={"function": body})

    def function_status(self, project, name, kind=None, selector=None):
        return self.api_call(
            "GET", f"projects/{project}/functions/{name}/status")

    def deploy_nuclio_function(self, func=None, builder_env=None):
        """Node-local replacement: "nuclio deploy" = start a local
        serving host for the function (reference deploys via the
        nuclio dashboard)."""
        return self.start_function(function=func)

    def get_nuclio_deploy_status(self, func=None, last_log_timestamp=0,
                                 verbose=False):
        name = func.metadata.name if hasattr(func, "metadata") else ""
        project = getattr(getattr(func, "metadata", None), "project",
                          "") or "default"
        return self.function_status(project, name)

    # --------------------------------------------------- api gateways
    def store_api_gateway(self, api_gateway, project=""):
        body = api_gateway.to_dict() if hasattr(api_gateway, "to_dict") \
            else api_gateway
        name = body.get("metadata", body).get("name", "") or \
            body.get("name", "")
        return self.api_call(
            "PUT", f"projects/{project or 'default'}/api-gateways/{name}",
            json_body=body)

    def get_api_gateway(self, name, project=""):
        return self.api_call(
            "GET", f"projects/{project or 'default'}/api-gateways/{name}")

    def list_api_gateways(self, project=""):
        return self.api_call(
            "GET", f"projects/{project or 'default'}/api-gateways"
        ).get("api_gateways", [])

    def delete_api_gateway(self, name, project=""):
        self.api_call(
            "DELETE",
            f"projects/{project or 'default'}/api-gateways/{name}")

    # ------------------------------------------------------- pipelines
    def submit_pipeline(self, project, pipeline, arguments=None,
                        experiment=None, run=None, namespace=None,
                        artifact_path=None, ops=None, ttl=None):
        """Local workflow runner analog of KFP submit."""
        name = pipeline if isinstance(pipeline, str) else \
            getattr(pipeline, "name", "workflow")
        return self.submit_workflow(project, name,
                                    arguments=arguments)

    def get_pipeline(self, run_id, namespace=None, timeout=30,
                     format_=None, project=None):
        return self.api_call(
            "GET", f"projects/{project or 'default'}/pipelines/{run_id}")

    def get_workflow_id(self, project, name, run_id, engine=None):
        return {"workflow_id": run_id}

    # ------------------------------------------------ model monitoring
    def enable_model_monitoring(self, project, base_period=10,
                                image="mlrun/mlrun", **kwargs):
        """Start the node-local monitoring controller for a project
        (reference deploys controller + writer nuclio functions)."""
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
        from ..model_monitoring import (HistogramDataDriftApplication,
                                        enable_model_monitoring as _en)

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
        """Node-local stores need no credentials (reference wires v3io/
        TSDB secrets)."""
        return True

    # ------------------------------------------------------ pagination
    def paginated_api_call(self, method, path, params=None, **kwargs):
        """Generator over token-paginated list endpoints (reference
        paginated_api_call)."""
        params = dict(params or {})
        params.setdefault("page_size", 200)
        while True:
            resp = self.api_call(method, path, params=params, **kwargs)
            yield resp
            token = (resp.get("pagination") or {}).get("page_token")
            if not token:
                return
            params = {"page_token": token}

    @staticmethod
    def process_paginated_responses(responses, key):
        items = []
        for resp in responses:
            items.extend(resp.get(key) or [])
        return items

    # --------------------------------------------------------- k8s-only
    def warn_on_s3_and_ecr_permissions_conflict(self, func):
        """N/A node-locally (reference checks AWS ECR pull secrets)."""
        return None

    def get_datastore_profile(self, name, project=""):
        return self.api_call(
            "GET",
            f"projects/{project or 'default'}/datastore-profiles/{name}")

    def delete_datastore_profile(self, name, project=""):
        self.api_call(
            "DELETE",
            f"projects/{project or 'default'}/datastore-profiles/{name}")

    def get_project_summary(self, project):
        return self.api_call("GET", f"project-summaries/{project}")

    def list_project_summaries(self):
        return self.api_call("GET", "project-summaries").get(
            "project_summaries", [])
