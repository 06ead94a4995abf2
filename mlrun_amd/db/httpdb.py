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
                 raw: bool = False, timeout: int = 45):
        url = f"{self.base_url}/api/v1/{path.lstrip('/')}"
        headers = {}
        if self.token:
            headers["Authorization"] = f"Bearer {self.token}"
        resp = self.session.request(
            method, url, params=params, data=body, json=json_body,
            headers=headers, timeout=timeout)
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
        return self.api_call("GET", "runs", params=params).get("runs", [])

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
        return self.api_call("GET", "artifacts", params=params).get(
            "artifacts", [])

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
