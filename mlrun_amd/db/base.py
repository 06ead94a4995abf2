# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Run-DB interface (parity target: reference mlrun/db/base.py:33
RunDBInterface).  All backends (SQLite-local, HTTP, Nop) implement this."""

import abc


class RunDBInterface(abc.ABC):
    kind = ""

    def connect(self, secrets=None):
        return self

    # --- runs ---
    @abc.abstractmethod
    def store_run(self, struct: dict, uid: str, project: str = "", iter: int = 0):
        ...

    @abc.abstractmethod
    def update_run(self, updates: dict, uid: str, project: str = "", iter: int = 0):
        ...

    @abc.abstractmethod
    def read_run(self, uid: str, project: str = "", iter: int = 0) -> dict:
        ...

    @abc.abstractmethod
    def list_runs(self, name="", uid=None, project="", labels=None, state=None,
                  sort=True, last=0, iter=False, start_time_from=None,
                  start_time_to=None) -> list:
        ...

    @abc.abstractmethod
    def del_run(self, uid: str, project: str = "", iter: int = 0):
        ...

    def del_runs(self, name="", project="", labels=None, state=None, days_ago=0):
        pass

    def abort_run(self, uid: str, project: str = "", iter: int = 0,
                  status_text: str = ""):
        pass

    # --- logs ---
    def store_log(self, uid: str, project: str = "", body: bytes = None,
                  append: bool = False):
        pass

    def get_log(self, uid: str, project: str = "", offset: int = 0,
                size: int = 0):
        return "", b""

    def watch_log(self, uid: str, project: str = "", watch: bool = True,
                  offset: int = 0):
        state, text = self.get_log(uid, project, offset=offset)
        if text:
            print(text.decode(errors="replace"))
        return state

    # --- artifacts ---
    @abc.abstractmethod
    def store_artifact(self, key: str, artifact: dict, uid=None, iter=None,
                       tag="", project="", tree=None):
        ...

    @abc.abstractmethod
    def read_artifact(self, key: str, tag="", iter=None, project="",
                      tree=None, uid=None) -> dict:
        ...

    @abc.abstractmethod
    def list_artifacts(self, name="", project="", tag="", labels=None,
                       since=None, until=None, kind=None, category=None,
                       iter=None, tree=None) -> list:
        ...

    @abc.abstractmethod
    def del_artifact(self, key: str, tag="", project="", uid=None, tree=None):
        ...

    def del_artifacts(self, name="", project="", tag="", labels=None):
        pass

    # --- functions ---
    @abc.abstractmethod
    def store_function(self, function: dict, name: str, project: str = "",
                       tag: str = "", versioned: bool = False) -> str:
        ...

    @abc.abstractmethod
    def get_function(self, name: str, project: str = "", tag: str = "",
                     hash_key: str = "") -> dict:
        ...

    @abc.abstractmethod
    def list_functions(self, name=None, project="", tag="", labels=None) -> list:
        ...

    def delete_function(self, name: str, project: str = ""):
        pass

    # --- projects ---
    def create_project(self, project) -> dict:
        raise NotImplementedError

    def get_project(self, name: str) -> dict:
        raise NotImplementedError

    def list_projects(self, owner=None, format_=None, labels=None, state=None) -> list:
        return []

    def delete_project(self, name: str, deletion_strategy=None):
        pass

    def store_project(self, name: str, project) -> dict:
        raise NotImplementedError

    # --- schedules ---
    def create_schedule(self, project: str, schedule: dict):
        pass

    def update_schedule(self, project: str, name: str, schedule: dict):
        pass

    def get_schedule(self, project: str, name: str) -> dict:
        raise NotImplementedError

    def list_schedules(self, project: str, name: str = "") -> list:
        return []

    def delete_schedule(self, project: str, name: str):
        pass

    def invoke_schedule(self, project: str, name: str):
        pass

    # --- feature store ---
    def store_feature_set(self, feature_set: dict, name=None, project="",
                          tag=None, versioned=False):
        pass

    def get_feature_set(self, name: str, project: str = "", tag: str = None) -> dict:
        raise NotImplementedError

    def list_feature_sets(self, project="", name=None, tag=None, labels=None) -> list:
        return []

    def delete_feature_set(self, name: str, project: str = "", tag=None):
        pass

    def store_feature_vector(self, feature_vector: dict, name=None, project="",
                             tag=None, versioned=False):
        pass

    def get_feature_vector(self, name: str, project: str = "", tag: str = None) -> dict:
        raise NotImplementedError

    def list_feature_vectors(self, project="", name=None, tag=None, labels=None) -> list:
        return []

    def delete_feature_vector(self, name: str, project: str = "", tag=None):
        pass

    # --- model endpoints (monitoring) ---
    def store_model_endpoint(self, project: str, endpoint_id: str, endpoint: dict):
        pass

    def get_model_endpoint(self, project: str, endpoint_id: str) -> dict:
        raise NotImplementedError

    def list_model_endpoints(self, project: str, model: str = None,
                             function: str = None, labels=None) -> list:
        return []

    def delete_model_endpoint(self, project: str, endpoint_id: str):
        pass

    # --- background tasks ---
    def store_background_task(self, project: str, task: dict):
        pass

    def get_background_task(self, project: str, name: str) -> dict:
        raise NotImplementedError

    def list_background_tasks(self, project: str) -> list:
        return []

    # --- misc ---
    def submit_job(self, runspec, schedule=None):
        raise NotImplementedError

    def remote_builder(self, func, with_mlrun=True):
        raise NotImplementedError

    def api_call(self, method, path, **kwargs):
        raise NotImplementedError

    def get_builder_status(self, func, offset=0, logs=True):
        raise NotImplementedError

    def deploy_function(self, func, **kwargs):
        raise NotImplementedError

    def store_alert_config(self, project, name, alert: dict):
        pass

    def get_alert_config(self, project, name) -> dict:
        raise NotImplementedError

    def list_alert_configs(self, project) -> list:
        return []

    def delete_alert_config(self, project, name):
        pass

    def generate_event(self, project, name, event: dict):
        pass
