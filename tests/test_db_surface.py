# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Round-2 HTTPRunDB surface tests: the method families added to close
the client gap vs reference httpdb.py."""

import pytest

from mlrun_amd.db.httpdb import HTTPRunDB
from mlrun_amd.db.sqldb import SQLRunDB


@pytest.fixture()
def client(tmp_path):
    from fastapi.testclient import TestClient

    from mlrun_amd.api.main import create_app

    db = SQLRunDB(str(tmp_path / "api.db"))
    app = create_app(db, with_scheduler=False)
    with TestClient(app) as c:
        c.db = db
        yield c


@pytest.fixture()
def httpdb(client):
    db = HTTPRunDB("http://testserver")
    orig_request = client.request

    def request(method, url, params=None, data=None, json=None,
                headers=None, timeout=None):
        return orig_request(method, url, params=params, content=data,
                            json=json, headers=headers)

    db.session = client
    db.session.request = request
    return db


class TestOperations:
    def test_trigger_migrations(self, httpdb):
        from mlrun_amd.db.sqldb import SCHEMA_VERSION

        result = httpdb.trigger_migrations()
        assert result["schema_version"] == SCHEMA_VERSION

    def test_url_helpers(self, httpdb):
        assert httpdb.get_api_path_prefix() == "api/v1"
        assert httpdb.verify_authorization() is True


class TestRunsSurface:
    def test_del_runs_and_notifications(self, httpdb):
        httpdb.store_run({"metadata": {"name": "r1", "uid": "u1"},
                          "status": {"state": "completed"}}, "u1", "p")
        httpdb.set_run_notifications("p", "u1", [{"kind": "console"}])
        run = httpdb.read_run("u1", "p")
        assert run["spec"]["notifications"] == [{"kind": "console"}]
        httpdb.del_runs(project="p")
        assert httpdb.list_runs(project="p") == []

    def test_runs_partition_over_http(self, httpdb):
        for i in range(4):
            for name in ("a", "b"):
                httpdb.store_run(
                    {"metadata": {"name": name, "uid": f"{name}{i}"},
                     "status": {"state": "completed"}},
                    f"{name}{i}", "p")
        runs = httpdb.api_call(
            "GET", "runs", params={"project": "p",
                                   "partition_by": "name",
                                   "rows_per_partition": 1})["runs"]
        assert len(runs) == 2

    def test_paginated_api_call(self, httpdb):
        for i in range(25):
            httpdb.store_artifact(f"k{i:02d}",
                                  {"kind": "model",
                                   "metadata": {"key": f"k{i:02d}"}},
                                  uid=f"t{i}", project="p")
        pages = list(httpdb.paginated_api_call(
            "GET", "artifacts", params={"project": "p",
                                        "page_size": 10}))
        items = httpdb.process_paginated_responses(pages, "artifacts")
        assert len(items) == 25
        assert len(pages) == 3


class TestFeatureSurface:
    def test_create_patch_list_features(self, httpdb):
        fset = {"metadata": {"name": "fs1"},
                "spec": {"entities": [{"name": "uid"}],
                         "features": [{"name": "spend"},
                                      {"name": "clicks"}]}}
        httpdb.create_feature_set(fset, project="p")
        httpdb.patch_feature_set(
            "fs1", {"spec": {"description": "patched"}}, project="p")
        got = httpdb.get_feature_set("fs1", "p")
        assert got["spec"]["description"] == "patched"
        assert got["spec"]["features"]  # additive patch kept features
        features = httpdb.list_features("p", name="spend")
        assert len(features) == 1
        assert features[0]["feature"]["name"] == "spend"
        entities = httpdb.list_entities("p")
        assert entities[0]["entity"]["name"] == "uid"

    def test_feature_vector_patch(self, httpdb):
        httpdb.create_feature_vector(
            {"metadata": {"name": "v1"},
             "spec": {"features": ["fs1.spend"]}}, project="p")
        httpdb.patch_feature_vector("v1", {"spec": {"label_feature":
                                                    "y"}}, project="p")
        got = httpdb.get_feature_vector("v1", "p")
        assert got["spec"]["label_feature"] == "y"
        assert got["spec"]["features"] == ["fs1.spend"]


class TestHubAndGateways:
    def test_hub_source_crud(self, httpdb, tmp_path):
        httpdb.store_hub_source("extra", {"spec": {"path": str(tmp_path),
                                                   "order": 5}})
        got = httpdb.get_hub_source("extra")
        assert got["spec"]["path"] == str(tmp_path)
        names = [s["name"] for s in httpdb.list_hub_sources()]
        assert "extra" in names
        httpdb.delete_hub_source("extra")
        with pytest.raises(Exception):
            httpdb.get_hub_source("extra")

    def test_api_gateway_crud(self, httpdb):
        httpdb.store_api_gateway(
            {"metadata": {"name": "gw1"},
             "spec": {"functions": ["f1"], "project": "p"}},
            project="p")
        got = httpdb.get_api_gateway("gw1", "p")
        assert got["spec"]["functions"] == ["f1"]
        assert len(httpdb.list_api_gateways("p")) == 1
        httpdb.delete_api_gateway("gw1", "p")
        assert httpdb.list_api_gateways("p") == []


class TestModelEndpointSurface:
    def test_create_and_patch(self, httpdb):
        httpdb.create_model_endpoint("p", "ep1", {
            "kind": "model-endpoint",
            "metadata": {"uid": "ep1"},
            "spec": {"model": "m"}, "status": {"state": "ready"}})
        httpdb.patch_model_endpoint("p", "ep1",
                                    {"status": {"drift_status":
                                                "no_drift"}})
        got = httpdb.get_model_endpoint("p", "ep1")
        assert got["status"]["drift_status"] == "no_drift"
        assert got["status"]["state"] == "ready"


class TestFunctionLifecycle:
    def test_remote_builder_and_status(self, httpdb):
        func = {"kind": "job", "metadata": {"name": "bf",
                                            "project": "p"},
                "spec": {}}
        httpdb.store_function(func, "bf", "p")
        resp = httpdb.remote_builder(func, with_mlrun=False)
        assert resp["background_task"] == "build-bf"
        import time

        for _ in range(50):
            status = httpdb.get_builder_status(func)
            if status["state"] in ("succeeded", "failed"):
                break
            time.sleep(0.1)
        assert status["state"] == "succeeded", status

    def test_function_status(self, httpdb):
        httpdb.store_function(
            {"kind": "job", "metadata": {"name": "fs", "project": "p"},
             "status": {"state": "ready"}}, "fs", "p")
        status = httpdb.function_status("p", "fs")
        assert status["status"]["state"] == "ready"


class TestSecretsAndBackgroundAliases:
    def test_secrets_aliases(self, httpdb):
        httpdb.create_project_secrets("p", secrets={"KEY": "VAL"})
        assert "KEY" in httpdb.list_project_secrets("p")["secrets"]
        with pytest.raises(NotImplementedError):
            httpdb.create_user_secrets("me", secrets={"a": "b"})

    def test_runtime_resources(self, httpdb):
        resources = httpdb.list_runtime_resources("default")
        assert "gpu" in resources


class TestProjectSummaries:
    def test_summary_counts(self, httpdb):
        httpdb.create_project({"metadata": {"name": "sums"}})
        httpdb.store_run({"metadata": {"name": "r", "uid": "u1"},
                          "status": {"state": "completed"}}, "u1",
                         "sums")
        httpdb.store_run({"metadata": {"name": "r", "uid": "u2"},
                          "status": {"state": "error"}}, "u2", "sums")
        httpdb.store_artifact("m1", {"kind": "model",
                                     "metadata": {"key": "m1"}},
                              uid="t1", project="sums")
        summary = httpdb.get_project_summary("sums")
        assert summary["runs_completed_recent_count"] == 1
        assert summary["runs_failed_recent_count"] == 1
        assert summary["models_count"] == 1
        all_summaries = httpdb.list_project_summaries()
        assert any(s["name"] == "sums" for s in all_summaries)


class TestLogHelpers:
    def test_watch_log_and_size(self, httpdb, capsys):
        httpdb.store_run({"metadata": {"name": "wl", "uid": "w1"},
                          "status": {"state": "completed"}}, "w1", "p")
        httpdb.store_log("w1", "p", b"hello logs")
        assert httpdb.get_log_size("w1", "p") == len(b"hello logs")
        state = httpdb.watch_log("w1", project="p")
        assert state == "completed"
        assert "hello logs" in capsys.readouterr().out


class TestApiCallRetry:
    def test_get_retries_on_connection_error(self, httpdb):
        calls = {"n": 0}
        orig = httpdb.session.request

        def flaky(method, url, **kw):
            calls["n"] += 1
            if calls["n"] < 3:
                import requests as _requests

                raise _requests.ConnectionError("transient")
            return orig(method, url, **kw)

        httpdb.store_project("rp", {"metadata": {"name": "rp"}})
        httpdb.session.request = flaky
        try:
            result = httpdb.get_project("rp")
            assert result["metadata"]["name"] == "rp"
            assert calls["n"] == 3  # two retries then success
        finally:
            httpdb.session.request = orig

    def test_post_does_not_retry(self, httpdb):
        import requests as _requests

        calls = {"n": 0}

        def always_fail(method, url, **kw):
            calls["n"] += 1
            raise _requests.ConnectionError("down")

        orig = httpdb.session.request
        httpdb.session.request = always_fail
        try:
            with pytest.raises(_requests.ConnectionError):
                httpdb.api_call("POST", "runs")
            assert calls["n"] == 1  # non-idempotent: no retry
        finally:
            httpdb.session.request = orig


class TestLocalDbSurfaceParity:
    """The LOCAL SQLRunDB exposes the same client surface as the HTTP
    client (drop-in local mode — RunDBExtras mixin)."""

    def test_no_method_gap(self):
        from mlrun_amd.db.httpdb import HTTPRunDB
        from mlrun_amd.db.sqldb import SQLRunDB

        http_methods = {m for m in dir(HTTPRunDB)
                        if not m.startswith("_")}
        local_methods = {m for m in dir(SQLRunDB)
                         if not m.startswith("_")}
        assert not (http_methods - local_methods)

    def test_local_extras_behave(self, tmp_path):
        from mlrun_amd.db.sqldb import SQLRunDB

        db = SQLRunDB(str(tmp_path / "x.db"))
        db.store_feature_set({"metadata": {"name": "fs"},
                              "spec": {"features": []}},
                             name="fs", project="p")
        db.patch_feature_set("fs", {"spec": {"description": "d"}},
                             project="p")
        assert db.get_feature_set("fs", "p")["spec"][
            "description"] == "d"
        db.store_run({"metadata": {"name": "r", "uid": "u"},
                      "status": {"state": "completed"}}, "u", "p")
        db.set_run_notifications("p", "u", [{"kind": "console"}])
        assert db.read_run("u", "p")["spec"]["notifications"]
        db.store_log("u", "p", b"12345")
        assert db.get_log_size("u", "p") == 5
        summary = db.get_project_summary("p")
        assert summary["runs_completed_recent_count"] == 1
