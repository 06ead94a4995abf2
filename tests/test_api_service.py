# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""API-service tests: FastAPI TestClient + fresh SQLite per test
(mirrors reference tests/api/conftest.py seam) — including the
client->server round trip through HTTPRunDB."""

import datetime

import pytest
from fastapi.testclient import TestClient

from mlrun_amd.api import create_app, CronTrigger
from mlrun_amd.db.sqldb import SQLRunDB


@pytest.fixture
def client(tmp_path):
    db = SQLRunDB(str(tmp_path / "api.db"))
    app = create_app(db, with_scheduler=False)
    with TestClient(app) as test_client:
        test_client.db = db
        yield test_client


class TestEndpoints:
    def test_healthz(self, client):
        resp = client.get("/api/v1/healthz")
        assert resp.status_code == 200
        assert resp.json()["status"] == "ok"

    def test_runs_crud(self, client):
        run = {"metadata": {"name": "r1", "uid": "u1"},
               "status": {"state": "running"}}
        assert client.post("/api/v1/run/p/u1", json=run).status_code == 200
        resp = client.get("/api/v1/run/p/u1")
        assert resp.json()["data"]["metadata"]["name"] == "r1"
        client.patch("/api/v1/run/p/u1",
                     json={"status.state": "completed"})
        runs = client.get("/api/v1/runs",
                          params={"project": "p"}).json()["runs"]
        assert len(runs) == 1 and runs[0]["status"]["state"] == "completed"
        client.post("/api/v1/run/p/u1/abort", json={})
        assert client.get("/api/v1/run/p/u1").json()[
            "data"]["status"]["state"] == "completed"  # terminal wins
        client.delete("/api/v1/run/p/u1")
        assert client.get("/api/v1/run/p/u1").status_code == 404

    def test_artifact_and_function_routes(self, client):
        artifact = {"kind": "model", "metadata": {"key": "m"},
                    "spec": {"target_path": "/tmp/m"}}
        client.post("/api/v1/artifact/p/m", json=artifact,
                    params={"tree": "t1"})
        assert client.get("/api/v1/artifact/p/m").json()[
            "data"]["spec"]["target_path"] == "/tmp/m"
        assert len(client.get("/api/v1/artifacts",
                              params={"project": "p"}).json()[
            "artifacts"]) == 1
        func = {"kind": "job", "metadata": {"name": "f"}}
        hash_key = client.post("/api/v1/func/p/f", json=func).json()[
            "hash_key"]
        assert hash_key
        assert client.get("/api/v1/func/p/f").json()["func"]["kind"] == \
            "job"

    def test_logs(self, client):
        client.post("/api/v1/log/p/u1", content=b"line one\n")
        client.post("/api/v1/log/p/u1", content=b"line two",
                    params={"append": 1})
        resp = client.get("/api/v1/log/p/u1")
        assert resp.content == b"line one\nline two"

    def test_projects_and_schedules(self, client):
        client.post("/api/v1/projects",
                    json={"metadata": {"name": "proj"}})
        assert client.get("/api/v1/projects/proj").status_code == 200
        sched = {"name": "s1", "kind": "job",
                 "cron_trigger": "0 * * * *", "task": {}}
        client.post("/api/v1/projects/proj/schedules", json=sched)
        listed = client.get("/api/v1/projects/proj/schedules").json()
        assert len(listed["schedules"]) == 1
        client.delete("/api/v1/projects/proj/schedules/s1")
        assert client.get("/api/v1/projects/proj/schedules").json()[
            "schedules"] == []

    def test_submit_job(self, client, tmp_path):
        task = {
            "metadata": {"name": "apirun", "project": "default"},
            "spec": {"handler": None, "function": ""},
        }
        resp = client.post("/api/v1/submit_job", json={"task": task})
        assert resp.status_code == 200
        # the run executed server-side and is in the DB
        data = resp.json()["data"]
        assert data["status"]["state"] in ("completed", "error")

    def test_error_mapping(self, client):
        assert client.get("/api/v1/run/p/missing").status_code == 404
        assert client.get("/api/v1/projects/missing").status_code == 404

    def test_alerts_and_events(self, client):
        alert = {"name": "a1", "summary": "drift!", "severity": "high",
                 "trigger": {"events": ["model-drift"]},
                 "criteria": {"count": 2},
                 "notifications": [{"kind": "console"}]}
        client.put("/api/v1/projects/p/alerts/a1", json=alert)
        # first event: below criteria count
        resp = client.post("/api/v1/projects/p/events/model-drift",
                           json={"endpoint": "e1"})
        assert resp.json()["alerts_fired"] == []
        resp = client.post("/api/v1/projects/p/events/model-drift",
                           json={"endpoint": "e1"})
        assert resp.json()["alerts_fired"] == ["a1"]
        # unrelated event kind does not fire
        resp = client.post("/api/v1/projects/p/events/other",
                           json={})
        assert resp.json()["alerts_fired"] == []


class TestHTTPRunDBClient:
    """Client round trip against the in-process app (reference seam:
    config.dbpath -> TestClient)."""

    def test_client_roundtrip(self, client, monkeypatch):
        from mlrun_amd.db.httpdb import HTTPRunDB

        httpdb = HTTPRunDB("http://testserver")
        # route the requests session through the TestClient
        httpdb.session = client
        orig_request = client.request

        def request(method, url, params=None, data=None, json=None,
                    headers=None, timeout=None):
            return orig_request(method, url, params=params, content=data,
                                json=json, headers=headers)

        httpdb.session.request = request
        httpdb.store_run({"metadata": {"name": "rr", "uid": "uu"},
                          "status": {"state": "running"}}, "uu", "p")
        run = httpdb.read_run("uu", "p")
        assert run["metadata"]["name"] == "rr"
        assert len(httpdb.list_runs(project="p")) == 1
        httpdb.store_log("uu", "p", b"log text")
        state, log = httpdb.get_log("uu", "p")
        assert log == b"log text"
        hash_key = httpdb.store_function({"kind": "job"}, "fn", "p")
        assert httpdb.get_function("fn", "p")["kind"] == "job"


class TestCron:
    def test_parse_and_match(self):
        trigger = CronTrigger("*/5 * * * *")
        dt = datetime.datetime(2026, 9, 12, 10, 5)
        assert trigger.matches(dt)
        assert not trigger.matches(dt.replace(minute=7))
        nxt = trigger.next_fire_time(dt)
        assert nxt.minute == 10

    def test_invalid(self):
        from mlrun_amd.errors import MLRunInvalidArgumentError

        with pytest.raises(MLRunInvalidArgumentError):
            CronTrigger("* * *")

    def test_min_interval(self):
        assert CronTrigger("*/5 * * * *").min_interval_seconds() == 300
        assert CronTrigger("0 * * * *").min_interval_seconds() == 3600

    def test_schedule_execution(self, tmp_path):
        """Invoke a stored schedule and verify the run executed."""
        from mlrun_amd.api import Scheduler
        from mlrun_amd.db.sqldb import SQLRunDB

        db = SQLRunDB(str(tmp_path / "sched.db"))
        task = {"metadata": {"name": "sched-run", "project": "default"},
                "spec": {}}
        db.create_schedule("default", {"name": "s1", "kind": "job",
                                       "cron_trigger": "0 0 1 1 *",
                                       "task": task})
        scheduler = Scheduler(db)
        scheduler.invoke("default", "s1")
        runs = db.list_runs(project="default")
        assert len(runs) == 1
        sched = db.get_schedule("default", "s1")
        assert sched["last_run_uri"]


class TestRuntimeResources:
    def test_listing_with_fake_allocator(self, client):
        """The reference tests k8s handlers against a mocked clientset;
        the analog seam here is a fake GPU allocator."""
        from mlrun_amd.parallel.scheduler import (
            GpuAllocator, set_gpu_allocator, get_gpu_allocator)

        fake = GpuAllocator(total=4)
        set_gpu_allocator(fake)
        try:
            lease = fake.acquire(2, owner="run-1")
            resp = client.get(
                "/api/v1/projects/default/runtime-resources").json()
            assert resp["gpu"]["total"] == 4
            assert len(resp["gpu"]["in_use"]) == 2
            assert resp["gpu"]["available"] == [2, 3]
            lease.release()
            resp = client.get(
                "/api/v1/projects/default/runtime-resources").json()
            assert resp["gpu"]["in_use"] == {}
        finally:
            set_gpu_allocator(GpuAllocator())

    def test_pagination(self, client):
        for i in range(7):
            client.post(f"/api/v1/run/p/u{i}",
                        json={"metadata": {"name": f"r{i}", "uid": f"u{i}"},
                              "status": {"state": "completed"}})
        resp = client.get("/api/v1/runs",
                          params={"project": "p", "page": 2,
                                  "page_size": 3}).json()
        assert len(resp["runs"]) == 3
        assert resp["pagination"]["total"] == 7


def test_memory_report(client):
    resp = client.get("/api/v1/monitoring/memory")
    assert resp.status_code == 200
    body = resp.json()
    assert body["rss_bytes"] > 0 and body["threads"] >= 1


class TestHubSecretsPipelinesAPI:
    def test_hub_endpoints(self, client):
        resp = client.get("/api/v1/hub/sources")
        assert resp.status_code == 200
        names = [s["name"] for s in resp.json()["sources"]]
        assert "builtin" in names
        resp = client.get("/api/v1/hub/sources/builtin/items")
        items = [i["name"] for i in resp.json()["catalog"]]
        assert "describe" in items and "batch-infer" in items
        resp = client.get("/api/v1/hub/sources/builtin/items/describe")
        assert resp.json()["spec"]["kind"] == "job"
        resp = client.get("/api/v1/hub/sources/builtin/items/nope")
        assert resp.status_code == 404

    def test_secrets_endpoints(self, client):
        resp = client.post("/api/v1/projects/p1/secrets",
                           json={"secrets": {"TOKEN": "abc",
                                             "KEY2": "v"}})
        assert resp.status_code == 200
        keys = client.get("/api/v1/projects/p1/secret-keys"
                          ).json()["secret_keys"]
        assert keys == ["KEY2", "TOKEN"]
        # values are never exposed over the list API
        assert "abc" not in client.get(
            "/api/v1/projects/p1/secret-keys").text
        client.delete("/api/v1/projects/p1/secrets?secrets=TOKEN")
        keys = client.get("/api/v1/projects/p1/secret-keys"
                          ).json()["secret_keys"]
        assert keys == ["KEY2"]

    def test_pipelines_endpoints(self, client, tmp_path):
        import mlrun_amd

        project = mlrun_amd.new_project("pipeproj",
                                        context=str(tmp_path))
        code = tmp_path / "wf_fn.py"
        code.write_text("def handler(context):\n"
                        "    context.log_result('ok', 1)\n")
        project.set_function(str(code), name="step1", kind="job")

        def workflow(project=None, **kw):
            from mlrun_amd.projects.operations import run_function

            run_function("step1", handler="handler",
                         project_object=project)

        status = project.run(workflow_handler=workflow)
        resp = client.get("/api/v1/projects/pipeproj/pipelines")
        runs = resp.json()["runs"]
        assert any(r["run_id"] == status.run_id for r in runs)
        resp = client.get(
            f"/api/v1/projects/pipeproj/pipelines/{status.run_id}")
        assert resp.json()["state"] == "completed"
        assert client.get(
            "/api/v1/projects/pipeproj/pipelines/zzz").status_code == 404


class TestSchedulerMisfire:
    def test_misfire_catch_up_invokes_missed_schedule(self, client,
                                                      rundb, tmp_path):
        """A schedule whose stored next_run_time passed while the
        service was down is invoked on scheduler start (APScheduler
        misfire analog, reference scheduler.py)."""
        import datetime

        from mlrun_amd.api.scheduler import Scheduler

        code = tmp_path / "sched_fn.py"
        code.write_text("def handler(context):\n"
                        "    context.log_result('ran', True)\n")
        task = {"metadata": {"name": "missed", "project": "default"},
                "spec": {"function": str(code), "handler": "handler"}}
        missed_at = (datetime.datetime.now() -
                     datetime.timedelta(minutes=5)).isoformat()
        rundb.create_schedule("default", {
            "name": "missed", "kind": "job",
            "cron_trigger": "*/10 * * * *", "task": task,
            "next_run_time": missed_at})
        scheduler = Scheduler(rundb, tick_seconds=3600)
        scheduler.start()
        try:
            sched = rundb.get_schedule("default", "missed")
            assert sched.get("last_run_uri"), "missed schedule not run"
        finally:
            scheduler.stop()

    def test_too_old_misfire_is_skipped(self, client, rundb, tmp_path):
        import datetime

        from mlrun_amd.api.scheduler import Scheduler

        code = tmp_path / "old_fn.py"
        code.write_text("def handler(context):\n    pass\n")
        task = {"metadata": {"name": "old", "project": "default"},
                "spec": {"function": str(code), "handler": "handler"}}
        stale = (datetime.datetime.now() -
                 datetime.timedelta(hours=3)).isoformat()
        rundb.create_schedule("default", {
            "name": "old", "kind": "job",
            "cron_trigger": "*/10 * * * *", "task": task,
            "next_run_time": stale})
        scheduler = Scheduler(rundb, tick_seconds=3600)
        scheduler.start()
        try:
            sched = rundb.get_schedule("default", "old")
            assert not sched.get("last_run_uri")  # beyond grace window
        finally:
            scheduler.stop()


def test_runs_pagination(client):
    for i in range(7):
        client.db.store_run(
            {"metadata": {"name": f"r{i}", "uid": f"u{i}"},
             "status": {"state": "completed"}}, f"u{i}", "default")
    resp = client.get("/api/v1/runs?page=1&page_size=3")
    body = resp.json()
    assert len(body["runs"]) == 3
    assert body["pagination"]["page"] == 1
    resp2 = client.get("/api/v1/runs?page=3&page_size=3")
    assert len(resp2.json()["runs"]) == 1  # 7 = 3 + 3 + 1
    uids = {r["metadata"]["uid"] for r in body["runs"]} | \
        {r["metadata"]["uid"]
         for r in client.get("/api/v1/runs?page=2&page_size=3"
                             ).json()["runs"]} | \
        {r["metadata"]["uid"] for r in resp2.json()["runs"]}
    assert len(uids) == 7  # pages partition the set


def test_workflow_submit_endpoint(client, tmp_path):
    import time

    import mlrun_amd

    project = mlrun_amd.new_project("wfapi", context=str(tmp_path),
                                    save=False)
    code = tmp_path / "wf_step.py"
    code.write_text("def handler(context):\n"
                    "    context.log_result('done', 1)\n")
    project.set_function(str(code), name="s1", kind="job")
    wf = tmp_path / "flow.py"
    wf.write_text(
        "def pipeline(project=None, **kw):\n"
        "    from mlrun_amd.projects.operations import run_function\n"
        "    run_function('s1', handler='handler',\n"
        "                 project_object=project)\n")
    project.set_workflow("main", str(wf), handler="pipeline")
    client.db.store_project("wfapi", project.to_dict())

    resp = client.get("/api/v1/projects/wfapi/workflows")
    assert resp.json()["workflows"][0]["name"] == "main"

    resp = client.post("/api/v1/projects/wfapi/workflows/main/submit",
                       json={})
    assert resp.status_code == 200
    task = resp.json()["background_task"]
    deadline = time.time() + 30
    state = "running"
    while time.time() < deadline:
        state = client.get(
            f"/api/v1/projects/wfapi/background-tasks/{task}"
        ).json().get("status", {}).get("state", "running")
        if state in ("succeeded", "failed"):
            break
        time.sleep(0.3)
    assert state == "succeeded"


def test_httpdb_new_surfaces_roundtrip(client, tmp_path, monkeypatch):
    """HTTPRunDB client methods against the live app (secrets, hub,
    workflows, pipelines)."""
    from mlrun_amd.db.httpdb import HTTPRunDB

    db = HTTPRunDB("http://testserver")
    # route the client's HTTP through the TestClient
    def api_call(method, path, params=None, json_body=None, **kw):
        url = f"/api/v1/{path}"
        resp = client.request(method, url, params=params,
                              json=json_body)
        assert resp.status_code < 500, resp.text
        try:
            return resp.json()
        except ValueError:
            return {}
    monkeypatch.setattr(db, "api_call", api_call)

    db.store_project_secrets("hp", {"A": "1", "B": "2"})
    assert db.list_project_secret_keys("hp") == ["A", "B"]
    db.delete_project_secrets("hp", ["A"])
    assert db.list_project_secret_keys("hp") == ["B"]
    items = [i["name"] for i in db.get_hub_catalog("builtin")]
    assert "llm-serving" in items
    assert db.list_pipelines("hp") == []
    assert db.list_workflows("hp") == []


def test_alert_reset_policies(client):
    db = client.db
    db.store_alert_config("rp", "auto-a", {
        "name": "auto-a", "trigger": {"events": ["e"]},
        "criteria": {"count": 2}, "reset_policy": "auto",
        "notifications": [{"kind": "console"}]})
    db.store_alert_config("rp", "manual-a", {
        "name": "manual-a", "trigger": {"events": ["e"]},
        "criteria": {"count": 2}, "reset_policy": "manual",
        "notifications": [{"kind": "console"}]})
    from mlrun_amd.api.events import process_event

    assert process_event("rp", "e", {}, db=db) == []
    assert sorted(process_event("rp", "e", {}, db=db)) == \
        ["auto-a", "manual-a"]
    # auto rearmed (count reset); manual stays above threshold
    assert process_event("rp", "e", {}, db=db) == ["manual-a"]
    resp = client.post("/api/v1/projects/rp/alerts/manual-a/reset")
    assert resp.status_code == 200
    # 4th event: auto-a reaches its threshold again (1 -> 2); the
    # freshly reset manual-a is back at 1
    assert process_event("rp", "e", {}, db=db) == ["auto-a"]


def test_runs_monitor_aborts_stuck_runs(client, monkeypatch):
    import datetime

    from mlrun_amd.api.main import check_stuck_runs
    from mlrun_amd.config import config as cfg

    db = client.db
    old = (datetime.datetime.now(datetime.timezone.utc) -
           datetime.timedelta(hours=30)).isoformat()
    fresh = datetime.datetime.now(datetime.timezone.utc).isoformat()
    db.store_run({"metadata": {"name": "stuck", "uid": "s1"},
                  "status": {"state": "running", "start_time": old}},
                 "s1", "default")
    db.store_run({"metadata": {"name": "live", "uid": "s2"},
                  "status": {"state": "running", "start_time": fresh}},
                 "s2", "default")
    aborted = check_stuck_runs(db)
    assert aborted == ["s1"]
    assert db.read_run("s1", "default")["status"]["state"] == "aborted"
    assert db.read_run("s2", "default")["status"]["state"] == "running"


def test_schedule_payload_validated(client):
    # missing cron_trigger -> pydantic rejects before reaching the DB
    resp = client.post("/api/v1/projects/vp/schedules",
                       json={"name": "bad"})
    assert resp.status_code in (400, 422, 500)
    assert client.db.list_schedules("vp") == []
    resp = client.post("/api/v1/projects/vp/schedules",
                       json={"name": "ok",
                             "cron_trigger": "*/5 * * * *",
                             "task": {"metadata": {"name": "t"}}})
    assert resp.status_code == 200
    assert client.db.get_schedule("vp", "ok")["cron_trigger"] == \
        "*/5 * * * *"


def test_tags_and_files_endpoints(client, tmp_path):
    db = client.db
    db.store_artifact("modelA", {"kind": "model",
                                 "metadata": {"key": "modelA",
                                              "tree": "t1"},
                                 "spec": {}},
                      tree="t1", project="tp")
    resp = client.put("/api/v1/projects/tp/tags/prod",
                      json={"identifiers": [{"key": "modelA",
                                             "tree": "t1"}]})
    assert resp.status_code == 200
    tags = client.get("/api/v1/projects/tp/tags?key=modelA"
                      ).json()["tags"]
    assert "prod" in tags
    client.delete("/api/v1/projects/tp/tags/prod?key=modelA")
    tags = client.get("/api/v1/projects/tp/tags?key=modelA"
                      ).json()["tags"]
    assert "prod" not in tags

    from mlrun_amd.config import config

    blob = tmp_path / "blob.bin"
    blob.write_bytes(b"0123456789")
    saved = config.httpdb.files_allowed_paths
    config.httpdb.files_allowed_paths = str(tmp_path)
    try:
        resp = client.get(f"/api/v1/files?path={blob}")
        assert resp.content == b"0123456789"
        resp = client.get(f"/api/v1/files?path={blob}&offset=3&size=4")
        assert resp.content == b"3456"
        assert client.get(f"/api/v1/filestat?path={blob}"
                          ).json()["size"] == 10
        assert client.get(f"/api/v1/files?path={tmp_path}/nope"
                          ).status_code == 404
        # outside every allowed prefix -> denied, not served
        assert client.get("/api/v1/files?path=/nope/x"
                          ).status_code == 403
    finally:
        config.httpdb.files_allowed_paths = saved


def test_alert_templates(client):
    body = {"summary": "high drift on {{endpoint}}",
            "severity": "high",
            "trigger": {"events": ["model-drift"]},
            "criteria": {"count": 3}}
    assert client.put("/api/v1/alert-templates/drift-high",
                      json=body).status_code == 200
    got = client.get("/api/v1/alert-templates/drift-high").json()
    assert got["criteria"]["count"] == 3
    names = [t["template_name"] for t in
             client.get("/api/v1/alert-templates").json()["templates"]]
    assert "drift-high" in names
    client.delete("/api/v1/alert-templates/drift-high")
    assert client.get("/api/v1/alert-templates/drift-high"
                      ).status_code in (404, 500)


def test_datastore_profiles(client):
    prof = {"name": "my-s3", "type": "s3",
            "public": {"endpoint_url": "http://minio:9000",
                       "bucket": "data"}}
    assert client.put("/api/v1/projects/dp/datastore-profiles",
                      json=prof).status_code == 200
    got = client.get("/api/v1/projects/dp/datastore-profiles/my-s3"
                     ).json()
    assert got["public"]["bucket"] == "data"
    assert len(client.get("/api/v1/projects/dp/datastore-profiles"
                          ).json()["profiles"]) == 1
    client.delete("/api/v1/projects/dp/datastore-profiles/my-s3")
    assert client.get("/api/v1/projects/dp/datastore-profiles"
                      ).json()["profiles"] == []


def test_sweep_service_caches(tmp_path):
    """Pagination-cache + background-task TTL hygiene (reference
    cleanup loops)."""
    import datetime

    from mlrun_amd.api.main import sweep_service_caches
    from mlrun_amd.db.sqldb import SQLRunDB

    db = SQLRunDB(str(tmp_path / "h.db"))
    for i in range(30):
        db.store_artifact(f"a{i}", {"kind": "model",
                                    "metadata": {"key": f"a{i}"}},
                          tree=f"t{i}", project="h")
    _, token = db.paginated_list("list_artifacts", project="h",
                                 page_size=10)
    assert token
    db.store_background_task("h", {
        "metadata": {"name": "old-task", "project": "h",
                     "updated": "2020-01-01T00:00:00"},
        "status": {"state": "succeeded"}})
    db.store_background_task("h", {
        "metadata": {"name": "live-task", "project": "h",
                     "updated": "2020-01-01T00:00:00"},
        "status": {"state": "running"}})
    future = datetime.datetime.now() + datetime.timedelta(days=400)
    result = sweep_service_caches(db, now=future)
    assert "h/old-task" in result["background_tasks"]
    assert "h/live-task" not in result["background_tasks"]
    # the idle pagination token is gone
    import pytest as _pytest

    from mlrun_amd.errors import MLRunNotFoundError

    with _pytest.raises(MLRunNotFoundError):
        db.paginated_list("list_artifacts", page_token=token)


class TestWireValidation:
    """Expanded pydantic wire schemas: malformed shapes 422
    (reference: per-endpoint pydantic bodies)."""

    def test_feature_set_validation(self, client):
        bad = {"metadata": {"name": "fs"},
               "spec": {"entities": [{"no_name": True}]}}
        assert client.put("/api/v1/projects/p/feature-sets/fs",
                          json=bad).status_code == 422
        good = {"metadata": {"name": "fs"},
                "spec": {"entities": [{"name": "k"}],
                         "aggregations": [
                             {"name": "a", "column": "v",
                              "operations": ["sum"],
                              "windows": ["1h"]}]}}
        assert client.put("/api/v1/projects/p/feature-sets/fs",
                          json=good).status_code == 200

    def test_aggregation_requires_ops_and_windows(self, client):
        bad = {"metadata": {"name": "fs2"},
               "spec": {"aggregations": [{"name": "a",
                                          "column": "v"}]}}
        assert client.put("/api/v1/projects/p/feature-sets/fs2",
                          json=bad).status_code == 422

    def test_alert_validation(self, client):
        bad = {"name": "a2", "severity": "catastrophic"}
        assert client.put("/api/v1/projects/p/alerts/a2",
                          json=bad).status_code == 422
        bad_notification = {"name": "a2", "severity": "high",
                            "notifications": [{"kind": "carrier-pigeon"}]}
        assert client.put("/api/v1/projects/p/alerts/a2",
                          json=bad_notification).status_code == 422

    def test_datastore_profile_validation(self, client):
        assert client.put("/api/v1/projects/p/datastore-profiles",
                          json={"type": "s3"}).status_code == 422
        assert client.put("/api/v1/projects/p/datastore-profiles",
                          json={"name": "s3main",
                                "type": "s3"}).status_code == 200

    def test_model_endpoint_validation(self, client):
        bad = {"metadata": {"name": "ep"},
               "spec": {"monitoring_mode": ["not-a-string"]}}
        assert client.put(
            "/api/v1/projects/p/model-endpoints/ep",
            json=bad).status_code == 422
