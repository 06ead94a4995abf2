# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""CLI tests (python -m mlrun_amd ...)."""

import pytest

from mlrun_amd.__main__ import main


class TestCLI:
    def test_version(self, capsys):
        main(["version"])
        assert "version" in capsys.readouterr().out

    def test_config(self, capsys):
        main(["config"])
        assert "dbpath" in capsys.readouterr().out

    def test_run_and_get_and_logs(self, tmp_path, capsys, rundb):
        script = tmp_path / "job.py"
        script.write_text(
            "import mlrun_amd\n"
            "ctx = mlrun_amd.get_or_create_ctx('cli')\n"
            "ctx.log_result('out', ctx.get_param('x', 0) * 2)\n"
            "print('hello from job')\n"
            "ctx.commit(completed=True)\n")
        main(["run", str(script), "--name", "clirun", "-p", "x=21",
              "--kind", "local"])
        out = capsys.readouterr().out
        assert "completed" in out and "out: 42" in out

        main(["get", "runs"])
        out = capsys.readouterr().out
        assert "clirun" in out

        runs = rundb.list_runs(project="default", name="clirun")
        uid = runs[0]["metadata"]["uid"]
        main(["logs", uid])
        out = capsys.readouterr().out
        assert "hello from job" in out

    def test_run_failure_exit_code(self, tmp_path):
        script = tmp_path / "bad.py"
        script.write_text("raise SystemExit(2)\n")
        with pytest.raises(SystemExit):
            main(["run", str(script), "--name", "bad", "--kind", "local"])

    def test_clean(self, tmp_path, capsys, rundb):
        rundb.store_run({"metadata": {"name": "x", "uid": "u9"},
                         "status": {"state": "completed"}}, "u9", "default")
        main(["clean"])
        assert rundb.list_runs(project="default") == []

    def test_build_and_run_image(self, tmp_path, capsys, rundb):
        code = tmp_path / "fn.py"
        code.write_text("def handler(context):\n"
                        "    context.log_result('v', 3)\n")
        main(["build", "--name", "cli-img", "--command", str(code)])
        out = capsys.readouterr().out
        assert "image built at" in out
        import os

        from mlrun_amd.config import config

        image = os.path.join(config.base_dir, "images", "default",
                             "cli-img", "latest")
        assert os.path.isfile(os.path.join(image, "run.sh"))

    def test_get_schedules_and_projects(self, capsys, rundb):
        import mlrun_amd

        mlrun_amd.new_project("cli-proj")
        main(["get", "projects"])
        out = capsys.readouterr().out
        assert "cli-proj" in out

    def test_hyperparam_run(self, tmp_path, capsys, rundb):
        code = tmp_path / "hp.py"
        code.write_text("def handler(context, p1=0):\n"
                        "    context.log_result('r', p1 * 2)\n")
        main(["run", str(code), "--handler", "handler", "--name", "hp",
              "--hyperparam", "p1=[1,2,3]", "--local"])
        out = capsys.readouterr().out
        assert "finished: completed" in out
        runs = rundb.list_runs(project="default", name="hp")
        iters = runs[0]["status"].get("iterations") or []
        results = runs[0]["status"].get("results", {})
        assert results.get("best_iteration") or iters

    def test_db_service_boots_over_http(self, tmp_path):
        """`mlrun_amd db` boots the real uvicorn service; check
        healthz + one CRUD roundtrip over actual HTTP."""
        import os
        import socket
        import subprocess
        import sys
        import time

        import requests

        with socket.socket() as sock:
            sock.bind(("127.0.0.1", 0))
            port = sock.getsockname()[1]
        env = dict(os.environ)
        env["MLRUN_BASE_DIR"] = str(tmp_path)
        repo = os.path.dirname(os.path.dirname(os.path.abspath(
            __file__)))
        env["PYTHONPATH"] = repo + os.pathsep + env.get("PYTHONPATH", "")
        proc = subprocess.Popen(
            [sys.executable, "-m", "mlrun_amd", "db", "--port",
             str(port)], env=env)
        base = f"http://127.0.0.1:{port}"
        try:
            deadline = time.time() + 60
            while time.time() < deadline:
                try:
                    resp = requests.get(base + "/api/v1/healthz",
                                        timeout=2)
                    if resp.status_code == 200:
                        break
                except Exception:
                    time.sleep(0.3)
            else:
                raise TimeoutError("service did not become healthy")
            run = {"metadata": {"name": "httprun", "uid": "h1"},
                   "status": {"state": "completed"}}
            assert requests.post(base + "/api/v1/run/default/h1",
                                 json=run, timeout=10
                                 ).status_code == 200
            got = requests.get(base + "/api/v1/run/default/h1",
                               timeout=10).json()["data"]
            assert got["metadata"]["name"] == "httprun"
        finally:
            proc.terminate()
            proc.wait(timeout=10)

    def test_remote_submit_through_live_service(self, tmp_path):
        """Reference stack 3.2: client fn.run(local=False) ->
        HTTPRunDB.submit_job over real HTTP -> ServerSideLauncher
        executes -> client reads the run back."""
        import os
        import socket
        import subprocess
        import sys
        import time

        import requests

        with socket.socket() as sock:
            sock.bind(("127.0.0.1", 0))
            port = sock.getsockname()[1]
        env = dict(os.environ)
        env["MLRUN_BASE_DIR"] = str(tmp_path / "server")
        repo = os.path.dirname(os.path.dirname(os.path.abspath(
            __file__)))
        env["PYTHONPATH"] = repo + os.pathsep + env.get("PYTHONPATH", "")
        proc = subprocess.Popen(
            [sys.executable, "-m", "mlrun_amd", "db", "--port",
             str(port)], env=env)
        base = f"http://127.0.0.1:{port}"
        code = tmp_path / "remote_fn.py"
        code.write_text("def handler(context, x=1):\n"
                        "    context.log_result('doubled', x * 2)\n")
        client_code = tmp_path / "client.py"
        client_code.write_text(f"""
import mlrun_amd
from mlrun_amd.config import config
config.dbpath = {base!r}
fn = mlrun_amd.new_function(name="remote-job", kind="job",
                            command={str(code)!r})
run = fn.run(handler="handler", params={{"x": 21}}, local=False,
             watch=True)
assert run.status.state == "completed", run.status.state
assert run.outputs["doubled"] == 42, run.outputs
print("REMOTE_OK", run.metadata.uid)
""")
        try:
            deadline = time.time() + 60
            while time.time() < deadline:
                try:
                    if requests.get(base + "/api/v1/healthz",
                                    timeout=2).status_code == 200:
                        break
                except Exception:
                    time.sleep(0.3)
            out = subprocess.run(
                [sys.executable, str(client_code)], env=env,
                capture_output=True, text=True, timeout=120)
            assert "REMOTE_OK" in out.stdout, out.stdout + out.stderr
        finally:
            proc.terminate()
            proc.wait(timeout=10)

    def test_get_artifacts_and_schedules(self, capsys, rundb, tmp_path):
        rundb.store_artifact("art1", {"kind": "file",
                                      "metadata": {"key": "art1"},
                                      "spec": {}}, tree="t1",
                            project="default")
        code = tmp_path / "s.py"
        code.write_text("def handler(context):\n    pass\n")
        rundb.create_schedule("default", {
            "name": "nightly", "kind": "job",
            "cron_trigger": "0 3 * * *",
            "task": {"metadata": {"name": "nightly"},
                     "spec": {"function": str(code),
                              "handler": "handler"}}})
        main(["get", "artifacts"])
        assert "art1" in capsys.readouterr().out
        main(["get", "schedules"])
        out = capsys.readouterr().out
        assert "nightly" in out and "0 3 * * *" in out


def test_cli_migrate_and_summary(tmp_path, monkeypatch, capsys):
    import json as _json

    from mlrun_amd.__main__ import main
    from mlrun_amd.config import config

    monkeypatch.setattr(config, "base_dir", str(tmp_path))
    main(["migrate"])
    out = capsys.readouterr().out
    assert _json.loads(out)["schema_version"] >= 4

    from mlrun_amd.db import get_run_db

    db = get_run_db()
    db.store_project("cliproj", {"metadata": {"name": "cliproj"}})
    db.store_run({"metadata": {"name": "r", "uid": "u"},
                  "status": {"state": "completed"}}, "u", "cliproj")
    main(["summary", "cliproj"])
    out = capsys.readouterr().out
    summary = _json.loads(out)
    assert summary["runs_completed_recent_count"] == 1
