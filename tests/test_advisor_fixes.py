# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Regression tests for the round-1 advisor findings."""

import datetime
import importlib
import sys

import pytest

from mlrun_amd.errors import MLRunInvalidArgumentError
from mlrun_amd.utils.safe_eval import UnsafeExpressionError, safe_eval


class TestAliasMetaPath:
    """mlrun.* imports must resolve to the SAME module object as
    mlrun_amd.* even for paths not yet loaded (ADVICE medium #1)."""

    def test_fresh_submodule_shares_instance(self):
        import mlrun  # noqa: F401

        # pick modules and drop them from sys.modules to simulate
        # a not-yet-loaded path
        for name in ("mlrun.data_types", "mlrun_amd.data_types"):
            sys.modules.pop(name, None)
        a = importlib.import_module("mlrun.data_types")
        b = importlib.import_module("mlrun_amd.data_types")
        assert a is b

    def test_from_import_shares_classes(self):
        import mlrun  # noqa: F401

        sys.modules.pop("mlrun.secrets", None)
        sys.modules.pop("mlrun_amd.secrets", None)
        from mlrun.secrets import SecretsStore as A
        from mlrun_amd.secrets import SecretsStore as B
        assert A is B

    def test_nonexistent_path_raises_import_error(self):
        import mlrun  # noqa: F401

        with pytest.raises(ImportError):
            importlib.import_module("mlrun.does_not_exist_xyz")


class TestFilesEndpointAuthorization:
    """GET /files must not serve paths outside the allowed data
    prefixes (ADVICE medium #2)."""

    @pytest.fixture()
    def client(self, tmp_path, monkeypatch):
        from fastapi.testclient import TestClient

        from mlrun_amd.config import config
        from mlrun_amd.api.main import create_app
        from mlrun_amd.db.sqldb import SQLRunDB

        monkeypatch.setattr(config.httpdb, "dirpath", str(tmp_path))
        monkeypatch.setattr(config, "base_dir", str(tmp_path))
        monkeypatch.setattr(config, "artifact_path", "")
        monkeypatch.setattr(config.httpdb, "files_allowed_paths", "")
        app = create_app(SQLRunDB(str(tmp_path / "db.sqlite")),
                         with_scheduler=False)
        with TestClient(app) as c:
            yield c

    def test_etc_passwd_denied(self, client):
        resp = client.get("/api/v1/files", params={"path": "/etc/passwd"})
        assert resp.status_code == 403
        resp = client.get("/api/v1/filestat",
                          params={"path": "/etc/passwd"})
        assert resp.status_code == 403

    def test_traversal_denied(self, client, tmp_path):
        sneaky = str(tmp_path / ".." / ".." / "etc" / "passwd")
        resp = client.get("/api/v1/files", params={"path": sneaky})
        assert resp.status_code == 403

    def test_allowed_prefix_served(self, client, tmp_path):
        target = tmp_path / "data.txt"
        target.write_text("hello")
        resp = client.get("/api/v1/files", params={"path": str(target)})
        assert resp.status_code == 200
        assert resp.content == b"hello"
        resp = client.get("/api/v1/filestat",
                          params={"path": str(target)})
        assert resp.status_code == 200

    def test_configured_extra_prefix(self, client, tmp_path, monkeypatch,
                                     tmp_path_factory):
        from mlrun_amd.config import config

        other = tmp_path_factory.mktemp("extra")
        (other / "x.txt").write_text("x")
        monkeypatch.setattr(config.httpdb, "files_allowed_paths",
                            str(other))
        resp = client.get("/api/v1/files",
                          params={"path": str(other / "x.txt")})
        assert resp.status_code == 200


class TestCronWeekday:
    """Cron DOW numbering: Sunday is 0 (and 7), Monday is 1
    (ADVICE low #3)."""

    def _trigger(self, expr):
        from mlrun_amd.api.scheduler import CronTrigger

        return CronTrigger(expr)

    def test_sunday_zero(self):
        trig = self._trigger("0 9 * * 0")
        sunday = datetime.datetime(2026, 9, 13, 9, 0)  # a Sunday
        monday = datetime.datetime(2026, 9, 14, 9, 0)
        assert trig.matches(sunday)
        assert not trig.matches(monday)

    def test_sunday_seven(self):
        trig = self._trigger("0 9 * * 7")
        sunday = datetime.datetime(2026, 9, 13, 9, 0)
        assert trig.matches(sunday)

    def test_monday_one(self):
        trig = self._trigger("0 9 * * 1")
        monday = datetime.datetime(2026, 9, 14, 9, 0)
        sunday = datetime.datetime(2026, 9, 13, 9, 0)
        assert trig.matches(monday)
        assert not trig.matches(sunday)

    def test_range_mon_fri(self):
        trig = self._trigger("0 9 * * 1-5")
        friday = datetime.datetime(2026, 9, 18, 9, 0)
        saturday = datetime.datetime(2026, 9, 19, 9, 0)
        assert trig.matches(friday)
        assert not trig.matches(saturday)

    def test_out_of_range_raises(self):
        with pytest.raises(MLRunInvalidArgumentError):
            self._trigger("0 9 * * 8")
        with pytest.raises(MLRunInvalidArgumentError):
            self._trigger("99 9 * * *")


class TestSafeEval:
    """ast-based expression evaluation (ADVICE low #4)."""

    def test_comparisons_and_bool(self):
        assert safe_eval("accuracy > 0.9 and loss < 0.1",
                         {"accuracy": 0.95, "loss": 0.05}) is True
        assert safe_eval("a == 1 or b == 2", {"a": 0, "b": 2}) is True

    def test_subscript_and_attr(self):
        class Event:
            body = {"url": "http://x/y"}

        assert safe_eval('event.body["url"]', {"event": Event()}) == \
            "http://x/y"

    def test_method_call(self):
        assert safe_eval('event.get("k", 5)', {"event": {}}) == 5

    def test_fstring(self):
        assert safe_eval('f"http://h/{event[\'p\']}"',
                         {"event": {"p": "z"}}) == "http://h/z"

    def test_dunder_blocked(self):
        with pytest.raises(UnsafeExpressionError):
            safe_eval("().__class__", {})
        with pytest.raises(UnsafeExpressionError):
            safe_eval('x.__globals__', {"x": safe_eval})

    def test_import_blocked(self):
        with pytest.raises(UnsafeExpressionError):
            safe_eval('__import__("os").system("true")', {})

    def test_unknown_name_blocked(self):
        with pytest.raises(UnsafeExpressionError):
            safe_eval("open('/etc/passwd')", {})

    def test_lambda_blocked(self):
        with pytest.raises(UnsafeExpressionError):
            safe_eval("(lambda: 1)()", {})

    def test_stop_condition_path(self):
        from mlrun_amd.runtimes.generators import GridGenerator
        from mlrun_amd.model import HyperParamOptions

        gen = GridGenerator({"p": [1, 2]},
                            HyperParamOptions(
                                stop_condition="accuracy > 0.9"))
        assert gen.eval_stop_condition({"accuracy": 0.95}) is True
        assert gen.eval_stop_condition({"accuracy": 0.5}) is False
        # malicious condition evaluates to False, not code execution
        evil = GridGenerator({"p": [1]}, HyperParamOptions(
            stop_condition="().__class__.__mro__"))
        assert evil.eval_stop_condition({}) is False
