# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Core spine tests: config, model objects, DB, datastore, artifacts."""

import json
import os

import pytest

from mlrun_amd.config import Config, read_env
from mlrun_amd.errors import MLRunNotFoundError
from mlrun_amd.model import (
    RunObject,
    RunSpec,
    RunStates,
    RunTemplate,
    new_task,
)


class TestConfig:
    def test_defaults(self, config_test_base):
        cfg = config_test_base
        assert cfg.default_project == "default"
        assert cfg.gpu.arch == "gfx950"
        assert int(cfg.gpu.devices_per_node) == 8

    def test_env_override(self):
        overrides = read_env({"MLRUN_HTTPDB__PORT": "9999",
                              "MLRUN_LOG_LEVEL": '"DEBUG"',
                              "MLRUN_DBPATH": "http://localhost:8080"})
        assert overrides["httpdb"]["port"] == 9999
        assert overrides["log_level"] == "DEBUG"
        assert overrides["dbpath"] == "http://localhost:8080"

    def test_nested_set(self):
        cfg = Config({"a": {"b": 1}})
        assert cfg.a.b == 1
        cfg.update({"a": {"c": 2}})
        assert cfg.a.b == 1 and cfg.a.c == 2


class TestModelObjects:
    def test_task_roundtrip(self):
        task = new_task(name="t1", project="p1", params={"x": 5},
                        inputs={"data": "/tmp/x.csv"})
        struct = task.to_dict()
        back = RunTemplate.from_dict(struct)
        assert back.metadata.name == "t1"
        assert back.spec.parameters == {"x": 5}
        assert back.spec.inputs == {"data": "/tmp/x.csv"}

    def test_run_object_outputs(self):
        run = RunObject.from_template(new_task(name="r"))
        run.status.results = {"acc": 0.9}
        run.status.artifacts = [
            {"kind": "model", "metadata": {"key": "m", "project": "p",
                                           "tree": "abc"},
             "spec": {"target_path": "/tmp/m", "db_key": "m"}}]
        assert run.output("acc") == 0.9
        assert run.output("m").startswith("store://")
        outputs = run.outputs
        assert set(outputs) == {"acc", "m"}

    def test_hyper_params(self):
        task = new_task(name="h").with_hyper_params(
            {"p1": [1, 2]}, selector="max.acc", strategy="grid")
        assert task.spec.hyperparams == {"p1": [1, 2]}
        assert task.spec.hyper_param_options.strategy == "grid"

    def test_states(self):
        assert RunStates.is_terminal(RunStates.completed)
        assert not RunStates.is_terminal(RunStates.running)


class TestSQLRunDB:
    def test_runs_crud(self, rundb):
        struct = {"metadata": {"name": "r1", "uid": "u1"},
                  "status": {"state": "running"}}
        rundb.store_run(struct, "u1", "proj")
        run = rundb.read_run("u1", "proj")
        assert run["metadata"]["name"] == "r1"
        rundb.update_run({"status.state": "completed"}, "u1", "proj")
        assert rundb.read_run("u1", "proj")["status"]["state"] == "completed"
        runs = rundb.list_runs(project="proj")
        assert len(runs) == 1
        runs = rundb.list_runs(project="proj", state="completed")
        assert len(runs) == 1
        rundb.del_run("u1", "proj")
        with pytest.raises(MLRunNotFoundError):
            rundb.read_run("u1", "proj")

    def test_list_runs_filters(self, rundb):
        for i in range(5):
            rundb.store_run(
                {"metadata": {"name": f"run{i}",
                              "labels": {"kind": "job" if i % 2 else "local"}},
                 "status": {"state": "completed"}}, f"uid{i}", "p")
        assert len(rundb.list_runs(project="p")) == 5
        assert len(rundb.list_runs(project="p", labels={"kind": "job"})) == 2
        assert len(rundb.list_runs(project="p", last=3)) == 3
        assert len(rundb.list_runs(project="p", name="run1")) == 1

    def test_artifacts_crud(self, rundb):
        artifact = {"kind": "artifact", "metadata": {"key": "a1"},
                    "spec": {"target_path": "/tmp/a1"}}
        rundb.store_artifact("a1", artifact, project="p", tree="t1",
                             tag="v1")
        read = rundb.read_artifact("a1", project="p", tag="v1")
        assert read["spec"]["target_path"] == "/tmp/a1"
        # latest tag also resolves
        read = rundb.read_artifact("a1", project="p")
        assert read["metadata"]["key"] == "a1"
        assert len(rundb.list_artifacts(project="p")) == 1
        rundb.del_artifact("a1", project="p")
        with pytest.raises(MLRunNotFoundError):
            rundb.read_artifact("a1", project="p")

    def test_functions_crud(self, rundb):
        hash_key = rundb.store_function(
            {"kind": "job", "metadata": {"name": "f1"}}, "f1", "p",
            tag="latest", versioned=True)
        assert hash_key
        func = rundb.get_function("f1", "p")
        assert func["kind"] == "job"
        func = rundb.get_function("f1", "p", hash_key=hash_key)
        assert func["kind"] == "job"
        assert len(rundb.list_functions(project="p")) == 1
        rundb.delete_function("f1", "p")
        with pytest.raises(MLRunNotFoundError):
            rundb.get_function("f1", "p")

    def test_projects_crud(self, rundb):
        rundb.create_project({"metadata": {"name": "p1"}})
        assert rundb.get_project("p1")["metadata"]["name"] == "p1"
        from mlrun_amd.errors import MLRunConflictError

        with pytest.raises(MLRunConflictError):
            rundb.create_project({"metadata": {"name": "p1"}})
        rundb.delete_project("p1")
        with pytest.raises(MLRunNotFoundError):
            rundb.get_project("p1")

    def test_logs(self, rundb):
        rundb.store_log("u1", "p", b"hello ")
        rundb.store_log("u1", "p", b"world", append=True)
        _, log = rundb.get_log("u1", "p")
        assert log == b"hello world"

    def test_schedules(self, rundb):
        rundb.create_schedule("p", {"name": "s1", "kind": "job",
                                    "cron_trigger": "*/5 * * * *"})
        sched = rundb.get_schedule("p", "s1")
        assert sched["cron_trigger"] == "*/5 * * * *"
        assert len(rundb.list_schedules("p")) == 1
        rundb.delete_schedule("p", "s1")
        assert rundb.list_schedules("p") == []

    def test_feature_sets(self, rundb):
        rundb.store_feature_set({"metadata": {"name": "fs1"},
                                 "spec": {"entities": ["id"]}}, project="p")
        feature_set = rundb.get_feature_set("fs1", "p")
        assert feature_set["spec"]["entities"] == ["id"]
        assert len(rundb.list_feature_sets("p")) == 1


class TestDatastore:
    def test_file_store(self, tmp_path):
        from mlrun_amd.datastore import store_manager

        path = tmp_path / "x.txt"
        item = store_manager.object(str(path))
        item.put("hello")
        assert item.get(encoding="utf-8") == "hello"
        assert item.stat()["size"] == 5
        assert item.local() == str(path)

    def test_memory_store(self):
        from mlrun_amd.datastore import store_manager

        item = store_manager.object("memory://buf1")
        item.put(b"data")
        assert item.get() == b"data"
        item.delete()

    def test_dataframe(self, tmp_path):
        import pandas as pd

        from mlrun_amd.datastore import store_manager

        df = pd.DataFrame({"a": [1, 2], "b": [3.0, 4.0]})
        path = tmp_path / "d.csv"
        df.to_csv(path, index=False)
        loaded = store_manager.object(str(path)).as_df()
        assert list(loaded.columns) == ["a", "b"]
        assert len(loaded) == 2

    def test_store_uri_parse(self):
        from mlrun_amd.datastore import parse_store_uri

        kind, project, key, tag, tree, it = parse_store_uri(
            "store://artifacts/proj/mykey:v2")
        assert (kind, project, key, tag) == ("artifacts", "proj", "mykey",
                                             "v2")
        kind, project, key, tag, tree, it = parse_store_uri(
            "store://models/proj/m@tree123")
        assert tree == "tree123"


class TestArtifacts:
    def test_store_and_resolve(self, rundb, tmp_path):
        from mlrun_amd.artifacts import ArtifactManager, ArtifactProducer
        from mlrun_amd.datastore import store_manager

        manager = ArtifactManager(db=rundb)
        producer = ArtifactProducer("run", "p", "myrun", uid="tree1")
        item = manager.log_artifact(producer, "results", body="abc",
                                    artifact_path=str(tmp_path))
        assert os.path.isfile(item.spec.target_path)
        # resolve through store://
        data_item = store_manager.object("store://artifacts/p/results")
        assert data_item.get() == b"abc"

    def test_log_model_layout(self, rundb, tmp_path):
        from mlrun_amd.artifacts import (
            ArtifactManager, ArtifactProducer, get_model)

        manager = ArtifactManager(db=rundb)
        producer = ArtifactProducer("run", "p", "myrun", uid="tree2")
        model = manager.log_model(
            producer, "mymodel", body=b"\x00weights", framework="torch",
            parameters={"layers": 2}, artifact_path=str(tmp_path),
            extra_data={"config.json": json.dumps({"a": 1})})
        target = model.spec.target_path.rstrip("/")
        assert os.path.isfile(os.path.join(target, "model_spec.yaml"))
        assert os.path.isfile(os.path.join(target, "mymodel.bin"))
        assert os.path.isfile(os.path.join(target, "config.json"))
        # resolve via store uri
        model_file, spec, extra = get_model(model.uri)
        assert spec.framework == "torch"
        assert os.path.basename(model_file) == "mymodel.bin"
        assert "config.json" in extra


class TestFsspecStore:
    def test_local_protocol_roundtrip(self, tmp_path):
        from mlrun_amd.datastore import FsspecStore

        store = FsspecStore(None, "t", "file")
        path = str(tmp_path / "blob.bin")
        store.put(path, b"hello fsspec")
        assert store.get(path) == b"hello fsspec"
        assert store.stat(path)["size"] == 12
        store.put(path, b"!", append=True)
        assert store.get(path).endswith(b"fsspec!")
        store.rm(path)
        import os

        assert not os.path.exists(path)

    def test_memory_protocol(self):
        from mlrun_amd.datastore import FsspecStore

        store = FsspecStore(None, "t", "memory")
        store.put("/m/x", b"abc")
        assert store.get("/m/x") == b"abc"

    def test_scheme_registry_covers_object_stores(self):
        from mlrun_amd.datastore import FsspecStore, schema_to_store

        for scheme in ("s3", "gcs", "az", "http", "hdfs"):
            assert schema_to_store(scheme) is FsspecStore


class TestSQLRunDBConcurrency:
    def test_concurrent_writers_and_readers(self, rundb):
        import threading

        errors = []

        def writer(tid):
            try:
                for i in range(30):
                    uid = f"c{tid}-{i}"
                    rundb.store_run(
                        {"metadata": {"name": f"r{tid}", "uid": uid},
                         "status": {"state": "completed"}},
                        uid, "default")
                    rundb.read_run(uid, "default")
            except Exception as exc:
                errors.append(exc)

        threads = [threading.Thread(target=writer, args=(t,))
                   for t in range(8)]
        [t.start() for t in threads]
        [t.join(timeout=60) for t in threads]
        assert not errors, errors[:2]
        assert len(rundb.list_runs(project="default",
                                   last=1000)) >= 240


class TestUtilsHelpers:
    """Reference utils/helpers.py surface (curated public set)."""

    def test_serialization_helpers(self):
        from mlrun_amd.utils import (dict_to_json, dict_to_str,
                                     dict_to_yaml)

        assert "a: 1" in dict_to_yaml({"a": 1})
        assert dict_to_json({"a": 1}) == '{"a": 1}'
        assert dict_to_str({"a": 1, "b": 2}) == "a=1,b=2"

    def test_uri_helpers(self):
        from mlrun_amd.utils import (generate_artifact_uri,
                                     generate_object_uri,
                                     parse_artifact_uri)

        assert generate_artifact_uri("p", "k", tag="t", iter=1) == \
            "p/k#1:t"
        assert generate_object_uri("p", "f", hash_key="abc") == \
            "p/f@abc"
        assert parse_artifact_uri("p/key#2:tag@tree") == \
            ("p", "key", 2, "tag", "tree")
        assert parse_artifact_uri("key", "dflt")[0] == "dflt"

    def test_object_hash_stable(self):
        from mlrun_amd.utils import fill_object_hash

        a = {"metadata": {"name": "x", "updated": "t1"},
             "spec": {"v": 1}, "status": {"state": "ready"}}
        b = {"metadata": {"name": "x", "updated": "t2"},
             "spec": {"v": 1}, "status": {"state": "error"}}
        assert fill_object_hash(a) == fill_object_hash(b)
        c = {"metadata": {"name": "x"}, "spec": {"v": 2}}
        assert fill_object_hash(c) != fill_object_hash(a)

    def test_dynamic_loading(self):
        from mlrun_amd.utils import get_class, get_function

        assert get_class("mlrun_amd.artifacts.Artifact").kind == \
            "artifact"
        fn = get_function("mlrun_amd.utils.dict_to_json")
        assert fn({"x": 1}) == '{"x": 1}'

    def test_time_and_chunk_helpers(self):
        import pandas as pd

        from mlrun_amd.utils import (iterate_list_by_chunks,
                                     str_to_timestamp)

        assert str_to_timestamp("now - 1h") < pd.Timestamp.now()
        assert str_to_timestamp("2026-01-01").year == 2026
        assert list(iterate_list_by_chunks([1, 2, 3, 4, 5], 2)) == \
            [[1, 2], [3, 4], [5]]

    def test_path_safety_and_templates(self):
        from mlrun_amd.utils import (StorePrefix, is_safe_path,
                                     template_artifact_path)

        assert is_safe_path("/tmp/base", "/tmp/base/sub/x")
        assert not is_safe_path("/tmp/base", "/tmp/base/../etc")
        assert StorePrefix.kind_to_prefix("model") == "models"
        assert template_artifact_path(
            "/data/{{project}}/{{run.uid}}", "p", "u1") == "/data/p/u1"

    def test_retry_until_successful(self):
        from mlrun_amd.utils import retry_until_successful

        calls = {"n": 0}

        def flaky():
            calls["n"] += 1
            if calls["n"] < 3:
                raise RuntimeError("not yet")
            return "done"

        assert retry_until_successful(0.01, 5, None, False,
                                      flaky) == "done"
        assert calls["n"] == 3


class TestChildContext:
    """get_child_context / update_child_iterations / mark_as_best
    (reference execution.py:223-291)."""

    def _parent(self, tmp_path):
        import mlrun_amd.db as db_mod
        from mlrun_amd.db.sqldb import SQLRunDB
        from mlrun_amd.execution import MLClientCtx

        db = SQLRunDB(str(tmp_path / "cc.db"))
        db_mod.set_run_db(db)
        ctx = MLClientCtx.from_dict(
            {"metadata": {"name": "parent", "project": "p"},
             "spec": {"parameters": {"base": 1}}}, rundb=db)
        return ctx, db

    def test_children_and_best(self, tmp_path):
        import mlrun_amd.db as db_mod

        ctx, db = self._parent(tmp_path)
        try:
            best_acc = 0
            for i, lr in enumerate([0.1, 0.2, 0.3]):
                with ctx.get_child_context(lr=lr) as child:
                    acc = 1 - abs(lr - 0.2)  # best at 0.2
                    child.log_result("accuracy", acc)
                    if acc > best_acc:
                        child.mark_as_best()
                        best_acc = acc
                    assert child.get_param("lr") == lr
                    assert child._iteration == i + 1
            ctx.update_child_iterations(best_run=2)
            assert ctx._results["best_iteration"] == 2
            assert ctx._results["accuracy"] == 1.0
            rows = ctx._iteration_results
            assert rows[0][:2] == ["state", "iter"]
            assert "param.lr" in rows[0]
            assert len(rows) == 4
        finally:
            db_mod._run_db = None
            db_mod._run_db_pinned = False

    def test_with_parent_params_and_nesting_guard(self, tmp_path):
        import pytest as _pytest

        import mlrun_amd.db as db_mod
        from mlrun_amd.errors import MLRunInvalidArgumentError

        ctx, db = self._parent(tmp_path)
        try:
            child = ctx.get_child_context(with_parent_params=True,
                                          extra=2)
            assert child.get_param("base") == 1
            assert child.get_param("extra") == 2
            with _pytest.raises(MLRunInvalidArgumentError):
                child.get_child_context()
        finally:
            db_mod._run_db = None
            db_mod._run_db_pinned = False

    def test_child_error_marks_failed(self, tmp_path):
        import mlrun_amd.db as db_mod

        ctx, db = self._parent(tmp_path)
        try:
            with pytest.raises(ValueError):
                with ctx.get_child_context(lr=1) as child:
                    raise ValueError("boom")
            assert child._state == "error"
        finally:
            db_mod._run_db = None
            db_mod._run_db_pinned = False


class TestRunTemplateParamFile:
    def test_with_param_file(self):
        from mlrun_amd.model import new_task

        task = new_task("grid").with_param_file(
            "/tmp/params.json", selector="max.accuracy",
            strategy="grid")
        opts = task.spec.hyper_param_options
        assert opts.param_file == "/tmp/params.json"
        assert opts.selector == "max.accuracy"
        assert opts.strategy == "grid"


class TestSecretsToDict:
    def test_to_dict(self):
        from mlrun_amd.secrets import SecretsStore

        store = SecretsStore()
        store.add_source("inline", {"K": "V"})
        struct = store.to_dict()
        assert struct["secret_sources"][0]["source"]["K"] == "V"


class TestModelAuxClasses:
    """Reference model.py aux objects (ObjectDict/ObjectList,
    Credentials, ImageBuilder, entrypoints, TargetPathObject)."""

    def test_object_list(self):
        from mlrun_amd.feature_store.feature_set import Feature
        from mlrun_amd.model import ObjectList

        features = ObjectList.from_list(
            Feature, [{"name": "a", "value_type": "float"},
                      {"name": "b", "value_type": "str"}])
        assert len(features) == 2
        assert features["a"].value_type == "float"
        assert features[1].name == "b"
        assert [f.name for f in features] == ["a", "b"]
        assert features.to_dict()[0]["name"] == "a"

    def test_object_dict_kind_dispatch(self):
        from mlrun_amd.model import ObjectDict
        from mlrun_amd.serving.states import TaskStep, classes_map

        steps = ObjectDict(classes_map, default_kind="task")
        steps["s1"] = {"kind": "task", "handler": "h"}
        assert isinstance(steps["s1"], TaskStep)
        assert steps.to_dict()["s1"]["handler"] == "h"

    def test_target_path_object(self):
        from mlrun_amd.model import TargetPathObject

        tp = TargetPathObject("/data/out", run_id="r1")
        assert "{run_id}" in tp.get_templated_path()
        assert tp.get_absolute_path() == "/data/out/r1/"
        tpf = TargetPathObject("/data/out/f.pq", run_id="r2",
                               is_single_file=True)
        assert tpf.get_absolute_path() == "/data/out/r2/f.pq"

    def test_builder_and_entrypoints(self):
        from mlrun_amd.model import (Credentials, EntrypointParam,
                                     FunctionEntrypoint, ImageBuilder)

        builder = ImageBuilder(base_image="rocm/pytorch",
                               requirements=["einops"])
        back = ImageBuilder.from_dict(builder.to_dict())
        assert back.base_image == "rocm/pytorch"
        assert back.requirements == ["einops"]
        entry = FunctionEntrypoint(
            name="train", doc="trains",
            parameters=[EntrypointParam("lr", type="float",
                                        default=0.1).to_dict()])
        assert entry.to_dict()["parameters"][0]["name"] == "lr"
        assert Credentials.generate_access_key == "$generate"
