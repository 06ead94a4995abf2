# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Tests for the parity modules: sources/targets, secrets, data_types,
render, track, hub."""

import os

import numpy as np
import pandas as pd
import pytest

from mlrun_amd import feature_store as fstore
from mlrun_amd.datastore.sources import (
    CSVSource,
    DataFrameSource,
    ParquetSource,
    StreamSource,
)
from mlrun_amd.datastore.targets import (
    CSVTarget,
    NoSqlTarget,
    ParquetTarget,
    SQLTarget,
    get_target_from_spec,
)


@pytest.fixture(autouse=True)
def _reset_tables():
    fstore.reset_online_tables()
    yield
    fstore.reset_online_tables()


def _df(n=20):
    rng = np.random.default_rng(7)
    return pd.DataFrame({"id": [f"k{i % 4}" for i in range(n)],
                         "v": rng.normal(size=n)})


class TestSources:
    def test_csv_source(self, tmp_path):
        path = tmp_path / "in.csv"
        _df().to_csv(path, index=False)
        src = CSVSource(path=str(path))
        assert len(src.to_dataframe()) == 20

    def test_parquet_source_time_filter(self, tmp_path):
        df = _df()
        df["ts"] = pd.date_range("2026-01-01", periods=len(df), freq="h")
        path = tmp_path / "in.parquet"
        df.to_parquet(path)
        src = ParquetSource(path=str(path), time_field="ts")
        out = src.to_dataframe(start_time="2026-01-01 05:00")
        assert len(out) == 15

    def test_stream_source(self):
        src = StreamSource()
        src.push([{"a": 1}, {"a": 2}])
        src.push({"a": 3})
        df = src.to_dataframe()
        assert list(df["a"]) == [1, 2, 3]
        assert src.drain() == []

    def test_ingest_with_source_and_targets(self, tmp_path):
        df = _df()
        fset = fstore.FeatureSet("src-tgt", entities=["id"])
        out_parquet = tmp_path / "out.parquet"
        out_csv = tmp_path / "out.csv"
        fstore.ingest(fset, DataFrameSource(df),
                      targets=[ParquetTarget(path=str(out_parquet)),
                               CSVTarget(path=str(out_csv)),
                               NoSqlTarget()])
        assert os.path.isfile(out_parquet)
        assert os.path.isfile(out_csv)
        table = fstore.get_online_table(fset)
        assert table.get([{"id": "k0"}])[0]["v"] is not None


class TestTargets:
    def test_sql_target(self, tmp_path):
        fset = fstore.FeatureSet("sqlfs", entities=["id"])
        target = SQLTarget(path=str(tmp_path / "t.db"))
        target.write_dataframe(_df(), fset)
        import sqlite3

        conn = sqlite3.connect(target.path)
        count = conn.execute("SELECT COUNT(*) FROM sqlfs").fetchone()[0]
        assert count == 20

    def test_target_from_spec(self):
        assert get_target_from_spec("parquet").kind == "parquet"
        assert get_target_from_spec({"kind": "csv"}).kind == "csv"
        target = ParquetTarget(path="/x")
        assert get_target_from_spec(target) is target


class TestSecrets:
    def test_layers(self, monkeypatch, tmp_path):
        from mlrun_amd.secrets import SecretsStore, get_secret_or_env

        store = SecretsStore()
        store.add_source("inline", {"A": "1"})
        monkeypatch.setenv("MYENV", "2")
        store.add_source("env", "MYENV")
        secret_file = tmp_path / "s.env"
        secret_file.write_text("FILEKEY=3\n# comment\n")
        store.add_source("file", str(secret_file))
        assert store.get("A") == "1"
        assert store.get("MYENV") == "2"
        assert store.get("FILEKEY") == "3"
        monkeypatch.setenv("MLRUN_SECRET_TOK", "s3cr3t")
        assert get_secret_or_env("TOK") == "s3cr3t"
        with pytest.raises(Exception):
            store.add_source("vault", {})


class TestDataTypes:
    def test_infer_and_stats(self):
        from mlrun_amd.data_types import (
            ValueType, get_df_preview, get_df_stats, infer_schema_from_df,
            InferOptions)

        df = pd.DataFrame({"x": [1.0, 2.0, 3.0], "s": ["a", "b", "a"],
                           "n": [1, 2, 3]})
        schema = infer_schema_from_df(df)
        assert schema["features"]["x"] == ValueType.DOUBLE
        assert schema["features"]["s"] == ValueType.STRING
        stats = get_df_stats(df, InferOptions.all())
        assert stats["x"]["mean"] == 2.0
        assert "hist" in stats["x"]
        assert stats["s"]["unique"] == 2
        preview = get_df_preview(df)
        assert preview[0] == ["x", "s", "n"]


class TestRender:
    def test_tables(self):
        from mlrun_amd.render import artifacts_to_html, runs_to_html

        markup = runs_to_html([{"metadata": {"uid": "abc123", "name": "r"},
                                "status": {"state": "completed",
                                           "results": {"acc": 1}}}],
                              display=False)
        assert "<table" in markup and "completed" in markup
        markup = artifacts_to_html(
            [{"kind": "model", "metadata": {"key": "m", "tree": "t"},
              "spec": {"target_path": "/x"}}], display=False)
        assert "model" in markup


class TestTrack:
    def test_manager_noop_without_mlflow(self):
        from mlrun_amd.track import get_trackers_manager

        manager = get_trackers_manager()
        # no mlflow in the image: hooks are safe no-ops
        manager.pre_run(None)
        manager.post_run(None)


class TestHub:
    def test_hub_import(self, tmp_path, monkeypatch):
        import yaml

        import mlrun_amd
        from mlrun_amd import hub

        src_dir = tmp_path / "hubsrc" / "myfunc"
        src_dir.mkdir(parents=True)
        (src_dir / "function.yaml").write_text(yaml.safe_dump({
            "kind": "job", "metadata": {"name": "myfunc"},
            "spec": {"command": ""}}))
        hub.add_hub_source("testsrc", str(tmp_path / "hubsrc"))
        fn = mlrun_amd.import_function("hub://testsrc/myfunc")
        assert fn.metadata.name == "myfunc"
        catalog = hub.get_hub_catalog("testsrc")
        assert catalog[0]["name"] == "myfunc"


class TestFrameworksAuto:
    def test_detect_and_server_table(self):
        import torch

        from mlrun_amd.frameworks import (
            apply_mlrun, detect_framework, get_model_server_class)

        assert detect_framework(torch.nn.Linear(2, 2)) == "pytorch"
        from sklearn.linear_model import LinearRegression

        assert detect_framework(LinearRegression()) == "sklearn"
        assert get_model_server_class("llama").__name__ == "LlamaServer"
        assert get_model_server_class(
            "sklearn").__name__ == "SKLearnModelServer"

    def test_sklearn_interface(self, rundb):
        import numpy as np
        from sklearn.linear_model import LinearRegression

        import mlrun_amd
        from mlrun_amd.frameworks import apply_mlrun

        ctx = mlrun_amd.get_or_create_ctx("sk")
        model = LinearRegression()
        x = np.arange(10).reshape(-1, 1)
        y = 2 * np.arange(10)
        model.fit(x, y)
        iface = apply_mlrun(model, context=ctx)
        iface.log_model()
        results = iface.evaluate_and_log(
            x, y, [lambda a, b: float(np.abs(a - b).mean())])
        assert list(results.values())[0] < 1e-6

    def test_sklearn_gbdt_server_cpu(self):
        """SKLearn tree ensembles export to the tree kernel path."""
        import numpy as np
        from sklearn.ensemble import GradientBoostingRegressor

        import mlrun_amd
        from mlrun_amd.frameworks.tree import TreeEnsembleModel

        rng = np.random.default_rng(0)
        x = rng.normal(size=(200, 4))
        y = x[:, 0] * 2 + x[:, 1]
        model = GradientBoostingRegressor(n_estimators=20,
                                          max_depth=3).fit(x, y)
        gpu_model = TreeEnsembleModel.from_sklearn(model)
        ours = gpu_model.predict(x[:20]).numpy()
        theirs = model.predict(x[:20])
        assert np.abs(ours - theirs).max() < 1e-4


class TestPlatforms:
    def test_output_stream(self):
        from mlrun_amd.platforms import OutputStream

        stream = OutputStream("proj/events")
        stream.push({"a": 1})
        stream.push([{"a": 2}, {"a": 3}])
        same = OutputStream.get_stream("proj/events")
        assert len(same.drain()) == 3

    def test_mount_noops(self):
        import mlrun_amd

        fn = mlrun_amd.new_function(name="m", kind="job")
        assert mlrun_amd.auto_mount()(fn) is fn


class TestMlrunAlias:
    def test_import_mlrun(self):
        """Reference user code (`import mlrun`) runs unchanged."""
        import mlrun
        import mlrun.feature_store as fstore
        from mlrun.serving import V2ModelServer  # noqa: F401

        assert mlrun.get_version()
        run = mlrun.run_local(handler=lambda context: context.log_result(
            "ok", 1), name="alias-run")
        assert run.status.results["ok"] == 1
        assert fstore.FeatureSet("a", entities=["k"]).name == "a"


class TestBuiltinHub:
    def test_describe_function(self, tmp_path):
        import pandas as pd

        import mlrun_amd

        df = pd.DataFrame({"a": [1.0, 2.0, 3.0] * 10,
                           "label": [0, 1, 0] * 10})
        path = tmp_path / "d.parquet"
        df.to_parquet(path)
        fn = mlrun_amd.import_function("hub://describe")
        fn.deploy(watch=False)  # materialize embedded source
        run = fn.run(handler="describe", inputs={"table": str(path)},
                     params={"label_column": "label"}, local=True)
        assert run.status.state == "completed", run.status.error
        assert run.status.results["rows"] == 30

    def test_sklearn_trainer(self, tmp_path):
        import numpy as np
        import pandas as pd

        import mlrun_amd

        rng = np.random.default_rng(0)
        df = pd.DataFrame({"x1": rng.normal(size=200),
                           "x2": rng.normal(size=200)})
        df["label"] = df.x1 * 2 + df.x2
        path = tmp_path / "train.parquet"
        df.to_parquet(path)
        fn = mlrun_amd.import_function("hub://sklearn-trainer")
        fn.deploy(watch=False)
        run = fn.run(handler="train", inputs={"dataset": str(path)},
                     params={"n_estimators": 20}, local=True)
        assert run.status.state == "completed", run.status.error
        assert run.status.results["mae"] < 1.0
        assert run.output("model").startswith("store://")


class TestHubBuiltins:
    def test_batch_infer_end_to_end(self, rundb, tmp_path):
        import numpy as np
        import pandas as pd
        from sklearn.ensemble import GradientBoostingRegressor

        import mlrun_amd
        from mlrun_amd.frameworks import TreeEnsembleModel

        rng = np.random.default_rng(0)
        x = rng.normal(size=(64, 4)).astype(np.float32)
        y = (x[:, 0] * 2 + x[:, 1]).astype(np.float32)
        skl = GradientBoostingRegressor(n_estimators=10,
                                        max_depth=3).fit(x, y)
        model = TreeEnsembleModel.from_sklearn(skl)
        path = tmp_path / "model.npz"
        model.save(str(path))

        log_fn = mlrun_amd.new_function(name="log-model", kind="local")

        def log_handler(context):
            context.log_model("gbdt", body=open(path, "rb").read(),
                              model_file="model.npz",
                              framework="tree")

        run = log_fn.run(handler=log_handler, local=True)
        model_uri = run.outputs["gbdt"]

        fn = mlrun_amd.import_function("hub://batch-infer")
        df = pd.DataFrame(x, columns=[f"f{i}" for i in range(4)])
        namespace = {}
        exec(compile(fn.spec.build["functionSourceCode"],
                     "<hub>", "exec"), namespace)
        ctx_run = log_fn.run(handler=lambda context: namespace[
            "batch_infer"](context, model_uri=model_uri, dataset=df),
            local=True, name="bi")
        assert ctx_run.outputs["count"] == 64
        assert "predictions" in ctx_run.outputs

    def test_llm_serving_hub_spec_loads(self, rundb):
        import mlrun_amd

        fn = mlrun_amd.import_function("hub://llm-serving")
        assert fn.kind == "serving"
        graph = fn.spec.graph
        assert "llm" in graph.routes


class TestSecretsSources:
    def test_inline_env_file_layering(self, tmp_path, monkeypatch):
        from mlrun_amd.secrets import SecretsStore

        store = SecretsStore()
        store.add_source("inline", {"A": "1"})
        monkeypatch.setenv("FROM_ENV", "2")
        store.add_source("env", "FROM_ENV")
        secrets_file = tmp_path / "secrets.env"
        secrets_file.write_text("B=3\n# comment=skip\nC=4\n")
        store.add_source("file", str(secrets_file))
        assert store.get("A") == "1"
        assert store.get("FROM_ENV") == "2"
        assert store.get("B") == "3" and store.get("C") == "4"
        monkeypatch.setenv("MLRUN_SECRET_D", "5")
        assert store.get("D") == "5"  # env-prefix fallback
        assert store.get("missing", "dflt") == "dflt"

    def test_unavailable_providers_raise(self):
        import pytest as _pytest

        from mlrun_amd.errors import MLRunInvalidArgumentError
        from mlrun_amd.secrets import SecretsStore

        for kind in ("vault", "azure_vault", "kubernetes", "bogus"):
            with _pytest.raises(MLRunInvalidArgumentError):
                SecretsStore().add_source(kind, {})

    def test_get_secret_or_env(self, monkeypatch):
        from mlrun_amd.secrets import get_secret_or_env

        monkeypatch.setenv("SOME_TOKEN", "tok")
        assert get_secret_or_env("SOME_TOKEN") == "tok"
        assert get_secret_or_env("NOPE_X", default="d") == "d"


class TestAliasEquivalence:
    def test_mlrun_is_mlrun_amd(self):
        """`import mlrun` must expose the SAME objects (drop-in)."""
        import mlrun
        import mlrun_amd

        assert mlrun.new_function is mlrun_amd.new_function
        assert mlrun.code_to_function is mlrun_amd.code_to_function
        from mlrun.feature_store import FeatureSet as A
        from mlrun_amd.feature_store import FeatureSet as B

        assert A is B
        from mlrun.serving import V2ModelServer as SA
        from mlrun_amd.serving import V2ModelServer as SB

        assert SA is SB
        fn = mlrun.new_function(name="alias-fn", kind="local")
        assert type(fn).__module__.startswith("mlrun_amd")


class TestTorchCallbacks:
    """Reference pytorch callbacks analog: epoch hooks, checkpoints,
    early stopping (frameworks/pytorch/callbacks)."""

    def _train(self, ctx, callbacks, epochs=6):
        import torch

        from mlrun_amd.frameworks.torch_nn import apply_mlrun

        torch.manual_seed(3)
        model = torch.nn.Linear(4, 1)
        iface = apply_mlrun(model, context=ctx, auto_ddp=False)
        x, y = torch.randn(16, 4), torch.randn(16, 1)
        return iface.train([(x, y)], torch.nn.MSELoss(),
                           torch.optim.SGD(model.parameters(), lr=0.1),
                           epochs=epochs, callbacks=callbacks,
                           auto_log_model=False)

    def test_logging_and_checkpoint_callbacks(self, rundb):
        import mlrun_amd
        from mlrun_amd.frameworks.torch_nn import (
            CheckpointCallback,
            MLRunLoggingCallback,
        )

        ctx = mlrun_amd.get_or_create_ctx("cb-test")
        self._train(ctx, [MLRunLoggingCallback(log_model=True),
                          CheckpointCallback(every=3)], epochs=6)
        assert "epoch_0_loss" in ctx.results
        assert "epoch_5_loss" in ctx.results
        keys = {a.get("metadata", {}).get("key")
                for a in rundb.list_artifacts(
                    project=ctx.project or "default")}
        assert "model" in keys
        assert "checkpoint-epoch-2" in keys
        assert "checkpoint-epoch-5" in keys

    def test_early_stopping(self, rundb):
        import mlrun_amd
        from mlrun_amd.frameworks.torch_nn import EarlyStoppingCallback

        ctx = mlrun_amd.get_or_create_ctx("es-test")

        class _AlwaysWorse(EarlyStoppingCallback):
            def on_epoch_end(self, interface, epoch, results):
                results = dict(results, loss=1.0 + epoch)  # worsening
                super().on_epoch_end(interface, epoch, results)

        history = self._train(ctx, [_AlwaysWorse(patience=2)],
                              epochs=50)
        assert len(history["loss"]) < 50  # stopped early


class TestSignatureParity:
    """Lock in drop-in signature parity: every parameter of the
    reference's core entrypoints exists on ours (extra params are
    allowed; the reference source is parsed textually)."""

    CHECKS = [
        ("run.py", "new_function", None),
        ("run.py", "code_to_function", None),
        ("run.py", "get_or_create_ctx", None),
        ("run.py", "import_function", None),
        ("projects/project.py", "new_project", None),
        ("projects/project.py", "get_or_create_project", None),
    ]
    METHOD_CHECKS = [
        ("runtimes/base.py", "run",
         "mlrun_amd.runtimes.base.BaseRuntime"),
        ("projects/project.py", "set_function",
         "mlrun_amd.projects.project.MlrunProject"),
        ("projects/project.py", "run_function",
         "mlrun_amd.projects.project.MlrunProject"),
        ("projects/project.py", "log_model",
         "mlrun_amd.projects.project.MlrunProject"),
        ("feature_store/feature_set.py", "add_aggregation",
         "mlrun_amd.feature_store.feature_set.FeatureSet"),
        ("feature_store/api.py", "get_offline_features", None),
        ("feature_store/api.py", "get_online_feature_service", None),
        ("feature_store/api.py", "ingest", None),
        ("execution.py", "log_artifact",
         "mlrun_amd.execution.MLClientCtx"),
        ("execution.py", "log_model",
         "mlrun_amd.execution.MLClientCtx"),
    ]
    REF = "/root/reference/mlrun/"

    @staticmethod
    def _ref_params(path, name, method):
        import os
        import re

        ref_file = os.path.join("/root/reference/mlrun", path)
        if not os.path.isfile(ref_file):
            return None
        src = open(ref_file).read()
        pattern = (rf'\n    def {name}\(\s*self,?(.*?)\)( ->|:)'
                   if method else rf'\ndef {name}\((.*?)\)( ->|:)')
        match = re.search(pattern, src, re.S)
        if not match:
            return None
        text = re.sub(r"\s+", " ", match.group(1))
        out, depth, cur = set(), 0, ""
        for ch in text + ",":
            if ch in "([{":
                depth += 1
            if ch in ")]}":
                depth -= 1
            if ch == "," and depth == 0:
                token = cur.split("=")[0].split(":")[0].strip().lstrip("*")
                if token:
                    out.add(token)
                cur = ""
            else:
                cur += ch
        return out - {"self", "", "kwargs", "class_args"}

    def _ours(self, dotted, name):
        import importlib
        import inspect

        if dotted is None:
            import mlrun_amd

            target = getattr(mlrun_amd, name, None)
            if target is None:
                module = importlib.import_module(
                    "mlrun_amd.feature_store.api")
                target = getattr(module, name)
        else:
            module_name, _, cls_name = dotted.rpartition(".")
            cls = getattr(importlib.import_module(module_name), cls_name)
            target = getattr(cls, name)
        return set(inspect.signature(target).parameters)

    def test_core_entrypoints(self):
        import pytest as _pytest

        failures = []
        for path, name, method in self.CHECKS + self.METHOD_CHECKS:
            ref = self._ref_params(path, name, method)
            if ref is None:
                continue  # reference moved; skip silently
            ours = self._ours(method, name) | {"kwargs"}
            missing = ref - ours
            if missing:
                failures.append((name, sorted(missing)))
        assert not failures, failures


class TestTopLevelHelpers:
    def test_set_env_from_file(self, tmp_path):
        import os

        import mlrun

        env_file = tmp_path / "test.env"
        env_file.write_text("# comment\nMY_TEST_VAR=abc\nOTHER=1\n")
        result = mlrun.set_env_from_file(str(env_file), return_dict=True)
        assert result == {"MY_TEST_VAR": "abc", "OTHER": "1"}
        assert os.environ["MY_TEST_VAR"] == "abc"
        os.environ.pop("MY_TEST_VAR", None)
        os.environ.pop("OTHER", None)

    def test_set_env_from_file_errors(self, tmp_path):
        import pytest as _pytest

        import mlrun
        from mlrun_amd.errors import (MLRunInvalidArgumentError,
                                      MLRunNotFoundError)

        with _pytest.raises(MLRunNotFoundError):
            mlrun.set_env_from_file(str(tmp_path / "nope.env"))
        bad = tmp_path / "bad.env"
        bad.write_text("NOT A PAIR\n")
        with _pytest.raises(MLRunInvalidArgumentError):
            mlrun.set_env_from_file(str(bad))

    def test_get_sample_path(self):
        import mlrun

        assert mlrun.get_sample_path("data/x.csv").endswith("data/x.csv")
        base = mlrun.get_sample_path()
        assert base.startswith("http")

    def test_packager_exports(self):
        import mlrun

        assert hasattr(mlrun, "Packager")
        assert hasattr(mlrun, "DefaultPackager")
        assert hasattr(mlrun, "ArtifactType")


class TestExportSurfaceBatch:
    """Round-2 export-surface additions (reference subpackage
    __init__ exports)."""

    def test_serving_exports(self):
        from mlrun_amd.serving import (ErrorStep, MLModelServer,
                                       MonitoringApplicationStep,
                                       new_v1_model_server)

        step = ErrorStep(handler="h", name="err")
        step.before = ["echo"]
        d = step.to_dict()
        assert d["kind"] == "error_step" and d["before"] == ["echo"]
        from mlrun_amd.serving.states import step_from_dict

        back = step_from_dict(d)
        assert isinstance(back, ErrorStep) and back.before == ["echo"]
        assert MonitoringApplicationStep(
            handler="h").to_dict()["kind"] == "monitoring_application"
        assert MLModelServer is not None
        fn = new_v1_model_server("v1srv", model_class="MyModel")
        assert fn.kind == "serving"

    def test_error_handler_before_both_engines(self):
        import mlrun_amd

        def raising_step(x):
            raise ValueError("boom")

        def handle_error(event):
            event.body = {"handled": str(event.error)}
            return event

        def echo(x):
            return {"echo": x}

        namespace = {"raising_step": raising_step,
                     "handle_error": handle_error, "echo": echo}
        for engine in ("sync", "async"):
            fn = mlrun_amd.new_function("g", kind="serving")
            graph = fn.set_topology("flow", engine=engine)
            graph.to(name="raise", handler="raising_step") \
                .error_handler(name="error_catcher",
                               handler="handle_error",
                               full_event=True, before="echo")
            graph.add_step(name="echo", handler="echo",
                           after="raise").respond()
            server = fn.to_mock_server(namespace=namespace)
            resp = server.test("/", body={"a": 1})
            assert resp["echo"]["handled"] == "ValueError: boom", \
                (engine, resp)
            server.graph.shutdown()

    def test_artifacts_helpers(self, tmp_path):
        import pandas as pd

        from mlrun_amd.artifacts import (DirArtifact, TableArtifact,
                                         dict_to_artifact)

        table = TableArtifact("t1", df=pd.DataFrame({"a": [1, 2]}))
        assert "a" in table.get_body()
        art = dict_to_artifact({"kind": "dir",
                                "metadata": {"key": "d1"}})
        assert isinstance(art, DirArtifact)

    def test_update_model_and_dataset_meta(self, tmp_path):
        import mlrun_amd
        import mlrun_amd.db as db_mod
        from mlrun_amd.artifacts import update_model
        from mlrun_amd.db.sqldb import SQLRunDB

        db = SQLRunDB(str(tmp_path / "m.db"))
        prev = db_mod._run_db
        db_mod.set_run_db(db)
        try:
            from mlrun_amd.config import config as _cfg

            prev_ap = _cfg.artifact_path
            _cfg.artifact_path = str(tmp_path / "arts")
            ctx = mlrun_amd.get_or_create_ctx("upd", project="p")
            model_file = tmp_path / "model.bin"
            model_file.write_bytes(b"weights")
            model = ctx.log_model("mymodel",
                                  model_file=str(model_file))
            update_model(model.uri, metrics={"accuracy": 0.99},
                         labels={"stage": "prod"})
            stored = db.read_artifact("mymodel", project="p")
            assert stored["spec"]["metrics"]["accuracy"] == 0.99
            assert stored["metadata"]["labels"]["stage"] == "prod"
        finally:
            _cfg.artifact_path = prev_ap
            db_mod._run_db = prev
            db_mod._run_db_pinned = False

    def test_datastore_helpers(self):
        from mlrun_amd.datastore import (CSVSource, ParquetTarget,
                                         get_store_uri,
                                         get_stream_pusher,
                                         is_store_uri, parse_kafka_url)

        assert is_store_uri("store://artifacts/p/k")
        assert not is_store_uri("/tmp/x")
        assert get_store_uri("models", "p/m") == "store://models/p/m"
        topic, brokers = parse_kafka_url("kafka://b1:9092/topic")
        assert topic == "topic" and brokers == ["b1:9092"]
        pusher = get_stream_pusher("teststream")
        pusher.push({"x": 1})
        assert len(pusher.drain()) == 1
        assert CSVSource is not None and ParquetTarget is not None

    def test_feature_store_exports(self, tmp_path):
        import mlrun_amd.db as db_mod
        import mlrun_amd.feature_store as fstore
        from mlrun_amd.db.sqldb import SQLRunDB

        assert fstore.FixedWindowType.LastClosedWindow.value == 2
        config = fstore.RunConfig(local=True, parameters={"x": 1})
        config.with_secret("inline", {"k": "v"})
        assert config.secret_sources[0]["kind"] == "inline"
        db = SQLRunDB(str(tmp_path / "fs.db"))
        prev = db_mod._run_db
        db_mod.set_run_db(db)
        try:
            fset = fstore.FeatureSet("fs1", entities=["uid"])
            db.store_feature_set(fset.to_dict(), name="fs1",
                                 project="p")
            got = fstore.get_feature_set("p/fs1")
            assert got.metadata.name == "fs1"
            fstore.delete_feature_set("fs1", project="p", force=True)
        finally:
            db_mod._run_db = prev
            db_mod._run_db_pinned = False

    def test_model_monitoring_objects(self):
        from mlrun_amd.model_monitoring import (ModelEndpoint,
                                                TrackingPolicy,
                                                get_stream_path)

        ep = ModelEndpoint()
        ep.spec.function_uri = "p/f:latest"
        ep.spec.model = "m:v1"
        uid = ep.create_endpoint_id()
        assert len(uid) == 40
        back = ModelEndpoint.from_dict(ep.to_dict())
        assert back.spec.model == "m:v1"
        assert TrackingPolicy().base_period == 10
        assert get_stream_path("p").endswith("p/stream")

    def test_db_and_runtime_helpers(self):
        from mlrun_amd.db import RunDBError, get_or_set_dburl
        from mlrun_amd.runtimes import (RunError, ServingRuntime,
                                        is_local_runtime,
                                        new_model_server)

        assert issubclass(RunDBError, Exception)
        assert issubclass(RunError, Exception)
        assert is_local_runtime("local") and not is_local_runtime("job")
        assert get_or_set_dburl() is not None
        fn = new_model_server("msrv", model_class="MyServer",
                              models={"m": "/tmp/m"})
        assert fn.kind == "serving" and ServingRuntime is not None


class TestExportSurfaceBatch2:
    def test_get_store_resource_feature_set(self, tmp_path):
        import mlrun_amd.db as db_mod
        import mlrun_amd.feature_store as fstore
        from mlrun_amd.datastore import get_store_resource
        from mlrun_amd.db.sqldb import SQLRunDB

        db = SQLRunDB(str(tmp_path / "sr.db"))
        prev = db_mod._run_db
        db_mod.set_run_db(db)
        try:
            fset = fstore.FeatureSet("fs1", entities=["uid"])
            db.store_feature_set(fset.to_dict(), name="fs1",
                                 project="p")
            got = get_store_resource("store://feature-sets/p/fs1")
            assert got.metadata.name == "fs1"
        finally:
            db_mod._run_db = prev
            db_mod._run_db_pinned = False

    def test_in_memory_items(self):
        from mlrun_amd.datastore import (get_in_memory_items,
                                         set_in_memory_item)

        item = set_in_memory_item("box/data", b"abc123")
        assert item.get() == b"abc123"
        assert any(k.startswith("box/")
                   for k in get_in_memory_items())

    def test_plotly_artifact_rejects_non_figures(self):
        import pytest as _pytest

        from mlrun_amd.artifacts import PlotlyArtifact
        from mlrun_amd.errors import MLRunInvalidArgumentError

        with _pytest.raises(MLRunInvalidArgumentError):
            PlotlyArtifact(figure=object(), key="p1")

    def test_run_config_to_function(self):
        import mlrun_amd
        from mlrun_amd.feature_store import RunConfig

        fn = mlrun_amd.new_function("ing", kind="job")
        config = RunConfig(fn, parameters={"x": 1})
        materialized = config.to_function()
        assert materialized.kind == "job"
        config2 = RunConfig(kind="job", image="mlrun/mlrun")
        fn2 = config2.to_function()
        assert fn2.spec.image == "mlrun/mlrun"

    def test_runtime_aliases(self):
        from mlrun_amd.runtimes import (MPIJobCRDVersions, MpiRuntime,
                                        MpiRuntimeV1,
                                        new_v2_model_server)

        assert MpiRuntimeV1 is MpiRuntime
        assert MPIJobCRDVersions.default() == "v1"
        fn = new_v2_model_server("v2srv", model_class="S")
        assert fn.kind == "serving"

    def test_http_output_stream_and_source(self):
        import json
        import threading
        from http.server import BaseHTTPRequestHandler, HTTPServer

        from mlrun_amd.datastore import get_stream_pusher

        received = []

        class Handler(BaseHTTPRequestHandler):
            def do_POST(self):
                n = int(self.headers.get("content-length", 0))
                received.append(json.loads(self.rfile.read(n)))
                self.send_response(200)
                self.end_headers()

            def log_message(self, *a):
                pass

        httpd = HTTPServer(("127.0.0.1", 0), Handler)
        port = httpd.server_address[1]
        threading.Thread(target=httpd.serve_forever,
                         daemon=True).start()
        try:
            pusher = get_stream_pusher(f"http://127.0.0.1:{port}/s")
            pusher.push({"k": 1})
            assert received == [{"k": 1}]
        finally:
            httpd.shutdown()


class TestAPIGateway:
    """Node-local APIGateway proxy: reference-shape construction +
    run-DB persistence (reference nuclio/api_gateway.py:340)."""

    def test_from_schema_and_canary_validation(self):
        import pytest as _pytest

        from mlrun_amd.errors import MLRunInvalidArgumentError
        from mlrun_amd.runtimes import (APIGateway, APIGatewayMetadata,
                                        APIGatewaySpec)

        gw = APIGateway.from_schema(
            APIGatewayMetadata(name="gw"),
            APIGatewaySpec(functions=["http://127.0.0.1:1/a",
                                      "http://127.0.0.1:2/b"],
                           project="p", canary=[80, 20]))
        assert [u["percent"] for u in gw.upstreams] == [80, 20]
        with _pytest.raises(MLRunInvalidArgumentError):
            APIGateway(name="bad").with_canary(
                ["http://x", "http://y"], [60, 60])

    def test_save_and_delete_persistence(self, tmp_path):
        import mlrun_amd.db as db_mod
        from mlrun_amd.db.sqldb import SQLRunDB
        from mlrun_amd.runtimes import APIGateway

        db = SQLRunDB(str(tmp_path / "gw.db"))
        prev = db_mod._run_db
        db_mod.set_run_db(db)
        try:
            gw = APIGateway(name="gw1", project="p")
            gw.add_upstream("http://127.0.0.1:9/f1")
            gw.save()
            stored = db.get_api_gateway("p", "gw1")
            assert stored["spec"]["upstreams"][0]["address"] == \
                "http://127.0.0.1:9/f1"
            gw.delete()
            assert db.list_api_gateways("p") == []
        finally:
            db_mod._run_db = prev
            db_mod._run_db_pinned = False
