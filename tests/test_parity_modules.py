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
