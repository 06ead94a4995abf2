# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""RunList / ArtifactList rich wrappers (reference lists.py)."""

import pytest

from mlrun_amd.db.sqldb import SQLRunDB
from mlrun_amd.lists import ArtifactList, RunList


@pytest.fixture()
def db(tmp_path):
    return SQLRunDB(str(tmp_path / "lists.db"))


def _store_runs(db):
    for i, state in enumerate(["completed", "error", "completed"]):
        db.store_run({
            "metadata": {"name": f"train-{i}", "uid": f"u{i}",
                         "project": "p",
                         "labels": {"owner": "me", "idx": str(i)}},
            "spec": {"parameters": {"lr": 0.1 * (i + 1)}},
            "status": {"state": state,
                       "start_time": f"2026-09-1{i + 1}T00:00:00",
                       "results": {"accuracy": 0.9 + i / 100}},
        }, f"u{i}", "p")


class TestRunList:
    def test_list_runs_returns_runlist(self, db):
        _store_runs(db)
        runs = db.list_runs(project="p")
        assert isinstance(runs, RunList)
        assert len(runs) == 3

    def test_to_rows_and_df(self, db):
        _store_runs(db)
        runs = db.list_runs(project="p")
        rows = runs.to_rows()
        assert rows[0][0] == "project"  # header
        assert len(rows) == 4
        df = runs.to_df()
        assert set(df["state"]) == {"completed", "error"}
        assert df["start"].notna().all()
        flat = runs.to_df(flat=True, cache=False)
        assert "param.lr" in flat.columns
        assert "output.accuracy" in flat.columns

    def test_to_objects(self, db):
        _store_runs(db)
        objs = db.list_runs(project="p").to_objects()
        assert objs[0].metadata.project == "p"
        assert {o.status.state for o in objs} == {"completed", "error"}

    def test_show_and_compare(self, db, tmp_path):
        _store_runs(db)
        runs = db.list_runs(project="p")
        html = runs.show(display=False)
        assert "train-" in html
        out = tmp_path / "cmp.html"
        cmp_html = runs.compare(filename=str(out))
        assert "lr" in cmp_html
        assert out.exists()

    def test_extend_iterations(self):
        runs = RunList([{
            "metadata": {"name": "hp", "uid": "h1", "project": "p"},
            "status": {
                "state": "completed",
                "iterations": [
                    ["state", "iter", "param.lr", "output.acc"],
                    ["completed", 1, 0.1, 0.8],
                    ["completed", 2, 0.2, 0.9],
                ]},
        }])
        rows = runs.to_rows(extend_iterations=True)
        assert len(rows) == 3  # header + 2 iterations
        from mlrun_amd.lists import run_list_header

        iter_col = run_list_header.index("iter")
        assert {r[iter_col] for r in rows[1:]} == {1, 2}


class TestArtifactList:
    def test_list_artifacts_returns_artifactlist(self, db):
        db.store_artifact("model1", {
            "kind": "model",
            "metadata": {"key": "model1", "tree": "t1",
                         "project": "p",
                         "updated": "2026-09-12T00:00:00"},
            "spec": {"target_path": "/tmp/m.bin"}}, uid="t1",
            project="p")
        artifacts = db.list_artifacts(project="p")
        assert isinstance(artifacts, ArtifactList)
        rows = artifacts.to_rows()
        assert "uri" in rows[0]
        assert rows[1][rows[0].index("key")] == "model1"
        df = artifacts.to_df()
        assert df.loc[0, "path"] == "/tmp/m.bin"
        objs = artifacts.to_objects()
        assert objs[0].kind == "model"
        html = artifacts.show(display=False)
        assert "model1" in html

    def test_dataitems(self, db, tmp_path):
        target = tmp_path / "data.csv"
        target.write_text("a,b\n1,2\n")
        db.store_artifact("ds", {
            "kind": "dataset",
            "metadata": {"key": "ds", "project": "p"},
            "spec": {"target_path": str(target)}}, uid="t2",
            project="p")
        items = db.list_artifacts(project="p").dataitems()
        assert len(items) == 1
        assert "a,b" in items[0].get().decode()


class TestParallelCoordinates:
    def test_compare_run_objects_svg(self, db, tmp_path):
        from mlrun_amd.frameworks.parallel_coordinates import (
            compare_run_objects)

        _store_runs(db)
        runs = db.list_runs(project="p").to_objects()
        out = tmp_path / "cmp.html"
        html = compare_run_objects(runs, filename=str(out),
                                   extend_iterations=False)
        assert "<svg" in html and "polyline" in html
        assert "param.lr" in html
        assert out.exists()

    def test_compare_db_runs(self, db):
        import mlrun_amd.db as db_mod
        from mlrun_amd.frameworks.parallel_coordinates import (
            compare_db_runs)

        _store_runs(db)
        prev = db_mod._run_db
        db_mod.set_run_db(db)
        try:
            html = compare_db_runs(project_name="p",
                                   run_name="~train")
        finally:
            db_mod._run_db = prev
            db_mod._run_db_pinned = False
        assert "<svg" in html

    def test_categorical_axis(self):
        import pandas as pd

        from mlrun_amd.frameworks.parallel_coordinates import (
            gen_pcp_plot)

        df = pd.DataFrame({"param.optim": ["sgd", "adam", "sgd"],
                           "output.acc": [0.8, 0.9, 0.85]})
        html = gen_pcp_plot(df, "iter")
        assert "adam" in html and "polyline" in html
