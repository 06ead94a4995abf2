# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""The golden path, end to end in one test:

project -> code_to_function -> train run (logs a model artifact in
the model_spec.yaml layout) -> serving function loading THAT artifact
-> HTTP inference with monitoring -> drift event -> alert fired.

This is the composition the reference's system tests exercise against
a live cluster (SURVEY §4 tier 4), run here node-locally."""

import numpy as np
import pytest
import torch

import mlrun_amd


TRAIN_CODE = '''
import numpy as np

from mlrun_amd.frameworks import TreeEnsembleModel


def train(context, n_samples: int = 256, n_features: int = 8):
    from sklearn.ensemble import GradientBoostingRegressor

    rng = np.random.default_rng(7)
    x = rng.normal(size=(n_samples, n_features)).astype(np.float32)
    y = (2 * x[:, 0] - x[:, 1]).astype(np.float32)
    skl = GradientBoostingRegressor(n_estimators=20,
                                    max_depth=3).fit(x, y)
    context.log_result("train_score", float(skl.score(x, y)))
    model = TreeEnsembleModel.from_sklearn(skl)
    import tempfile, os

    tmp = tempfile.mkdtemp()
    path = os.path.join(tmp, "model.npz")
    model.save(path)
    context.log_model("regressor", body=open(path, "rb").read(),
                      model_file="model.npz", framework="tree",
                      metrics={"r2": float(skl.score(x, y))})
'''


class TestGoldenPath:
    def test_project_train_serve_monitor_alert(self, rundb, tmp_path):
        # 1. project + function
        project = mlrun_amd.new_project("golden", context=str(tmp_path))
        code = tmp_path / "train.py"
        code.write_text(TRAIN_CODE)
        fn = project.set_function(str(code), name="trainer", kind="job")

        # 2. training run logs the model artifact
        run = fn.run(handler="train", local=True,
                     params={"n_samples": 128})
        assert run.status.state == "completed"
        assert run.outputs["train_score"] > 0.5
        model_uri = run.outputs["regressor"]
        assert model_uri.startswith("store://")

        # 3. serving function loads THAT artifact (store:// URI)
        serving = mlrun_amd.new_function(name="golden-srv",
                                         kind="serving")
        serving.add_model(
            "reg",
            class_name="mlrun_amd.frameworks.tree."
                       "TreeEnsembleModelServer",
            model_path=model_uri)
        serving.set_tracking()

        # 4. alert on drift events for this project
        from mlrun_amd.api.events import AlertConfig

        rundb.store_alert_config("golden", "drift-alert", AlertConfig(
            project="golden", name="drift-alert",
            summary="model drifted", severity="high",
            trigger={"events": ["model-drift"]},
            notifications=[{"kind": "console"}]).to_dict())

        address = serving.deploy()
        try:
            import requests

            rng = np.random.default_rng(3)
            x = rng.normal(size=(8, 8)).astype(np.float32)
            resp = requests.post(
                address + "/v2/models/reg/infer",
                json={"inputs": x.tolist()}, timeout=60)
            assert resp.status_code == 200
            preds = resp.json()["outputs"]
            assert len(preds) == 8

            # predictions must match the logged model exactly
            from mlrun_amd.artifacts import get_model
            from mlrun_amd.frameworks import TreeEnsembleModel

            model_file, spec, _extra = get_model(model_uri)
            model = TreeEnsembleModel.load(model_file)
            expect = model.predict(torch.from_numpy(x)).tolist()
            assert preds == pytest.approx(expect, rel=1e-5)

            # 5. monitoring recorded the traffic
            from mlrun_amd.model_monitoring import get_stream_processor

            processor = get_stream_processor("default")
            stats = processor.endpoint_stats("reg")
            assert stats["300"]["count"] >= 1  # 5m window saw traffic
            assert stats["300"]["error_count"] == 0
        finally:
            serving.stop()

        # 6. drift event fires the alert -> console notification +
        # alert state recorded
        from mlrun_amd.api.events import process_event
        from mlrun_amd.model_monitoring.drift import (
            compute_feature_drift)

        ref = np.random.default_rng(0).normal(0, 1, 1000)
        cur = np.random.default_rng(1).normal(4, 1, 1000)
        drift = compute_feature_drift(ref, cur)
        assert drift["drift_score"] > 0.5
        fired = process_event("golden", "model-drift",
                              {"endpoint": "reg",
                               "drift_score": drift["drift_score"]},
                              db=rundb)
        assert fired == ["drift-alert"]
