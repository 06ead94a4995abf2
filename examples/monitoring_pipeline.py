#!/usr/bin/env python3
# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Model-monitoring pipeline example: serve traffic -> sliding-window
stats -> drift application -> alert notification.

    python examples/monitoring_pipeline.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(
    __file__))))

import numpy as np

import mlrun_amd
from mlrun_amd.db import get_run_db
from mlrun_amd.model_monitoring import (
    MonitoringController,
    get_stream_processor,
)
from mlrun_amd.serving.v2_serving import V2ModelServer


class MeanModel(V2ModelServer):
    def load(self):
        pass

    def predict(self, request):
        return [float(np.mean(request["inputs"]))]


def main():
    project = "mon-demo"
    db = get_run_db()
    db.store_alert_config(project, "drift-alert", {
        "name": "drift-alert", "project": project,
        "summary": "data drift detected on {{endpoint}}",
        "severity": "high",
        "trigger": {"events": ["model-drift"]},
        "notifications": [{"kind": "console"}],
    })

    fn = mlrun_amd.new_function("scorer", kind="serving",
                                project=project)
    fn.set_topology("router")
    fn.add_model("scorer", class_name=MeanModel, model_path=".")
    fn.set_tracking()
    server = fn.to_mock_server(track_models=True)

    rng = np.random.default_rng(0)

    def serve(mean, n):
        for _ in range(n):
            server.test("/v2/models/scorer/infer",
                        body={"inputs": rng.normal(mean, 1, 8).tolist()})

    controller = MonitoringController(project, db=db)
    serve(0.0, 50)  # healthy traffic
    processor = get_stream_processor(project)
    endpoint_id = next(iter(processor._endpoint_ids))
    controller.set_reference(endpoint_id, rng.normal(0, 1, 2000))

    results = controller.run_iteration()
    print("healthy sweep:", results[endpoint_id]["status"],
          f"(score {results[endpoint_id]['drift_score']:.3f})")

    serve(6.0, 200)  # drifted traffic
    results = controller.run_iteration()
    print("after shift:", results[endpoint_id]["status"],
          f"(score {results[endpoint_id]['drift_score']:.3f})")
    assert results[endpoint_id]["status"] == "drift_detected"

    stats = processor.endpoint_stats(endpoint_id)
    print("5m sliding window:", stats["300"])
    app_results = controller.writer.read_results(endpoint_id)
    print(f"writer log rows: {len(app_results)}")
    print("OK — drift detected, alert fired (console notification "
          "above), results persisted")


if __name__ == "__main__":
    main()
