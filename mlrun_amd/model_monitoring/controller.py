# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Monitoring controller: the periodic job that sweeps model endpoints,
computes drift vs. reference feature stats, stores results and fires
events.

Parity target: reference mlrun/model_monitoring/controller.py:265
MonitoringApplicationController (the k8s cron job becomes a node-local
background thread).
"""

import threading

from ..utils import logger, now_iso
from .drift import compute_feature_drift, drift_status
from .stream import get_stream_processor


class MonitoringController:
    def __init__(self, project: str = "default", base_period: int = 600,
                 db=None):
        self.project = project
        self.base_period = base_period
        self._db = db
        self._reference_samples: dict = {}
        self._thread = None
        self._stop = threading.Event()

    def _get_db(self):
        if self._db is None:
            from ..db import get_run_db

            self._db = get_run_db()
        return self._db

    def set_reference(self, endpoint_id: str, samples):
        """Register reference (training-time) feature samples."""
        self._reference_samples[endpoint_id] = samples

    def run_iteration(self) -> dict:
        """One sweep: for each endpoint with reference data, compute
        drift on the recent sample window and update its record."""
        processor = get_stream_processor(self.project)
        processor.record_tsdb()
        results = {}
        for endpoint_id, reference in self._reference_samples.items():
            current = processor.feature_samples(endpoint_id)
            if not current:
                continue
            import numpy as np

            cur = np.asarray([np.ravel(np.asarray(s, dtype=np.float64))
                              for s in current if s is not None])
            metrics = compute_feature_drift(np.ravel(reference), cur.ravel())
            metrics["status"] = drift_status(metrics["drift_score"])
            results[endpoint_id] = metrics
            processor.update_endpoint_record(
                endpoint_id, extra={"drift_metrics": metrics,
                                    "drift_status": metrics["status"]})
            if metrics["status"] == "drift_detected":
                self._fire_event(endpoint_id, metrics)
        return results

    def _fire_event(self, endpoint_id: str, metrics: dict):
        logger.warning("model drift detected", endpoint=endpoint_id,
                       score=metrics["drift_score"])
        try:
            from ..api.events import process_event

            process_event(self.project, "model-drift", {
                "endpoint_id": endpoint_id, "metrics": metrics,
                "time": now_iso()})
        except Exception:
            pass

    def start(self):
        if self._thread is not None:
            return
        self._stop.clear()

        def _loop():
            while not self._stop.wait(self.base_period):
                try:
                    self.run_iteration()
                except Exception as exc:
                    logger.warning("monitoring iteration failed",
                                   error=str(exc))

        self._thread = threading.Thread(target=_loop, daemon=True,
                                        name="monitoring-controller")
        self._thread.start()

    def stop(self):
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=5)
            self._thread = None


_controllers: dict = {}
_lock = threading.Lock()


def enable_model_monitoring(project, base_period: int = 10,
                            start: bool = False) -> MonitoringController:
    name = project.name if hasattr(project, "name") else str(project)
    with _lock:
        if name not in _controllers:
            _controllers[name] = MonitoringController(
                name, base_period=base_period * 60)
        controller = _controllers[name]
    if start:
        controller.start()
    return controller
