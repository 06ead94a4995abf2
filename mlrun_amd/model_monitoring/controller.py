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
                 db=None, applications: list = None):
        from .apps import HistogramDataDriftApplication
        from .writer import ModelMonitoringWriter

        self.project = project
        self.base_period = base_period
        self._db = db
        self._reference_samples: dict = {}
        self._thread = None
        self._stop = threading.Event()
        # the application set run each sweep (reference: per-app nuclio
        # functions driven by the controller job); the histogram drift
        # app ships by default like the reference's
        self.applications = applications if applications is not None \
            else [HistogramDataDriftApplication()]
        self.writer = ModelMonitoringWriter(self.project, db=db)

    def add_application(self, application):
        self.applications.append(application)
        return application

    def _get_db(self):
        if self._db is None:
            from ..db import get_run_db

            self._db = get_run_db()
        return self._db

    def set_reference(self, endpoint_id: str, samples):
        """Register reference (training-time) feature samples."""
        self._reference_samples[endpoint_id] = samples

    def run_iteration(self) -> dict:
        """One sweep (reference controller.py:265): run every
        registered monitoring application for every endpoint with
        reference data, persist results through the writer, fire
        events for detections.  Also keeps the legacy drift_metrics
        record shape for compatibility."""
        from .apps import MonitoringApplicationContext, ResultStatusApp

        processor = get_stream_processor(self.project)
        processor.record_tsdb()
        results = {}
        for endpoint_id, reference in self._reference_samples.items():
            current = processor.feature_samples(endpoint_id)
            if not current:
                continue
            window_stats = processor.endpoint_stats(endpoint_id)
            app_context = MonitoringApplicationContext(
                self.project, endpoint_id,
                sample_features=current,
                reference_features=reference,
                window_stats=window_stats)
            endpoint_results = {}
            for application in self.applications:
                try:
                    app_results, _ = application.do(app_context)
                except Exception as exc:
                    logger.warning("monitoring app failed",
                                   app=application.name, error=str(exc))
                    continue
                self.writer.write(endpoint_id, application.name,
                                  app_results,
                                  window_stats=window_stats)
                endpoint_results[application.name] = [
                    r.to_dict() for r in app_results]
                for result in app_results:
                    if getattr(result, "status", 0) == \
                            ResultStatusApp.detected:
                        self._fire_event(endpoint_id, {
                            "application": application.name,
                            "result_name": result.name,
                            "drift_score": result.value,
                            **result.extra_data})
            # legacy record shape (round-1 API): top-level drift metrics
            import numpy as np

            cur = np.asarray([np.ravel(np.asarray(s, dtype=np.float64))
                              for s in current if s is not None])
            metrics = compute_feature_drift(np.ravel(reference),
                                            cur.ravel())
            metrics["status"] = drift_status(metrics["drift_score"])
            metrics["app_results"] = endpoint_results
            results[endpoint_id] = metrics
            processor.update_endpoint_record(
                endpoint_id, extra={"drift_metrics": {
                    k: v for k, v in metrics.items()
                    if k != "app_results"},
                    "drift_status": metrics["status"]})
        self.writer.flush()
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
