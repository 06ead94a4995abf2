# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Model-monitoring application framework.

Parity targets (reference):
- mlrun/model_monitoring/applications/base.py
  ``ModelMonitoringApplicationBase`` (``do`` -> ``do_tracking``)
- mlrun/model_monitoring/applications/results.py
  ``ModelMonitoringApplicationResult`` / ``...Metric``
- mlrun/model_monitoring/applications/histogram_data_drift.py
  ``HistogramDataDriftApplication`` (TVD/Hellinger/KL per feature)

Node-local redesign: applications run in-process inside the
MonitoringController sweep (the reference schedules them as nuclio
jobs); results flow through the ModelMonitoringWriter into the
model-endpoint record + a parquet results log.
"""

import dataclasses
import typing
from abc import ABC, abstractmethod

import numpy as np

from ..utils import logger


class ResultKindApp:
    data_drift = "data_drift"
    concept_drift = "concept_drift"
    model_performance = "model_performance"
    system_performance = "system_performance"
    anomaly = "anomaly"


class ResultStatusApp:
    irrelevant = -1
    no_detection = 0
    potential_detection = 1
    detected = 2


@dataclasses.dataclass
class ModelMonitoringApplicationResult:
    name: str
    value: float
    kind: str = ResultKindApp.data_drift
    status: int = ResultStatusApp.no_detection
    extra_data: dict = dataclasses.field(default_factory=dict)

    def to_dict(self) -> dict:
        return {"result_name": self.name, "result_value": self.value,
                "result_kind": self.kind, "result_status": self.status,
                "result_extra_data": self.extra_data}


@dataclasses.dataclass
class ModelMonitoringApplicationMetric:
    name: str
    value: float

    def to_dict(self) -> dict:
        return {"metric_name": self.name, "metric_value": self.value}


class MonitoringApplicationContext:
    """Everything an application sees for one (endpoint, sweep):
    current feature samples, reference samples, windowed serving stats
    (from the GPU/CPU ring), and the endpoint identity."""

    def __init__(self, project: str, endpoint_id: str, model: str = "",
                 sample_features=None, reference_features=None,
                 window_stats: dict = None, logger_=None):
        self.project = project
        self.endpoint_id = endpoint_id
        self.model = model
        self.sample_features = sample_features
        self.reference_features = reference_features
        self.window_stats = window_stats or {}
        self.logger = logger_ or logger
        self.artifacts: dict = {}

    def log_artifact(self, key, item):
        self.artifacts[key] = item

    def sample_array(self) -> np.ndarray:
        rows = [np.ravel(np.asarray(s, dtype=np.float64))
                for s in (self.sample_features or []) if s is not None]
        return np.concatenate(rows) if rows else np.empty(0)

    def reference_array(self) -> np.ndarray:
        if self.reference_features is None:
            return np.empty(0)
        return np.ravel(np.asarray(self.reference_features,
                                   dtype=np.float64))


class ModelMonitoringApplicationBase(ABC):
    """Inherit and implement ``do_tracking`` (reference base.py)."""

    kind = "monitoring_application"
    NAME = ""

    def do(self, monitoring_context: MonitoringApplicationContext
           ) -> typing.Tuple[list, MonitoringApplicationContext]:
        results = self.do_tracking(monitoring_context)
        if isinstance(results, dict):
            results = [ModelMonitoringApplicationMetric(name=k, value=v)
                       for k, v in results.items()]
        if not isinstance(results, list):
            results = [results]
        return results, monitoring_context

    @abstractmethod
    def do_tracking(self, monitoring_context: MonitoringApplicationContext):
        raise NotImplementedError

    @property
    def name(self) -> str:
        return self.NAME or type(self).__name__


class HistogramDataDriftApplication(ModelMonitoringApplicationBase):
    """Per-endpoint histogram drift: TVD / Hellinger / KL between
    reference and current feature distributions (reference
    histogram_data_drift.py; metrics in drift.py)."""

    NAME = "histogram-data-drift"

    def __init__(self, potential_threshold: float = 0.5,
                 detected_threshold: float = 0.7, bins: int = 20):
        self.potential_threshold = potential_threshold
        self.detected_threshold = detected_threshold
        self.bins = bins

    def do_tracking(self, monitoring_context: MonitoringApplicationContext):
        from .drift import compute_feature_drift

        reference = monitoring_context.reference_array()
        current = monitoring_context.sample_array()
        if reference.size == 0 or current.size == 0:
            return ModelMonitoringApplicationResult(
                name="general_drift", value=0.0,
                status=ResultStatusApp.irrelevant)
        metrics = compute_feature_drift(reference, current,
                                        bins=self.bins)
        score = metrics["drift_score"]
        if score >= self.detected_threshold:
            status = ResultStatusApp.detected
        elif score >= self.potential_threshold:
            status = ResultStatusApp.potential_detection
        else:
            status = ResultStatusApp.no_detection
        results = [
            ModelMonitoringApplicationResult(
                name="general_drift", value=score, status=status,
                extra_data={k: v for k, v in metrics.items()
                            if isinstance(v, float)}),
            ModelMonitoringApplicationMetric("tvd_mean", metrics["tvd"]),
            ModelMonitoringApplicationMetric("hellinger_mean",
                                             metrics["hellinger"]),
            ModelMonitoringApplicationMetric("kld_mean", metrics["kld"]),
        ]
        return results


class LatencyPerformanceApplication(ModelMonitoringApplicationBase):
    """System-performance app over the sliding-window serving stats:
    flags endpoints whose 5m average latency exceeds a threshold."""

    NAME = "latency-performance"

    def __init__(self, p_avg_latency_ms: float = 1000.0):
        self.p_avg_latency_ms = p_avg_latency_ms

    def do_tracking(self, monitoring_context: MonitoringApplicationContext):
        stats = monitoring_context.window_stats.get("300") or {}
        avg = float(stats.get("avg_latency_ms", 0.0))
        status = ResultStatusApp.detected \
            if avg > self.p_avg_latency_ms else ResultStatusApp.no_detection
        return ModelMonitoringApplicationResult(
            name="avg_latency", value=avg,
            kind=ResultKindApp.system_performance, status=status,
            extra_data=dict(stats))
