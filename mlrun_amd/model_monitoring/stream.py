# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Model-monitoring event stream processing.

Per model endpoint, serving steps push sampled prediction events
(V2ModelServer._ModelLogPusher analog).  The stream processor keeps
5m/1h tumbling-window aggregates (count, avg latency, error count,
feature stats) in memory, flushes raw events to parquet, and updates
the endpoint record in the run DB.

Parity target: reference mlrun/model_monitoring/stream_processing.py:45
EventStreamProcessor (storey graph: aggregate 5m/1h -> KV + TSDB +
parquet) rebuilt as an in-process engine; on-GPU window aggregation
reuses the feature-store HIP kernels when the endpoint is GPU-resident.
"""

import json
import os
import threading
import time
import typing
from collections import defaultdict, deque

from ..config import config
from ..utils import logger, now_iso


class ModelMonitoringEvent:
    __slots__ = ["endpoint_id", "model", "inputs", "outputs", "latency_ms",
                 "error", "timestamp", "request_id"]

    def __init__(self, endpoint_id, model, inputs=None, outputs=None,
                 latency_ms=0.0, error=None, timestamp=None, request_id=None):
        self.endpoint_id = endpoint_id
        self.model = model
        self.inputs = inputs
        self.outputs = outputs
        self.latency_ms = latency_ms
        self.error = error
        self.timestamp = timestamp or time.time()
        self.request_id = request_id

    def to_dict(self):
        return {slot: getattr(self, slot) for slot in self.__slots__}


class _WindowStats:
    """Tumbling-window aggregate for one (endpoint, window)."""

    def __init__(self, window_seconds: int):
        self.window_seconds = window_seconds
        self.window_start = 0.0
        self.count = 0
        self.error_count = 0
        self.latency_sum = 0.0
        self.latency_max = 0.0

    def add(self, event: ModelMonitoringEvent):
        bucket = event.timestamp - (event.timestamp % self.window_seconds)
        if bucket != self.window_start:
            self.window_start = bucket
            self.count = self.error_count = 0
            self.latency_sum = self.latency_max = 0.0
        self.count += 1
        if event.error:
            self.error_count += 1
        self.latency_sum += event.latency_ms
        self.latency_max = max(self.latency_max, event.latency_ms)

    def snapshot(self) -> dict:
        return {
            "window_seconds": self.window_seconds,
            "window_start": self.window_start,
            "count": self.count,
            "error_count": self.error_count,
            "avg_latency_ms": (self.latency_sum / self.count)
            if self.count else 0.0,
            "max_latency_ms": self.latency_max,
        }


class EventStreamProcessor:
    """In-process monitoring stream: push() from serving steps."""

    WINDOWS = (300, 3600)  # 5m / 1h, matching the reference defaults

    def __init__(self, project: str = "", parquet_dir: str = "",
                 parquet_batch: int = None, db=None):
        self.project = project or "default"
        self.parquet_dir = parquet_dir or os.path.join(
            config.base_dir, "monitoring", self.project)
        self.parquet_batch = parquet_batch or int(
            config.model_endpoint_monitoring.parquet_batching_max_events)
        self._db = db
        self._lock = threading.Lock()
        self._stats: dict = defaultdict(
            lambda: {w: _WindowStats(w) for w in self.WINDOWS})
        self._pending: list = []
        self._feature_samples: dict = defaultdict(lambda: deque(maxlen=4096))
        self._tsdb: dict = defaultdict(list)  # endpoint -> [(t, snapshot)]

    def _get_db(self):
        if self._db is None:
            from ..db import get_run_db

            self._db = get_run_db()
        return self._db

    def push(self, event: typing.Union[ModelMonitoringEvent, dict]):
        if isinstance(event, dict):
            event = ModelMonitoringEvent(**event)
        with self._lock:
            for win in self._stats[event.endpoint_id].values():
                win.add(event)
            self._pending.append(event.to_dict())
            if event.inputs is not None:
                self._feature_samples[event.endpoint_id].append(event.inputs)
            flush = len(self._pending) >= self.parquet_batch
        if flush:
            self.flush()

    def endpoint_stats(self, endpoint_id: str) -> dict:
        with self._lock:
            return {str(w): s.snapshot()
                    for w, s in self._stats[endpoint_id].items()}

    def feature_samples(self, endpoint_id: str) -> list:
        with self._lock:
            return list(self._feature_samples[endpoint_id])

    def record_tsdb(self):
        """Snapshot all endpoint windows into the in-memory TSDB."""
        now = time.time()
        with self._lock:
            for endpoint_id, windows in self._stats.items():
                self._tsdb[endpoint_id].append(
                    (now, {str(w): s.snapshot()
                           for w, s in windows.items()}))

    def tsdb_series(self, endpoint_id: str) -> list:
        with self._lock:
            return list(self._tsdb[endpoint_id])

    def flush(self):
        """Write pending events to a parquet file."""
        with self._lock:
            batch, self._pending = self._pending, []
        if not batch:
            return
        try:
            import pandas as pd

            os.makedirs(self.parquet_dir, exist_ok=True)
            df = pd.DataFrame(batch)
            for col in ("inputs", "outputs"):
                if col in df:
                    df[col] = df[col].map(
                        lambda v: json.dumps(v, default=str))
            path = os.path.join(self.parquet_dir,
                                f"events-{int(time.time() * 1000)}.parquet")
            df.to_parquet(path)
        except Exception as exc:
            logger.warning("monitoring parquet flush failed", error=str(exc))

    def update_endpoint_record(self, endpoint_id: str, model: str = "",
                               function_uri: str = "", extra: dict = None):
        stats = self.endpoint_stats(endpoint_id)
        record = {
            "kind": "model-endpoint",
            "metadata": {"project": self.project, "uid": endpoint_id},
            "spec": {"model": model, "function_uri": function_uri},
            "status": {"state": "ready", "last_request": now_iso(),
                       "stats": stats, **(extra or {})},
        }
        try:
            self._get_db().store_model_endpoint(self.project, endpoint_id,
                                                record)
        except Exception as exc:
            logger.warning("endpoint record update failed", error=str(exc))
        return record


_processors: dict = {}
_processors_lock = threading.Lock()


def get_stream_processor(project: str = "default") -> EventStreamProcessor:
    with _processors_lock:
        if project not in _processors:
            _processors[project] = EventStreamProcessor(project)
        return _processors[project]
