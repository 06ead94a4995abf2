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


class EventStreamProcessor:
    """In-process monitoring stream: push() from serving steps.

    Window stats are SLIDING (5m/1h over 60s period buckets), computed
    from the feature-store WindowRing — the same per-period-cell HIP
    kernels that power the online feature store serve the monitoring
    latency/error windows (GPU-resident when the endpoint runs on
    GPU)."""

    WINDOWS = (300, 3600)  # 5m / 1h, matching the reference defaults
    RING_PERIOD = 60       # one bucket per minute; ring spans 1h + 1

    def __init__(self, project: str = "", parquet_dir: str = "",
                 parquet_batch: int = None, db=None, device="cpu"):
        from ..feature_store.online import WindowRing

        self.project = project or "default"
        self.parquet_dir = parquet_dir or os.path.join(
            config.base_dir, "monitoring", self.project)
        self.parquet_batch = parquet_batch or int(
            config.model_endpoint_monitoring.parquet_batching_max_events)
        self._db = db
        self._lock = threading.Lock()
        n_periods = max(self.WINDOWS) // self.RING_PERIOD + 1
        self._latency_ring = WindowRing(self.RING_PERIOD, n_periods,
                                        device=device, capacity=64)
        self._error_ring = WindowRing(self.RING_PERIOD, n_periods,
                                      device=device, capacity=64)
        self._endpoint_ids: dict = {}
        self._ring_buffer: list = []   # (endpoint_idx, latency, err, ts)
        self._pending: list = []
        self._feature_samples: dict = defaultdict(lambda: deque(maxlen=4096))
        self._tsdb: dict = defaultdict(list)  # endpoint -> [(t, snapshot)]

    def _endpoint_idx(self, endpoint_id) -> int:
        idx = self._endpoint_ids.get(endpoint_id)
        if idx is None:
            idx = len(self._endpoint_ids)
            self._endpoint_ids[endpoint_id] = idx
            if idx >= self._latency_ring.capacity:
                self._latency_ring.grow(self._latency_ring.capacity * 2)
                self._error_ring.grow(self._error_ring.capacity * 2)
        return idx

    def _fold_rings_locked(self):
        """Batch-ingest buffered events into the window rings."""
        if not self._ring_buffer:
            return
        import torch

        idxs = torch.tensor([b[0] for b in self._ring_buffer])
        lats = torch.tensor([b[1] for b in self._ring_buffer])
        errs = torch.tensor([b[2] for b in self._ring_buffer])
        # f64: unix seconds in a float32 tensor quantize to ~128 s and
        # can land events in FUTURE buckets the reduce never covers
        ts = torch.tensor([b[3] for b in self._ring_buffer],
                          dtype=torch.float64)
        self._ring_buffer = []
        self._latency_ring.ingest(idxs, lats, ts)
        self._error_ring.ingest(idxs, errs, ts)

    def _get_db(self):
        if self._db is None:
            from ..db import get_run_db

            self._db = get_run_db()
        return self._db

    def push(self, event: typing.Union[ModelMonitoringEvent, dict]):
        if isinstance(event, dict):
            event = ModelMonitoringEvent(**event)
        with self._lock:
            idx = self._endpoint_idx(event.endpoint_id)
            self._ring_buffer.append(
                (idx, float(event.latency_ms),
                 1.0 if event.error else 0.0, float(event.timestamp)))
            if len(self._ring_buffer) >= 256:
                self._fold_rings_locked()
            self._pending.append(event.to_dict())
            if event.inputs is not None:
                self._feature_samples[event.endpoint_id].append(event.inputs)
            flush = len(self._pending) >= self.parquet_batch
        if flush:
            self.flush()

    def endpoint_stats(self, endpoint_id: str,
                       now_ts: float = None) -> dict:
        """Sliding-window serving stats from the rings: count, error
        count, avg/max/min latency per 5m/1h window."""
        now_ts = now_ts or time.time()
        with self._lock:
            self._fold_rings_locked()
            idx = self._endpoint_ids.get(endpoint_id)
            out = {}
            for window in self.WINDOWS:
                if idx is None:
                    out[str(window)] = {"window_seconds": window,
                                        "count": 0, "error_count": 0,
                                        "avg_latency_ms": 0.0,
                                        "max_latency_ms": 0.0}
                    continue
                lat = self._latency_ring.window_values(window, now_ts)
                err = self._error_ring.window_values(window, now_ts)

                def _val(tensor):
                    value = float(tensor[idx])
                    return 0.0 if value != value else value

                out[str(window)] = {
                    "window_seconds": window,
                    "count": int(_val(lat["count"])),
                    "error_count": int(_val(err["sum"])),
                    "avg_latency_ms": _val(lat["avg"]),
                    "max_latency_ms": _val(lat["max"]),
                    "min_latency_ms": _val(lat["min"]),
                }
            return out

    def feature_samples(self, endpoint_id: str) -> list:
        with self._lock:
            return list(self._feature_samples[endpoint_id])

    def record_tsdb(self):
        """Snapshot all endpoint windows into the in-memory TSDB."""
        now = time.time()
        for endpoint_id in list(self._endpoint_ids):
            snapshot = self.endpoint_stats(endpoint_id, now)
            with self._lock:
                self._tsdb[endpoint_id].append((now, snapshot))

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
        record = {}
        try:  # merge into the stored record (the writer's app_results
            record = self._get_db().get_model_endpoint(  # must survive)
                self.project, endpoint_id) or {}
        except Exception:
            record = {}
        record.setdefault("kind", "model-endpoint")
        record.setdefault("metadata", {}).update(
            {"project": self.project, "uid": endpoint_id})
        spec = record.setdefault("spec", {})
        if model:
            spec["model"] = model
        if function_uri:
            spec["function_uri"] = function_uri
        record.setdefault("status", {}).update(
            {"state": "ready", "last_request": now_iso(),
             "stats": stats, **(extra or {})})
        try:
            self._get_db().store_model_endpoint(self.project, endpoint_id,
                                                record)
        except Exception as exc:
            logger.warning("endpoint record update failed", error=str(exc))
        return record


_processors: dict = {}
_processors_lock = threading.Lock()


def _monitoring_device() -> str:
    """Ring placement: config model_endpoint_monitoring.device —
    "cpu" (default; stats are tiny), "auto" (GPU when available —
    shares the feature-store HIP window kernels), or an explicit
    device string."""
    device = str(config.model_endpoint_monitoring.get("device", "cpu"))
    if device == "auto":
        import torch

        return "cuda:0" if torch.cuda.is_available() else "cpu"
    return device or "cpu"


def get_stream_processor(project: str = "default") -> EventStreamProcessor:
    with _processors_lock:
        if project not in _processors:
            _processors[project] = EventStreamProcessor(
                project, device=_monitoring_device())
        return _processors[project]
