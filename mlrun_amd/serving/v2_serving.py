# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""V2ModelServer — the model-serving step protocol.

KFServing-v2-style operations (infer/predict/explain/metadata/ready)
with load/preprocess/predict/postprocess hooks and monitoring push.
Parity target: reference mlrun/serving/v2_serving.py (V2ModelServer
:32, do_event :228 — the hot path, predict :381, _load_and_update_state
:124, _ModelLogPusher :429).

MI355X-native subclasses (mlrun_amd/frameworks) implement predict()
as HIP kernel launches; GPU servers should keep weights resident on
device and capture their per-batch kernel chain in a hipGraph
(see mlrun_amd/models/llama.py LlamaServer).
"""

import threading
import time
import traceback

from ..errors import MLRunInvalidArgumentError
from ..utils import logger


class V2ModelServer:
    """Base model-serving step.  Subclass and implement load() +
    predict(); optionally preprocess/postprocess/validate/explain."""

    def __init__(self, context=None, name: str = None, model_path: str = None,
                 model=None, protocol=None, input_path: str = None,
                 result_path: str = None, **class_args):
        self.name = name
        self.version = ""
        if name and ":" in name:
            self.name, self.version = name.split(":", 1)
        self.context = context
        self.ready = False
        self.error = ""
        self.protocol = protocol or "v2"
        self.model_path = model_path
        self.model_spec = None
        self._model_logger = None
        self._params = class_args
        self.metrics = {}
        self.labels = {}
        self.model = model
        self._load_lock = threading.Lock()
        self._stats_lock = threading.Lock()
        self.requests = 0
        self.error_count = 0
        self.latency_sum = 0.0

    def post_init(self, mode="sync"):
        """Called after graph init: load the model (sync mode) and wire
        the monitoring pusher."""
        stream = getattr(self.context, "stream", None) if self.context \
            else None
        if stream is not None:
            self._model_logger = _ModelLogPusher(self, stream)
        if mode == "sync":
            self._load_and_update_state()

    def _load_and_update_state(self):
        with self._load_lock:
            if self.ready:
                return
            try:
                self.load()
                self.ready = True
                self.error = ""
            except Exception as exc:
                self.error = f"{type(exc).__name__}: {exc}"
                logger.error(f"model {self.name} failed to load",
                             error=self.error,
                             tb=traceback.format_exc())
                raise

    # ------------------------------------------------------------- hooks
    def set_metric(self, name: str, value):
        """Set a real-time metric (rides the monitoring push —
        reference v2_serving.py:162)."""
        self.metrics[name] = value

    def get_param(self, key: str, default=None):
        if key in self._params:
            return self._params[key]
        if self.context is not None and hasattr(self.context, "get_param"):
            return self.context.get_param(key, default)
        return default

    def get_model(self, suffix=""):
        """Resolve (model_file, extra_data) from model_path — store://
        uri or directory (parity: reference v2_serving.py:166)."""
        from ..artifacts import get_model

        if not self.model_path:
            raise MLRunInvalidArgumentError(
                f"model {self.name} has no model_path")
        model_file, model_artifact, extra_data = get_model(self.model_path,
                                                           suffix)
        self.model_spec = model_artifact
        return model_file, extra_data

    def load(self):
        """Override: load the model into memory/GPU."""
        if self.model is None and self.model_path:
            raise NotImplementedError(
                "load() must be implemented by the model server subclass")

    def preprocess(self, request: dict, operation: str) -> dict:
        return request

    def postprocess(self, request: dict) -> dict:
        return request

    def validate(self, request: dict, operation: str) -> dict:
        if self.protocol == "v2" and operation in ("infer", "predict"):
            if not isinstance(request, dict) or "inputs" not in request:
                raise MLRunInvalidArgumentError(
                    'invalid request: expected {"inputs": [...]}')
        return request

    def predict(self, request: dict):
        """Override: run inference on request["inputs"]."""
        raise NotImplementedError

    def explain(self, request: dict):
        raise NotImplementedError(
            f"model {self.name} does not implement explain")

    # ---------------------------------------------------------- protocol
    def do_event(self, event):
        """Dispatch a serving event (HOT PATH).

        Path forms: .../infer | /predict | /explain | /ready |
        GET model -> metadata."""
        start = time.perf_counter()
        operation = _resolve_operation(event)
        request = event.body if isinstance(event.body, dict) else \
            ({} if event.body is None else {"inputs": event.body})
        try:
            if operation in ("infer", "predict", ""):
                if not self.ready:
                    self._load_and_update_state()
                request = self.preprocess(request, operation)
                request = self.validate(request, operation or "infer")
                outputs = self.predict(request)
                response = {
                    "id": event.id,
                    "model_name": self.name,
                    "outputs": outputs,
                }
                if self.version:
                    response["model_version"] = self.version
                response = self.postprocess(response)
                event.body = response
            elif operation == "explain":
                if not self.ready:
                    self._load_and_update_state()
                request = self.preprocess(request, operation)
                outputs = self.explain(request)
                event.body = {"id": event.id, "model_name": self.name,
                              "outputs": outputs}
            elif operation == "ready":
                if not self.ready:
                    self._load_and_update_state()
                event.body = {"name": self.name, "ready": self.ready}
            elif operation == "metadata":
                event.body = {
                    "name": self.name,
                    "version": self.version,
                    "inputs": getattr(self.model_spec, "inputs", []) or [],
                    "outputs": getattr(self.model_spec, "outputs", []) or [],
                }
            else:
                raise MLRunInvalidArgumentError(
                    f"unsupported operation {operation}")
        except Exception as exc:
            with self._stats_lock:
                self.error_count += 1
            if self._model_logger:
                self._model_logger.push(start, request, None,
                                        error=str(exc))
            raise
        latency_ms = (time.perf_counter() - start) * 1000.0
        with self._stats_lock:
            self.requests += 1
            self.latency_sum += latency_ms
        if self._model_logger and operation in ("infer", "predict", ""):
            self._model_logger.push(start, request, event.body)
        return event

    def logged_results(self, request: dict, response: dict, op: str):
        """Override to control what gets pushed to monitoring."""
        return request.get("inputs"), (response or {}).get("outputs")

    def stats(self) -> dict:
        with self._stats_lock:
            return {
                "requests": self.requests,
                "errors": self.error_count,
                "avg_latency_ms": (self.latency_sum / self.requests)
                if self.requests else 0.0,
            }


def _resolve_operation(event) -> str:
    path = event.path or ""
    if path.endswith("/"):
        path = path[:-1]
    last = path.rsplit("/", 1)[-1] if "/" in path else path
    if last in ("infer", "predict", "explain", "ready"):
        return last
    if event.method == "GET":
        return "metadata"
    # POST to .../models/<name> defaults to infer
    return "infer"


class _ModelLogPusher:
    """Sampled async push of prediction events to the monitoring stream
    (parity: reference v2_serving.py:429)."""

    def __init__(self, server: V2ModelServer, stream, sample_percent=None):
        from ..config import config

        self.server = server
        self.stream = stream
        self.sample_percent = sample_percent if sample_percent is not None \
            else float(config.model_endpoint_monitoring.sample_percent)
        self._counter = 0

    def push(self, start_time, request, response, error=None):
        self._counter += 1
        if self.sample_percent < 100 and \
                (self._counter * self.sample_percent) % 100 >= \
                self.sample_percent:
            return
        try:
            from ..model_monitoring import ModelMonitoringEvent

            inputs, outputs = self.server.logged_results(
                request or {}, response if isinstance(response, dict)
                else {}, "infer")
            self.stream.push(ModelMonitoringEvent(
                endpoint_id=self.server.name,
                model=self.server.model_path or self.server.name,
                inputs=inputs,
                outputs=outputs,
                latency_ms=(time.perf_counter() - start_time) * 1000.0,
                error=error))
        except Exception as exc:
            logger.warning("monitoring push failed", error=str(exc))
