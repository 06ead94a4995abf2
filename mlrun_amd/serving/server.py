# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""GraphServer: hosts a serving graph, feeds it events.

Parity target: reference mlrun/serving/server.py (GraphServer :86,
init_states :150, test :196, run :252 — the hot path, GraphContext
:493, MockEvent :445, create_graph_server :412).
"""

import inspect
import json
import socket
import time
import traceback
import typing
import uuid

from ..errors import MLRunInvalidArgumentError
from ..model import ModelObj
from ..utils import logger
from .states import (RootFlowStep, RouterStep, graph_root_setter, step_from_dict)


class Event:
    """A serving event (fast __slots__ object — per-request allocation
    is on the hot path)."""

    __slots__ = ["id", "body", "path", "method", "headers", "content_type",
                 "error", "terminated", "responded", "origin_state", "time",
                 "_recovery"]

    def __init__(self, body=None, id=None, path="", method="POST",
                 headers=None, content_type=None, time_=None):
        self.id = id or uuid.uuid4().hex
        self.body = body
        self.path = path
        self.method = method
        self.headers = headers or {}
        self.content_type = content_type
        self.error = None
        self.terminated = False
        self.responded = False
        self.origin_state = None
        self.time = time_ or time.time()

    def copy(self):
        import copy as _copy

        return _copy.copy(self)

    def __repr__(self):
        return f"Event(id={self.id!r}, path={self.path!r}, body={self.body!r})"


MockEvent = Event  # reference exposes MockEvent for tests


class GraphContext:
    """Context handed to graph step classes (parity: server.py:493)."""

    def __init__(self, server=None, logger_=None, verbose=False):
        self.server = server
        self.logger = logger_ or logger
        self.verbose = verbose
        self.root = None
        self.project = ""
        self.current_function = ""
        self.stream = None
        self._rundb = None

    @property
    def rundb(self):
        if self._rundb is None:
            from ..db import get_run_db

            self._rundb = get_run_db()
        return self._rundb

    def get_param(self, key: str, default=None):
        if self.server and key in (self.server.parameters or {}):
            return self.server.parameters[key]
        return default

    def get_secret(self, key: str, default=None):
        import os

        return os.environ.get(key, default)

    def get_store_resource(self, uri):
        from ..datastore import store_manager

        return store_manager.object(uri, project=self.project)

    def get_remote_endpoint(self, name, external=False):
        """Address of a deployed child-function host ("" = run
        locally).  A function never forwards to itself."""
        if self.server is None:
            return ""
        if name == (self.server._current_function or ""):
            return ""
        return (self.server.child_endpoints or {}).get(name, "")

    def push_error(self, event, message, source=None, **kwargs):
        """Log AND publish failed events to the configured error
        stream (reference GraphContext.push_error -> error stream;
        node-locally an OutputStream queue)."""
        self.logger.error(f"graph error from {source}: {message}",
                          event_id=getattr(event, "id", None))
        stream = getattr(self.server, "_error_stream_object", None) \
            if self.server else None
        if stream is None and self.server and \
                getattr(self.server, "error_stream", None):
            from ..platforms import OutputStream

            stream = OutputStream(self.server.error_stream)
            self.server._error_stream_object = stream
        if stream is not None:
            try:
                stream.push([{
                    "error": message, "source": source,
                    "event_id": getattr(event, "id", None),
                    "path": getattr(event, "path", ""),
                    "body": getattr(event, "body", None)}])
            except Exception as exc:
                # a broken error stream must never mask the original
                # failure (reference test_push_error contract)
                self.logger.error("error-stream push failed",
                                  error=str(exc))


class GraphServer(ModelObj):
    kind = "server"

    def __init__(self, graph=None, parameters=None, load_mode=None,
                 function_uri=None, verbose=False, version=None,
                 functions=None, graph_initializer=None, error_stream=None,
                 track_models=None, secret_sources=None,
                 default_content_type=None):
        self._graph = None
        self.graph = graph
        self.parameters = parameters or {}
        self.load_mode = load_mode or "sync"
        self.function_uri = function_uri
        self.verbose = verbose
        self.version = version or "v2"
        self.functions = functions or []
        self.graph_initializer = graph_initializer
        self.error_stream = error_stream
        self.track_models = track_models
        self.secret_sources = secret_sources
        self.default_content_type = default_content_type
        self.context: typing.Optional[GraphContext] = None
        self.http_trigger = True
        self._namespace = None
        self._current_function = None
        # child-function name -> deployed host address (multi-function
        # graphs; reference _deploy_function_refs + queue links)
        self.child_endpoints: dict = {}

    @property
    def graph(self) -> typing.Union[RootFlowStep, RouterStep]:
        return self._graph

    @graph.setter
    def graph(self, graph):
        if graph is None:
            self._graph = None
            return
        self._graph = graph_root_setter(self, graph)

    def set_current_function(self, function):
        self._current_function = function

    def init_states(self, context=None, namespace=None, logger_=None,
                    is_mock=False, monitoring=None):
        """Initialize the graph: build step objects, wire monitoring.

        Parity: reference server.py:150.
        """
        self.context = context or GraphContext(server=self,
                                               verbose=self.verbose)
        self.context.verbose = self.verbose
        self.context.project = (self.function_uri or "").split("/")[0] \
            if self.function_uri else ""
        self.context.stream = monitoring
        if self.track_models and monitoring is None:
            from ..model_monitoring import get_stream_processor

            self.context.stream = get_stream_processor(
                self.context.project or "default")
        self._namespace = namespace or {}
        if self.graph_initializer:
            initializer = self.graph_initializer
            if isinstance(initializer, str):
                from .states import _resolve_handler

                initializer = _resolve_handler(initializer, self._namespace)
            initializer(self)
        if self._graph is None:
            raise MLRunInvalidArgumentError("server has no graph")
        self.context.root = self._graph
        self._graph.init_object(self.context, self._namespace,
                                self.load_mode)
        return self

    def init_object(self, namespace):
        self._graph.init_object(self.context, namespace, self.load_mode)

    def run(self, event: Event, context=None, get_body=False,
            start_step: str = None):
        """Feed one event through the graph (HOT PATH — parity:
        reference server.py:252).  start_step: enter the flow at a
        named step (child-function hosts receiving forwarded
        events)."""
        server_context = self.context
        body = event.body
        if isinstance(body, (str, bytes)) and body and \
                (event.content_type in (None, "application/json")):
            try:
                event.body = json.loads(body)
            except (ValueError, TypeError):
                pass
        try:
            if start_step and hasattr(self._graph, "run_from_step"):
                response = self._graph.run_from_step(start_step, event)
            else:
                response = self._graph.run(event)
        except Exception as exc:
            if server_context and server_context.verbose:
                logger.error("graph run failed",
                             error=str(exc), tb=traceback.format_exc())
            message = f"{type(exc).__name__}: {exc}"
            if self.error_stream:
                server_context.push_error(event, message, source="server")
            return _ErrorResponse(message)
        if response is None:
            return None
        if getattr(response, "error", None):
            if self.error_stream and server_context:
                server_context.push_error(
                    event, response.error,
                    source=getattr(response, "origin_state", None))
            return _ErrorResponse(response.error,
                                  origin=getattr(response, "origin_state",
                                                 None))
        if get_body:
            return response.body
        return response

    def test(self, path: str = "/", body=None, method: str = "POST",
             headers: dict = None, content_type: str = None,
             silent: bool = False, get_body: bool = True, event_id=None):
        """Feed a synthetic request (the mock-server test entry —
        parity: reference server.py:196; this is also the benchmark
        harness shape of hack/benchmarks/model_serving_benchmark_local.py)."""
        if not self._graph:
            raise MLRunInvalidArgumentError("no graph in server")
        event = Event(body=body, path=path, method=method, headers=headers,
                      content_type=content_type, id=event_id)
        response = self.run(event, get_body=False)
        if isinstance(response, _ErrorResponse):
            if silent:
                return response
            raise RuntimeError(f"error in serving graph: {response.body}")
        if response is None:
            return None
        return response.body if get_body else response

    def wait_for_completion(self):
        if self._graph:
            self._graph.wait_for_completion()

    def to_dict(self, fields=None, exclude=None, strip=False):
        return {
            "parameters": self.parameters,
            "load_mode": self.load_mode,
            "function_uri": self.function_uri,
            "verbose": self.verbose,
            "version": self.version,
            "track_models": self.track_models,
            "graph": self._graph.to_dict() if self._graph else None,
        }

    @classmethod
    def from_dict(cls, struct=None, fields=None, deprecated_fields=None):
        struct = dict(struct or {})
        graph = struct.pop("graph", None)
        server = cls(**{k: v for k, v in struct.items()
                        if k in ("parameters", "load_mode", "function_uri",
                                 "verbose", "version", "track_models",
                                 "graph_initializer", "error_stream",
                                 "default_content_type")})
        if graph:
            server.graph = step_from_dict(graph) if isinstance(graph, dict) \
                else graph
        return server


class _ErrorResponse:
    def __init__(self, error, origin=None, status_code=None):
        self.body = {"error": error, "origin_state": origin}
        self.error = error
        self.status_code = status_code or _status_from_error(error)


def _status_from_error(error_text: str) -> int:
    """Map a step error (formatted "ExcClass: message") back to an
    HTTP status (reference: unknown model -> 404, bad request ->
    400).  The class is resolved from the errors module so subclasses
    (e.g. MLRunInvalidArgumentError -> 400) map correctly."""
    from .. import errors as errors_mod

    cls_name = str(error_text).split(":", 1)[0]
    exc_cls = getattr(errors_mod, cls_name, None)
    if isinstance(exc_cls, type) and issubclass(exc_cls, BaseException):
        for code, base in errors_mod.STATUS_ERRORS.items():
            if issubclass(exc_cls, base):
                return code
    if cls_name in ("ValueError", "KeyError", "TypeError"):
        return 400
    return 500

    def __repr__(self):
        return f"ErrorResponse({self.error!r})"


def create_graph_server(parameters=None, load_mode=None, graph=None,
                        verbose=False, current_function=None,
                        **kwargs) -> GraphServer:
    """Create a standalone graph server (for tests / ingestion graphs;
    parity: reference server.py:412)."""
    server = GraphServer(graph=graph, parameters=parameters or {},
                         load_mode=load_mode, verbose=verbose, **kwargs)
    server.set_current_function(current_function)
    return server


class GraphServerHost:
    """HTTP host for a graph server — the nuclio-worker replacement.

    Runs uvicorn+FastAPI in a background thread; POST /<anything>
    feeds the graph.  The reference deploys to nuclio pods
    (v2_serving_init server.py:315); here deploy == start this host.
    """

    def __init__(self, server: GraphServer, host="127.0.0.1", port=0):
        self.server = server
        self.host = host
        self.port = port or _free_port()
        self._uvicorn_server = None
        self._thread = None

    @property
    def address(self) -> str:
        return f"http://{self.host}:{self.port}"

    def _build_app(self):
        from fastapi import FastAPI, Request, Response

        app = FastAPI(title="mlrun-amd-serving")
        graph_server = self.server

        @app.get("/healthz")
        async def healthz():
            return {"status": "ok"}

        # Prometheus scrape endpoint (reference: `scrape_metrics` run
        # label + grafana integration; here native counters/latency
        # histograms per graph host)
        try:
            from prometheus_client import (CONTENT_TYPE_LATEST,
                                           CollectorRegistry, Counter,
                                           Histogram, generate_latest)

            registry = CollectorRegistry()
            self._metric_requests = Counter(
                "mlrun_serving_requests_total",
                "Requests handled by this graph host",
                ["path", "status"], registry=registry)
            self._metric_latency = Histogram(
                "mlrun_serving_request_seconds",
                "End-to-end request latency", ["path"],
                registry=registry,
                buckets=(.005, .01, .025, .05, .1, .25, .5, 1, 2.5, 5,
                         10, 30))

            @app.get("/metrics")
            async def metrics():
                return Response(generate_latest(registry),
                                media_type=CONTENT_TYPE_LATEST)
        except ImportError:
            self._metric_requests = self._metric_latency = None

        @app.api_route("/{full_path:path}",
                       methods=["GET", "POST", "PUT", "DELETE"])
        async def handle(full_path: str, request: Request):
            import time as _time

            handle_start = _time.perf_counter()
            body = await request.body()
            event = Event(body=body, path="/" + full_path,
                          method=request.method,
                          headers=dict(request.headers),
                          content_type=request.headers.get("content-type"))
            import anyio

            start_step = request.headers.get("x-mlrun-step")
            response = await anyio.to_thread.run_sync(
                lambda: graph_server.run(event, get_body=False,
                                         start_step=start_step))
            status = str(response.status_code) \
                if isinstance(response, _ErrorResponse) \
                else "200"
            if self._metric_requests is not None:
                route = "/" + full_path.split("/")[0]
                self._metric_requests.labels(path=route,
                                             status=status).inc()
                self._metric_latency.labels(path=route).observe(
                    _time.perf_counter() - handle_start)
            if isinstance(response, _ErrorResponse):
                return Response(json.dumps(response.body),
                                status_code=response.status_code,
                                media_type="application/json")
            body_out = response.body if response is not None else ""
            if inspect.isgenerator(body_out) or (
                    hasattr(body_out, "__next__") and
                    not isinstance(body_out, (bytes, str))):
                from fastapi.responses import StreamingResponse

                return StreamingResponse(
                    body_out, media_type="application/x-ndjson")
            if isinstance(body_out, (dict, list)):
                return Response(json.dumps(body_out, default=str),
                                media_type="application/json")
            if isinstance(body_out, bytes):
                return Response(body_out)
            return Response(str(body_out))

        return app

    def start(self, wait_ready=True, timeout=20):
        import threading

        import uvicorn

        app = self._build_app()
        uv_config = uvicorn.Config(app, host=self.host, port=self.port,
                                   log_level="warning", access_log=False)
        self._uvicorn_server = uvicorn.Server(uv_config)
        self._thread = threading.Thread(target=self._uvicorn_server.run,
                                        daemon=True, name="serving-host")
        self._thread.start()
        if wait_ready:
            deadline = time.monotonic() + timeout
            import requests

            while time.monotonic() < deadline:
                try:
                    requests.get(f"{self.address}/healthz", timeout=1)
                    return self
                except Exception:
                    time.sleep(0.05)
            raise TimeoutError("serving host did not start")
        return self

    def stop(self):
        if self._uvicorn_server is not None:
            self._uvicorn_server.should_exit = True
        if self._thread is not None:
            self._thread.join(timeout=5)
        graph = getattr(self.server, "graph", None)
        if graph is not None and hasattr(graph, "shutdown"):
            graph.shutdown()  # async controller / queue workers


def _free_port() -> int:
    sock = socket.socket()
    sock.bind(("127.0.0.1", 0))
    port = sock.getsockname()[1]
    sock.close()
    return port
