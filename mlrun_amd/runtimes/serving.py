# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""ServingRuntime — the serving-graph function kind.

Parity target: reference mlrun/runtimes/nuclio/serving.py
(ServingRuntime :232, set_topology :245, add_model :356, set_tracking
:308, add_child_function :447, deploy :580, to_mock_server :668).

deploy() starts a node-local HTTP host (GraphServerHost) instead of a
nuclio pod; child functions become additional local hosts linked by
queue steps.
"""

import typing

from ..errors import MLRunInvalidArgumentError
from ..serving.server import GraphServer, GraphServerHost, create_graph_server
from ..serving.states import (
    FlowStep,
    RootFlowStep,
    RouterStep,
    TaskStep,
    step_from_dict,
)
from ..utils import logger
from .base import BaseRuntime, FunctionSpec


class ServingSpec(FunctionSpec):
    def __init__(self, command=None, args=None, image=None, mode=None,
                 build=None, entry_points=None, description=None,
                 workdir=None, default_handler=None, pythonpath=None,
                 graph=None, parameters=None, load_mode=None, verbose=None,
                 graph_initializer=None, error_stream=None, track_models=None,
                 function_refs=None, default_content_type=None,
                 secret_sources=None, disable_auto_mount=None,
                 allow_empty_resources=None, resources=None,
                 min_replicas=None, max_replicas=None):
        super().__init__(command, args, image, mode, build, entry_points,
                         description, workdir, default_handler, pythonpath,
                         disable_auto_mount, allow_empty_resources, resources)
        self._graph = None
        self.graph = graph
        self.parameters = parameters or {}
        self.load_mode = load_mode
        self.verbose = verbose
        self.graph_initializer = graph_initializer
        self.error_stream = error_stream
        self.track_models = track_models
        self.function_refs = function_refs or {}
        self.default_content_type = default_content_type
        self.secret_sources = secret_sources or []
        # worker-process replicas (nuclio min/max_replicas analog)
        self.min_replicas = min_replicas
        self.max_replicas = max_replicas

    @property
    def graph(self):
        return self._graph

    @graph.setter
    def graph(self, graph):
        if isinstance(graph, dict):
            graph = step_from_dict(graph)
        self._graph = graph

    def to_dict(self, fields=None, exclude=None, strip=False):
        struct = super().to_dict(fields, exclude=["graph"])
        if self._graph is not None:
            struct["graph"] = self._graph.to_dict()
        for key in ("parameters", "load_mode", "verbose", "track_models",
                    "default_content_type"):
            value = getattr(self, key, None)
            if value:
                struct[key] = value
        return struct


class ServingRuntime(BaseRuntime):
    kind = "serving"

    def __init__(self, metadata=None, spec=None):
        super().__init__(metadata, spec)
        self._spec: ServingSpec
        self._server: typing.Optional[GraphServer] = None
        self._host: typing.Optional[GraphServerHost] = None
        self._mock_server: typing.Optional[GraphServer] = None

    @property
    def spec(self) -> ServingSpec:
        return self._spec

    @spec.setter
    def spec(self, value):
        self._spec = self._verify_dict(value, "spec", ServingSpec)

    # ---------------------------------------------------------- topology
    def set_topology(self, topology=None, class_name=None, engine=None,
                     exist_ok=False, **class_args):
        """topology: "router" (default) or "flow"."""
        topology = topology or "router"
        if self.spec.graph is not None and not exist_ok:
            raise MLRunInvalidArgumentError(
                "graph topology already set (pass exist_ok=True to replace)")
        if topology == "router":
            self.spec.graph = RouterStep(class_name=class_name,
                                         class_args=class_args)
        elif topology == "flow":
            self.spec.graph = RootFlowStep()
            self.spec.graph.engine = engine or "sync"
        else:
            raise MLRunInvalidArgumentError(
                f"unsupported topology {topology}")
        return self.spec.graph

    @property
    def graph(self):
        return self.spec.graph

    def add_model(self, key: str, model_path: str = None, class_name=None,
                  model_url=None, handler=None, router_step=None,
                  child_function=None, **class_args):
        """Add a model route to the router topology."""
        graph = self.spec.graph
        if graph is None:
            graph = self.set_topology()
        if isinstance(graph, RouterStep):
            router = graph
        elif isinstance(graph, FlowStep):
            router = None
            target = router_step
            for step in graph.steps.values():
                if isinstance(step, RouterStep) and (
                        not target or step.name == target):
                    router = step
                    break
            if router is None:
                raise MLRunInvalidArgumentError(
                    "no router step found in flow topology")
        else:
            raise MLRunInvalidArgumentError("graph has no router")
        if model_url:
            # remote model endpoint route (reference
            # new_remote_endpoint): proxy infer calls over HTTP
            from ..serving.remote import RemoteStep

            route = RemoteStep(url=model_url, name=key, **class_args)
            from ..serving.states import TaskStep as _TaskStep

            step = _TaskStep(route, name=key)
            return router.add_route(key, route=step)
        if class_name is None and handler is None:
            raise MLRunInvalidArgumentError(
                "class_name (a V2ModelServer subclass) or handler is "
                "required")
        if model_path:
            class_args = dict(class_args)
            class_args["model_path"] = model_path
        if isinstance(class_name, type):
            # store the dotted path when it resolves back — so the
            # spec survives serialization (worker processes /
            # save-load); locally-defined classes stay as objects
            dotted = f"{class_name.__module__}." \
                     f"{class_name.__qualname__}"
            try:
                import importlib

                module = importlib.import_module(class_name.__module__)
                resolved = module
                for part in class_name.__qualname__.split("."):
                    resolved = getattr(resolved, part)
                if resolved is class_name:
                    class_name = dotted
            except (ImportError, AttributeError):
                pass
        route = TaskStep(class_name, class_args, handler=handler, name=key,
                         function=child_function)
        return router.add_route(key, route=route)

    def set_tracking(self, stream_path=None, batch=None, sample=None,
                     tracking_policy=None, enable_tracking: bool = True,
                     stream_args: dict = None):
        self.spec.track_models = bool(enable_tracking)
        if stream_args:
            self.spec.parameters = {**(self.spec.parameters or {}),
                                    "tracking_stream_args": stream_args}
        return self

    def add_child_function(self, name, url=None, image=None, requirements=None,
                           kind=None):
        self.spec.function_refs[name] = {"url": url, "image": image,
                                         "kind": kind or "serving"}
        return self

    def list_child_functions(self) -> list:
        """Distinct child-function names referenced by graph steps
        (reference states.py list_child_functions)."""
        graph = self.spec.graph
        names = []
        if graph is None or not hasattr(graph, "steps"):
            return names
        for step in graph.steps.values():
            function = getattr(step, "function", None)
            if function and function != "*" and function not in names:
                names.append(function)
        return names

    def _deploy_child_functions(self, parent_server, namespace=None):
        """Start one host per child function named by graph steps and
        register the addresses (reference _deploy_function_refs + the
        queue/stream links, serving.py:512): forwarded events enter
        the child's graph AT the annotated step (x-mlrun-step)."""
        names = self.list_child_functions()
        if not names:
            return
        self._child_hosts = getattr(self, "_child_hosts", [])
        endpoints = {}
        for name in names:
            child = self._build_server(namespace, current_function=name,
                                       fresh_graph=True)
            host = GraphServerHost(child)
            host.start()
            self._child_hosts.append(host)
            endpoints[name] = host.address
            logger.info("child function deployed", function=name,
                        address=host.address)
        # every server (parent + children) can reach every child;
        # get_remote_endpoint refuses self-forwarding by name
        parent_server.child_endpoints.update(endpoints)
        for host in self._child_hosts:
            host.server.child_endpoints.update(endpoints)

    def remove_states(self, keys: list):
        graph = self.spec.graph
        if isinstance(graph, RouterStep):
            graph.clear_children(keys)
        elif isinstance(graph, FlowStep):
            for key in keys:
                graph.steps.pop(key, None)

    # ---------------------------------------------------------- serving
    def _build_server(self, namespace=None, current_function=None,
                      fresh_graph=False) -> GraphServer:
        graph = self.spec.graph
        if fresh_graph and graph is not None:
            # child hosts need their OWN step instances/controller —
            # rebuild from the serialized spec (class objects stored
            # as dotted paths survive; add_model guarantees that)
            graph = step_from_dict(graph.to_dict())
        server = create_graph_server(
            parameters=self.spec.parameters,
            load_mode=self.spec.load_mode,
            graph=graph,
            verbose=bool(self.spec.verbose),
            graph_initializer=self.spec.graph_initializer,
            error_stream=self.spec.error_stream,
            track_models=self.spec.track_models,
            function_uri=f"{self.metadata.project or 'default'}/"
                         f"{self.metadata.name}",
            current_function=current_function,
        )
        server.init_states(namespace=namespace or _caller_namespace())
        return server

    def to_mock_server(self, namespace=None, current_function=None,
                       track_models=False, **kwargs) -> GraphServer:
        """In-process test server (parity: reference serving.py:668).
        Also the benchmark-harness entry."""
        if track_models:
            self.spec.track_models = True
        self._mock_server = self._build_server(namespace)
        return self._mock_server

    def deploy(self, project="", tag="", verbose=False, auth_info=None,
               builder_env=None, force_build=False, with_mlrun=None,
               namespace=None, workers: int = 0) -> str:
        """Start the node-local HTTP serving host; returns its address.
        workers > 1: N worker PROCESSES behind an L4 round-robin proxy
        (nuclio min_replicas analog; removes the single-process GIL
        cap — serving/workers.py)."""
        workers = workers or int(self.spec.min_replicas or 0)
        max_workers = int(self.spec.max_replicas or 0)
        if workers > 1 or max_workers > max(workers, 1):
            from ..serving.workers import WorkerPool

            self._worker_pool = WorkerPool(self.to_dict(),
                                           max(workers, 1),
                                           max_workers=max_workers)
            address = self._worker_pool.start()
            self.status.state = "ready"
            self.status.address = address
            self.status.external_invocation_urls = [address]
            logger.info("serving function deployed (worker pool)",
                        address=address, workers=workers)
            return address
        server = self._build_server(namespace)
        self._server = server
        self._host = GraphServerHost(server)
        self._host.start()
        self._deploy_child_functions(server, namespace)
        self.status.state = "ready"
        self.status.address = self._host.address
        self.status.external_invocation_urls = [self._host.address]
        # register model endpoints for monitoring
        if self.spec.track_models:
            self._register_model_endpoints()
        try:
            self.save()
        except Exception:
            pass
        logger.info("serving function deployed", address=self._host.address)
        return self._host.address

    def _register_model_endpoints(self):
        from ..model_monitoring import get_stream_processor

        graph = self.spec.graph
        routes = {}
        if isinstance(graph, RouterStep):
            routes = graph.routes
        elif isinstance(graph, FlowStep):
            for step in graph.steps.values():
                if isinstance(step, RouterStep):
                    routes.update(step.routes)
        processor = get_stream_processor(self.metadata.project or "default")
        for key, route in routes.items():
            processor.update_endpoint_record(
                key, model=route.class_args.get("model_path", "")
                if hasattr(route, "class_args") else "",
                function_uri=self.uri)

    def invoke(self, path: str, body=None, method="POST", headers=None):
        """Call the deployed function (HTTP) or the mock server."""
        if self._host is not None:
            import requests

            data = body
            if isinstance(body, (dict, list)):
                import json as _json

                data = _json.dumps(body, default=str)
            resp = requests.request(method,
                                    self._host.address + path, data=data,
                                    headers=headers or
                                    {"content-type": "application/json"},
                                    timeout=60)
            try:
                return resp.json()
            except ValueError:
                return resp.content
        if self._mock_server is None:
            self._mock_server = self.to_mock_server()
        return self._mock_server.test(path, body=body, method=method,
                                      headers=headers)

    def stop(self):
        if getattr(self, "_worker_pool", None) is not None:
            self._worker_pool.stop()
            self._worker_pool = None
        for host in getattr(self, "_child_hosts", []):
            host.stop()
        self._child_hosts = []
        if self._host is not None:
            self._host.stop()
            self._host = None

    def with_secrets(self, kind, source):
        self.spec.secret_sources.append({"kind": kind, "source": source})
        return self

    def with_replicas(self, min_replicas: int, max_replicas: int = None):
        """Worker-process count (reference nuclio min/max_replicas):
        the pool starts min_replicas workers and AUTOSCALES up to
        max_replicas when live connections exceed the per-worker
        budget (serving/workers.py WorkerPool._scale_loop)."""
        self.spec.min_replicas = min_replicas
        self.spec.max_replicas = max_replicas or min_replicas
        return self


def _caller_namespace():
    """Find user classes in the __main__ / caller globals."""
    import inspect
    import sys

    namespace = {}
    main = sys.modules.get("__main__")
    if main is not None:
        namespace.update(vars(main))
    # walk a few frames up past this module
    frame = inspect.currentframe()
    try:
        for _ in range(8):
            if frame is None:
                break
            module = frame.f_globals.get("__name__", "")
            if not module.startswith("mlrun_amd"):
                namespace.update(frame.f_globals)
            frame = frame.f_back
    finally:
        del frame
    return namespace
