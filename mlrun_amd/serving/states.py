# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Serving-graph step DAG.

Step kinds: task (run a class/handler), router (choose a route),
queue (buffered hand-off between branches), flow (DAG of steps).
Parity target: reference mlrun/serving/states.py (BaseStep :102,
TaskStep :398, RouterStep :671, QueueStep :801, FlowStep :892 with the
sync hot loop :1279-1320, error_handler/respond semantics :556).

Design difference from the reference: instead of compiling to the
external storey asyncio engine, the flow compiles to a flat next-step
chain interpreted by a tight sync loop (per-event overhead is a few
dict lookups), with an optional thread-pool async facade in
server.py.  GPU steps (V2ModelServer subclasses with HIP kernels)
batch events and capture their kernel chain in a hipGraph — see
mlrun_amd/serving/v2_serving.py and mlrun_amd/models/llama.py.
"""

import copy
import inspect
import queue as queue_mod
import threading
import traceback
import typing

from ..errors import MLRunInvalidArgumentError
from ..utils import logger

callable_prefix = "_"
path_splitter = "/"


class GraphError(Exception):
    pass


class StepKinds:
    router = "router"
    task = "task"
    flow = "flow"
    queue = "queue"
    choice = "choice"
    root = "root"


_task_step_fields = ["kind", "class_name", "class_args", "handler", "name",
                     "after", "function", "comment", "shape", "full_event",
                     "on_error", "responder", "input_path", "result_path"]


class _StepBase:
    kind = "step"
    has_children = False

    def __init__(self, name: str = None, after: list = None, shape=None):
        self.name = name or ""
        self.after = after or []
        self.shape = shape
        self.comment = ""
        self.context = None
        self.parent: typing.Optional["FlowStep"] = None
        self.on_error: typing.Optional[str] = None
        self._next: typing.List[str] = []
        self._on_error_step = None
        self.responder = False

    @property
    def fullname(self) -> str:
        name = self.name
        parent = self.parent
        while parent is not None and parent.name:
            name = f"{parent.name}.{name}"
            parent = parent.parent
        return name

    def to(self, class_name=None, name=None, handler=None, graph_shape=None,
           function=None, full_event=None, input_path=None, result_path=None,
           **class_args) -> "_StepBase":
        """Add a downstream step and return it (chaining sugar)."""
        if self.parent is None:
            raise GraphError(
                f"step {self.name} must be part of a graph before .to()")
        step = self.parent.add_step(
            class_name, name=name, handler=handler, after=[self.name],
            function=function, full_event=full_event, input_path=input_path,
            result_path=result_path, **class_args)
        return step

    def error_handler(self, name: str = None, class_name=None, handler=None,
                      before=None, function=None, full_event=None,
                      **class_args):
        """Route step errors to a named (or new ErrorStep) step
        (reference states.py:155); ``before`` names the step(s) the
        flow continues with after the handler runs."""
        if class_name or handler:
            step = ErrorStep(class_name, class_args=class_args or None,
                             handler=handler, name=name,
                             full_event=full_event, function=function)
            step.base_step = getattr(self, "name", None)
            parent = self.parent if getattr(self, "parent", None) \
                is not None else self
            parent._attach(step)
            if before:
                step.before = [before] if isinstance(before, str) \
                    else list(before)
            self.on_error = step.name
        else:
            self.on_error = name
        return self

    def respond(self):
        """Mark this step as the responder: its output becomes the
        event response and downstream continues async."""
        self.responder = True
        return self

    def init_object(self, context, namespace, mode="sync", reset=False):
        self.context = context

    def run(self, event):
        raise NotImplementedError

    def to_dict(self):
        struct = {"kind": self.kind, "name": self.name}
        if self.after:
            struct["after"] = self.after
        if self.on_error:
            struct["on_error"] = self.on_error
        if self.responder:
            struct["responder"] = True
        return struct

    def __repr__(self):
        return f"{self.__class__.__name__}({self.name!r})"


class BaseStep(_StepBase):
    pass


def _resolve_class(class_name, namespace):
    if class_name is None or not isinstance(class_name, str):
        return class_name
    if namespace and class_name in namespace:
        return namespace[class_name]
    if "." in class_name:
        module_name, _, cls = class_name.rpartition(".")
        import importlib

        module = importlib.import_module(module_name)
        return getattr(module, cls)
    # search loaded modules (globals of the server module)
    import builtins

    if hasattr(builtins, class_name):
        return getattr(builtins, class_name)
    raise MLRunInvalidArgumentError(f"class {class_name} not found")


class TaskStep(BaseStep):
    """Run a class instance (with do(event)) or a handler function."""

    kind = "task"

    def __init__(self, class_name=None, class_args=None, handler=None,
                 name=None, after=None, full_event=None, function=None,
                 responder=False, input_path=None, result_path=None):
        super().__init__(name, after)
        self.class_name = class_name if isinstance(class_name, str) else (
            class_name.__name__ if inspect.isclass(class_name) else None)
        self._class_object = class_name if inspect.isclass(class_name) \
            else None
        self.class_args = class_args or {}
        self.handler = handler if isinstance(handler, str) else None
        self._handler_fn = handler if callable(handler) else None
        self.full_event = full_event
        self.function = function
        self.responder = responder
        self.input_path = input_path
        self.result_path = result_path
        self._object = None
        if class_name is not None and not isinstance(class_name, str) and \
                not inspect.isclass(class_name):
            # a pre-built instance
            self._object = class_name
            self.class_name = type(class_name).__name__

    @property
    def object(self):
        return self._object

    def init_object(self, context, namespace, mode="sync", reset=False):
        self.context = context
        if self._object is not None and not reset:
            self._post_init(mode)
            return
        if self.class_name:
            cls = self._class_object or _resolve_class(self.class_name,
                                                       namespace)
            args = dict(self.class_args)
            sig = inspect.signature(cls.__init__)
            if "context" in sig.parameters:
                args["context"] = context
            if "name" in sig.parameters:
                args.setdefault("name", self.name)
            if "graph_step" in sig.parameters:
                args["graph_step"] = self
            self._object = cls(**args)
            if hasattr(self._object, "context"):
                self._object.context = context
        elif self.handler and self._handler_fn is None:
            self._handler_fn = _resolve_handler(self.handler, namespace)
        self._post_init(mode)

    def _post_init(self, mode):
        obj = self._object
        if obj is not None and hasattr(obj, "post_init"):
            obj.post_init(mode)

    def run(self, event):
        # input_path/result_path: operate on a body subfield
        # (reference states.py TaskStep event-path semantics)
        saved_body = None
        if self.input_path and isinstance(event.body, dict):
            from ..utils import get_in

            saved_body = event.body
            event.body = get_in(saved_body, self.input_path)
        if self._object is not None:
            if self.handler:
                result = getattr(self._object, self.handler)(event)
            else:
                result = self._object.do_event(event) if hasattr(
                    self._object, "do_event") else self._object.do(event)
            event = result if result is not None else event
        elif self._handler_fn is not None:
            if self.full_event:
                result = self._handler_fn(event)
                event = result if result is not None else event
            else:
                event.body = self._handler_fn(event.body)
        if saved_body is not None:
            from ..utils import update_in

            if self.result_path:
                update_in(saved_body, self.result_path, event.body)
            elif self.input_path:
                update_in(saved_body, self.input_path, event.body)
            event.body = saved_body
        elif self.result_path and isinstance(event.body, dict) is False:
            pass
        return event

    def to_dict(self):
        struct = super().to_dict()
        struct.update({k: v for k, v in {
            "class_name": self.class_name,
            "class_args": _safe_args(self.class_args),
            "handler": self.handler,
            "full_event": self.full_event,
            "function": self.function,
        }.items() if v})
        return struct


def _safe_args(args: dict) -> dict:
    return {k: v for k, v in (args or {}).items()
            if isinstance(v, (str, int, float, bool, list, dict, type(None)))}


def _resolve_handler(handler: str, namespace):
    if namespace and handler in namespace:
        return namespace[handler]
    if "." in handler:
        module_name, _, fn = handler.rpartition(".")
        import importlib

        module = importlib.import_module(module_name)
        return getattr(module, fn)
    raise MLRunInvalidArgumentError(f"handler {handler} not found")


class RouterStep(TaskStep):
    """A task step whose object is a router holding named routes."""

    kind = "router"
    has_children = True

    def __init__(self, class_name=None, class_args=None, routes=None,
                 name=None, after=None, function=None):
        super().__init__(class_name, class_args, name=name or "router",
                         after=after, function=function)
        self.routes: typing.Dict[str, BaseStep] = {}
        if routes:
            for key, route in routes.items():
                self.add_route(key, route=route)

    def add_route(self, key, route=None, class_name=None, handler=None,
                  function=None, **class_args) -> BaseStep:
        if route is None:
            route = TaskStep(class_name, class_args, handler=handler,
                             name=key, function=function)
        route.name = key
        route.parent = self
        self.routes[key] = route
        return route

    def clear_children(self, routes: list = None):
        if not routes:
            self.routes = {}
        else:
            for key in routes:
                self.routes.pop(key, None)

    def init_object(self, context, namespace, mode="sync", reset=False):
        self.class_name = self.class_name or "ModelRouter"
        if not self._object:
            from .routers import router_classes

            cls = None
            if self.class_name in router_classes:
                cls = router_classes[self.class_name]
            else:
                cls = self._class_object or _resolve_class(self.class_name,
                                                           namespace)
            args = dict(self.class_args)
            self._object = cls(context=context, name=self.name,
                               routes=self.routes, **args)
        else:
            self._object.routes = self.routes
            self._object.context = context
        for route in self.routes.values():
            route.init_object(context, namespace, mode, reset=reset)
        self._post_init(mode)
        self.context = context

    def run(self, event):
        return self._object.do_event(event)

    def to_dict(self):
        struct = super().to_dict()
        struct["routes"] = {k: r.to_dict() for k, r in self.routes.items()}
        return struct

    def __getitem__(self, key):
        return self.routes[key]


class QueueStep(BaseStep):
    """Buffered hand-off between graph branches.

    In the reference this maps to a Kafka/V3IO stream between child
    functions (states.py:801); node-locally it is an in-process
    bounded queue drained by a worker thread, preserving the
    same decoupling semantics.  For GPU-to-GPU hand-off the body may
    be a torch tensor — passed by reference on the same device, or
    (cross-process) via hipIpc in the mpijob runtime.
    """

    kind = "queue"

    def __init__(self, name=None, after=None, path=None, shards=1,
                 retention_in_hours=None, trigger_args=None, max_size=None,
                 **options):
        super().__init__(name, after)
        self.path = path
        self.shards = shards
        self.retention_in_hours = retention_in_hours
        self.options = options
        self.max_size = max_size
        self._queue: typing.Optional[queue_mod.Queue] = None
        self._worker: typing.Optional[threading.Thread] = None
        self._stop = False

    def init_object(self, context, namespace, mode="sync", reset=False):
        from ..config import config

        self.context = context
        if self.parent is not None and \
                getattr(self.parent, "engine", "sync") == "async":
            # async engine: the step's bounded inbox IS the buffer —
            # no thread-backed queue (see async_flow.py)
            return
        if self._queue is None:
            self._queue = queue_mod.Queue(
                maxsize=self.max_size or int(config.serving.max_queue))
        if self._next and self._worker is None:
            self._worker = threading.Thread(target=self._drain, daemon=True,
                                            name=f"queue-{self.name}")
            self._worker.start()

    def _drain(self):
        while not self._stop:
            try:
                event = self._queue.get(timeout=0.2)
            except queue_mod.Empty:
                continue
            try:
                for next_name in self._next:
                    step = self.parent[next_name]
                    self.parent._run_from(step, copy.copy(event))
            except Exception as exc:
                logger.error("queue consumer failed", error=str(exc))
            finally:
                self._queue.task_done()

    def run(self, event):
        if self.parent is not None and \
                getattr(self.parent, "engine", "sync") == "async":
            # async engine: passthrough — downstream steps have their
            # own bounded inboxes; a pathed queue with no local
            # consumers publishes to its stream instead (reference
            # _init_async_objects stream-target behavior)
            if self.path and not self._next:
                from ..platforms import OutputStream

                OutputStream(self.path).push([event.body])
                event.terminated = True
            return event
        if self._queue is None:
            self.init_object(self.context, None)
        self._queue.put(event)
        event.terminated = True
        return event

    def to_dict(self):
        struct = super().to_dict()
        if self.path:
            struct["path"] = self.path
        return struct


class FlowStep(BaseStep):
    """A DAG of steps executed by the sync interpreter."""

    kind = "flow"
    has_children = True

    def __init__(self, name=None, steps=None, after=None, engine=None,
                 final_step=None):
        super().__init__(name, after)
        self.steps: typing.Dict[str, BaseStep] = {}
        self.engine = engine or "sync"
        self.final_step = final_step
        self._start_steps: typing.List[BaseStep] = []
        self._controller = None  # async engine (engine="async")
        if steps:
            for key, step in steps.items():
                step.name = step.name or key
                self._attach(step)

    def _attach(self, step: BaseStep):
        step.parent = self
        self.steps[step.name] = step
        return step

    def __getitem__(self, name) -> BaseStep:
        return self.steps[name]

    def __contains__(self, name) -> bool:
        return name in self.steps

    def add_step(self, class_name=None, name=None, handler=None, after=None,
                 before=None, function=None, full_event=None,
                 input_path=None, result_path=None, **class_args) -> BaseStep:
        name = name or (class_name if isinstance(class_name, str) and
                        "." not in class_name else None) or \
            (getattr(class_name, "__name__", None)
             if inspect.isclass(class_name) else None) or \
            (handler if isinstance(handler, str) else
             getattr(handler, "__name__", None)) or f"step{len(self.steps)}"
        if isinstance(class_name, BaseStep):
            step = class_name
            step.name = step.name or name
        elif class_name == "$queue" or class_args.pop("_queue", False):
            step = QueueStep(name, path=class_args.pop("path", None),
                             max_size=class_args.pop("max_size", None),
                             **class_args)
            class_args = {}
        elif isinstance(class_name, str) and class_name.startswith("*"):
            # "*" = router step ("*RouterClass" names the router class —
            # reference states.py new_model_endpoint/"*" convention)
            step = RouterStep(class_name=class_name[1:] or None,
                              class_args=class_args, name=name,
                              function=function)
        else:
            step = TaskStep(class_name, class_args, handler=handler,
                            name=name, full_event=full_event,
                            function=function, input_path=input_path,
                            result_path=result_path)
        if after:
            step.after = [after] if isinstance(after, str) else list(after)
            if "$prev" in step.after:
                prev = list(self.steps.values())[-1].name if self.steps \
                    else None
                step.after = [a for a in step.after if a != "$prev"] + \
                    ([prev] if prev else [])
        elif self.steps:
            # default chaining: after the previously added step
            step.after = [list(self.steps.values())[-1].name]
        if self.engine == "async" and not isinstance(step, QueueStep):
            # a step consuming from a queue runs in a CHILD function —
            # it must name one (reference async-flow contract, tested
            # by test_async_error_on_missing_function_parameter)
            for upstream in step.after or []:
                if upstream in self.steps and \
                        isinstance(self.steps[upstream], QueueStep) and \
                        not (function or getattr(step, "function", None)):
                    raise MLRunInvalidArgumentError(
                        f"step '{step.name}' must specify a function, "
                        f"because it follows a queue step")
        self._attach(step)
        if before:
            self.steps[before].after = [step.name]
        return step

    def to(self, class_name=None, name=None, handler=None, **class_args):
        """First step of the flow (root-level .to chaining)."""
        if not self.steps:
            step = self.add_step(class_name, name=name, handler=handler,
                                 **class_args)
            step.after = []
            return step
        return super().to(class_name, name=name, handler=handler,
                          **class_args)

    def add_route(self, key, route=None, class_name=None, handler=None,
                  **class_args):
        """Sugar: add a route to the (single) router child."""
        for step in self.steps.values():
            if isinstance(step, RouterStep):
                return step.add_route(key, route=route, class_name=class_name,
                                      handler=handler, **class_args)
        raise GraphError("no router step in flow")

    def init_object(self, context, namespace, mode="sync", reset=False):
        self.context = context
        self._build_links()
        for step in self.steps.values():
            step.init_object(context, namespace, mode, reset=reset)
        if self.engine == "async" and self._controller is None:
            from ..config import config
            from .async_flow import AsyncFlowController

            self._controller = AsyncFlowController(
                self, max_queue=int(config.serving.max_queue))
            self._controller.start()

    def _build_links(self):
        for step in self.steps.values():
            step._next = []
        # steps referenced as on_error targets are never start steps
        error_targets = {s.on_error for s in self.steps.values()
                         if s.on_error}
        if self.on_error:
            error_targets.add(self.on_error)
        self._start_steps = []
        for step in self.steps.values():
            if step.after:
                for upstream in step.after:
                    if upstream not in self.steps:
                        raise GraphError(
                            f"step {step.name} is after unknown step "
                            f"{upstream}")
                    self.steps[upstream]._next.append(step.name)
            elif step.name not in error_targets:
                self._start_steps.append(step)
        # ErrorStep.before: the flow continues with those steps after
        # the handler runs (reference error_handler `before` param)
        for step in self.steps.values():
            for next_name in getattr(step, "before", None) or []:
                if next_name not in self.steps:
                    raise GraphError(
                        f"error step {step.name} is before unknown "
                        f"step {next_name}")
                if next_name not in step._next:
                    step._next.append(next_name)
        if self.final_step and self.final_step in self.steps:
            pass

    def run(self, event):
        if not self._start_steps:
            self._build_links()
        if self._controller is not None:
            # async engine: submit + block on the per-event future
            # (async callers use run_async for a non-blocking await)
            return self._controller.emit(event, timeout=600)
        for step in self._start_steps:
            result = self._run_from(step, event)
            event = result if result is not None else event
        return event

    def run_from_step(self, step_name: str, event):
        """Run starting at a NAMED step (child-function hosts entering
        mid-graph — reference queue/stream links deliver into the
        consumer step)."""
        if step_name not in self.steps:
            raise GraphError(f"unknown start step {step_name}")
        if not self._start_steps:
            self._build_links()
        if self._controller is not None:
            return self._controller.emit(event, timeout=600,
                                         start_step=step_name)
        return self._run_from(self.steps[step_name], event)

    async def run_async(self, event):
        """Awaitable entry for async hosts (FastAPI): wraps the
        per-event future."""
        if self._controller is None:
            return self.run(event)
        import asyncio as _asyncio

        return await _asyncio.wrap_future(
            self._controller.emit_nowait(event))

    def _run_from(self, step: BaseStep, event):
        """The sync hot loop: walk the next-chain from a step
        (parity: reference states.py:1293-1320)."""
        from ..utils import tracing

        trace = tracing.is_enabled()
        hops = 0
        while step is not None:
            hops += 1
            if hops > 10000:  # cycle guard: a mis-wired graph must
                raise GraphError(  # error, not hang the server
                    f"flow exceeded 10000 hops at step "
                    f"{step.fullname} — graph cycle?")
            try:
                if trace:
                    with tracing.trace_step(step.fullname):
                        event = step.run(event)
                else:
                    event = step.run(event)
            except Exception as exc:
                event = self._handle_error(step, event, exc)
                if event is None or getattr(event, "error", None):
                    return event
                return event
            if event is None or getattr(event, "terminated", False):
                return event
            if step.responder:
                event.responded = True
            nxt = step._next
            if not nxt:
                return event
            if len(nxt) == 1:
                step = self.steps[nxt[0]]
            else:
                # fan-out: run each branch with a shallow-copied event
                last = None
                for name in nxt:
                    branch_event = copy.copy(event)
                    last = self._run_from(self.steps[name], branch_event)
                return last
        return event

    def _handle_error(self, step: BaseStep, event, exc: Exception):
        handler_name = step.on_error or self.on_error
        error_text = f"{type(exc).__name__}: {exc}"
        if self.context and getattr(self.context, "verbose", False):
            logger.error(f"step {step.fullname} failed",
                         error=error_text,
                         tb=traceback.format_exc())
        if handler_name and handler_name in self.steps:
            event.error = error_text
            event.origin_state = step.fullname
            result = self._run_from(self.steps[handler_name], event)
            if result is not None and getattr(result, "error", None) == \
                    error_text:
                # the handler consumed the error: its output is a normal
                # response (async recovery-step semantics; origin_state
                # stays readable)
                result.error = None
            return result
        event.error = error_text
        event.origin_state = step.fullname
        return event

    def wait_for_completion(self):
        if self._controller is not None:
            self._controller.wait_for_completion()
        for step in self.steps.values():
            if isinstance(step, QueueStep) and step._queue is not None:
                step._queue.join()

    def shutdown(self):
        """Stop the async controller (and any queue workers)."""
        if self._controller is not None:
            self._controller.stop()
            self._controller = None
        for step in self.steps.values():
            if isinstance(step, QueueStep):
                step._stop = True

    def to_dict(self):
        struct = super().to_dict()
        struct["steps"] = {k: s.to_dict() for k, s in self.steps.items()}
        struct["engine"] = self.engine
        return struct

    def plot(self, filename=None, format=None, **kw):
        lines = [f"{s.name} -> {n}" for s in self.steps.values()
                 for n in s._next]
        text = "\n".join(lines) or "(empty flow)"
        if filename:
            with open(filename, "w") as fp:
                fp.write(text)
        return text


class RootFlowStep(FlowStep):
    """Top-level flow (parity: reference states.py:1405)."""

    kind = "root"


class ErrorStep(TaskStep):
    """Error-handler execution step (reference states.py:635): a task
    step attached via ``step.error_handler(...)``; ``before`` names
    steps whose errors it handles, ``base_step`` the one it resumes."""

    kind = "error_step"

    def __init__(self, class_name=None, class_args=None, handler=None,
                 name=None, after=None, full_event=None, function=None,
                 responder=False, input_path=None, result_path=None):
        super().__init__(class_name=class_name, class_args=class_args,
                         handler=handler, name=name, after=after,
                         full_event=full_event, function=function,
                         responder=responder, input_path=input_path,
                         result_path=result_path)
        self.before = None
        self.base_step = None

    def to_dict(self):
        struct = super().to_dict()
        if self.before:
            struct["before"] = self.before
        if self.base_step:
            struct["base_step"] = self.base_step
        return struct


class MonitoringApplicationStep(TaskStep):
    """Model-monitoring application step (reference states.py:602):
    a task step running a monitoring app class."""

    kind = "monitoring_application"


classes_map = {
    "task": TaskStep,
    "router": RouterStep,
    "flow": FlowStep,
    "queue": QueueStep,
    "error_step": ErrorStep,
    "monitoring_application": MonitoringApplicationStep,
}


def graph_root_setter(server, graph):
    """Build the graph object from a spec dict / step object."""
    if isinstance(graph, (RootFlowStep, FlowStep)):
        root = RootFlowStep(steps=graph.steps) if not isinstance(
            graph, RootFlowStep) else graph
    elif isinstance(graph, RouterStep):
        root = RootFlowStep()
        graph.name = graph.name or "router"
        root._attach(graph)
    elif isinstance(graph, dict):
        root = step_from_dict({"kind": "root", **graph})
    else:
        raise MLRunInvalidArgumentError("unsupported graph object")
    return root


def step_from_dict(struct: dict) -> BaseStep:
    kind = struct.get("kind", "task")
    if kind in ("flow", "root"):
        flow = RootFlowStep(name=struct.get("name")) if kind == "root" \
            else FlowStep(name=struct.get("name"))
        flow.engine = struct.get("engine", "sync")
        for name, child in (struct.get("steps") or {}).items():
            child.setdefault("name", name)
            step = step_from_dict(child)
            flow._attach(step)
        flow.on_error = struct.get("on_error")
        return flow
    if kind == "router":
        router = RouterStep(class_name=struct.get("class_name"),
                            class_args=struct.get("class_args"),
                            name=struct.get("name"))
        for key, route in (struct.get("routes") or {}).items():
            route.setdefault("name", key)
            router.add_route(key, route=step_from_dict(route))
        router.after = struct.get("after", [])
        router.on_error = struct.get("on_error")
        return router
    if kind == "queue":
        step = QueueStep(name=struct.get("name"), path=struct.get("path"))
        step.after = struct.get("after", [])
        return step
    step_cls = classes_map.get(kind, TaskStep)
    step = step_cls(class_name=struct.get("class_name"),
                    class_args=struct.get("class_args"),
                    handler=struct.get("handler"),
                    name=struct.get("name"),
                    full_event=struct.get("full_event"))
    step.after = struct.get("after", [])
    step.on_error = struct.get("on_error")
    step.responder = struct.get("responder", False)
    if isinstance(step, ErrorStep):
        step.before = struct.get("before")
        step.base_step = struct.get("base_step")
    return step
