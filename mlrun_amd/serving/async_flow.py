# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Async flow engine for serving graphs (``engine="async"``).

The reference compiles async flows to the external storey asyncio
engine (states.py:1190 ``_build_async_flow``, :1622
``_init_async_objects``); this is the in-process MI355X-native
equivalent with the same observable semantics:

- **awaitable per-event results** — every emitted event gets a future
  resolved by the responder step (``.respond()``) or, if none, when the
  event has fully drained through the DAG
- **responder mid-flow** — the responder's output becomes the response
  while downstream branches keep running in the background
- **back-pressure** — each step owns a BOUNDED inbox; a slow step makes
  upstream ``put``s await, propagating pressure to the source
- **pipelining** — each step drains its inbox on its own worker; sync
  step bodies run on a per-step single-thread executor (per-step event
  order preserved), so different steps process different events
  concurrently — the asyncio analog of storey's event pipelining
- **error handlers** — a step error sets ``event.error`` +
  ``event.origin_state`` and routes to the step's (or flow's) error
  handler, whose downstream then continues; with no handler the event
  resolves as an error response
- **multi-function graphs** — steps annotated with ``function=`` that
  follow a queue step are forwarded to that child function's host over
  HTTP when one is registered (``context.get_remote_endpoint``); in
  mock servers they run locally (the reference's ``context.is_mock``
  skip-stream behavior, states.py:1632)
"""

import asyncio
import concurrent.futures
import copy
import threading
import traceback
import typing

from ..utils import logger


class _Envelope:
    """Per-event bookkeeping: result future + outstanding branch count.
    All mutation happens on the loop thread — no locking."""

    __slots__ = ("future", "outstanding", "last_event")

    def __init__(self, future: concurrent.futures.Future):
        self.future = future
        self.outstanding = 1
        self.last_event = None

    def fan_out(self, n: int):
        """This branch forks into n branches (n may be 0: branch dies)."""
        self.outstanding += n - 1

    def finish_branch(self, event):
        self.outstanding -= 1
        if event is not None:
            self.last_event = event
        if self.outstanding <= 0 and not self.future.done():
            # no responder fired: resolve with the last terminal event
            self.future.set_result(self.last_event)

    def respond(self, event):
        if not self.future.done():
            self.future.set_result(event)


class AsyncFlowController:
    """Drives a FlowStep DAG on a background asyncio loop."""

    def __init__(self, flow, max_queue: int = 128):
        self.flow = flow
        self.max_queue = max_queue
        self._loop: typing.Optional[asyncio.AbstractEventLoop] = None
        self._thread: typing.Optional[threading.Thread] = None
        self._inboxes: typing.Dict[str, asyncio.Queue] = {}
        self._workers: typing.List[asyncio.Task] = []
        self._executors: typing.Dict[
            str, concurrent.futures.ThreadPoolExecutor] = {}
        self._active = 0            # queued + currently-processing items
        self._drained: typing.Optional[asyncio.Event] = None
        self._started = threading.Event()
        self._stopping = False

    # ------------------------------------------------------- lifecycle
    def start(self):
        if self._thread is not None:
            return
        self._thread = threading.Thread(target=self._run_loop,
                                        daemon=True,
                                        name=f"async-flow-{self.flow.name}")
        self._thread.start()
        if not self._started.wait(timeout=10):
            raise RuntimeError("async flow loop failed to start")

    def _run_loop(self):
        self._loop = asyncio.new_event_loop()
        asyncio.set_event_loop(self._loop)
        self._drained = asyncio.Event()
        self._drained.set()
        for step in self.flow.steps.values():
            self._inboxes[step.name] = asyncio.Queue(
                maxsize=self._step_maxsize(step))
            self._workers.append(self._loop.create_task(
                self._worker(step)))
        self._started.set()
        try:
            self._loop.run_forever()
        finally:
            pending = [t for t in self._workers if not t.done()]
            for task in pending:
                task.cancel()
            if pending:
                self._loop.run_until_complete(
                    asyncio.gather(*pending, return_exceptions=True))
            self._loop.close()

    def _step_maxsize(self, step) -> int:
        if getattr(step, "kind", "") == "queue" and \
                getattr(step, "max_size", None):
            return int(step.max_size)
        return self.max_queue

    def stop(self):
        if self._loop is None or self._stopping:
            return
        self._stopping = True
        self._loop.call_soon_threadsafe(self._loop.stop)
        if self._thread is not None:
            self._thread.join(timeout=5)
            self._thread = None
        for pool in self._executors.values():
            pool.shutdown(wait=False)
        self._executors = {}

    # ---------------------------------------------------------- submit
    def emit(self, event, timeout: float = None, start_step: str = None):
        """Submit one event; block until its future resolves (sync
        callers — GraphServer.run)."""
        return self.emit_nowait(event,
                                start_step=start_step).result(
            timeout=timeout)

    def emit_nowait(self, event,
                    start_step: str = None) -> concurrent.futures.Future:
        """Submit one event; return its result future (the awaitable
        per-event result contract).  start_step: enter mid-graph
        (forwarded child-function events)."""
        self.start()
        future: concurrent.futures.Future = concurrent.futures.Future()
        envelope = _Envelope(future)

        async def _inject():
            starts = ([self.flow.steps[start_step]] if start_step
                      else self.flow._start_steps)
            if not starts:
                envelope.finish_branch(event)
                return
            envelope.fan_out(len(starts))
            for step in starts:
                await self._route_to(step, copy.copy(event), envelope)

        asyncio.run_coroutine_threadsafe(_inject(), self._loop)
        return future

    # --------------------------------------------------------- routing
    async def _route_to(self, step, event, envelope):
        """Queue an event into a step's inbox (awaits when the inbox is
        full — back-pressure propagates upstream)."""
        self._active += 1
        self._drained.clear()
        await self._inboxes[step.name].put((event, envelope))

    def _executor_for(self, step):
        pool = self._executors.get(step.name)
        if pool is None:
            pool = concurrent.futures.ThreadPoolExecutor(
                max_workers=1, thread_name_prefix=f"step-{step.name}")
            self._executors[step.name] = pool
        return pool

    async def _worker(self, step):
        inbox = self._inboxes[step.name]
        while True:
            event, envelope = await inbox.get()
            try:
                await self._process(step, event, envelope)
            except Exception as exc:  # must never kill the worker
                logger.error("async flow worker error",
                             step=step.name, error=str(exc),
                             tb=traceback.format_exc())
                envelope.finish_branch(event)
            finally:
                inbox.task_done()
                self._active -= 1
                if self._active <= 0:
                    self._drained.set()

    async def _run_step_body(self, step, event):
        """Execute a step body without blocking the loop: coroutine
        handlers are awaited in-loop; sync bodies run on the step's
        single-thread executor (order-preserving, pipelined)."""
        result = None
        obj = getattr(step, "_object", None)
        target = None
        if obj is not None and hasattr(obj, "do_event") and \
                asyncio.iscoroutinefunction(obj.do_event):
            target = obj.do_event
        if target is not None:
            result = await target(event)
        else:
            loop = asyncio.get_running_loop()
            result = await loop.run_in_executor(
                self._executor_for(step), step.run, event)
            if asyncio.iscoroutine(result):
                result = await result
        return result if result is not None else event

    async def _process(self, step, event, envelope):
        flow = self.flow
        remote = self._remote_endpoint(step)
        try:
            if remote is not None:
                event = await self._run_remote(step, event, remote)
            else:
                event = await self._run_step_body(step, event)
        except Exception as exc:
            handled = await self._handle_error(step, event, envelope, exc)
            if not handled:
                envelope.finish_branch(event)
            return
        if getattr(event, "_recovery", False):
            # the error-handler step ran: the error is consumed and the
            # handler's output is a NORMAL response (reference async
            # recovery-step semantics — origin_state stays readable)
            event.error = None
            event._recovery = False
        if event is None or getattr(event, "terminated", False):
            envelope.finish_branch(event)
            return
        if step.responder:
            event.responded = True
            envelope.respond(copy.copy(event))
        nexts = [flow.steps[n] for n in step._next]
        if not nexts:
            envelope.finish_branch(event)
            return
        if envelope.last_event is None:
            envelope.last_event = event
        envelope.fan_out(len(nexts))
        for next_step in nexts:
            await self._route_to(next_step, copy.copy(event), envelope)

    async def _handle_error(self, step, event, envelope, exc) -> bool:
        """Reference error semantics (states.py:556): set error +
        origin_state, then run the error-handler chain if any."""
        flow = self.flow
        error_text = f"{type(exc).__name__}: {exc}"
        event.error = error_text
        event.origin_state = step.fullname
        handler_name = step.on_error or flow.on_error
        context = getattr(flow, "context", None)
        if context is not None and getattr(context, "verbose", False):
            logger.error(f"async step {step.fullname} failed",
                         error=error_text)
        if handler_name and handler_name in flow.steps and \
                handler_name != step.name:
            event._recovery = True
            await self._route_to(flow.steps[handler_name], event,
                                 envelope)
            return True
        # no handler: the error IS the response
        if not envelope.future.done():
            envelope.future.set_result(event)
        return False

    # ------------------------------------------------ remote functions
    def _remote_endpoint(self, step):
        function = getattr(step, "function", None)
        if not function or function == "*":
            return None
        context = getattr(self.flow, "context", None)
        if context is None or not hasattr(context, "get_remote_endpoint"):
            return None
        try:
            endpoint = context.get_remote_endpoint(function)
        except Exception:
            return None
        return endpoint or None

    async def _run_remote(self, step, event, endpoint):
        """POST the event body to the child function's host; the
        response body continues through this step's downstream
        (pipeline decomposition across processes — the reference's
        queue/stream → child-function links, states.py:1231)."""

        def _post():
            import json as _json

            import requests

            url = endpoint.rstrip("/") + (event.path or "/")
            body = event.body
            data = _json.dumps(body, default=str) if isinstance(
                body, (dict, list)) else body
            resp = requests.post(url, data=data, timeout=60,
                                 headers={"content-type":
                                          "application/json",
                                          "x-mlrun-step": step.name})
            resp.raise_for_status()
            try:
                return resp.json()
            except ValueError:
                return resp.text

        loop = asyncio.get_running_loop()
        result = await loop.run_in_executor(self._executor_for(step),
                                            _post)
        event.body = result
        return event

    # --------------------------------------------------------- waiting
    def wait_for_completion(self, timeout: float = 30.0):
        """Block until every queued AND in-flight item has finished
        (reference graph.wait_for_completion)."""
        if self._loop is None:
            return

        async def _wait():
            await self._drained.wait()

        asyncio.run_coroutine_threadsafe(
            _wait(), self._loop).result(timeout=timeout)
