# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Real-time HTTP function runtimes (nuclio replacement).

RemoteRuntime (kind="remote"/"nuclio"): a python handler behind a
node-local HTTP server.  ApplicationRuntime (kind="application"): a
long-lived app process.  Parity target: reference
mlrun/runtimes/nuclio/function.py:253 RemoteRuntime (deploy :551,
invoke) and application/application.py:182.
"""

import json
import subprocess
import sys
import typing

from ..errors import MLRunInvalidArgumentError
from ..serving.server import Event, GraphServer, GraphServerHost
from ..serving.states import RootFlowStep
from ..utils import logger
from .base import BaseRuntime


class RemoteRuntime(BaseRuntime):
    """A single python handler served over HTTP."""

    kind = "remote"

    def __init__(self, metadata=None, spec=None):
        super().__init__(metadata, spec)
        self._host: typing.Optional[GraphServerHost] = None
        self._handler_fn = None
        self._triggers: list = []
        self._trigger_threads: list = []
        self._trigger_stop = None

    def with_http(self, workers=None, port=0, host=None, **kwargs):
        self.spec.build["port"] = port
        return self

    def add_trigger(self, name: str, spec) -> "RemoteRuntime":
        """Attach a trigger (reference RemoteRuntime.add_trigger):
        cron triggers invoke the deployed handler on schedule — the
        nuclio cron-trigger analog, node-local."""
        if isinstance(spec, str):
            spec = {"kind": "cron", "interval": spec}
        self._triggers.append({"name": name, **spec})
        return self

    def add_cron_trigger(self, name: str, interval: str = "",
                         body=None, path: str = "/"):
        """Sugar: interval like "10s"/"5m" (nuclio cron interval
        form)."""
        return self.add_trigger(name, {"kind": "cron",
                                       "interval": interval,
                                       "body": body, "path": path})

    def _start_triggers(self):
        import threading

        from ..feature_store.feature_set import parse_span

        if not self._triggers:
            return
        self._trigger_stop = threading.Event()

        def runner(trigger):
            interval = trigger.get("interval", "1m")
            seconds = parse_span(interval) if not str(
                interval).replace(".", "").isdigit() else float(interval)
            while not self._trigger_stop.wait(seconds):
                try:
                    self.invoke(trigger.get("path", "/"),
                                body=trigger.get("body"))
                except Exception as exc:
                    logger.warning("cron trigger invoke failed",
                                   trigger=trigger["name"],
                                   error=str(exc))

        for trigger in self._triggers:
            if trigger.get("kind") != "cron":
                continue
            thread = threading.Thread(target=runner, args=(trigger,),
                                      daemon=True,
                                      name=f"trigger-{trigger['name']}")
            thread.start()
            self._trigger_threads.append(thread)

    def _resolve_handler(self):
        if self._handler_fn is not None:
            return self._handler_fn
        handler = self.spec.default_handler or "handler"
        command = self.spec.command
        if command:
            from .local import load_module

            module = load_module(command)
            self._handler_fn = getattr(module, handler)
        else:
            source = self.spec.build.get("functionSourceCode")
            if not source:
                raise MLRunInvalidArgumentError(
                    "remote function needs command or embedded source")
            namespace: dict = {}
            exec(compile(source, "<function source>", "exec"), namespace)
            self._handler_fn = namespace[handler]
        return self._handler_fn

    def deploy(self, project="", tag="", verbose=False, auth_info=None,
               builder_env=None) -> str:
        handler_fn = self._resolve_handler()

        # wrap the nuclio-style handler(context, event) in a 1-step graph
        class _HandlerStep:
            def __init__(self, context=None, name=None):
                self.context = context

            def do_event(self, event):
                result = handler_fn(self.context, event)
                if result is not None and not isinstance(result, Event):
                    event.body = result
                return event

        graph = RootFlowStep()
        graph.add_step(_HandlerStep, name="handler")
        server = GraphServer(graph=graph)
        server.init_states(namespace={"_HandlerStep": _HandlerStep})
        self._host = GraphServerHost(server,
                                     port=self.spec.build.get("port", 0))
        self._host.start()
        self.status.state = "ready"
        self.status.address = self._host.address
        self.status.external_invocation_urls = [self._host.address]
        self._start_triggers()
        logger.info("remote function deployed", address=self._host.address)
        return self._host.address

    def invoke(self, path: str = "/", body=None, method="POST", headers=None,
               dashboard="", force_external_address=False):
        if self._host is None:
            self.deploy()
        import requests

        data = body
        if isinstance(body, (dict, list)):
            data = json.dumps(body, default=str)
        resp = requests.request(method, self._host.address + path, data=data,
                                headers=headers or
                                {"content-type": "application/json"},
                                timeout=60)
        try:
            return resp.json()
        except ValueError:
            return resp.content

    def stop(self):
        if self._trigger_stop is not None:
            self._trigger_stop.set()
            for thread in self._trigger_threads:
                thread.join(timeout=3)
            self._trigger_threads = []
        if self._host is not None:
            self._host.stop()
            self._host = None


class ApplicationRuntime(BaseRuntime):
    """A long-lived application process (reverse-proxy-less analog of
    the reference's sidecar model)."""

    kind = "application"

    def __init__(self, metadata=None, spec=None):
        super().__init__(metadata, spec)
        self._process: typing.Optional[subprocess.Popen] = None

    def deploy(self, project="", tag="", verbose=False, **kwargs):
        if not self.spec.command:
            raise MLRunInvalidArgumentError(
                "application runtime needs a command to run")
        args = [str(a) for a in (self.spec.args or [])]
        cmd = self.spec.command
        if cmd.endswith(".py"):
            full = [sys.executable, "-u", cmd] + args
        else:
            full = [cmd] + args
        self._process = subprocess.Popen(full)
        self.status.state = "ready"
        logger.info("application started", pid=self._process.pid)
        return f"pid://{self._process.pid}"

    def is_running(self) -> bool:
        return self._process is not None and self._process.poll() is None

    def stop(self):
        if self._process is not None and self._process.poll() is None:
            self._process.terminate()
            try:
                self._process.wait(timeout=10)
            except subprocess.TimeoutExpired:
                self._process.kill()
        self._process = None
