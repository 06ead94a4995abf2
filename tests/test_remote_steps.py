# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""RemoteStep / BatchHttpRequests against a real local HTTP host, and
the ApplicationRuntime process lifecycle."""

import sys
import time

import mlrun_amd
from mlrun_amd.serving import GraphServer, GraphServerHost, MockEvent
from mlrun_amd.serving.remote import BatchHttpRequests, RemoteStep
from mlrun_amd.serving.states import RootFlowStep


class _Echo:
    def __init__(self, context=None, name=None):
        pass

    def do_event(self, event):
        body = event.body if isinstance(event.body, dict) else {}
        body["echoed"] = True
        event.body = body
        return event


def _make_host():
    graph = RootFlowStep()
    graph.add_step(_Echo, name="echo")
    server = GraphServer(graph=graph)
    server.init_states(namespace={"_Echo": _Echo})
    host = GraphServerHost(server, port=0)
    host.start()
    return host


class TestRemoteStep:
    def test_calls_local_graph_host(self):
        host = _make_host()
        try:
            step = RemoteStep(url=host.address, name="r")
            event = MockEvent(body={"x": 1})
            out = step.do_event(event)
            assert out.body["echoed"] is True and out.body["x"] == 1
        finally:
            host.stop()

    def test_url_expression_and_subpath(self):
        host = _make_host()
        try:
            step = RemoteStep(url=host.address, subpath="$route",
                              name="r")
            out = step.do_event(MockEvent(body={"route": "sub/path",
                                                "y": 2}))
            assert out.body["y"] == 2
        finally:
            host.stop()

    def test_batch_requests_fan_out(self):
        host = _make_host()
        try:
            step = BatchHttpRequests(url=host.address, name="b",
                                     max_in_flight=4)
            out = step.do_event(MockEvent(body=[{"i": i}
                                                for i in range(6)]))
            assert len(out.body) == 6
            assert all(item["echoed"] for item in out.body)
            assert sorted(item["i"] for item in out.body) == list(range(6))
        finally:
            host.stop()


class TestApplicationRuntime:
    def test_deploy_and_stop(self, tmp_path):
        app = tmp_path / "app.py"
        app.write_text("import time\nwhile True:\n    time.sleep(0.2)\n")
        fn = mlrun_amd.new_function(name="app", kind="application",
                                    command=str(app))
        address = fn.deploy()
        assert address.startswith("pid://")
        assert fn.is_running()
        fn.stop()
        deadline = time.time() + 5
        while fn.is_running() and time.time() < deadline:
            time.sleep(0.1)
        assert not fn.is_running()


class TestWebhookNotificationLive:
    def test_webhook_posts_to_local_host(self):
        """Webhook + slack notifications against a real local HTTP
        endpoint (no external network)."""
        received = []

        class _Sink:
            def __init__(self, context=None, name=None):
                pass

            def do_event(self, event):
                received.append(event.body)
                event.body = {"ok": True}
                return event

        from mlrun_amd.serving.states import RootFlowStep

        graph = RootFlowStep()
        graph.add_step(_Sink, name="sink")
        server = GraphServer(graph=graph)
        server.init_states(namespace={"_Sink": _Sink})
        host = GraphServerHost(server, port=0)
        host.start()
        try:
            from mlrun_amd.utils.notifications import (
                get_notification_class)

            hook = get_notification_class("webhook")(
                params={"url": host.address + "/hook"})
            hook.push("run done", "info",
                      [{"metadata": {"name": "r1"},
                        "status": {"state": "completed"}}])
            slack = get_notification_class("slack")(
                params={"webhook": host.address + "/slack"})
            slack.push("run failed", "error",
                       [{"metadata": {"name": "r2"},
                         "status": {"state": "error"}}])
        finally:
            host.stop()
        assert received[0]["message"] == "run done"
        assert received[0]["runs"][0]["metadata"]["name"] == "r1"
        assert "*error* run failed" in received[1]["text"]


class TestCronTrigger:
    def test_cron_trigger_invokes_handler(self):
        """Nuclio cron-trigger analog: the deployed handler fires on
        the configured interval (reference RemoteRuntime.add_trigger)."""
        import time

        import mlrun_amd

        hits = []

        def handler(context, event):
            hits.append(event.body)
            return {"n": len(hits)}

        fn = mlrun_amd.new_function("cronfn", kind="remote")
        fn._handler_fn = handler
        fn.add_cron_trigger("tick", interval="1", body={"from": "cron"})
        fn.deploy()
        try:
            deadline = time.time() + 15
            while len(hits) < 2 and time.time() < deadline:
                time.sleep(0.2)
            assert len(hits) >= 2, hits
            assert {"from": "cron"} in hits
        finally:
            fn.stop()
        count = len(hits)
        time.sleep(1.5)
        assert len(hits) == count  # stopped cleanly


class TestRemoteStepResilience:
    def test_retries_then_succeeds(self):
        """RemoteStep retries transient failures (reference
        RemoteStep retries arg)."""
        import threading

        from http.server import BaseHTTPRequestHandler, HTTPServer

        import mlrun_amd
        from mlrun_amd.serving.remote import RemoteStep

        attempts = []

        class Handler(BaseHTTPRequestHandler):
            def do_POST(self):
                attempts.append(1)
                if len(attempts) < 3:
                    self.send_response(503)
                    self.end_headers()
                    return
                self.send_response(200)
                self.send_header("content-type", "application/json")
                self.end_headers()
                self.wfile.write(b'{"ok": true}')

            def log_message(self, *a):
                pass

        httpd = HTTPServer(("127.0.0.1", 0), Handler)
        port = httpd.server_address[1]
        threading.Thread(target=httpd.serve_forever,
                         daemon=True).start()
        try:
            step = RemoteStep(url=f"http://127.0.0.1:{port}/x",
                              retries=3)

            class _Ev:
                path = "/x"
                id = "1"
                body = {"q": 1}

            out = step.do_event(_Ev())
            assert out.body == {"ok": True}
            assert len(attempts) == 3
        finally:
            httpd.shutdown()

    def test_exhausted_retries_raise(self):
        from mlrun_amd.serving.remote import RemoteStep

        step = RemoteStep(url="http://127.0.0.1:9/never", retries=1,
                          timeout=1)

        class _Ev:
            path = "/never"
            id = "1"
            body = {}

        import pytest as _pytest

        with _pytest.raises(Exception):
            step.do_event(_Ev())
