# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""HTTP call steps: invoke an external (or child-function) endpoint as
part of a graph.  Parity: reference mlrun/serving/remote.py
(RemoteStep :39, BatchHttpRequests :241) without the storey dependency.
"""

import concurrent.futures
import json

from ..errors import MLRunRuntimeError


class RemoteStep:
    """Call url (or url_expression over the event) with the event body;
    the response becomes the new event body."""

    def __init__(self, context=None, name=None, url: str = None,
                 subpath: str = None, method: str = None, headers=None,
                 url_expression: str = None, body_expression: str = None,
                 return_json: bool = True, timeout: int = 30,
                 retries: int = 0, **kwargs):
        self.context = context
        self.name = name
        self.url = url
        self.subpath = subpath
        self.method = method or "POST"
        self.headers = headers or {}
        self.url_expression = url_expression
        self.body_expression = body_expression
        self.return_json = return_json
        self.timeout = timeout
        self.retries = retries
        self._session = None

    def post_init(self, mode="sync"):
        pass

    def _get_session(self):
        if self._session is None:
            import requests

            self._session = requests.Session()
        return self._session

    def _resolve_url(self, event) -> str:
        if self.url_expression:
            from ..utils.safe_eval import safe_eval

            return safe_eval(self.url_expression, {"event": event})
        url = self.url or ""
        if self.subpath:
            subpath = self.subpath
            if subpath.startswith("$"):
                subpath = str(event.body.get(subpath[1:], ""))
            url = url.rstrip("/") + "/" + subpath.lstrip("/")
        return url

    def do_event(self, event):
        url = self._resolve_url(event)
        body = event.body
        if self.body_expression:
            from ..utils.safe_eval import safe_eval

            body = safe_eval(self.body_expression, {"event": event})
        data = json.dumps(body, default=str) if isinstance(
            body, (dict, list)) else body
        last_error = None
        for _ in range(self.retries + 1):
            try:
                resp = self._get_session().request(
                    self.method, url, data=data,
                    headers={"content-type": "application/json",
                             **self.headers},
                    timeout=self.timeout)
                if resp.status_code >= 500:
                    last_error = MLRunRuntimeError(
                        f"remote step {url} returned {resp.status_code}")
                    continue
                event.body = resp.json() if self.return_json else resp.content
                return event
            except Exception as exc:
                last_error = exc
        raise MLRunRuntimeError(
            f"remote step call to {url} failed: {last_error}")

    do = do_event


class BatchHttpRequests(RemoteStep):
    """Fan a list body into concurrent HTTP calls; event body becomes
    the list of responses."""

    def __init__(self, *args, max_in_flight: int = 8, **kwargs):
        super().__init__(*args, **kwargs)
        self.max_in_flight = max_in_flight
        self._pool = None

    def do_event(self, event):
        items = event.body if isinstance(event.body, list) else [event.body]
        if self._pool is None:
            self._pool = concurrent.futures.ThreadPoolExecutor(
                max_workers=self.max_in_flight)
        url = self._resolve_url(event)

        def _one(item):
            resp = self._get_session().request(
                self.method, url,
                data=json.dumps(item, default=str),
                headers={"content-type": "application/json", **self.headers},
                timeout=self.timeout)
            return resp.json() if self.return_json else resp.content

        event.body = list(self._pool.map(_one, items))
        return event
