# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""API gateway: a node-local reverse proxy in front of deployed
serving/remote functions with basic auth and canary traffic split.

Parity target: reference mlrun/runtimes/nuclio/api_gateway.py (nuclio
API-gateway CRUD: auth, canary) — rebuilt as an in-process FastAPI
proxy instead of nuclio dashboard objects.
"""

import base64
import random
import threading
import typing

from ..errors import MLRunInvalidArgumentError
from ..model import ModelObj
from ..utils import logger


class APIGateway(ModelObj):
    """Route requests to one or two upstream function hosts.

    upstreams: [{"address": url, "percent": 80}, ...] — percents of a
    canary split (must sum to 100 when two upstreams are set).
    """

    kind = "api-gateway"

    def __init__(self, name=None, project=None, host="127.0.0.1", port=0,
                 upstreams=None, auth_mode="none", username=None,
                 password=None):
        self.name = name
        self.project = project
        self.host = host
        self.port = port
        self.upstreams = upstreams or []
        self.auth_mode = auth_mode  # none | basic
        self.username = username
        self.password = password
        self._server = None
        self._thread = None

    def with_basic_auth(self, username: str, password: str):
        self.auth_mode = "basic"
        self.username = username
        self.password = password
        return self

    def with_canary(self, functions: list, percents: typing.List[int]):
        if len(functions) != len(percents) or sum(percents) != 100:
            raise MLRunInvalidArgumentError(
                "canary needs matching functions/percents summing to 100")
        self.upstreams = []
        for fn, percent in zip(functions, percents):
            address = fn if isinstance(fn, str) else fn.status.address
            if not address:
                raise MLRunInvalidArgumentError(
                    "function has no deployed address")
            self.upstreams.append({"address": address, "percent": percent})
        return self

    def add_upstream(self, function_or_address, percent: int = 100):
        address = function_or_address if isinstance(
            function_or_address, str) else function_or_address.status.address
        self.upstreams.append({"address": address, "percent": percent})
        return self

    def _pick_upstream(self) -> str:
        if not self.upstreams:
            raise MLRunInvalidArgumentError("gateway has no upstreams")
        if len(self.upstreams) == 1:
            return self.upstreams[0]["address"]
        roll = random.uniform(0, 100)
        acc = 0.0
        for upstream in self.upstreams:
            acc += upstream["percent"]
            if roll <= acc:
                return upstream["address"]
        return self.upstreams[-1]["address"]

    def _check_auth(self, headers: dict) -> bool:
        if self.auth_mode != "basic":
            return True
        header = headers.get("authorization", "")
        if not header.lower().startswith("basic "):
            return False
        try:
            decoded = base64.b64decode(header.split(" ", 1)[1]).decode()
        except Exception:
            return False
        return decoded == f"{self.username}:{self.password}"

    @property
    def address(self) -> str:
        return f"http://{self.host}:{self.port}"

    def deploy(self, wait_ready=True) -> str:
        from fastapi import FastAPI, Request, Response

        import requests as requests_lib

        from ..serving.server import _free_port

        if not self.port:
            self.port = _free_port()
        app = FastAPI(title=f"gateway-{self.name}")
        gateway = self
        session = requests_lib.Session()

        @app.api_route("/{path:path}",
                       methods=["GET", "POST", "PUT", "DELETE"])
        async def proxy(path: str, request: Request):
            if not gateway._check_auth(dict(request.headers)):
                return Response("unauthorized", status_code=401)
            upstream = gateway._pick_upstream()
            body = await request.body()
            import anyio

            def _forward():
                return session.request(
                    request.method, f"{upstream}/{path}", data=body,
                    headers={k: v for k, v in request.headers.items()
                             if k.lower() not in ("host", "authorization")},
                    timeout=120)
            resp = await anyio.to_thread.run_sync(_forward)
            return Response(resp.content, status_code=resp.status_code,
                            media_type=resp.headers.get("content-type"))

        import uvicorn

        config_ = uvicorn.Config(app, host=self.host, port=self.port,
                                 log_level="warning", access_log=False)
        self._server = uvicorn.Server(config_)
        self._thread = threading.Thread(target=self._server.run,
                                        daemon=True,
                                        name=f"gateway-{self.name}")
        self._thread.start()
        if wait_ready:
            import time

            import requests

            deadline = time.monotonic() + 15
            while time.monotonic() < deadline:
                try:
                    requests.get(f"{self.address}/healthz", timeout=1)
                    break
                except Exception:
                    time.sleep(0.05)
        logger.info("api gateway deployed", address=self.address)
        return self.address

    def invoke(self, path="/", body=None, method="POST", headers=None,
               credentials: tuple = None):
        import json as json_lib

        import requests

        headers = dict(headers or {})
        if credentials:
            token = base64.b64encode(
                f"{credentials[0]}:{credentials[1]}".encode()).decode()
            headers["Authorization"] = f"Basic {token}"
        data = body
        if isinstance(body, (dict, list)):
            data = json_lib.dumps(body, default=str)
            headers.setdefault("content-type", "application/json")
        resp = requests.request(method, self.address + path, data=data,
                                headers=headers, timeout=60)
        try:
            return resp.status_code, resp.json()
        except ValueError:
            return resp.status_code, resp.content

    def stop(self):
        if self._server is not None:
            self._server.should_exit = True
        if self._thread is not None:
            self._thread.join(timeout=5)


class APIGatewayMetadata(ModelObj):
    """Reference-shape metadata object (api_gateway.py:115)."""

    def __init__(self, name: str = None, namespace: str = None,
                 labels: dict = None, annotations: dict = None):
        self.name = name
        self.namespace = namespace
        self.labels = labels or {}
        self.annotations = annotations or {}


class APIGatewaySpec(ModelObj):
    """Reference-shape spec object (api_gateway.py:145)."""

    def __init__(self, functions: list = None, project: str = None,
                 name: str = None, description: str = None,
                 path: str = "/", host: str = None, canary: list = None,
                 authentication_mode: str = "none"):
        self.functions = functions or []
        self.project = project
        self.name = name
        self.description = description
        self.path = path
        self.host = host
        self.canary = canary
        self.authentication_mode = authentication_mode


def _gateway_from_schema(metadata: APIGatewayMetadata,
                         spec: APIGatewaySpec) -> APIGateway:
    """Build a proxy gateway from the reference (metadata, spec) pair
    (reference APIGateway(metadata, spec) constructor form)."""
    gateway = APIGateway(name=metadata.name, project=spec.project)
    percents = spec.canary or ([100] if len(spec.functions) == 1
                               else None)
    for i, fn in enumerate(spec.functions):
        address = fn if isinstance(fn, str) and "://" in fn else \
            (spec.host or fn)
        gateway.add_upstream(address,
                             percents[i] if percents else 100)
    if spec.authentication_mode == "basicAuth":
        gateway.auth_mode = "basic"
    return gateway


def _gateway_save(self):
    """Persist this gateway as an api-gateway document in the run DB
    (reference: provision in nuclio; local: db api-gateway CRUD)."""
    import inspect as _inspect

    from ..db import get_run_db

    db = get_run_db()
    struct = {"metadata": {"name": self.name},
              "spec": {"project": self.project,
                       "upstreams": self.upstreams,
                       "auth_mode": self.auth_mode,
                       "host": self.host, "port": self.port}}
    params = list(_inspect.signature(db.store_api_gateway).parameters)
    if params and params[0] == "project":
        db.store_api_gateway(self.project or "default", self.name,
                             struct)
    else:
        db.store_api_gateway(struct, project=self.project or "default")
    return self


def _gateway_delete(self):
    """Remove the stored api-gateway document."""
    import inspect as _inspect

    from ..db import get_run_db

    db = get_run_db()
    params = list(_inspect.signature(db.delete_api_gateway).parameters)
    if params and params[0] == "project":
        db.delete_api_gateway(self.project or "default", self.name)
    else:
        db.delete_api_gateway(self.name, project=self.project or
                              "default")


APIGateway.save = _gateway_save
APIGateway.delete = _gateway_delete
APIGateway.from_schema = staticmethod(_gateway_from_schema)
