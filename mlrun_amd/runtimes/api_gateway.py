# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""API gateway in front of serving functions (reference
runtimes/nuclio/api_gateway.py:340 APIGateway).

The reference provisions a nuclio API gateway (auth + canary traffic
split) in front of deployed functions; the node-local equivalent is a
client object with the same spec surface whose ``invoke`` routes
requests to the functions' local HTTP hosts — weighted by the canary
percentages — and which persists through the run-DB's api-gateway
documents (db.store_api_gateway)."""

import random
import typing

from ..errors import MLRunInvalidArgumentError
from ..model import ModelObj


class APIGatewayMetadata(ModelObj):
    def __init__(self, name: str = None, namespace: str = None,
                 labels: dict = None, annotations: dict = None):
        self.name = name
        self.namespace = namespace
        self.labels = labels or {}
        self.annotations = annotations or {}


class APIGatewaySpec(ModelObj):
    def __init__(self, functions: list = None, project: str = None,
                 name: str = None, description: str = None,
                 path: str = "/", authentication_mode: str = "none",
                 canary: list = None, host: str = None,
                 username: str = None, password: str = None):
        self.functions = functions or []
        self.project = project
        self.name = name
        self.description = description
        self.path = path
        self.authentication_mode = authentication_mode
        self.canary = canary
        self.host = host
        self.username = username
        self.password = password


class APIGatewayStatus(ModelObj):
    def __init__(self, state: str = None):
        self.state = state


class APIGateway(ModelObj):
    """Gateway over one or two serving functions with optional basic
    auth and canary split (reference api_gateway.py:340)."""

    _dict_fields = ["metadata", "spec", "status"]

    def __init__(self, metadata=None, spec=None, status=None):
        self._metadata = None
        self.metadata = metadata or APIGatewayMetadata()
        self._spec = None
        self.spec = spec or APIGatewaySpec()
        self._status = None
        self.status = status or APIGatewayStatus()

    @property
    def metadata(self) -> APIGatewayMetadata:
        return self._metadata

    @metadata.setter
    def metadata(self, value):
        self._metadata = self._verify_dict(value, "metadata",
                                           APIGatewayMetadata)

    @property
    def spec(self) -> APIGatewaySpec:
        return self._spec

    @spec.setter
    def spec(self, value):
        self._spec = self._verify_dict(value, "spec", APIGatewaySpec)

    @property
    def status(self) -> APIGatewayStatus:
        return self._status

    @status.setter
    def status(self, value):
        self._status = self._verify_dict(value, "status",
                                         APIGatewayStatus)

    # ---------------------------------------------------- configuration
    def with_basic_auth(self, username: str, password: str):
        """Require basic auth on invocations (reference :514)."""
        self.spec.authentication_mode = "basicAuth"
        self.spec.username = username
        self.spec.password = password
        return self

    def with_canary(self, functions: list, canary: typing.List[int]):
        """Split traffic between exactly two functions by percentage
        (reference :530)."""
        if len(functions) != 2:
            raise MLRunInvalidArgumentError(
                "canary requires exactly two functions")
        if len(canary) != 2 or sum(canary) != 100 or any(
                p < 0 or p > 100 for p in canary):
            raise MLRunInvalidArgumentError(
                "canary percents must be two values summing to 100")
        self.spec.functions = [
            f if isinstance(f, str) else f.metadata.name
            for f in functions]
        self.spec.canary = list(canary)
        return self

    # -------------------------------------------------------- lifecycle
    def save(self):
        """Persist as an api-gateway document (reference: provision in
        nuclio; here: run-DB api-gateway CRUD)."""
        import inspect as _inspect

        from ..db import get_run_db

        db = get_run_db()
        params = list(_inspect.signature(
            db.store_api_gateway).parameters)
        if params and params[0] == "project":
            # local SQLRunDB signature: (project, name, body)
            db.store_api_gateway(self.spec.project or "default",
                                 self.metadata.name, self.to_dict())
        else:
            # HTTP client signature: (api_gateway, project=)
            db.store_api_gateway(self.to_dict(),
                                 project=self.spec.project or "default")
        self.status.state = "ready"
        return self

    def delete(self):
        from ..db import get_run_db

        get_run_db().delete_api_gateway(
            self.metadata.name, self.spec.project or "default")

    # ------------------------------------------------------- invocation
    def _pick_function(self) -> str:
        functions = self.spec.functions
        if not functions:
            raise MLRunInvalidArgumentError(
                "api gateway has no functions")
        if self.spec.canary and len(functions) == 2:
            roll = random.uniform(0, 100)
            return functions[0] if roll < self.spec.canary[0] \
                else functions[1]
        return functions[0]

    def invoke(self, method: str = "POST", headers: dict = None,
               credentials: tuple = None, path: str = None,
               body=None, func_url_resolver=None):
        """Route one request to a gateway function's local host
        (canary-weighted).  ``func_url_resolver(name) -> base_url``
        defaults to the project's deployed-function registry."""
        import json as _json

        import requests

        if self.spec.authentication_mode == "basicAuth":
            if credentials is None:
                credentials = (self.spec.username, self.spec.password)
            if tuple(credentials) != (self.spec.username,
                                      self.spec.password):
                from ..errors import MLRunAccessDeniedError

                raise MLRunAccessDeniedError(
                    "api gateway credentials rejected")
        name = self._pick_function()
        if func_url_resolver is not None:
            base = func_url_resolver(name)
        else:
            base = self.spec.host
        if not base:
            raise MLRunInvalidArgumentError(
                f"no url known for gateway function {name} "
                "(deploy it or pass func_url_resolver)")
        url = base.rstrip("/") + (path or self.spec.path or "/")
        data = _json.dumps(body, default=str) if isinstance(
            body, (dict, list)) else body
        return requests.request(
            method, url, data=data, timeout=60,
            headers={"content-type": "application/json",
                     **(headers or {})})

    @property
    def invoke_url(self) -> str:
        return (self.spec.host or "").rstrip("/") + \
            (self.spec.path or "/")
