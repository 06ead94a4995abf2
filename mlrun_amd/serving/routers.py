# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Model routers: path/body-based routing, voting ensembles, parallel
fan-out.

Parity target: reference mlrun/serving/routers.py (BaseModelRouter :43,
ModelRouter :167 with _handle_event :199, ParallelRun :245 w/ thread/
process executors :391-398, VotingEnsemble :480, EnrichmentModelRouter
:1118).
"""

import concurrent.futures
import copy
import json

from ..errors import MLRunInvalidArgumentError, MLRunNotFoundError


class BaseModelRouter:
    """Route events to child model steps by URL path or body."""

    def __init__(self, context=None, name=None, routes=None, protocol=None,
                 url_prefix=None, health_prefix=None, **kwargs):
        self.context = context
        self.name = name or "router"
        self.routes = routes or {}
        self.protocol = protocol or "v2"
        self.url_prefix = url_prefix or f"/{self.protocol}/models"
        self.health_prefix = health_prefix or f"/{self.protocol}/health"
        self.inputs_key = "inputs"
        self._params = kwargs

    def post_init(self, mode="sync"):
        pass

    # user-overridable hooks (reference routers.py preprocess/
    # postprocess/validate around _handle_event)
    def preprocess(self, event):
        return event

    def postprocess(self, event):
        return event

    def validate(self, event):
        return event

    def parse_event(self, event):
        if isinstance(event.body, (str, bytes)) and event.body:
            try:
                event.body = json.loads(event.body)
            except (ValueError, TypeError):
                pass
        return event

    def _resolve_route(self, event):
        """Model name from path (/v2/models/<name>/op) or body."""
        path = event.path or ""
        subpath = None
        model = ""
        if path.startswith(self.url_prefix):
            subpath = path[len(self.url_prefix):].strip("/")
        if subpath:
            parts = subpath.split("/")
            model = parts[0]
        elif isinstance(event.body, dict):
            model = event.body.get("model") or ""
        if not model and len(self.routes) == 1:
            model = next(iter(self.routes))
        return model

    def do_event(self, event):
        event = self.parse_event(event)
        path = event.path or ""
        if path.startswith(self.health_prefix):
            event.body = {"status": "ok"}
            return event
        if (path.rstrip("/") == self.url_prefix.rstrip("/")
                or not self.routes) and not self._resolve_route(event):
            # list models
            event.body = {"models": list(self.routes.keys())}
            return event
        event = self.preprocess(event)
        event = self.validate(event)
        event = self._handle_event(event)
        return self.postprocess(event)

    def _handle_event(self, event):
        model = self._resolve_route(event)
        if not model:
            event.body = {"models": list(self.routes.keys())}
            return event
        route = self.routes.get(model)
        if route is None:
            raise MLRunNotFoundError(  # -> HTTP 404 at the host
                f"model {model} not found in router "
                f"(available: {list(self.routes)})")
        return route.run(event)


class ModelRouter(BaseModelRouter):
    pass


class ParallelRun(BaseModelRouter):
    """Fan an event to ALL routes concurrently, merge results.

    executor_type: "thread" (pool) or "array" (serial).  The
    reference's "process" pool mode maps to "thread" node-locally
    (GPU model steps must share the device context; a process pool
    would re-initialize HIP per call).
    """

    def __init__(self, context=None, name=None, routes=None,
                 extend_event=None, executor_type="thread", **kwargs):
        super().__init__(context, name, routes, **kwargs)
        self.executor_type = executor_type
        self.extend_event = extend_event
        self._pool = None

    def _executor(self):
        if self._pool is None:
            self._pool = concurrent.futures.ThreadPoolExecutor(
                max_workers=max(len(self.routes), 1))
        return self._pool

    def merger(self, body: dict, results: dict) -> dict:
        """Override to customize merging; default: nested results dict."""
        body = body if isinstance(body, dict) else {}
        body["results"] = results
        return body

    def _handle_event(self, event):
        results = {}
        if self.executor_type in ("thread", "process"):
            futures = {}
            for key, route in self.routes.items():
                branch = copy.copy(event)
                branch.body = copy.deepcopy(event.body)
                futures[self._executor().submit(route.run, branch)] = key
            for future in concurrent.futures.as_completed(futures):
                key = futures[future]
                try:
                    out = future.result()
                    results[key] = out.body if out is not None else None
                except Exception as exc:
                    results[key] = {"error": str(exc)}
        else:  # array (serial)
            for key, route in self.routes.items():
                branch = copy.copy(event)
                branch.body = copy.deepcopy(event.body)
                try:
                    out = route.run(branch)
                    results[key] = out.body if out is not None else None
                except Exception as exc:
                    results[key] = {"error": str(exc)}
        event.body = self.merger(event.body, results)
        return event


class VotingTypes:
    classification = "classification"
    regression = "regression"


class VotingEnsemble(ParallelRun):
    """Run all models, combine predictions by majority vote
    (classification) or mean (regression).  Parity: reference
    routers.py:480."""

    def __init__(self, context=None, name=None, routes=None,
                 vote_type=None, weights=None, **kwargs):
        super().__init__(context, name, routes, **kwargs)
        self.vote_type = vote_type
        self.weights = weights or {}

    def extract_results(self, results: dict) -> dict:
        outputs = {}
        for key, body in results.items():
            if isinstance(body, dict) and "outputs" in body:
                outputs[key] = body["outputs"]
        return outputs

    def _infer_vote_type(self, outputs) -> str:
        if self.vote_type:
            return self.vote_type
        for preds in outputs.values():
            flat = preds if isinstance(preds, list) else [preds]
            for value in flat:
                if isinstance(value, float) and not float(value).is_integer():
                    return VotingTypes.regression
        return VotingTypes.classification

    def vote(self, outputs: dict):
        import numpy as np

        if not outputs:
            return []
        arrays = {k: np.asarray(v) for k, v in outputs.items()}
        vote_type = self._infer_vote_type(outputs)
        stacked = np.stack(list(arrays.values()))  # [models, n, ...]
        if vote_type == VotingTypes.regression:
            if self.weights:
                weights = np.asarray(
                    [self.weights.get(k, 1.0) for k in arrays])
                weights = weights / weights.sum()
                return np.tensordot(weights, stacked, axes=1).tolist()
            return stacked.mean(axis=0).tolist()
        # classification: per-sample majority
        votes = []
        n = stacked.shape[1] if stacked.ndim > 1 else 1
        for i in range(n):
            sample = stacked[:, i] if stacked.ndim > 1 else stacked
            values, counts = np.unique(sample, return_counts=True)
            votes.append(values[counts.argmax()].item())
        return votes

    def _handle_event(self, event):
        # route to a specific model if the path names one
        model = None
        path = event.path or ""
        if path.startswith(self.url_prefix):
            subpath = path[len(self.url_prefix):].strip("/")
            if subpath:
                candidate = subpath.split("/")[0]
                if candidate in self.routes:
                    model = candidate
        if model:
            return self.routes[model].run(event)
        event = super()._handle_event(event)
        results = event.body.pop("results", {})
        outputs = self.extract_results(results)
        event.body = {
            "id": event.id,
            "model_name": self.name,
            "outputs": self.vote(outputs),
            "model_results": outputs,
        }
        return event


class EnrichmentModelRouter(ModelRouter):
    """Router that enriches the request from the online feature
    service before routing (parity: reference routers.py:1118)."""

    def __init__(self, context=None, name=None, routes=None,
                 feature_vector_uri="", impute_policy=None, **kwargs):
        super().__init__(context, name, routes, **kwargs)
        self.feature_vector_uri = feature_vector_uri
        self.impute_policy = impute_policy or {}
        self._feature_service = None

    def post_init(self, mode="sync"):
        if self.feature_vector_uri:
            from ..feature_store import get_online_feature_service

            self._feature_service = get_online_feature_service(
                self.feature_vector_uri,
                impute_policy=self.impute_policy)

    def preprocess(self, event):
        if self._feature_service is not None and isinstance(event.body, dict):
            entity_rows = event.body.get("inputs")
            if entity_rows is not None:
                vectors = self._feature_service.get(
                    [row if isinstance(row, dict) else {"id": row}
                     for row in entity_rows], as_list=True)
                event.body["inputs"] = vectors
        return event

    def _handle_event(self, event):
        return super()._handle_event(self.preprocess(event))


router_classes = {
    "ModelRouter": ModelRouter,
    "BaseModelRouter": BaseModelRouter,
    "ParallelRun": ParallelRun,
    "VotingEnsemble": VotingEnsemble,
    "EnrichmentModelRouter": EnrichmentModelRouter,
    "*": ModelRouter,
}
