# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Legacy v1 model server (flask-style protocol).

Parity target: reference mlrun/serving/v1_serving.py:70 MLModelServer —
kept for users migrating v1 graphs; new code should subclass
V2ModelServer.
"""

import time
import typing

from ..utils import logger


class MLModelServer:
    """v1 protocol: /predict with {"instances": [...]} -> {"predictions":
    [...]}; subclass and implement load() + predict()."""

    def __init__(self, name: str = None, model_dir: str = None, model=None,
                 **kwargs):
        self.name = name
        self.model_dir = model_dir
        self.model = model
        self.ready = False
        self._params = kwargs

    def get_param(self, key, default=None):
        return self._params.get(key, default)

    def load(self):
        raise NotImplementedError

    def predict(self, request: dict) -> typing.List:
        raise NotImplementedError

    def preprocess(self, request: dict) -> dict:
        return request

    def postprocess(self, result) -> typing.Union[dict, list]:
        return result

    def explain(self, request: dict):
        raise NotImplementedError

    def do_event(self, event):
        if not self.ready:
            self.load()
            self.ready = True
        body = event.body if isinstance(event.body, dict) else \
            {"instances": event.body}
        start = time.perf_counter()
        request = self.preprocess(body)
        if str(event.path or "").endswith("/explain"):
            result = self.explain(request)
        else:
            result = self.predict(request)
        result = self.postprocess(result)
        if not isinstance(result, dict):
            result = {"predictions": result}
        result.setdefault("model_name", self.name)
        event.body = result
        logger.debug("v1 predict", model=self.name,
                     ms=(time.perf_counter() - start) * 1000.0)
        return event


def new_v1_model_server(name, model_class: str, models: dict = None,
                        filename="", protocol="", image="", endpoint="",
                        workers=8, canary=None):
    """Create a (legacy) v1 model-server function
    (reference v1_serving.py:33): a serving function whose routes use
    the flask-style MLModelServer protocol."""
    from ..run import new_function

    fn = new_function(name=name, kind="serving", command=filename,
                      image=image)
    fn.spec.parameters["protocol"] = protocol or "v1"
    for key, model_path in (models or {}).items():
        fn.add_model(key, model_path=model_path, class_name=model_class)
    return fn
