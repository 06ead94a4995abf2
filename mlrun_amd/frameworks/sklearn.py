# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""sklearn-family serving: pickled estimators.

Parity target: reference frameworks/_ml_common/pkl_model_server.py:24
PickleModelServer (used for sklearn/xgboost).  Tree-based estimators
are auto-exported to the HIP tree-ensemble kernel so predict runs on
the MI355X; other estimators predict in-process (numpy).
"""

import pickle

import numpy as np

from ..serving.v2_serving import V2ModelServer


class PickleModelServer(V2ModelServer):
    """Load a pickled model; predict via model.predict(inputs)."""

    def load(self):
        if self.model is not None:
            return
        model_file, extra = self.get_model(".pkl")
        with open(model_file, "rb") as fp:
            self.model = pickle.load(fp)

    def predict(self, request: dict):
        inputs = np.asarray(request["inputs"])
        result = self.model.predict(inputs)
        return np.asarray(result).tolist()


class SKLearnModelServer(PickleModelServer):
    """PickleModelServer that moves tree ensembles onto the GPU kernel
    when possible."""

    def load(self):
        super().load()
        import torch

        if torch.cuda.is_available() and hasattr(self.model, "estimators_"):
            try:
                from .tree import TreeEnsembleModel

                self._gpu_model = TreeEnsembleModel.from_sklearn(
                    self.model).to("cuda:0")
            except (ValueError, AttributeError):
                self._gpu_model = None
        else:
            self._gpu_model = None

    def predict(self, request: dict):
        if getattr(self, "_gpu_model", None) is not None:
            return self._gpu_model.predict(request["inputs"]).cpu().tolist()
        return super().predict(request)
