# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""ONNX model serving.

Parity target: reference mlrun/frameworks/onnx (model_server.py:27
ONNXModelServer backed by onnxruntime).  This image ships neither
onnx nor onnxruntime; the server is import-gated — it loads and runs
when onnxruntime (ROCm EP) is present, and raises a clear dependency
error otherwise.  The MI355X-native serving path for transformer /
tree models is the native LlamaServer / TreeEnsembleModelServer —
ONNX is interchange support, not the hot path.
"""

import torch

from ..errors import MLRunMissingDependencyError
from ..serving.v2_serving import V2ModelServer


class ONNXModelServer(V2ModelServer):
    """V2ModelServer running an ONNX graph through onnxruntime
    (ROCMExecutionProvider when available, else CPU)."""

    def load(self):
        try:
            import onnxruntime  # noqa: F401
        except ImportError:
            raise MLRunMissingDependencyError(
                "ONNXModelServer requires onnxruntime (not in this "
                "image); export the model to TorchScript and use "
                "PyTorchModelServer, or to a tree dump and use "
                "TreeEnsembleModelServer instead")
        providers = ["ROCMExecutionProvider", "CPUExecutionProvider"] \
            if torch.cuda.is_available() else ["CPUExecutionProvider"]
        model_file, _extra = self.get_model(".onnx")
        self.session = onnxruntime.InferenceSession(model_file,
                                                    providers=providers)
        self.input_name = self.session.get_inputs()[0].name

    def predict(self, request: dict):
        import numpy as np

        inputs = np.asarray(request["inputs"], dtype=np.float32)
        outputs = self.session.run(None, {self.input_name: inputs})
        return outputs[0].tolist()
