# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""TensorFlow/Keras serving + training hooks.

Parity target: reference mlrun/frameworks/tf_keras (mlrun_interface
auto-logging + model_server.py:26 TFKerasModelServer).  TensorFlow is
not shipped in this ROCm image (torch is the native DL stack); the
server is import-gated and the ``apply_mlrun`` hook degrades to a
clear dependency error.  Training on MI355X belongs on
frameworks.torch_nn (DDP over RCCL) — this module exists for users
migrating keras model artifacts.
"""

from ..errors import MLRunMissingDependencyError
from ..serving.v2_serving import V2ModelServer


def _require_tf():
    try:
        import tensorflow  # noqa: F401
        return tensorflow
    except ImportError:
        raise MLRunMissingDependencyError(
            "tensorflow is not available in this image; convert the "
            "model to torch (frameworks.torch_nn) for the "
            "MI355X-native path")


class TFKerasModelServer(V2ModelServer):
    """V2ModelServer for saved keras models (requires tensorflow)."""

    def load(self):
        tf = _require_tf()
        model_file, _extra = self.get_model()
        self.model = tf.keras.models.load_model(model_file)

    def predict(self, request: dict):
        import numpy as np

        inputs = np.asarray(request["inputs"], dtype=np.float32)
        return self.model.predict(inputs, verbose=0).tolist()


def apply_mlrun(model=None, context=None, **kwargs):
    """Reference tf_keras.apply_mlrun analog (auto-log epochs/metrics
    via a keras callback)."""
    tf = _require_tf()

    class _MLRunLogger(tf.keras.callbacks.Callback):
        def on_epoch_end(self, epoch, logs=None):
            if context is not None and logs:
                for key, value in logs.items():
                    context.log_result(f"epoch{epoch}_{key}",
                                       float(value))

    if model is not None:
        model._mlrun_callback = _MLRunLogger()
    return model
