# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Framework integrations: model servers + training interfaces.

Parity target: reference mlrun/frameworks (pytorch/sklearn/xgboost/
onnx/... auto-instrumentation + per-framework V2ModelServer
subclasses).  MI355X-native set:

- tree:     GBDT/forest models on the HIP tree-ensemble kernel
- torch_nn: torch modules (bf16, ROCm) incl. the DDP/RCCL train loop
- llama:    generative serving (mlrun_amd/models/llama.py LlamaServer)
- sklearn:  pickled sklearn estimators (CPU predict in the graph,
            or exported to the tree kernel when tree-based)
"""

from .tree import (  # noqa: F401
    TreeEnsembleModel,
    TreeEnsembleModelServer,
    random_forest_nodes,
)
from .sklearn import PickleModelServer, SKLearnModelServer  # noqa: F401
from .torch_nn import PyTorchModelServer  # noqa: F401
from .auto import apply_mlrun, detect_framework, get_model_server_class  # noqa: F401
