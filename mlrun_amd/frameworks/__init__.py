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
- lgbm/xgboost: portable model dumps parsed natively (no library
            needed at serving time) onto the HIP tree kernel
- onnx/tf_keras: import-gated interchange servers
"""

from .tree import (  # noqa: F401
    TreeEnsembleModel,
    TreeEnsembleModelServer,
    random_forest_nodes,
)
from .sklearn import PickleModelServer, SKLearnModelServer  # noqa: F401
from .lgbm import LGBMModelServer, model_from_lgbm, nodes_from_lgbm_text  # noqa: F401
from .xgboost import (  # noqa: F401
    XGBoostModelServer,
    model_from_xgboost,
    nodes_from_xgboost_dump,
)
from .onnx import ONNXModelServer  # noqa: F401
from .tf_keras import TFKerasModelServer  # noqa: F401
from .torch_nn import PyTorchModelServer  # noqa: F401
from .auto import apply_mlrun, detect_framework, get_model_server_class  # noqa: F401
from .huggingface import HuggingFaceModelServer  # noqa: F401
from .parallel_coordinates import (  # noqa: F401
    compare_db_runs,
    compare_run_objects,
)
