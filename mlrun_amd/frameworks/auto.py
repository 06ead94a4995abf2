# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Framework auto-detection: apply_mlrun(model) picks the integration.

Parity target: reference frameworks/auto_mlrun (AutoMLRun.apply_mlrun
auto-detects torch/tf/sklearn/xgboost).  Node-local detection set:
torch modules, sklearn estimators (incl. tree ensembles -> HIP
kernel), Llama engines, plain callables.
"""


from ..errors import MLRunInvalidArgumentError


def detect_framework(model) -> str:
    mro_names = [f"{cls.__module__}.{cls.__name__}"
                 for cls in type(model).__mro__]
    joined = " ".join(mro_names)
    if "torch.nn.modules.module.Module" in joined:
        return "pytorch"
    if "sklearn." in joined:
        return "sklearn"
    if "mlrun_amd.models.llama" in joined:
        return "llama"
    if "mlrun_amd.frameworks.tree" in joined:
        return "tree"
    if "lightgbm." in joined:
        return "lightgbm"
    if "xgboost." in joined:
        return "xgboost"
    raise MLRunInvalidArgumentError(
        f"cannot auto-detect framework for {type(model).__name__}")


def apply_mlrun(model=None, context=None, model_name: str = "model",
                **kwargs):
    """Attach the framework interface: returns a training/serving
    helper bound to the run context."""
    framework = detect_framework(model)
    if framework == "pytorch":
        from .torch_nn import apply_mlrun as torch_apply

        return torch_apply(model, context=context)
    if framework == "sklearn":
        return _SKLearnInterface(model, context, model_name)
    if framework in ("llama", "tree"):
        return model
    if framework == "lightgbm":
        from .lgbm import model_from_lgbm

        return model_from_lgbm(model)
    if framework == "xgboost":
        from .xgboost import model_from_xgboost

        return model_from_xgboost(model)
    raise MLRunInvalidArgumentError(f"unsupported framework {framework}")


def get_model_server_class(framework: str):
    """Model-server class per framework kind (the V2ModelServer
    subclasses table of reference SURVEY §2.1 frameworks row)."""
    from ..models.llama import LlamaServer
    from .sklearn import PickleModelServer, SKLearnModelServer
    from .torch_nn import PyTorchModelServer
    from .tree import TreeEnsembleModelServer

    from .lgbm import LGBMModelServer
    from .xgboost import XGBoostModelServer

    table = {
        "sklearn": SKLearnModelServer,
        "pickle": PickleModelServer,
        "xgboost": XGBoostModelServer,
        "lightgbm": LGBMModelServer,
        "tree": TreeEnsembleModelServer,
        "pytorch": PyTorchModelServer,
        "torch": PyTorchModelServer,
        "llama": LlamaServer,
        "llm": LlamaServer,
    }
    if framework == "huggingface":
        from .huggingface import HuggingFaceModelServer

        return HuggingFaceModelServer
    if framework not in table:
        raise MLRunInvalidArgumentError(
            f"no model server for framework {framework!r} "
            f"(available: {sorted(table) + ['huggingface']})")
    return table[framework]


class _SKLearnInterface:
    """Minimal sklearn instrumentation: fit logging + model artifact
    (reference frameworks/sklearn apply_mlrun)."""

    def __init__(self, model, context, model_name):
        self.model = model
        self.context = context
        self.model_name = model_name

    def log_model(self, **kwargs):
        import pickle

        if self.context is None:
            return None
        return self.context.log_model(
            self.model_name, body=pickle.dumps(self.model),
            framework="sklearn", **kwargs)

    def evaluate_and_log(self, x, y, metric_fns: list = None):
        results = {}
        predictions = self.model.predict(x)
        for fn in metric_fns or []:
            name = getattr(fn, "__name__", "metric")
            results[name] = float(fn(y, predictions))
        if self.context is not None and results:
            self.context.log_results(results)
        return results
