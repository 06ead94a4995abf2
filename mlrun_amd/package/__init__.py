# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Packager-lite: the ``mlrun.handler`` decorator + type-driven packing.

Parity target: reference mlrun/package (handler decorator, ArtifactType,
packagers for python-std/numpy/pandas).  The decorator maps handler
inputs from DataItems (parsing by type hint) and packs outputs to
results/artifacts by the ``outputs`` spec.
"""

import functools
import inspect
import json


class ArtifactType:
    DATASET = "dataset"
    MODEL = "model"
    FILE = "file"
    OBJECT = "object"
    PLOT = "plot"
    RESULT = "result"
    DEFAULT = "default"


def _parse_input(value, annotation):
    """Unpack a DataItem input into the hinted type."""
    from ..datastore import DataItem

    if not isinstance(value, DataItem):
        return value
    if annotation in (inspect.Parameter.empty, None, DataItem):
        return value
    try:
        import pandas as pd

        if annotation is pd.DataFrame:
            return value.as_df()
    except ImportError:
        pass
    try:
        import numpy as np

        if annotation is np.ndarray:
            import io

            return np.load(io.BytesIO(value.get()), allow_pickle=True)
    except ImportError:
        pass
    if annotation is str:
        return value.get(encoding="utf-8")
    if annotation is bytes:
        return value.get()
    if annotation is dict:
        return json.loads(value.get(encoding="utf-8"))
    if annotation is list:
        return json.loads(value.get(encoding="utf-8"))
    return value


def handler(labels: dict = None, outputs: list = None, inputs=True):
    """Decorator: auto-parse DataItem inputs by type hints and log
    returned values per ``outputs`` (parity: reference mlrun.handler)."""

    def decorator(fn):
        @functools.wraps(fn)
        def wrapper(*args, **kwargs):
            context = _find_context(args, kwargs)
            if inputs:
                sig = inspect.signature(fn)
                bound = sig.bind_partial(*args, **kwargs)
                new_kwargs = {}
                for name, value in bound.arguments.items():
                    annotation = sig.parameters[name].annotation
                    new_kwargs[name] = _parse_input(value, annotation) \
                        if inputs else value
                result = fn(**new_kwargs)
            else:
                result = fn(*args, **kwargs)
            if context is not None and labels:
                for key, value in labels.items():
                    context.set_label(key, value)
            if context is not None and outputs:
                values = result if isinstance(result, tuple) else (result,)
                from ..runtimes.local import _parse_return_key, _log_single

                for i, value in enumerate(values):
                    spec = outputs[i] if i < len(outputs) else None
                    key, kind = _parse_return_key(spec, i, len(values))
                    _log_single(context, key, kind, value)
                return result
            return result

        wrapper._mlrun_handler = True
        wrapper._mlrun_outputs = outputs
        return wrapper

    return decorator


def _find_context(args, kwargs):
    from ..execution import MLClientCtx

    for value in list(args) + list(kwargs.values()):
        if isinstance(value, MLClientCtx):
            return value
    return None


class Packager:
    """Extensible packager base (reference package/packager.py:344)."""

    PACKABLE_OBJECT_TYPE: type = None

    @classmethod
    def is_packable(cls, obj, artifact_type=None) -> bool:
        return cls.PACKABLE_OBJECT_TYPE is not None and \
            isinstance(obj, cls.PACKABLE_OBJECT_TYPE)

    @classmethod
    def handles_type(cls, hint) -> bool:
        return cls.PACKABLE_OBJECT_TYPE is hint

    @classmethod
    def pack(cls, obj, key: str, context, artifact_type=None):
        raise NotImplementedError

    @classmethod
    def unpack(cls, data_item, artifact_type=None):
        raise NotImplementedError


class DefaultPackager(Packager):
    PACKABLE_OBJECT_TYPE = object

    @classmethod
    def pack(cls, obj, key, context, artifact_type=None):
        from ..runtimes.local import _log_single

        _log_single(context, key, artifact_type, obj)

    @classmethod
    def unpack(cls, data_item, artifact_type=None):
        return data_item


from .packagers import (  # noqa: E402,F401
    NumPyPackager,
    PackagersManager,
    PandasPackager,
    PythonObjectPackager,
    TorchTensorPackager,
    default_packagers_manager,
)
