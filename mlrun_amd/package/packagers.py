# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Concrete packagers + manager.

Parity target: reference mlrun/package/packagers_manager.py and
packagers/{python_standard_library,numpy,pandas}_packagers.py — the
type-driven registry that packs handler return values into artifacts
and unpacks DataItems back into typed objects.  Native addition: a
torch.Tensor packager (safetensors-free .pt roundtrip)."""

import io
import json
import typing

from . import ArtifactType, Packager


class PackagersManager:
    """Ordered registry: first packager whose ``is_packable`` accepts
    the object wins (reference packagers_manager.py)."""

    def __init__(self):
        self._packagers: typing.List[type] = []

    def register(self, packager: type, first: bool = True):
        if first:
            self._packagers.insert(0, packager)
        else:
            self._packagers.append(packager)

    def resolve(self, obj, artifact_type=None) -> type:
        for packager in self._packagers:
            if packager.is_packable(obj, artifact_type):
                return packager
        raise TypeError(f"no packager accepts {type(obj).__name__}")

    def pack(self, obj, key, context, artifact_type=None):
        return self.resolve(obj, artifact_type).pack(
            obj, key, context, artifact_type)

    def unpack(self, data_item, hint: type):
        for packager in self._packagers:
            if packager.handles_type(hint):
                return packager.unpack(data_item)
        return data_item


class PythonObjectPackager(Packager):
    """json-serializable std types -> result or file artifact."""

    PACKABLE_OBJECT_TYPE = (dict, list, str, int, float, bool)

    @classmethod
    def handles_type(cls, hint):
        return hint in cls.PACKABLE_OBJECT_TYPE

    @classmethod
    def pack(cls, obj, key, context, artifact_type=None):
        if artifact_type in (None, ArtifactType.RESULT):
            context.log_result(key, obj)
            return obj
        context.log_artifact(key, body=json.dumps(obj, default=str),
                             format="json")
        return obj

    @classmethod
    def unpack(cls, data_item, artifact_type=None):
        raw = data_item.get() if hasattr(data_item, "get") else data_item
        if isinstance(raw, bytes):
            raw = raw.decode()
        try:
            return json.loads(raw)
        except (ValueError, TypeError):
            return raw


class NumPyPackager(Packager):
    """np.ndarray <-> .npy artifact."""

    @classmethod
    def _np(cls):
        import numpy as np

        return np

    @classmethod
    def is_packable(cls, obj, artifact_type=None):
        return isinstance(obj, cls._np().ndarray)

    @classmethod
    def handles_type(cls, hint):
        return hint is cls._np().ndarray

    @classmethod
    def pack(cls, obj, key, context, artifact_type=None):
        np = cls._np()
        buf = io.BytesIO()
        np.save(buf, obj, allow_pickle=False)
        context.log_artifact(key, body=buf.getvalue(), format="npy")
        return obj

    @classmethod
    def unpack(cls, data_item, artifact_type=None):
        np = cls._np()
        raw = data_item.get() if hasattr(data_item, "get") else data_item
        return np.load(io.BytesIO(raw), allow_pickle=False)


class PandasPackager(Packager):
    """pd.DataFrame <-> dataset artifact (parquet)."""

    @classmethod
    def _pd(cls):
        import pandas as pd

        return pd

    @classmethod
    def is_packable(cls, obj, artifact_type=None):
        return isinstance(obj, cls._pd().DataFrame)

    @classmethod
    def handles_type(cls, hint):
        return hint is cls._pd().DataFrame

    @classmethod
    def pack(cls, obj, key, context, artifact_type=None):
        context.log_dataset(key, df=obj)
        return obj

    @classmethod
    def unpack(cls, data_item, artifact_type=None):
        if hasattr(data_item, "as_df"):
            return data_item.as_df()
        pd = cls._pd()
        raw = data_item.get() if hasattr(data_item, "get") else data_item
        return pd.read_parquet(io.BytesIO(raw))


class TorchTensorPackager(Packager):
    """torch.Tensor <-> .pt artifact (native addition — the reference
    ships no torch packager; tensors are first-class here)."""

    @classmethod
    def _torch(cls):
        import torch

        return torch

    @classmethod
    def is_packable(cls, obj, artifact_type=None):
        return isinstance(obj, cls._torch().Tensor)

    @classmethod
    def handles_type(cls, hint):
        return hint is cls._torch().Tensor

    @classmethod
    def pack(cls, obj, key, context, artifact_type=None):
        torch = cls._torch()
        buf = io.BytesIO()
        torch.save(obj.detach().cpu(), buf)
        context.log_artifact(key, body=buf.getvalue(), format="pt")
        return obj

    @classmethod
    def unpack(cls, data_item, artifact_type=None):
        torch = cls._torch()
        raw = data_item.get() if hasattr(data_item, "get") else data_item
        return torch.load(io.BytesIO(raw), map_location="cpu",
                          weights_only=True)


def default_packagers_manager() -> PackagersManager:
    manager = PackagersManager()
    manager.register(PythonObjectPackager, first=False)
    manager.register(NumPyPackager, first=False)
    manager.register(PandasPackager, first=False)
    manager.register(TorchTensorPackager, first=False)
    return manager
