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
        import inspect as _inspect

        for packager in self._packagers:
            if packager.handles_type(hint):
                sig = _inspect.signature(packager.unpack)
                if "hint" in sig.parameters:
                    return packager.unpack(data_item, hint=hint)
                return packager.unpack(data_item)
        return data_item


class PythonObjectPackager(Packager):
    """json-serializable std types -> result or file artifact
    (reference packagers/python_standard_library_packagers.py covers
    bool/bytes/bytearray/dict/float/frozenset/int/list/set/str/tuple —
    tuple/set/frozenset round-trip through lists, bytes through
    latin-1)."""

    PACKABLE_OBJECT_TYPE = (dict, list, str, int, float, bool, tuple,
                            set, frozenset, bytes, bytearray, type(None))

    @classmethod
    def handles_type(cls, hint):
        import typing as _typing

        origin = _typing.get_origin(hint)
        if origin is not None:  # typing.List[int] etc -> origin class
            hint = origin
        return hint in cls.PACKABLE_OBJECT_TYPE

    @classmethod
    def pack(cls, obj, key, context, artifact_type=None):
        serializable = obj
        if isinstance(obj, (tuple, set, frozenset)):
            serializable = list(obj)
        elif isinstance(obj, (bytes, bytearray)):
            serializable = bytes(obj).decode("latin-1")
        if artifact_type in (None, ArtifactType.RESULT):
            context.log_result(key, serializable)
            return obj
        context.log_artifact(key,
                             body=json.dumps(serializable, default=str),
                             format="json")
        return obj

    @classmethod
    def unpack(cls, data_item, artifact_type=None, hint=None):
        raw = data_item.get() if hasattr(data_item, "get") else data_item
        if isinstance(raw, bytes):
            raw = raw.decode()
        try:
            value = json.loads(raw)
        except (ValueError, TypeError):
            value = raw
        import typing as _typing

        origin = _typing.get_origin(hint) or hint
        if origin in (tuple, set, frozenset) and \
                isinstance(value, list):
            return origin(value)
        if origin in (bytes, bytearray) and isinstance(value, str):
            return origin(value.encode("latin-1"))
        return value


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


class NumPyScalarPackager(Packager):
    """np.number scalars -> results (reference
    NumPyNumberPackager)."""

    @classmethod
    def _np(cls):
        import numpy as np

        return np

    @classmethod
    def is_packable(cls, obj, artifact_type=None):
        return isinstance(obj, cls._np().number)

    @classmethod
    def handles_type(cls, hint):
        np = cls._np()
        return isinstance(hint, type) and issubclass(hint, np.number)

    @classmethod
    def pack(cls, obj, key, context, artifact_type=None):
        context.log_result(key, obj.item())
        return obj

    @classmethod
    def unpack(cls, data_item, artifact_type=None, hint=None):
        raw = data_item.get() if hasattr(data_item, "get") else data_item
        if isinstance(raw, bytes):
            raw = raw.decode()
        np = cls._np()
        return (hint or np.float64)(float(raw))


class NumPyArchivePackager(Packager):
    """dict/list of ndarrays <-> .npz artifact (reference
    NumPyNDArrayDictPackager / ...ListPackager)."""

    @classmethod
    def _np(cls):
        import numpy as np

        return np

    @classmethod
    def is_packable(cls, obj, artifact_type=None):
        np = cls._np()
        if isinstance(obj, dict) and obj and all(
                isinstance(v, np.ndarray) for v in obj.values()):
            return True
        return isinstance(obj, list) and bool(obj) and all(
            isinstance(v, np.ndarray) for v in obj)

    @classmethod
    def handles_type(cls, hint):
        return False  # unpack via NumPyPackager hints only

    @classmethod
    def pack(cls, obj, key, context, artifact_type=None):
        np = cls._np()
        buf = io.BytesIO()
        if isinstance(obj, dict):
            np.savez(buf, **obj)
        else:
            np.savez(buf, *obj)
        context.log_artifact(key, body=buf.getvalue(), format="npz")
        return obj

    @classmethod
    def unpack(cls, data_item, artifact_type=None, hint=None):
        np = cls._np()
        raw = data_item.get() if hasattr(data_item, "get") else data_item
        archive = np.load(io.BytesIO(raw), allow_pickle=False)
        return {k: archive[k] for k in archive.files}


class PandasSeriesPackager(Packager):
    """pd.Series <-> json-indexed artifact (reference
    PandasSeriesPackager)."""

    @classmethod
    def _pd(cls):
        import pandas as pd

        return pd

    @classmethod
    def is_packable(cls, obj, artifact_type=None):
        return isinstance(obj, cls._pd().Series)

    @classmethod
    def handles_type(cls, hint):
        return hint is cls._pd().Series

    @classmethod
    def pack(cls, obj, key, context, artifact_type=None):
        context.log_artifact(key, body=obj.to_json(orient="split"),
                             format="json")
        return obj

    @classmethod
    def unpack(cls, data_item, artifact_type=None, hint=None):
        pd = cls._pd()
        raw = data_item.get() if hasattr(data_item, "get") else data_item
        if isinstance(raw, bytes):
            raw = raw.decode()
        return pd.read_json(io.StringIO(raw), orient="split",
                            typ="series")


class PathPackager(Packager):
    """pathlib.Path -> file artifact; unpack -> local path (reference
    PathPackager file/directory handling; directories zip up)."""

    @classmethod
    def _pathlib(cls):
        import pathlib

        return pathlib

    @classmethod
    def is_packable(cls, obj, artifact_type=None):
        return isinstance(obj, cls._pathlib().Path)

    @classmethod
    def handles_type(cls, hint):
        pathlib = cls._pathlib()
        return hint in (pathlib.Path, pathlib.PosixPath)

    @classmethod
    def pack(cls, obj, key, context, artifact_type=None):
        import pathlib
        import zipfile

        path = pathlib.Path(obj)
        if path.is_dir():
            buf = io.BytesIO()
            with zipfile.ZipFile(buf, "w") as archive:
                for item in sorted(path.rglob("*")):
                    if item.is_file():
                        archive.write(item, item.relative_to(path))
            context.log_artifact(key, body=buf.getvalue(), format="zip")
        else:
            context.log_artifact(key, body=path.read_bytes(),
                                 format=path.suffix.lstrip("."))
        return obj

    @classmethod
    def unpack(cls, data_item, artifact_type=None, hint=None):
        import pathlib

        if hasattr(data_item, "local"):
            return pathlib.Path(data_item.local())
        return pathlib.Path(str(data_item))


class PicklePackager(Packager):
    """Catch-all: ANY object <-> cloudpickle artifact (reference
    DefaultPackager object artifact-type).  Registered LAST so typed
    packagers win."""

    @classmethod
    def _pickle(cls):
        try:
            import cloudpickle as pickle_mod
        except ImportError:  # pragma: no cover
            import pickle as pickle_mod
        return pickle_mod

    @classmethod
    def is_packable(cls, obj, artifact_type=None):
        return True

    @classmethod
    def handles_type(cls, hint):
        return hint is object

    @classmethod
    def pack(cls, obj, key, context, artifact_type=None):
        context.log_artifact(key, body=cls._pickle().dumps(obj),
                             format="pkl")
        return obj

    @classmethod
    def unpack(cls, data_item, artifact_type=None, hint=None):
        raw = data_item.get() if hasattr(data_item, "get") else data_item
        return cls._pickle().loads(raw)


def default_packagers_manager() -> PackagersManager:
    manager = PackagersManager()
    manager.register(NumPyScalarPackager, first=False)
    manager.register(NumPyArchivePackager, first=False)
    manager.register(PythonObjectPackager, first=False)
    manager.register(NumPyPackager, first=False)
    manager.register(PandasPackager, first=False)
    manager.register(PandasSeriesPackager, first=False)
    manager.register(TorchTensorPackager, first=False)
    manager.register(PathPackager, first=False)
    manager.register(PicklePackager, first=False)  # catch-all LAST
    return manager
