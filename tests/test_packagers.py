# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Packager breadth tests (reference mlrun/package/packagers/*)."""

import pathlib

import numpy as np
import pandas as pd
import pytest
import torch

import mlrun_amd
from mlrun_amd.package.packagers import default_packagers_manager


class _FakeItem:
    def __init__(self, raw):
        self.raw = raw

    def get(self):
        return self.raw


@pytest.fixture()
def ctx(rundb):
    return mlrun_amd.get_or_create_ctx("pack-test")


class TestPythonPackagers:
    def test_std_types_roundtrip_results(self, ctx):
        manager = default_packagers_manager()
        for key, value in [("s", "text"), ("i", 3), ("f", 2.5),
                           ("b", True), ("l", [1, 2]),
                           ("d", {"a": 1})]:
            manager.pack(value, key, ctx)
            assert ctx.results[key] == value

    def test_tuple_set_frozenset(self, ctx):
        manager = default_packagers_manager()
        manager.pack((1, 2, 3), "t", ctx)
        assert ctx.results["t"] == [1, 2, 3]
        manager.pack({4, 5}, "st", ctx)
        assert sorted(ctx.results["st"]) == [4, 5]
        # unpack with the hint restores the container type
        assert manager.unpack(_FakeItem("[1, 2]"), tuple) == (1, 2)
        assert manager.unpack(_FakeItem("[1, 2]"), frozenset) == \
            frozenset({1, 2})

    def test_bytes_roundtrip(self, ctx):
        manager = default_packagers_manager()
        manager.pack(b"\x01\x02", "by", ctx)
        out = manager.unpack(_FakeItem('"\\u0001\\u0002"'), bytes)
        assert out == b"\x01\x02"

    def test_typing_generics_resolve(self):
        import typing

        manager = default_packagers_manager()
        assert manager.unpack(_FakeItem("[1, 2]"),
                              typing.List[int]) == [1, 2]
        assert manager.unpack(_FakeItem('{"a": 1}'),
                              typing.Dict[str, int]) == {"a": 1}


class TestNumpyPackagers:
    def test_scalar_logs_result(self, ctx):
        manager = default_packagers_manager()
        manager.pack(np.float32(1.5), "score", ctx)
        assert ctx.results["score"] == 1.5
        out = manager.unpack(_FakeItem("1.5"), np.float32)
        assert isinstance(out, np.float32) and out == 1.5

    def test_array_dict_npz(self, ctx):
        import io

        from mlrun_amd.package.packagers import NumPyArchivePackager

        manager = default_packagers_manager()
        arrays = {"x": np.arange(4), "y": np.ones((2, 2))}
        # a dict of arrays resolves to the npz packager, not json
        assert manager.resolve(arrays) is NumPyArchivePackager
        manager.pack(arrays, "arrs", ctx)
        buf = io.BytesIO()
        np.savez(buf, **arrays)
        out = NumPyArchivePackager.unpack(_FakeItem(buf.getvalue()))
        assert np.array_equal(out["x"], arrays["x"])
        assert np.array_equal(out["y"], arrays["y"])


class TestPandasSeries:
    def test_series_roundtrip(self, ctx):
        from mlrun_amd.package.packagers import PandasSeriesPackager

        series = pd.Series([1.0, 2.0, 3.0], name="v")
        PandasSeriesPackager.pack(series, "ser", ctx)
        raw = series.to_json(orient="split")
        out = PandasSeriesPackager.unpack(_FakeItem(raw))
        assert list(out) == [1.0, 2.0, 3.0]


class TestPathPackager:
    def test_file_and_directory(self, ctx, tmp_path):
        from mlrun_amd.package.packagers import PathPackager

        target = tmp_path / "data.txt"
        target.write_text("hello")
        PathPackager.pack(target, "file", ctx)
        directory = tmp_path / "dir"
        directory.mkdir()
        (directory / "a.txt").write_text("a")
        (directory / "b.txt").write_text("b")
        PathPackager.pack(directory, "dir", ctx)

        class _Local:
            def local(self):
                return str(target)

        assert PathPackager.unpack(_Local()) == target


class TestPicklePackager:
    def test_catch_all(self, ctx):
        manager = default_packagers_manager()

        class Custom:
            def __init__(self, x):
                self.x = x

        manager.pack(Custom(7), "obj", ctx)
        from mlrun_amd.package.packagers import PicklePackager

        raw = PicklePackager._pickle().dumps(Custom(7))
        out = manager.unpack(_FakeItem(raw), object)
        assert out.x == 7
