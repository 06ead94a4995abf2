# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Value types + schema/statistics inference.

Parity target: reference mlrun/data_types (ValueType data_types.py:22,
infer.py schema+stats inference, used by feature-store ingest and
dataset artifacts).
"""



class ValueType:
    UNKNOWN = ""
    BOOL = "bool"
    INT8 = "int8"
    INT16 = "int16"
    INT32 = "int32"
    INT64 = "int"
    UINT8 = "uint8"
    UINT16 = "uint16"
    UINT32 = "uint32"
    UINT64 = "uint"
    FLOAT = "float32"
    DOUBLE = "float"
    BFLOAT16 = "bfloat16"
    STRING = "str"
    BYTES = "bytes"
    DATETIME = "datetime"


_NUMPY_KIND_MAP = {
    "b": ValueType.BOOL,
    "i": ValueType.INT64,
    "u": ValueType.UINT64,
    "f": ValueType.DOUBLE,
    "M": ValueType.DATETIME,
    "O": ValueType.STRING,
    "U": ValueType.STRING,
    "S": ValueType.BYTES,
}


def python_type_to_value_type(value) -> str:
    import datetime

    import numpy as np

    if isinstance(value, bool):
        return ValueType.BOOL
    if isinstance(value, int):
        return ValueType.INT64
    if isinstance(value, float):
        return ValueType.DOUBLE
    if isinstance(value, str):
        return ValueType.STRING
    if isinstance(value, bytes):
        return ValueType.BYTES
    if isinstance(value, datetime.datetime):
        return ValueType.DATETIME
    if isinstance(value, np.generic):
        return _NUMPY_KIND_MAP.get(value.dtype.kind, ValueType.UNKNOWN)
    return ValueType.UNKNOWN


def pd_dtype_to_value_type(dtype) -> str:
    return _NUMPY_KIND_MAP.get(getattr(dtype, "kind", "O"),
                               ValueType.UNKNOWN)


class InferOptions:
    Null = 0
    Entities = 1
    Features = 2
    Index = 4
    Stats = 8
    Histogram = 16
    Preview = 32
    Schema = Entities | Features | Index
    default = Schema | Stats | Preview

    @staticmethod
    def all():
        return InferOptions.Schema | InferOptions.Stats | \
            InferOptions.Histogram | InferOptions.Preview


def infer_schema_from_df(df, options: int = InferOptions.Schema) -> dict:
    """Infer {column: value_type} + entity candidates from a DataFrame
    (reference data_types/infer.py)."""
    schema = {
        "features": {col: pd_dtype_to_value_type(dtype)
                     for col, dtype in df.dtypes.items()},
        "index": [str(name) for name in df.index.names
                  if name is not None],
    }
    return schema


def get_df_stats(df, options: int = InferOptions.Stats,
                 num_bins: int = 20) -> dict:
    """Per-column statistics (+ histograms) for dataset artifacts and
    model-monitoring reference data."""
    import numpy as np

    stats: dict = {}
    for col in df.columns:
        series = df[col]
        entry: dict = {"count": int(series.count())}
        if series.dtype.kind in "if":
            desc = series.describe()
            for key in ("mean", "std", "min", "max"):
                if key in desc:
                    value = desc[key]
                    entry[key] = None if value != value else float(value)
            if options & InferOptions.Histogram and entry["count"] > 0:
                hist, edges = np.histogram(series.dropna(), bins=num_bins)
                entry["hist"] = [hist.tolist(), edges.tolist()]
        else:
            entry["unique"] = int(series.nunique())
        stats[str(col)] = entry
    return stats


def get_df_preview(df, preview_lines: int = 20) -> list:
    head = df.head(preview_lines)
    return [list(head.columns)] + head.values.tolist()
