# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Built-in feature-store transform steps (batched, dataframe-in/
dataframe-out — the MI355X engine ingests in batches, so every step
transforms a whole pandas DataFrame instead of per-event dicts).

Parity target: reference mlrun/feature_store/steps.py (MLRunStep :40,
FeaturesetValidator :94, MapValues :152, Imputer :377, OneHotEncoder
:427, DateExtractor :516, SetEventMetadata :635, DropFeatures :699).
"""




class MLRunStep:
    """Base transform step: do(df) -> df."""

    def __init__(self, context=None, name=None, **kwargs):
        self.context = context
        self.name = name

    def do(self, df):
        return df

    def do_event(self, event):
        event.body = self.do(event.body)
        return event


class FeaturesetValidator(MLRunStep):
    """Validate values against per-feature validators
    (``Feature.validator`` — MinMax/MinMaxLen/Regex, reference
    steps.py FeaturesetValidator) and feature value_type; violations
    are logged (severity-tagged) and optionally dropped."""

    def __init__(self, context=None, name=None, featureset=None,
                 columns=None, drop_invalid=True, **kwargs):
        super().__init__(context, name)
        self.featureset = featureset
        self.columns = columns
        self.drop_invalid = drop_invalid
        self.violations: list = []

    def _validators(self, df):
        out = {}
        if self.featureset is None:
            return out
        for feature in self.featureset.spec.features:
            validator = getattr(feature, "validator", None)
            if validator is not None and feature.name in df.columns and \
                    (not self.columns or feature.name in self.columns):
                validator.set_feature(feature)
                out[feature.name] = validator
        return out

    def do(self, df):
        import pandas as pd

        validators = self._validators(df)
        columns = self.columns or []
        if self.featureset is not None and not columns:
            columns = [f.name for f in self.featureset.spec.features
                       if not f.aggregate and f.name in df.columns]
        bad_mask = None
        for col, validator in validators.items():
            checked = df[col].map(lambda v: validator.check(v)[0])
            col_bad = ~checked
            for idx in df.index[col_bad]:
                ok, info = validator.check(df.at[idx, col])
                violation = {"feature": col,
                             "severity": validator.severity, **info}
                self.violations.append(violation)
                from ..utils import logger

                logger.warning("feature validation failed", **{
                    ("detail" if k == "message" else k): str(v)
                    for k, v in violation.items()})
            bad_mask = col_bad if bad_mask is None else (bad_mask | col_bad)
        for col in columns:
            if col in validators or col not in df.columns:
                continue
            numeric = pd.to_numeric(df[col], errors="coerce")
            col_bad = numeric.isna()
            bad_mask = col_bad if bad_mask is None else (bad_mask | col_bad)
        if bad_mask is not None and self.drop_invalid:
            df = df[~bad_mask]
        return df


class MapValues(MLRunStep):
    """Map/bucket column values (reference :152).

    mapping: {column: {old: new, ...}} or range maps
    {column: {"ranges": {"low": [0, 5], "high": [5, "inf"]}}}."""

    def __init__(self, context=None, name=None, mapping: dict = None,
                 with_original_features: bool = False, suffix: str = "mapped",
                 **kwargs):
        super().__init__(context, name)
        self.mapping = mapping or {}
        self.with_original_features = with_original_features
        self.suffix = suffix

    def do(self, df):
        df = df.copy()
        for col, spec in self.mapping.items():
            if col not in df.columns:
                continue
            target = f"{col}_{self.suffix}" if self.with_original_features \
                else col
            if isinstance(spec, dict) and "ranges" in spec:
                import numpy as np

                values = df[col].astype(float)
                out = df[col].copy().astype(object)
                for label, (low, high) in spec["ranges"].items():
                    low_v = -np.inf if low in ("-inf", None) else float(low)
                    high_v = np.inf if high in ("inf", None) else float(high)
                    out[(values >= low_v) & (values < high_v)] = label
                df[target] = out
            else:
                df[target] = df[col].map(lambda v: spec.get(v, v))
        return df


class Imputer(MLRunStep):
    """Fill missing values by method or per-column default
    (reference :377)."""

    def __init__(self, context=None, name=None, method: str = "avg",
                 default_value=None, mapping: dict = None, **kwargs):
        super().__init__(context, name)
        self.method = method
        self.default_value = default_value
        self.mapping = mapping or {}

    def do(self, df):
        df = df.copy()
        for col in df.columns:
            if col in self.mapping:
                df[col] = df[col].fillna(self.mapping[col])
            elif df[col].isna().any():
                if self.default_value is not None:
                    df[col] = df[col].fillna(self.default_value)
                elif self.method == "avg" and \
                        df[col].dtype.kind in "if":
                    df[col] = df[col].fillna(df[col].mean())
        return df


class OneHotEncoder(MLRunStep):
    """Expand categorical columns into 0/1 columns (reference :427)."""

    def __init__(self, context=None, name=None, mapping: dict = None,
                 **kwargs):
        super().__init__(context, name)
        self.mapping = mapping or {}

    def do(self, df):
        df = df.copy()
        for col, categories in self.mapping.items():
            if col not in df.columns:
                continue
            for cat in categories:
                safe = str(cat).replace(" ", "_").replace("-", "_")
                df[f"{col}_{safe}"] = (df[col] == cat).astype(int)
            df = df.drop(columns=[col])
        return df


class DateExtractor(MLRunStep):
    """Extract datetime parts into new columns (reference :516)."""

    def __init__(self, context=None, name=None, parts: list = None,
                 timestamp_col: str = None, **kwargs):
        super().__init__(context, name)
        self.parts = parts or ["day_of_week", "hour"]
        self.timestamp_col = timestamp_col

    def do(self, df):
        import pandas as pd

        col = self.timestamp_col
        if col is None or col not in df.columns:
            return df
        df = df.copy()
        ts = pd.to_datetime(df[col])
        for part in self.parts:
            attr = {"day_of_week": "dayofweek", "day_of_year": "dayofyear",
                    "week_of_year": "isocalendar"}.get(part, part)
            if attr == "isocalendar":
                df[f"{col}_{part}"] = ts.dt.isocalendar().week.astype(int)
            else:
                df[f"{col}_{part}"] = getattr(ts.dt, attr)
        return df


class DropFeatures(MLRunStep):
    """Drop columns (reference :699)."""

    def __init__(self, context=None, name=None, features: list = None,
                 **kwargs):
        super().__init__(context, name)
        self.features = features or []

    def do(self, df):
        return df.drop(columns=[c for c in self.features
                                if c in df.columns])


class SetEventMetadata(MLRunStep):
    """No-op in batch mode (the reference sets per-event id/key/time;
    batch ingestion derives these from columns)."""

    def __init__(self, context=None, name=None, id_path=None, key_path=None,
                 time_path=None, **kwargs):
        super().__init__(context, name)
