# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""FeatureSet: schema + transform graph + window aggregations.

Parity target: reference mlrun/feature_store/feature_set.py
(FeatureSet :320, add_aggregation :715 — ops count/sum/sqr/max/min/
first/last/avg/stdvar/stddev, windows + period with the
period-divides-window rule).
"""

import re
import typing

from ..errors import MLRunInvalidArgumentError
from ..model import ModelObj
from ..serving.states import RootFlowStep
from ..utils import now_iso

AGGREGATION_OPS = ["count", "sum", "sqr", "max", "min", "first", "last",
                   "avg", "stdvar", "stddev"]

_span_re = re.compile(r"^(\d+)\s*(s|m|h|d)$")
_SPAN_SECONDS = {"s": 1, "m": 60, "h": 3600, "d": 86400}


def parse_span(span: str) -> int:
    """'10m' -> 600 seconds."""
    match = _span_re.match(str(span).strip())
    if not match:
        raise MLRunInvalidArgumentError(
            f"invalid time span {span!r} (use e.g. '30s', '10m', '1h', '1d')")
    return int(match.group(1)) * _SPAN_SECONDS[match.group(2)]


class Entity(ModelObj):
    def __init__(self, name=None, value_type=None, description=None,
                 labels=None):
        self.name = name
        self.value_type = value_type or "str"
        self.description = description
        self.labels = labels or {}


class Feature(ModelObj):
    _dict_fields = ["name", "value_type", "description", "aggregate",
                    "labels", "validator"]

    def __init__(self, value_type=None, description=None, name=None,
                 aggregate=None, labels=None, validator=None):
        self.name = name
        self.value_type = value_type or "float"
        self.description = description
        self.aggregate = aggregate
        self.labels = labels or {}
        self._validator = None
        if validator is not None:
            self.validator = validator

    @property
    def validator(self):
        return self._validator

    @validator.setter
    def validator(self, validator):
        """Accepts a Validator instance or its dict form (reference
        features.py:125)."""
        if isinstance(validator, dict):
            from ..features import validator_from_dict

            validator = validator_from_dict(validator)
        if validator is not None:
            validator.set_feature(self)
        self._validator = validator


class FeatureAggregation(ModelObj):
    """One aggregation spec: column + ops + windows (+ period)."""

    def __init__(self, name=None, column=None, operations=None, windows=None,
                 period=None, step_name=None, after=None):
        self.name = name
        self.column = column
        self.operations = operations or []
        self.windows = windows if isinstance(windows, list) else \
            ([windows] if windows else [])
        self.period = period
        self.step_name = step_name
        self.after = after

    def validate(self):
        for op in self.operations:
            if op not in AGGREGATION_OPS:
                raise MLRunInvalidArgumentError(
                    f"unsupported aggregation op {op!r} "
                    f"(supported: {AGGREGATION_OPS})")
        if not self.windows:
            raise MLRunInvalidArgumentError("aggregation needs windows")
        if self.period:
            period_s = parse_span(self.period)
            for window in self.windows:
                window_s = parse_span(window)
                if window_s % period_s != 0:
                    raise MLRunInvalidArgumentError(
                        f"period {self.period} must divide window {window} "
                        f"(reference feature_set.py:715 rule)")


class FeatureSetMetadata(ModelObj):
    def __init__(self, name=None, project=None, tag=None, labels=None,
                 updated=None):
        self.name = name
        self.project = project
        self.tag = tag
        self.labels = labels or {}
        self.updated = updated


class FeatureSetSpec(ModelObj):
    def __init__(self, entities=None, features=None, timestamp_key=None,
                 description=None, aggregations=None, targets=None,
                 engine=None, label_column=None, source=None):
        self.entities = [e if isinstance(e, Entity) else
                         (Entity.from_dict(e) if isinstance(e, dict)
                          else Entity(e)) for e in (entities or [])]
        self.features = [f if isinstance(f, Feature) else
                         Feature.from_dict(f) for f in (features or [])]
        self.timestamp_key = timestamp_key
        self.description = description
        self.aggregations = [a if isinstance(a, FeatureAggregation) else
                             FeatureAggregation.from_dict(a)
                             for a in (aggregations or [])]
        self.targets = targets or []
        self.engine = engine or "local"
        self.label_column = label_column
        self.source = source

    def to_dict(self, fields=None, exclude=None, strip=False):
        return {
            "entities": [e.to_dict() for e in self.entities],
            "features": [f.to_dict() for f in self.features],
            "timestamp_key": self.timestamp_key,
            "description": self.description,
            "aggregations": [a.to_dict() for a in self.aggregations],
            "targets": self.targets,
            "engine": self.engine,
            "label_column": self.label_column,
        }


class FeatureSetStatus(ModelObj):
    def __init__(self, state=None, targets=None, stats=None, preview=None):
        self.state = state or "created"
        self.targets = targets or []
        self.stats = stats or {}
        self.preview = preview


class FeatureSet(ModelObj):
    kind = "FeatureSet"

    def __init__(self, name=None, description=None, entities=None,
                 timestamp_key=None, engine=None, label_column=None,
                 project=None):
        self.metadata = FeatureSetMetadata(name=name, project=project)
        self.spec = FeatureSetSpec(entities=entities,
                                   timestamp_key=timestamp_key,
                                   description=description, engine=engine,
                                   label_column=label_column)
        self.status = FeatureSetStatus()
        self.graph = RootFlowStep(name="ingest")
        self._last_df = None

    @property
    def name(self):
        return self.metadata.name

    @property
    def uri(self):
        project = self.metadata.project or "default"
        return f"store://feature-sets/{project}/{self.metadata.name}"

    @property
    def fullname(self):
        return f"{self.metadata.project or 'default'}/{self.metadata.name}"

    def add_entity(self, name, value_type=None, description=None,
                   labels: dict = None):
        entity = Entity(name, value_type, description)
        if labels:
            entity.labels = labels
        self.spec.entities.append(entity)
        return self

    def add_feature(self, feature: Feature, name=None):
        if name:
            feature.name = name
        self.spec.features.append(feature)
        return self

    def add_aggregation(self, column, operations, windows, period=None,
                        name=None, step_name=None, after=None,
                        before=None, emit_policy=None):
        """Register sliding/tumbling window aggregations on a column
        (parity: reference add_aggregation feature_set.py:715)."""
        agg = FeatureAggregation(name=name or column, column=column,
                                 operations=operations, windows=windows,
                                 period=period, step_name=step_name,
                                 after=after)
        agg.validate()
        self.spec.aggregations.append(agg)
        for op in agg.operations:
            for window in agg.windows:
                self.spec.features.append(Feature(
                    name=f"{agg.name}_{op}_{window}", value_type="float",
                    aggregate=True))
        return self

    def entity_names(self) -> list:
        return [e.name for e in self.spec.entities]

    def feature_names(self) -> list:
        return [f.name for f in self.spec.features]

    def get_target_path(self, name="parquet") -> typing.Optional[str]:
        for target in self.status.targets:
            if target.get("name") == name or target.get("kind") == name:
                return target.get("path")
        return None

    def set_targets(self, targets=None, with_defaults=True):
        self.spec.targets = targets or (["parquet", "nosql"]
                                        if with_defaults else [])
        return self

    def to_dict(self, fields=None, exclude=None, strip=False):
        return {
            "kind": self.kind,
            "metadata": self.metadata.to_dict(),
            "spec": self.spec.to_dict(),
            "status": self.status.to_dict(),
        }

    @classmethod
    def from_dict(cls, struct=None, fields=None, deprecated_fields=None):
        struct = struct or {}
        fset = cls()
        meta = struct.get("metadata", {})
        fset.metadata = FeatureSetMetadata.from_dict(meta)
        fset.spec = FeatureSetSpec(**{
            k: v for k, v in struct.get("spec", {}).items()
            if k in ("entities", "features", "timestamp_key", "description",
                     "aggregations", "targets", "engine", "label_column")})
        status = struct.get("status")
        if status:
            fset.status = FeatureSetStatus.from_dict(status)
        return fset

    def save(self, tag="", versioned=False):
        from ..db import get_run_db

        self.metadata.updated = now_iso()
        get_run_db().store_feature_set(
            self.to_dict(), name=self.metadata.name,
            project=self.metadata.project or "default", tag=tag)
        return self

    def plot(self, *args, **kwargs):
        return self.graph.plot(*args, **kwargs)
