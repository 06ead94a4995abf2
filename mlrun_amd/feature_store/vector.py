# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""FeatureVector + OnlineVectorService.

Parity target: reference mlrun/feature_store/feature_vector.py
(FeatureVector :468, OnlineVectorService :910 with get :975).
The online get() is a batched table lookup (one ring reduce per
aggregation serves the whole request batch) instead of per-row
asyncio emits.
"""

import enum
import typing

from ..errors import MLRunInvalidArgumentError
from ..model import ModelObj
from ..utils import now_iso


def parse_feature_string(feature: str):
    """'set_name.feature [as alias]' -> (set, feature, alias)."""
    feature = feature.strip()
    alias = None
    if " as " in feature:
        feature, alias = feature.split(" as ", 1)
    if "." not in feature:
        raise MLRunInvalidArgumentError(
            f"feature {feature!r} must be '<featureset>.<name>'")
    set_name, _, name = feature.partition(".")
    return set_name.strip(), name.strip(), (alias.strip() if alias else None)


class FeatureVectorMetadata(ModelObj):
    def __init__(self, name=None, project=None, tag=None, labels=None,
                 updated=None):
        self.name = name
        self.project = project
        self.tag = tag
        self.labels = labels or {}
        self.updated = updated


class FeatureVectorSpec(ModelObj):
    def __init__(self, features=None, description=None, label_feature=None,
                 with_indexes=None):
        self.features = features or []
        self.description = description
        self.label_feature = label_feature
        self.with_indexes = with_indexes


class FeatureVector(ModelObj):
    kind = "FeatureVector"

    def __init__(self, name=None, features=None, label_feature=None,
                 description=None, with_indexes=None, project=None,
                 join_graph=None, relations: dict = None,
                 entity_fields=None, entity_source=None,
                 timestamp_field: str = None, function=None,
                 analysis=None, graph=None):
        self.metadata = FeatureVectorMetadata(name=name, project=project)
        self.spec = FeatureVectorSpec(features=features,
                                      description=description,
                                      label_feature=label_feature,
                                      with_indexes=with_indexes)
        # reference spec extras: join_graph/relations drive the
        # offline merger; the rest are enrichment metadata
        self.spec.join_graph = join_graph
        self.spec.relations = relations or {}
        self.spec.entity_fields = entity_fields or []
        self.spec.entity_source = entity_source
        self.spec.timestamp_field = timestamp_field
        self.spec.function = function
        self.spec.analysis = analysis
        self.spec.graph = graph
        self.status = ModelObj()

    @property
    def uri(self):
        project = self.metadata.project or "default"
        return f"store://feature-vectors/{project}/{self.metadata.name}"

    def grouped_features(self):
        """Yield (featureset_name, [columns], {rename map}) groups."""
        groups: dict = {}
        renames: dict = {}
        for feature in self.spec.features:
            set_name, name, alias = parse_feature_string(feature)
            groups.setdefault(set_name, []).append(name)
            if alias:
                renames.setdefault(set_name, {})[name] = alias
        for set_name, columns in groups.items():
            if "*" in columns:
                columns = ["*"]
            yield set_name, columns, renames.get(set_name, {})

    def to_dict(self, fields=None, exclude=None, strip=False):
        return {
            "kind": self.kind,
            "metadata": self.metadata.to_dict(),
            "spec": {
                "features": self.spec.features,
                "description": self.spec.description,
                "label_feature": self.spec.label_feature,
            },
        }

    @classmethod
    def from_dict(cls, struct=None, fields=None, deprecated_fields=None):
        struct = struct or {}
        vec = cls()
        vec.metadata = FeatureVectorMetadata.from_dict(
            struct.get("metadata", {}))
        spec = struct.get("spec", {})
        vec.spec = FeatureVectorSpec(
            features=spec.get("features"),
            description=spec.get("description"),
            label_feature=spec.get("label_feature"))
        return vec

    def save(self, tag="", versioned=False):
        from ..db import get_run_db

        self.metadata.updated = now_iso()
        get_run_db().store_feature_vector(
            self.to_dict(), name=self.metadata.name,
            project=self.metadata.project or "default", tag=tag)
        return self

    @classmethod
    def resolve(cls, ref) -> "FeatureVector":
        if isinstance(ref, cls):
            return ref
        if isinstance(ref, str):
            from ..db import get_run_db

            name, project = ref, "default"
            if ref.startswith("store://feature-vectors/"):
                body = ref[len("store://feature-vectors/"):]
                project, _, name = body.partition("/")
            elif "/" in ref:
                project, _, name = ref.partition("/")
            name = name.split(":")[0]
            struct = get_run_db().get_feature_vector(name, project)
            return cls.from_dict(struct)
        raise MLRunInvalidArgumentError("cannot resolve feature vector")


class JoinGraph:
    """Per-feature-set join specification for offline merges
    (reference feature_vector.py:204): chain ``inner/left/outer``
    calls in merge order; ``get_offline_features(join_graph=...)``
    applies the requested join type per set instead of the default
    inner join."""

    def __init__(self, first_feature_set: str = None):
        self.steps: list = []
        if first_feature_set:
            self.steps.append((_name_of(first_feature_set), "first"))

    def _join(self, other, how: str) -> "JoinGraph":
        self.steps.append((_name_of(other), how))
        return self

    def inner(self, other) -> "JoinGraph":
        return self._join(other, "inner")

    def left(self, other) -> "JoinGraph":
        return self._join(other, "left")

    def outer(self, other) -> "JoinGraph":
        return self._join(other, "outer")

    def how_for(self, feature_set_name: str, default: str = "inner"):
        for name, how in self.steps:
            if name == feature_set_name and how != "first":
                return how
        return default

    def order(self) -> list:
        return [name for name, _ in self.steps]


def _name_of(feature_set) -> str:
    name = getattr(feature_set, "metadata", None)
    if name is not None:
        return feature_set.metadata.name
    return str(feature_set).split("/")[-1].split(":")[0]


class OnlineVectorService:
    """Online lookups over the vector's feature sets (reference
    OnlineVectorService.get :975 — here batched)."""

    def __init__(self, vector: FeatureVector, tables: dict,
                 impute_policy: dict = None):
        self.vector = vector
        self._tables = tables
        self.impute_policy = dict(impute_policy or {})
        self._groups = list(vector.grouped_features())
        self._resolve_stat_imputes()

    def _resolve_stat_imputes(self):
        """"$mean"/"$max"/"$min" impute values resolve from the
        feature-set statistics captured at ingest (reference
        impute_policy semantics)."""
        stat_keys = {"$mean": "mean", "$max": "max", "$min": "min"}
        pending = {k: v for k, v in self.impute_policy.items()
                   if isinstance(v, str) and v in stat_keys}
        if not pending:
            return
        for set_name, columns, aliases in self._groups:
            fset = self._tables[set_name].feature_set
            stats = fset.status.stats or {}
            for col in columns:
                name = aliases.get(col, col)
                if name in pending and col in stats:
                    stat = stats[col].get(stat_keys[pending[name]])
                    if stat is not None:
                        self.impute_policy[name] = float(stat)

    def _all_aggregates(self) -> bool:
        for set_name, cols, _ in self._groups:
            fset = self._tables[set_name].feature_set
            agg_names = {f.name for f in fset.spec.features if f.aggregate}
            if cols == ["*"] or any(c not in agg_names for c in cols):
                return False
        return True

    @property
    def status(self):
        return "ready"

    def get(self, entity_rows: typing.List[dict], as_list: bool = False):
        """entity_rows: [{entity: value, ...}, ...] -> feature records."""
        if isinstance(entity_rows, dict):
            entity_rows = [entity_rows]
        if as_list and self._all_aggregates():
            # columnar fast path: fancy-indexed ring reductions
            columns = []
            for set_name, cols, aliases in self._groups:
                table = self._tables[set_name]
                matrix = table.get_agg_matrix(entity_rows, cols)
                columns.append(matrix)
            import numpy as np

            matrix = np.concatenate(columns, axis=1)
            out = matrix.astype(object)
            # imputation by feature name order
            names = [aliases.get(c, c) for _, cols, aliases in self._groups
                     for c in cols]
            default = self.impute_policy.get("*")
            for j, name in enumerate(names):
                fill = self.impute_policy.get(name, default)
                col = out[:, j]
                col[np.isnan(matrix[:, j])] = fill
            return out.tolist()
        results = [dict() for _ in entity_rows]
        ordered_names: typing.List[str] = []
        for set_name, columns, aliases in self._groups:
            table = self._tables[set_name]
            records = table.get(entity_rows)
            fset = table.feature_set
            if columns == ["*"]:
                columns = [f.name for f in fset.spec.features]
            for out, record in zip(results, records):
                for col in columns:
                    name = aliases.get(col, col)
                    value = record.get(col)
                    out[name] = value
            for col in columns:
                name = aliases.get(col, col)
                if name not in ordered_names:
                    ordered_names.append(name)
        # imputation ("*" = default for every feature — reference
        # impute_policy wildcard)
        default = self.impute_policy.get("*")
        for out in results:
            for key, value in list(out.items()):
                if value is None and key in self.impute_policy:
                    out[key] = self.impute_policy[key]
                elif value is None and default is not None:
                    out[key] = default
        if as_list:
            return [[out.get(name) for name in ordered_names]
                    for out in results]
        return results

    def close(self):
        pass

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()


class FixedWindowType(enum.Enum):
    """Which fixed window the online service reads (reference
    feature_vector.py:449): the still-open current window or the last
    closed one."""

    CurrentOpenWindow = 1
    LastClosedWindow = 2
