# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""LightGBM models on the HIP tree-ensemble kernel.

Parity target: reference mlrun/frameworks/lgbm (apply_mlrun hooks +
lgbm/model_server.py:26 LGBMModelServer).  MI355X-native design: the
booster's portable TEXT dump (``Booster.model_to_string()`` /
``save_model`` output) is parsed directly — no lightgbm import needed
at serving time — into the flat SoA node table that the
``tree_ensemble`` HIP kernel walks in one launch per batch.

LightGBM semantics: numeric splits route LEFT when
``feature <= threshold``; the kernel compares with ``<``, so
thresholds are nudged with nextafter(thr, +inf) — exact in float32.
Leaf values already include learning rate and are summed over trees.
"""

import math
import typing

import numpy as np
import torch

from ..errors import MLRunInvalidArgumentError
from .tree import TreeEnsembleModel, V2ModelServer


def _section_trees(model_text: str) -> typing.List[dict]:
    """Split a LightGBM model dump into per-tree key→value dicts."""
    trees, current = [], None
    for line in model_text.splitlines():
        line = line.strip()
        if line.startswith("Tree="):
            current = {}
            trees.append(current)
            continue
        if current is None or "=" not in line:
            if line == "end of trees":
                current = None
            continue
        key, _, value = line.partition("=")
        current[key] = value
    return [t for t in trees if "leaf_value" in t]


def _floats(s: str) -> typing.List[float]:
    return [float(x) for x in s.split()] if s else []


def _ints(s: str) -> typing.List[int]:
    return [int(x) for x in s.split()] if s else []


def nodes_from_lgbm_text(model_text: str) -> typing.Tuple[dict, float,
                                                          str]:
    """Parse a LightGBM text model dump into the SoA node table.

    Returns (nodes, base_score, link).  Internal nodes of tree t are
    numbered 0..num_leaves-2 in the dump; a child index c >= 0 points
    at internal node c, c < 0 at leaf ~c (i.e. -(c)-1).  Leaves are
    appended after the internal nodes in the flat table.
    """
    sections = _section_trees(model_text)
    if not sections:
        raise MLRunInvalidArgumentError(
            "no trees found in LightGBM model text")
    link = "sigmoid" if "objective=binary" in model_text else "identity"
    fidx, thr, left, right, leaf, offsets = [], [], [], [], [], [0]
    for tree in sections:
        leaf_values = _floats(tree.get("leaf_value", ""))
        n_leaves = len(leaf_values)
        split_feature = _ints(tree.get("split_feature", ""))
        thresholds = _floats(tree.get("threshold", ""))
        left_child = _ints(tree.get("left_child", ""))
        right_child = _ints(tree.get("right_child", ""))
        n_internal = len(split_feature)
        start = offsets[-1]

        def flat(child: int) -> int:
            if child >= 0:
                return start + child
            return start + n_internal + (-child - 1)

        for i in range(n_internal):
            fidx.append(split_feature[i])
            # lightgbm routes left on <=; kernel compares with < —
            # nextafter makes (f < thr') ⟺ (f <= thr) exactly in f32
            t32 = np.float32(thresholds[i])
            thr.append(float(np.nextafter(t32, np.float32(math.inf))))
            left.append(flat(left_child[i]))
            right.append(flat(right_child[i]))
            leaf.append(0.0)
        for value in leaf_values:
            fidx.append(-1)
            thr.append(0.0)
            left.append(0)
            right.append(0)
            leaf.append(value)
        offsets.append(start + n_internal + n_leaves)
    nodes = {
        "feature_idx": torch.tensor(fidx, dtype=torch.int32),
        "threshold": torch.tensor(thr, dtype=torch.float32),
        "left": torch.tensor(left, dtype=torch.int32),
        "right": torch.tensor(right, dtype=torch.int32),
        "leaf_value": torch.tensor(leaf, dtype=torch.float32),
        "tree_offsets": torch.tensor(offsets, dtype=torch.int32),
    }
    return nodes, 0.0, link


def model_from_lgbm(model) -> TreeEnsembleModel:
    """Convert a fitted lightgbm Booster / LGBMModel (requires the
    lightgbm package only for this call, never for serving)."""
    booster = getattr(model, "booster_", model)
    text = booster.model_to_string()
    nodes, base, link = nodes_from_lgbm_text(text)
    return TreeEnsembleModel(nodes, base, link=link)


class LGBMModelServer(V2ModelServer):
    """V2ModelServer for LightGBM models: loads a ``.txt`` model dump
    (or an already-converted ``.npz`` node table) and predicts with
    the HIP tree-ensemble kernel."""

    def load(self):
        device = self.get_param(
            "device", "cuda:0" if torch.cuda.is_available() else "cpu")
        if isinstance(self.model, TreeEnsembleModel):
            self.model = self.model.to(device)
            return
        model_file, _extra = self.get_model()
        if str(model_file).endswith(".npz"):
            self.model = TreeEnsembleModel.load(model_file).to(device)
            return
        with open(model_file) as stream:
            nodes, base, link = nodes_from_lgbm_text(stream.read())
        self.model = TreeEnsembleModel(nodes, base, link=link).to(device)

    def predict(self, request: dict):
        return self.model.predict(request["inputs"]).cpu().tolist()
