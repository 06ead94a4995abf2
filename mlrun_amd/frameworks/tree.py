# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Tree-ensemble (GBDT / random-forest) models on the HIP tree kernel.

The model format is a flat SoA node table (feature_idx / threshold /
left / right / leaf_value / tree_offsets) saved as .npz — loadable
from sklearn/xgboost exports or synthesized.  Serving runs the whole
batch through one tree_ensemble kernel launch (reference analog: the
user's xgboost predict inside V2ModelServer.predict, v2_serving.py:381).
"""

import io
import typing

import numpy as np
import torch

from ..serving.v2_serving import V2ModelServer

NODE_KEYS = ["feature_idx", "threshold", "left", "right", "leaf_value",
             "tree_offsets"]


def random_forest_nodes(n_trees: int, depth: int, n_features: int,
                        seed: int = 0) -> dict:
    """Synthesize a complete-binary-tree ensemble (for tests/benchmarks;
    there is no network to fetch real models)."""
    rng = np.random.default_rng(seed)
    nodes_per_tree = (1 << (depth + 1)) - 1
    n_internal = (1 << depth) - 1
    total = n_trees * nodes_per_tree
    feature_idx = np.empty(total, dtype=np.int32)
    threshold = np.zeros(total, dtype=np.float32)
    left = np.zeros(total, dtype=np.int32)
    right = np.zeros(total, dtype=np.int32)
    leaf_value = np.zeros(total, dtype=np.float32)
    offsets = np.arange(0, (n_trees + 1) * nodes_per_tree, nodes_per_tree,
                        dtype=np.int32)
    for t in range(n_trees):
        base = t * nodes_per_tree
        for i in range(nodes_per_tree):
            if i < n_internal:
                feature_idx[base + i] = rng.integers(0, n_features)
                threshold[base + i] = rng.normal()
                left[base + i] = base + 2 * i + 1
                right[base + i] = base + 2 * i + 2
            else:
                feature_idx[base + i] = -1
                leaf_value[base + i] = rng.normal() * 0.1
    return {
        "feature_idx": torch.from_numpy(feature_idx),
        "threshold": torch.from_numpy(threshold),
        "left": torch.from_numpy(left),
        "right": torch.from_numpy(right),
        "leaf_value": torch.from_numpy(leaf_value),
        "tree_offsets": torch.from_numpy(offsets),
    }


def nodes_from_sklearn(model) -> typing.Tuple[dict, float]:
    """Flatten a fitted sklearn tree ensemble (GradientBoosting /
    RandomForest regressor) into the SoA node table."""
    estimators = getattr(model, "estimators_", None)
    if estimators is None:
        raise ValueError("model has no estimators_ (not a tree ensemble)")
    flat = np.asarray(estimators).ravel()
    scale = 1.0
    base = 0.0
    if hasattr(model, "learning_rate"):  # gradient boosting
        scale = model.learning_rate
        init = getattr(model, "init_", None)
        if init is not None and hasattr(init, "constant_"):
            base = float(np.ravel(init.constant_)[0])
    elif hasattr(model, "n_estimators"):  # forest: average
        scale = 1.0 / len(flat)
    fidx, thr, left, right, leaf, offsets = [], [], [], [], [], [0]
    for est in flat:
        tree = est.tree_
        start = offsets[-1]
        for i in range(tree.node_count):
            if tree.children_left[i] == -1:
                fidx.append(-1)
                thr.append(0.0)
                left.append(0)
                right.append(0)
                leaf.append(float(tree.value[i].ravel()[0]) * scale)
            else:
                fidx.append(int(tree.feature[i]))
                thr.append(float(tree.threshold[i]))
                left.append(start + int(tree.children_left[i]))
                right.append(start + int(tree.children_right[i]))
                leaf.append(0.0)
        offsets.append(start + tree.node_count)
    nodes = {
        "feature_idx": torch.tensor(fidx, dtype=torch.int32),
        "threshold": torch.tensor(thr, dtype=torch.float32),
        "left": torch.tensor(left, dtype=torch.int32),
        "right": torch.tensor(right, dtype=torch.int32),
        "leaf_value": torch.tensor(leaf, dtype=torch.float32),
        "tree_offsets": torch.tensor(offsets, dtype=torch.int32),
    }
    return nodes, base


class TreeEnsembleModel:
    """A tree ensemble + device placement + predict."""

    def __init__(self, nodes: dict, base_score: float = 0.0,
                 n_features: int = None, link: str = "identity"):
        self.nodes = nodes
        self.base_score = base_score
        self.n_features = n_features
        self.link = link  # identity | sigmoid

    def to(self, device) -> "TreeEnsembleModel":
        self.nodes = {k: v.to(device) for k, v in self.nodes.items()}
        return self

    @property
    def device(self):
        return self.nodes["feature_idx"].device

    def predict(self, features) -> torch.Tensor:
        from .. import ops

        if not isinstance(features, torch.Tensor):
            features = torch.as_tensor(np.asarray(features,
                                                  dtype=np.float32))
        features = features.to(self.device, dtype=torch.float32).contiguous()
        margins = ops.tree_ensemble_predict(features, self.nodes,
                                            self.base_score)
        if self.link == "sigmoid":
            margins = torch.sigmoid(margins)
        return margins

    def save(self, path: str):
        arrays = {k: v.cpu().numpy() for k, v in self.nodes.items()}
        arrays["base_score"] = np.float32(self.base_score)
        arrays["link"] = np.bytes_(self.link.encode())
        np.savez(path, **arrays)

    @classmethod
    def load(cls, path_or_bytes) -> "TreeEnsembleModel":
        if isinstance(path_or_bytes, bytes):
            path_or_bytes = io.BytesIO(path_or_bytes)
        data = np.load(path_or_bytes, allow_pickle=False)
        nodes = {k: torch.from_numpy(np.ascontiguousarray(data[k]))
                 for k in NODE_KEYS}
        base = float(data["base_score"]) if "base_score" in data else 0.0
        link = bytes(data["link"]).decode() if "link" in data else "identity"
        return cls(nodes, base, link=link)

    @classmethod
    def from_sklearn(cls, model) -> "TreeEnsembleModel":
        nodes, base = nodes_from_sklearn(model)
        return cls(nodes, base)


class TreeEnsembleModelServer(V2ModelServer):
    """V2ModelServer over the HIP tree-ensemble kernel (baseline
    config 2: classic model on 1 MI355X)."""

    def load(self):
        import torch as _torch

        device = self.get_param(
            "device", "cuda:0" if _torch.cuda.is_available() else "cpu")
        if self.model is not None and isinstance(self.model,
                                                 TreeEnsembleModel):
            self.model = self.model.to(device)
            return
        if self.model_path:
            model_file, extra = self.get_model(".npz")
            self.model = TreeEnsembleModel.load(model_file).to(device)
            return
        # synthetic model (benchmarks / tests without artifacts)
        self.model = TreeEnsembleModel(
            random_forest_nodes(
                n_trees=int(self.get_param("n_trees", 100)),
                depth=int(self.get_param("depth", 6)),
                n_features=int(self.get_param("n_features", 32)),
                seed=int(self.get_param("seed", 0))),
            base_score=float(self.get_param("base_score", 0.0))).to(device)

    def predict(self, request: dict):
        result = self.model.predict(request["inputs"])
        return result.cpu().tolist()
