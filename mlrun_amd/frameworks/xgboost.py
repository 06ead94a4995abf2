# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""XGBoost models on the HIP tree-ensemble kernel.

Parity target: reference mlrun/frameworks/xgboost (apply_mlrun +
PickleModelServer used for xgboost, _ml_common/pkl_model_server.py:24).
MI355X-native design: the booster's portable per-tree JSON dump
(``Booster.get_dump(dump_format="json")``) is parsed directly — no
xgboost import needed at serving time — into the flat SoA node table
walked by the ``tree_ensemble`` HIP kernel.

XGBoost semantics match the kernel exactly: route to the "yes" child
when ``feature < split_condition``; leaves sum over trees on top of
base_score; binary:logistic applies sigmoid to the summed margin.
"""

import json
import typing

import torch

from ..errors import MLRunInvalidArgumentError
from .tree import TreeEnsembleModel, V2ModelServer


def nodes_from_xgboost_dump(tree_dumps: typing.Sequence,
                            base_score: float = 0.5,
                            objective: str = "") -> typing.Tuple[dict,
                                                                 float,
                                                                 str]:
    """Parse xgboost's JSON tree dumps (list of per-tree JSON strings
    or already-decoded dicts) into the SoA node table."""
    if not tree_dumps:
        raise MLRunInvalidArgumentError("empty xgboost dump")
    link = "identity"
    if "logistic" in objective:
        link = "sigmoid"
        # base_score is stored in probability space for logistic
        b = min(max(float(base_score), 1e-7), 1 - 1e-7)
        base = float(torch.logit(torch.tensor(b)))
    else:
        base = float(base_score)
    fidx, thr, left, right, leaf, offsets = [], [], [], [], [], [0]
    for dump in tree_dumps:
        root = json.loads(dump) if isinstance(dump, str) else dump
        start = offsets[-1]
        # assign flat slots in dump order (dfs), then wire children
        flat_of: typing.Dict[int, int] = {}
        stack, order = [root], []
        while stack:
            node = stack.pop()
            flat_of[node["nodeid"]] = start + len(order)
            order.append(node)
            for child in reversed(node.get("children", [])):
                stack.append(child)
        for node in order:
            if "leaf" in node:
                fidx.append(-1)
                thr.append(0.0)
                left.append(0)
                right.append(0)
                leaf.append(float(node["leaf"]))
            else:
                feat = node["split"]
                feat_i = int(feat[1:]) if isinstance(feat, str) and \
                    feat.startswith("f") else int(feat)
                fidx.append(feat_i)
                thr.append(float(node["split_condition"]))
                left.append(flat_of[node["yes"]])
                right.append(flat_of[node["no"]])
                leaf.append(0.0)
        offsets.append(start + len(order))
    nodes = {
        "feature_idx": torch.tensor(fidx, dtype=torch.int32),
        "threshold": torch.tensor(thr, dtype=torch.float32),
        "left": torch.tensor(left, dtype=torch.int32),
        "right": torch.tensor(right, dtype=torch.int32),
        "leaf_value": torch.tensor(leaf, dtype=torch.float32),
        "tree_offsets": torch.tensor(offsets, dtype=torch.int32),
    }
    return nodes, base, link


def model_from_xgboost(model) -> TreeEnsembleModel:
    """Convert a fitted xgboost Booster / XGBModel (requires the
    xgboost package only for this call, never for serving)."""
    booster = model.get_booster() if hasattr(model, "get_booster") \
        else model
    dumps = booster.get_dump(dump_format="json")
    cfg = json.loads(booster.save_config())
    learner = cfg.get("learner", {})
    objective = learner.get("objective", {}).get("name", "")
    base_score = float(learner.get("learner_model_param", {})
                       .get("base_score", 0.5))
    nodes, base, link = nodes_from_xgboost_dump(dumps, base_score,
                                                objective)
    return TreeEnsembleModel(nodes, base, link=link)


class XGBoostModelServer(V2ModelServer):
    """V2ModelServer for XGBoost models: loads a ``.json`` dump file
    (list of per-tree dumps + optional meta) or a converted ``.npz``
    node table; predicts with the HIP tree-ensemble kernel."""

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
            payload = json.load(stream)
        if isinstance(payload, dict):  # {"trees": [...], "base_score"..}
            trees = payload["trees"]
            base_score = float(payload.get("base_score", 0.5))
            objective = payload.get("objective", "")
        else:
            trees, base_score, objective = payload, 0.5, ""
        nodes, base, link = nodes_from_xgboost_dump(trees, base_score,
                                                    objective)
        self.model = TreeEnsembleModel(nodes, base, link=link).to(device)

    def predict(self, request: dict):
        return self.model.predict(request["inputs"]).cpu().tolist()
