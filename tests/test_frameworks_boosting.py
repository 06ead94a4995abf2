# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""LightGBM / XGBoost dump parsers onto the tree-ensemble kernel.

Fixtures are hand-written portable dumps (the formats written by
Booster.model_to_string() / get_dump(dump_format="json")); expected
outputs are computed by hand from the tree semantics.
"""

import json

import numpy as np
import pytest
import torch

from mlrun_amd.frameworks import (
    LGBMModelServer,
    XGBoostModelServer,
    nodes_from_lgbm_text,
    nodes_from_xgboost_dump,
)
from mlrun_amd.frameworks.tree import TreeEnsembleModel
from mlrun_amd.serving import GraphServer  # noqa: F401  (import check)

# Two trees.  Tree 0: root splits f0 <= 1.5 -> leaves [10, 20];
# tree 1: root splits f1 <= 0.0, right child splits f0 <= -1.0
# -> leaves [1, 2, 3].
LGBM_TEXT = """tree
version=v4
objective=regression
feature_names=f0 f1

Tree=0
num_leaves=2
num_cat=0
split_feature=0
threshold=1.5
decision_type=2
left_child=-1
right_child=-2
leaf_value=10 20
internal_value=0

Tree=1
num_leaves=3
num_cat=0
split_feature=1 0
threshold=0.0 -1.0
decision_type=2 2
left_child=-1 -2
right_child=1 -3
leaf_value=1 2 3
internal_value=0 0

end of trees
"""

# Same shapes in xgboost json-dump form ("<" routing to "yes").
XGB_TREE0 = {
    "nodeid": 0, "split": "f0", "split_condition": 1.5,
    "yes": 1, "no": 2, "missing": 1,
    "children": [{"nodeid": 1, "leaf": 10.0},
                 {"nodeid": 2, "leaf": 20.0}],
}
XGB_TREE1 = {
    "nodeid": 0, "split": "f1", "split_condition": 0.0,
    "yes": 1, "no": 2, "missing": 1,
    "children": [
        {"nodeid": 1, "leaf": 1.0},
        {"nodeid": 2, "split": "f0", "split_condition": -1.0,
         "yes": 3, "no": 4, "missing": 3,
         "children": [{"nodeid": 3, "leaf": 2.0},
                      {"nodeid": 4, "leaf": 3.0}]},
    ],
}

FEATS = [[1.0, -1.0],   # t0: f0<=1.5 -> 10 ; t1: f1<=0 -> 1   => 11
         [2.0, 1.0],    # t0: 20 ; t1: f1>0, f0>-1 -> 3        => 23
         [-2.0, 0.5],   # t0: 10 ; t1: f1>0, f0<=-1 -> 2       => 12
         [1.5, 0.0]]    # t0: f0<=1.5 (boundary) -> 10; t1: 1  => 11


class TestLGBMParser:
    def test_predict_matches_hand_eval(self):
        nodes, base, link = nodes_from_lgbm_text(LGBM_TEXT)
        assert link == "identity" and base == 0.0
        model = TreeEnsembleModel(nodes, base, link=link)
        out = model.predict(torch.tensor(FEATS))
        assert out.tolist() == [11.0, 23.0, 12.0, 11.0]

    def test_boundary_le_semantics(self):
        # f0 == threshold must go LEFT (lightgbm <=); the nextafter
        # adjustment makes the kernel's < behave as <=
        nodes, base, link = nodes_from_lgbm_text(LGBM_TEXT)
        model = TreeEnsembleModel(nodes, base, link=link)
        out = model.predict(torch.tensor([[1.5, 5.0]]))
        assert out[0].item() == pytest.approx(10.0 + 3.0)

    def test_binary_objective_sets_sigmoid(self):
        text = LGBM_TEXT.replace("objective=regression",
                                 "objective=binary sigmoid:1")
        nodes, base, link = nodes_from_lgbm_text(text)
        model = TreeEnsembleModel(nodes, base, link=link)
        out = model.predict(torch.tensor([[1.0, -1.0]]))
        assert out[0].item() == pytest.approx(
            torch.sigmoid(torch.tensor(11.0)).item(), rel=1e-5)

    def test_server_loads_txt_dump(self, tmp_path):
        path = tmp_path / "model.txt"
        path.write_text(LGBM_TEXT)
        server = LGBMModelServer(None, name="m", model_path=str(path))
        server.load()
        assert server.predict({"inputs": FEATS}) == [11.0, 23.0, 12.0,
                                                     11.0]


class TestXGBoostParser:
    def test_predict_matches_hand_eval(self):
        dumps = [json.dumps(XGB_TREE0), json.dumps(XGB_TREE1)]
        nodes, base, link = nodes_from_xgboost_dump(dumps,
                                                    base_score=0.0)
        model = TreeEnsembleModel(nodes, base, link=link)
        out = model.predict(torch.tensor(FEATS))
        # boundary row differs: xgboost < routes both boundary splits RIGHT
        assert out.tolist() == [11.0, 23.0, 12.0, 23.0]

    def test_logistic_link_and_base(self):
        nodes, base, link = nodes_from_xgboost_dump(
            [XGB_TREE0], base_score=0.5, objective="binary:logistic")
        assert link == "sigmoid" and base == pytest.approx(0.0)
        model = TreeEnsembleModel(nodes, base, link=link)
        out = model.predict(torch.tensor([[0.0, 0.0]]))
        assert out[0].item() == pytest.approx(
            torch.sigmoid(torch.tensor(10.0)).item(), rel=1e-5)

    def test_server_loads_json_dump(self, tmp_path):
        path = tmp_path / "model.json"
        path.write_text(json.dumps({"trees": [XGB_TREE0, XGB_TREE1],
                                    "base_score": 0.0}))
        server = XGBoostModelServer(None, name="m",
                                    model_path=str(path))
        server.load()
        out = server.predict({"inputs": FEATS})
        assert out == [11.0, 23.0, 12.0, 23.0]

    def test_matches_cpu_reference_on_random_ensemble(self):
        # cross-check the flat table against an independent recursive
        # walk of the dump for random inputs
        rng = np.random.default_rng(3)
        dumps = [json.dumps(XGB_TREE0), json.dumps(XGB_TREE1)]
        nodes, base, _ = nodes_from_xgboost_dump(dumps, base_score=0.25)
        model = TreeEnsembleModel(nodes, base)
        feats = rng.normal(size=(64, 2)).astype(np.float32)

        def walk(node, row):
            while "leaf" not in node:
                fi = int(node["split"][1:])
                nxt = node["yes"] if row[fi] < node["split_condition"] \
                    else node["no"]
                node = next(c for c in node["children"]
                            if c["nodeid"] == nxt)
            return node["leaf"]

        expect = [0.25 + walk(XGB_TREE0, r) + walk(XGB_TREE1, r)
                  for r in feats]
        got = model.predict(torch.from_numpy(feats)).tolist()
        assert got == pytest.approx(expect, rel=1e-6)


class TestAutoDetectRouting:
    def test_server_table_routes_boosting_kinds(self):
        from mlrun_amd.frameworks.auto import get_model_server_class

        assert get_model_server_class("lightgbm") is LGBMModelServer
        assert get_model_server_class("xgboost") is XGBoostModelServer

    def test_gated_servers_raise_clearly(self):
        from mlrun_amd.errors import MLRunMissingDependencyError
        from mlrun_amd.frameworks import ONNXModelServer, TFKerasModelServer

        with pytest.raises(MLRunMissingDependencyError):
            ONNXModelServer(None, name="m").load()
        with pytest.raises(MLRunMissingDependencyError):
            TFKerasModelServer(None, name="m").load()
