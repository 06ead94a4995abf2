# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Hyperparameter iteration generators (grid/random/list) + selection.

Parity target: reference mlrun/runtimes/generators.py (GridGenerator
:111, RandomGenerator :146, ListGenerator :166, selector logic).
"""

import itertools
import random
import typing

from ..errors import MLRunInvalidArgumentError
from ..model import HyperParamOptions, HyperParamStrategies, RunObject


def get_generator(spec, execution=None, param_file_body=None):
    """Return a TaskGenerator if the run has hyperparams, else None."""
    hyperparams = spec.hyperparams
    options = spec.hyper_param_options or HyperParamOptions()
    strategy = options.strategy or spec.strategy
    if options.param_file and param_file_body is None:
        import json
        import os

        path = options.param_file
        if os.path.isfile(path):
            with open(path) as fp:
                if path.endswith(".json"):
                    param_file_body = json.load(fp)
                else:
                    import csv

                    rows = list(csv.DictReader(fp))
                    param_file_body = {
                        k: [_parse(r[k]) for r in rows] for k in rows[0]
                    } if rows else {}
    if param_file_body:
        hyperparams = param_file_body
    if not hyperparams:
        return None
    if not options.selector and spec.selector:
        options.selector = spec.selector
    if strategy in (None, HyperParamStrategies.grid):
        return GridGenerator(hyperparams, options)
    if strategy == HyperParamStrategies.random:
        return RandomGenerator(hyperparams, options)
    if strategy == HyperParamStrategies.list:
        return ListGenerator(hyperparams, options)
    raise MLRunInvalidArgumentError(f"unsupported strategy {strategy}")


def _parse(value: str):
    import json

    try:
        return json.loads(value)
    except (ValueError, TypeError):
        return value


class TaskGenerator:
    def __init__(self, hyperparams: dict, options: HyperParamOptions):
        self.hyperparams = hyperparams
        self.options = options

    @property
    def max_errors(self):
        return self.options.max_errors or 0

    def use_parallel(self) -> bool:
        return bool(self.options.parallel_runs and self.options.parallel_runs > 1)

    def generate(self, run: RunObject) -> typing.Iterator[RunObject]:
        for i, params in enumerate(self.param_sets(), start=1):
            task = run.copy()
            task.metadata.iteration = i
            task.spec.hyperparams = None
            task.spec.hyper_param_options = None
            task.spec.parameters = dict(run.spec.parameters or {})
            task.spec.parameters.update(params)
            yield task

    def param_sets(self) -> typing.Iterator[dict]:
        raise NotImplementedError

    def eval_stop_condition(self, results: dict) -> bool:
        condition = self.options.stop_condition
        if not condition:
            return False
        from ..utils.safe_eval import safe_eval

        try:
            return bool(safe_eval(condition, dict(results)))
        except Exception:
            return False


class GridGenerator(TaskGenerator):
    def param_sets(self):
        keys = list(self.hyperparams.keys())
        values = [v if isinstance(v, (list, tuple)) else [v]
                  for v in self.hyperparams.values()]
        for combo in itertools.product(*values):
            yield dict(zip(keys, combo))


class RandomGenerator(TaskGenerator):
    def param_sets(self):
        iterations = self.options.max_iterations or 10
        keys = list(self.hyperparams.keys())
        values = [v if isinstance(v, (list, tuple)) else [v]
                  for v in self.hyperparams.values()]
        for _ in range(iterations):
            yield {k: random.choice(v) for k, v in zip(keys, values)}


class ListGenerator(TaskGenerator):
    def param_sets(self):
        values = {k: (v if isinstance(v, (list, tuple)) else [v])
                  for k, v in self.hyperparams.items()}
        length = max(len(v) for v in values.values())
        for i in range(length):
            yield {k: v[i] for k, v in values.items() if i < len(v)}


def selector(results: list, criteria: str):
    """Pick the best iteration: criteria 'max.<key>' or 'min.<key>'.

    results: list of run dicts (children).  Returns (best_iteration_id,
    best_task_dict)."""
    if not criteria:
        return 0, None
    op = "max"
    if "." in criteria:
        op, criteria = criteria.split(".", 1)
    best_id, best_item, best_val = 0, None, None
    for item in results:
        value = item.get("status", {}).get("results", {}).get(criteria)
        if value is None:
            continue
        if best_val is None or (op == "max" and value > best_val) or \
                (op == "min" and value < best_val):
            best_val = value
            best_item = item
            best_id = item.get("metadata", {}).get("iteration", 0)
    return best_id, best_item
