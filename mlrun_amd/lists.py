# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Rich list wrappers returned by ``list_runs`` / ``list_artifacts``
(reference lists.py:49 ``RunList``, :165 ``ArtifactList``): plain
``list`` subclasses that add tabular/DataFrame/HTML views and
object-materialisation so notebook workflows (``runs.to_df()``,
``runs.show()``, ``artifacts.dataitems()``) work unchanged."""

import copy
import typing

from .utils import get_in

run_list_header = [
    "project", "uid", "iter", "start", "state", "kind", "name", "labels",
    "inputs", "parameters", "results", "artifacts", "artifact_uris",
    "error",
]

_iter_index = run_list_header.index("iter")
_state_index = run_list_header.index("state")
_parameters_index = run_list_header.index("parameters")
_results_index = run_list_header.index("results")


def _flatten(df, col, prefix=""):
    """Expand a dict-valued column into prefixed scalar columns."""
    import pandas as pd

    expanded = pd.DataFrame(
        [x if isinstance(x, dict) else {} for x in df[col]])
    expanded.columns = [prefix + str(c) for c in expanded.columns]
    expanded.index = df.index
    return pd.concat([df.drop(columns=[col]), expanded], axis=1)


class RunList(list):
    """List of run dicts with table/df/show helpers
    (reference lists.py:49)."""

    def to_rows(self, extend_iterations: bool = False) -> list:
        """Flatten each run into a row; with ``extend_iterations`` each
        hyper-param iteration becomes its own row."""
        rows = []
        for run in self:
            iterations = get_in(run, "status.iterations", "")
            row = [
                get_in(run, "metadata.project", ""),
                get_in(run, "metadata.uid", ""),
                get_in(run, "metadata.iteration", ""),
                get_in(run, "status.start_time", ""),
                get_in(run, "status.state", ""),
                get_in(run, "step_kind", get_in(run, "kind", "")),
                get_in(run, "metadata.name", ""),
                get_in(run, "metadata.labels", ""),
                get_in(run, "spec.inputs", ""),
                get_in(run, "spec.parameters", ""),
                get_in(run, "status.results", ""),
                get_in(run, "status.artifacts", []),
                get_in(run, "status.artifact_uris", {}),
                get_in(run, "status.error", ""),
            ]
            if extend_iterations and iterations:
                header = iterations[0]
                param_cols = {key[len("param."):]: i
                              for i, key in enumerate(header)
                              if key.startswith("param.")}
                result_cols = {key[len("output."):]: i
                               for i, key in enumerate(header)
                               if key.startswith("output.")}
                for iteration in iterations[1:]:
                    row[_state_index] = iteration[0]
                    row[_iter_index] = iteration[1]
                    row[_parameters_index] = {
                        key: iteration[col]
                        for key, col in param_cols.items()}
                    row[_results_index] = {
                        key: iteration[col]
                        for key, col in result_cols.items()}
                    rows.append(copy.copy(row))
            else:
                rows.append(row)
        return [run_list_header] + rows

    def to_df(self, flat: bool = False, extend_iterations: bool = False,
              cache: bool = True):
        """Convert to a pandas DataFrame (cached unless cache=False)."""
        import pandas as pd

        if getattr(self, "_df", None) is not None and cache:
            return self._df
        rows = self.to_rows(extend_iterations=extend_iterations)
        df = pd.DataFrame(rows[1:], columns=rows[0])
        df["start"] = pd.to_datetime(df["start"], errors="coerce",
                                     format="mixed")
        if flat:
            df = _flatten(df, "labels")
            df = _flatten(df, "parameters", "param.")
            df = _flatten(df, "results", "output.")
        self._df = df
        return df

    def show(self, display: bool = True, classes=None, short: bool = False,
             extend_iterations: bool = False):
        """Render an HTML table (notebooks); returns markup when
        display=False."""
        from .render import runs_to_html

        html = runs_to_html(list(self), display=display)
        if not display:
            return html

    def to_objects(self) -> list:
        """Materialise as RunObject instances."""
        from .model import RunObject

        return [RunObject.from_dict(run) for run in self]

    def compare(self, hide_identical: bool = True, exclude: list = None,
                show: bool = None, extend_iterations: bool = True,
                filename: str = None, colorscale: str = None):
        """Parallel-coordinates plot + table comparing the runs
        (reference lists.py:136 → frameworks.parallel_coordinates)."""
        from .frameworks.parallel_coordinates import compare_run_objects

        return compare_run_objects(
            self, hide_identical=hide_identical, exclude=exclude,
            show=show, extend_iterations=extend_iterations,
            filename=filename, colorscale=colorscale)


class ArtifactList(list):
    """List of artifact dicts with table/df/object helpers
    (reference lists.py:165)."""

    def __init__(self, *args):
        super().__init__(*args)
        self.tag = ""

    _head = {
        "tree": "metadata.tree",
        "key": "metadata.key",
        "iter": "metadata.iter",
        "kind": "kind",
        "path": "spec.target_path",
        "hash": "metadata.hash",
        "viewer": "spec.viewer",
        "updated": "metadata.updated",
        "description": "metadata.description",
        "producer": "spec.producer",
        "sources": "spec.sources",
        "labels": "metadata.labels",
        "uri": "uri",
    }

    def to_rows(self) -> list:
        from .artifacts import Artifact

        rows = []
        for artifact in self:
            row = [get_in(artifact, path, "")
                   for path in self._head.values()]
            row[-1] = Artifact.from_dict(artifact).uri
            rows.append(row)
        return [list(self._head.keys())] + rows

    def to_df(self, flat: bool = False):
        import pandas as pd

        rows = self.to_rows()
        df = pd.DataFrame(rows[1:], columns=rows[0])
        df["updated"] = pd.to_datetime(df["updated"], errors="coerce",
                                       format="mixed")
        if flat:
            df = _flatten(df, "producer", "prod_")
            df = _flatten(df, "sources", "src_")
        return df

    def show(self, display: bool = True, classes=None):
        from .render import artifacts_to_html

        html = artifacts_to_html(list(self), display=display)
        if not display:
            return html

    def to_objects(self) -> list:
        """Materialise as typed Artifact instances."""
        from .artifacts import Artifact

        return [Artifact.from_dict(artifact) for artifact in self]

    def objects(self) -> list:
        return self.to_objects()

    def dataitems(self) -> list:
        """DataItem per artifact target (reference lists.py:225)."""
        from .model import get_artifact_target
        from .run import get_dataitem

        items = []
        for artifact in self:
            target = get_artifact_target(artifact)
            if target:
                items.append(get_dataitem(target))
        return items


__all__ = ["RunList", "ArtifactList", "run_list_header"]
