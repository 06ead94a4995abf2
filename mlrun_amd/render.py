# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Notebook/HTML rendering of runs & artifacts.

Parity target: reference mlrun/render.py (run/artifact HTML tables for
Jupyter display).
"""

import html
import typing


def _table(headers: list, rows: list) -> str:
    head = "".join(f"<th>{html.escape(str(h))}</th>" for h in headers)
    body = "".join(
        "<tr>" + "".join(f"<td>{html.escape(str(c))}</td>" for c in row) +
        "</tr>" for row in rows)
    return (f"<table border='1' class='mlrun-table'>"
            f"<thead><tr>{head}</tr></thead><tbody>{body}</tbody></table>")


def runs_to_html(runs: typing.List[dict], display: bool = True) -> str:
    rows = []
    for run in runs:
        meta = run.get("metadata", {})
        status = run.get("status", {})
        results = status.get("results", {}) or {}
        rows.append([
            (meta.get("uid") or "")[:8],
            meta.get("name", ""),
            meta.get("project", ""),
            status.get("state", ""),
            status.get("start_time", ""),
            ", ".join(f"{k}={v}" for k, v in list(results.items())[:5]),
        ])
    markup = _table(["uid", "name", "project", "state", "start", "results"],
                    rows)
    if display:
        _display(markup)
    return markup


def artifacts_to_html(artifacts: typing.List[dict],
                      display: bool = True) -> str:
    rows = []
    for artifact in artifacts:
        meta = artifact.get("metadata", {})
        spec = artifact.get("spec", {})
        rows.append([meta.get("key", ""), artifact.get("kind", ""),
                     meta.get("tree", "")[:8], spec.get("target_path", "")])
    markup = _table(["key", "kind", "tree", "target"], rows)
    if display:
        _display(markup)
    return markup


def run_to_html(run: dict, display: bool = True) -> str:
    return runs_to_html([run], display)


def _display(markup: str):
    try:
        from IPython.display import HTML, display as ip_display

        ip_display(HTML(markup))
    except ImportError:
        pass
