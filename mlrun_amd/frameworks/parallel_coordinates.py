# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Run comparison: parallel-coordinates plot + table
(reference frameworks/parallel_coordinates.py:240 compare_run_objects,
:290 compare_db_runs).

The reference renders through plotly; this build emits a dependency-
free inline-SVG parallel-coordinates plot (one vertical axis per
hyper-param / result column, one polyline per run) plus an HTML table,
so it works in any notebook or file export without plotly installed."""

import html as _html
import typing

PLOT_WIDTH = 960
PLOT_HEIGHT = 360
MARGIN_X = 80
MARGIN_Y = 40

_LINE_COLORS = ["#1f77b4", "#ff7f0e", "#2ca02c", "#d62728", "#9467bd",
                "#8c564b", "#e377c2", "#7f7f7f", "#bcbd22", "#17becf"]


def _axis_scale(values: list) -> typing.Tuple[list, list]:
    """Map raw column values to [0,1] positions + tick labels.
    Numeric columns scale linearly; categorical columns get evenly
    spaced slots."""
    numeric = []
    for value in values:
        try:
            numeric.append(float(value))
        except (TypeError, ValueError):
            numeric = None
            break
    if numeric is not None:
        low, high = min(numeric), max(numeric)
        span = (high - low) or 1.0
        pos = [(v - low) / span for v in numeric]
        ticks = [(0.0, _fmt(low)), (1.0, _fmt(high))]
        if low != high:
            ticks.insert(1, (0.5, _fmt((low + high) / 2)))
        return pos, ticks
    categories = sorted({str(v) for v in values})
    if len(categories) == 1:
        slots = {categories[0]: 0.5}
    else:
        step = 1.0 / (len(categories) - 1)
        slots = {c: i * step for i, c in enumerate(categories)}
    return ([slots[str(v)] for v in values],
            [(p, c) for c, p in slots.items()])


def _fmt(value: float) -> str:
    if value == int(value) and abs(value) < 1e15:
        return str(int(value))
    return f"{value:.4g}"


def gen_pcp_plot(source_df, index_col: str = "iter",
                 hide_identical: bool = True, exclude: list = None,
                 colorscale: str = None) -> str:
    """Render a parallel-coordinates SVG for the param.*/output.*
    columns of source_df (reference gen_pcp_plot — plotly-free)."""
    df = source_df
    for col in exclude or []:
        for name in (col, f"param.{col}"):
            if name in df.columns:
                df = df.drop(columns=[name])
    axes = [c for c in df.columns
            if c.startswith("param.") or c.startswith("output.")]
    if hide_identical and len(df) > 1:
        axes = [c for c in axes if df[c].astype(str).nunique() > 1]
    if not axes or not len(df):
        return "<p>no comparable columns</p>"
    inner_w = PLOT_WIDTH - 2 * MARGIN_X
    inner_h = PLOT_HEIGHT - 2 * MARGIN_Y
    n_axes = len(axes)
    xs = [MARGIN_X + (inner_w * i // max(1, n_axes - 1))
          for i in range(n_axes)] if n_axes > 1 else \
        [MARGIN_X + inner_w // 2]
    positions, ticks_per_axis = [], []
    for col in axes:
        pos, ticks = _axis_scale(list(df[col]))
        positions.append(pos)
        ticks_per_axis.append(ticks)
    parts = [f'<svg xmlns="http://www.w3.org/2000/svg" '
             f'width="{PLOT_WIDTH}" height="{PLOT_HEIGHT}" '
             f'font-family="sans-serif" font-size="11">']
    # axes + labels + ticks
    for i, col in enumerate(axes):
        x = xs[i]
        parts.append(f'<line x1="{x}" y1="{MARGIN_Y}" x2="{x}" '
                     f'y2="{PLOT_HEIGHT - MARGIN_Y}" stroke="#888"/>')
        label = _html.escape(col)
        parts.append(f'<text x="{x}" y="{MARGIN_Y - 14}" '
                     f'text-anchor="middle" font-weight="bold">'
                     f'{label}</text>')
        for frac, tick in ticks_per_axis[i]:
            y = PLOT_HEIGHT - MARGIN_Y - frac * inner_h
            parts.append(f'<text x="{x - 6}" y="{y + 4}" '
                         f'text-anchor="end" fill="#555">'
                         f'{_html.escape(str(tick))}</text>')
    # one polyline per run
    for row_idx in range(len(df)):
        points = []
        for axis_idx in range(n_axes):
            x = xs[axis_idx]
            y = PLOT_HEIGHT - MARGIN_Y - \
                positions[axis_idx][row_idx] * inner_h
            points.append(f"{x},{y:.1f}")
        color = _LINE_COLORS[row_idx % len(_LINE_COLORS)]
        parts.append(f'<polyline points="{" ".join(points)}" '
                     f'fill="none" stroke="{color}" stroke-width="2" '
                     f'opacity="0.75"/>')
    parts.append("</svg>")
    return "".join(parts)


def _runs_df(runs_list, extend_iterations: bool):
    df = runs_list.to_df(flat=True,
                         extend_iterations=extend_iterations,
                         cache=False)
    keep = [c for c in df.columns
            if c in ("iter", "uid", "name", "state")
            or c.startswith("param.") or c.startswith("output.")]
    return df[keep]


def _show_and_export(plot_html: str, table_html: str, show, filename):
    page = (plot_html + "<br/>" + table_html)
    if filename:
        with open(filename, "w") as f:
            f.write(page)
    if show:
        from ..render import _display

        _display(page)
    return page


def compare_run_objects(runs_list, hide_identical: bool = True,
                        exclude: list = None, show: bool = None,
                        extend_iterations: bool = True,
                        filename: str = None,
                        colorscale: str = None) -> str:
    """Parallel-coordinates plot + table comparing RunObjects
    (reference parallel_coordinates.py:240)."""
    from ..lists import RunList
    from ..model import RunObject

    if isinstance(runs_list, RunObject):
        runs_list = [runs_list]
    if isinstance(runs_list, list) and not isinstance(runs_list, RunList):
        runs_list = RunList(
            [run.to_dict() if hasattr(run, "to_dict") else run
             for run in runs_list])
    df = _runs_df(runs_list, extend_iterations)
    plot = gen_pcp_plot(df, index_col="iter",
                        hide_identical=hide_identical, exclude=exclude,
                        colorscale=colorscale)
    return _show_and_export(plot, df.to_html(index=False), show,
                            filename)


def compare_db_runs(project_name: str = None, run_name: str = None,
                    labels=None, iter: bool = False,
                    start_time_from=None, hide_identical: bool = True,
                    exclude: list = None, show=None,
                    colorscale: str = "Blues", filename: str = None,
                    **query_args) -> str:
    """Query runs from the run DB and compare them
    (reference parallel_coordinates.py:290)."""
    from ..db import get_run_db

    runs_list = get_run_db().list_runs(
        name=run_name or "", project=project_name or "",
        labels=labels, iter=iter, start_time_from=start_time_from,
        **query_args)
    df = _runs_df(runs_list, extend_iterations=iter)
    plot = gen_pcp_plot(df, index_col="iter",
                        hide_identical=hide_identical, exclude=exclude,
                        colorscale=colorscale)
    return _show_and_export(plot, df.to_html(index=False), show,
                            filename)
