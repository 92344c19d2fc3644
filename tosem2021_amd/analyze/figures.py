"""Figure generation replacing the reference's R plots (RQs/RQ3/Rplot*.pdf,
properties_rq3.pdf, strategy_rq3.pdf) with dependency-free SVG charts."""
from __future__ import annotations

import html
import os
from typing import Dict, Sequence

import pandas as pd

from tosem2021_amd.analyze.tables import (
    rq3_properties_by_repo, rq3_strategies_by_repo, rq4_test_methods)

_COLORS = ["#4e79a7", "#f28e2b", "#e15759", "#76b7b2", "#59a14f", "#edc948",
           "#b07aa1", "#ff9da7", "#9c755f", "#bab0ac"]


def hbar_svg(labels: Sequence[str], values: Sequence[float], title: str,
             unit: str = "%", width: int = 860) -> str:
    n = len(labels)
    row_h, pad_l, pad_t = 22, 260, 46
    height = pad_t + n * row_h + 16
    vmax = max(max(values), 1e-9)
    bar_w = width - pad_l - 120
    parts = [
        f'<svg xmlns="http://www.w3.org/2000/svg" width="{width}" '
        f'height="{height}" font-family="Helvetica,Arial,sans-serif">',
        f'<text x="12" y="24" font-size="16" font-weight="bold">'
        f'{html.escape(title)}</text>']
    for i, (lab, val) in enumerate(zip(labels, values)):
        y = pad_t + i * row_h
        w = val / vmax * bar_w
        c = _COLORS[i % len(_COLORS)]
        parts.append(
            f'<text x="{pad_l-8}" y="{y+14}" font-size="11" '
            f'text-anchor="end">{html.escape(str(lab))}</text>'
            f'<rect x="{pad_l}" y="{y+3}" width="{w:.1f}" height="{row_h-8}" '
            f'fill="{c}"/>'
            f'<text x="{pad_l+w+6:.1f}" y="{y+14}" font-size="11">'
            f'{val:.2f}{unit}</text>')
    parts.append("</svg>")
    return "".join(parts)


def grouped_svg(df: pd.DataFrame, title: str, width: int = 1100) -> str:
    """One row per index label, one colored bar per column (grouped)."""
    rows = list(df.index)
    cols = list(df.columns)
    group_h = 14 * len(cols) + 10
    pad_l, pad_t = 240, 70
    height = pad_t + len(rows) * group_h + 20
    vmax = max(float(df.to_numpy().max()), 1e-9)
    bar_w = width - pad_l - 120
    parts = [
        f'<svg xmlns="http://www.w3.org/2000/svg" width="{width}" '
        f'height="{height}" font-family="Helvetica,Arial,sans-serif">',
        f'<text x="12" y="24" font-size="16" font-weight="bold">'
        f'{html.escape(title)}</text>']
    for j, c in enumerate(cols):
        parts.append(
            f'<rect x="{12+j*110}" y="34" width="10" height="10" '
            f'fill="{_COLORS[j % len(_COLORS)]}"/>'
            f'<text x="{26+j*110}" y="43" font-size="10">{html.escape(str(c))}'
            f'</text>')
    for i, r in enumerate(rows):
        y0 = pad_t + i * group_h
        parts.append(f'<text x="{pad_l-8}" y="{y0+group_h/2}" font-size="11" '
                     f'text-anchor="end">{html.escape(str(r))}</text>')
        for j, c in enumerate(cols):
            v = float(df.loc[r, c])
            w = v / vmax * bar_w
            parts.append(
                f'<rect x="{pad_l}" y="{y0 + j*14}" width="{w:.1f}" '
                f'height="10" fill="{_COLORS[j % len(_COLORS)]}"/>')
    parts.append("</svg>")
    return "".join(parts)


def write_figures(df: pd.DataFrame, out_dir: str) -> Dict[str, str]:
    os.makedirs(os.path.join(out_dir, "RQ3"), exist_ok=True)
    os.makedirs(os.path.join(out_dir, "RQ4"), exist_ok=True)
    paths: Dict[str, str] = {}

    props = rq3_properties_by_repo(df)
    p = os.path.join(out_dir, "RQ3", "properties_rq3.svg")
    with open(p, "w") as f:
        f.write(grouped_svg(props, "Tested quality properties per project (%)"))
    paths["fig_properties"] = p

    strat = rq3_strategies_by_repo(df)
    p = os.path.join(out_dir, "RQ3", "strategy_rq3.svg")
    with open(p, "w") as f:
        f.write(grouped_svg(strat, "Test strategies per project (%)"))
    paths["fig_strategies"] = p

    m = rq4_test_methods(df)
    p = os.path.join(out_dir, "RQ4", "tests_methods.svg")
    with open(p, "w") as f:
        f.write(hbar_svg(m["Test_methods"].tolist(),
                         m["percentage"].tolist(),
                         "Test methods (% of labeled rows)"))
    paths["fig_methods"] = p
    return paths
