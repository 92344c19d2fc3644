"""Human-readable summary report over a taxonomy table."""
from __future__ import annotations

from collections import Counter

import pandas as pd

from tosem2021_amd.analyze.taxonomy import (
    row_method, row_properties, row_stage, row_strategies)


def summary_report(df: pd.DataFrame) -> str:
    lines = [f"taxonomy rows: {len(df)}", ""]
    lines.append("rows per repo:")
    for repo, n in df["Repo"].value_counts().items():
        lines.append(f"  {repo:<16}{n:>7}")
    methods = Counter(row_method(df))
    lines.append("")
    lines.append("test methods:")
    for m, n in methods.most_common():
        lines.append(f"  {m:<16}{n:>7}  ({n/len(df)*100:.2f}%)")
    strat = Counter(s for ss in row_strategies(df) for s in ss)
    lines.append("")
    lines.append("top strategies:")
    for s, n in strat.most_common(10):
        lines.append(f"  {s:<28}{n:>7}")
    props = Counter(p for ps in row_properties(df) for p in ps)
    lines.append("")
    lines.append("top properties:")
    for p, n in props.most_common(10):
        lines.append(f"  {p:<40}{n:>7}")
    stages = Counter(row_stage(df))
    lines.append("")
    lines.append("workflow stages:")
    for s, n in stages.most_common():
        lines.append(f"  {s:<20}{n:>7}")
    return "\n".join(lines)
