"""Project-selection funnel (the reference's L1 layer).

Mirrors selection/Reposition/: GitHub-metadata candidate tables refined in
rounds (Repos_metrics_v3.csv 312 -> v2 226 -> v4 28 -> Repos_l.csv 14 -> 9
studied).  The reference ships only the round OUTPUTS; this module provides
the funnel machinery — load a metadata table (17-column schema at
Repos_metrics_v3.csv:1), apply configurable screening criteria, and emit the
next round's table — plus the paper-style defaults.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

import pandas as pd

METRICS_COLUMNS = [
    "Repos", "topic", "commits", "contributors", "issues", "pulls",
    "releases", "size", "stars", "forks", "open_issues", "archived",
    "created_at", "updated_at", "language",
    "homepage", "description",
]

NUMERIC_COLUMNS = ["commits", "contributors", "issues", "pulls", "releases",
                   "size", "stars", "forks", "open_issues"]


@dataclass
class FunnelCriteria:
    """One screening round (all thresholds inclusive minimums)."""
    min_stars: int = 0
    min_commits: int = 0
    min_contributors: int = 0
    min_releases: int = 0
    exclude_archived: bool = True
    languages: Optional[List[str]] = None     # keep these languages only
    require_description: bool = False
    name: str = "round"


def load_metrics(path: str) -> pd.DataFrame:
    df = pd.read_csv(path, encoding="utf-8-sig")
    # reference sheets carry trailing unnamed columns — drop them
    df = df[[c for c in df.columns if not str(c).startswith("Unnamed")]]
    missing = [c for c in METRICS_COLUMNS[:15] if c not in df.columns]
    if missing:
        raise ValueError(f"metrics CSV missing columns: {missing}")
    for c in NUMERIC_COLUMNS:
        df[c] = pd.to_numeric(df[c], errors="coerce")
    return df


def apply_criteria(df: pd.DataFrame, c: FunnelCriteria) -> pd.DataFrame:
    keep = pd.Series(True, index=df.index)
    keep &= df["stars"].fillna(0) >= c.min_stars
    keep &= df["commits"].fillna(0) >= c.min_commits
    keep &= df["contributors"].fillna(0) >= c.min_contributors
    keep &= df["releases"].fillna(0) >= c.min_releases
    if c.exclude_archived:
        arch = df["archived"].astype(str).str.lower().isin(("true", "1"))
        keep &= ~arch
    if c.languages:
        keep &= df["language"].astype(str).isin(c.languages)
    if c.require_description:
        keep &= df["description"].astype(str).str.len() > 0
    return df[keep].reset_index(drop=True)


def run_funnel(df: pd.DataFrame, rounds: List[FunnelCriteria]
               ) -> List[pd.DataFrame]:
    """Apply rounds successively; returns each round's surviving table."""
    out = []
    cur = df
    for c in rounds:
        cur = apply_criteria(cur, c)
        out.append(cur)
    return out


# Defaults approximating the paper's screening narrative (engineered ML
# systems: active, multi-contributor, released, unarchived).
DEFAULT_ROUNDS = [
    FunnelCriteria(name="screen", exclude_archived=True,
                   require_description=True),
    FunnelCriteria(name="activity", min_commits=1000, min_contributors=20,
                   min_stars=1000, min_releases=1),
]
