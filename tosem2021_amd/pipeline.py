"""End-to-end mining pipeline: corpus -> extract -> classify -> taxonomy CSV
-> RQ tables (the reference package's L0->L4 data flow)."""
from __future__ import annotations

import csv
import os
from typing import List, Optional, Sequence

from tosem2021_amd.classify.rules import classify_case
from tosem2021_amd.corpus import walker
from tosem2021_amd.corpus.registry import PROJECTS, project_root
from tosem2021_amd.extract.python_extractor import extract_file
from tosem2021_amd.extract.schema import TAXONOMY_COLUMNS, TestCaseRow
from tosem2021_amd.utils.metrics import get_metrics
from tosem2021_amd.utils.trace import trace


def mine_project(key: str, corpus_root: Optional[str] = None,
                 languages: Sequence[str] = ("python",),
                 max_files: int = 0) -> List[TestCaseRow]:
    """Mine one subject project into taxonomy rows."""
    proj = PROJECTS[key]
    root = project_root(proj, corpus_root)
    if not os.path.isdir(root):
        raise FileNotFoundError(f"project snapshot not found: {root}")
    rows: List[TestCaseRow] = []
    with trace("mine_project", project=key):
        files = walker.test_files(root, languages)
        if max_files:
            files = files[:max_files]
        for file_id, f in enumerate(files):
            if f.language == "python":
                cases = extract_file(f.path, f.rel)
            elif f.language == "cpp":
                from tosem2021_amd.extract.gtest_extractor import extract_gtest_file
                cases = extract_gtest_file(f.path, f.rel)
            elif f.language == "ts":
                from tosem2021_amd.extract.ts_extractor import extract_ts_file
                cases = extract_ts_file(f.path, f.rel)
            else:
                continue
            for case in cases:
                rows.extend(classify_case(case, repo=proj.name, file_id=file_id))
    get_metrics().inc("rows_mined", len(rows), repo=proj.name)
    for i, r in enumerate(rows):
        r.index = i + 1
    return rows


def write_taxonomy_csv(rows: List[TestCaseRow], path: str) -> str:
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    with open(path, "w", newline="") as f:
        w = csv.writer(f)
        w.writerow(TAXONOMY_COLUMNS)
        for r in rows:
            w.writerow(r.to_csv_row())
    return path


def mine(projects: Sequence[str], out_csv: str,
         corpus_root: Optional[str] = None,
         languages: Sequence[str] = ("python", "cpp", "ts"),
         workers: int = 0) -> str:
    """Mine several projects into one taxonomy CSV.

    workers > 0 uses the fault-tolerant parallel pool (parallel/pool.py).
    """
    rows: List[TestCaseRow] = []
    if workers and len(projects) > 1:
        from tosem2021_amd.parallel.pool import run_tasks
        results = run_tasks(
            [(mine_project, (k, corpus_root, tuple(languages))) for k in projects],
            workers=workers)
        for rs in results:
            rows.extend(rs)
    else:
        for k in projects:
            rows.extend(mine_project(k, corpus_root, languages))
    for i, r in enumerate(rows):
        r.index = i + 1
    return write_taxonomy_csv(rows, out_csv)
