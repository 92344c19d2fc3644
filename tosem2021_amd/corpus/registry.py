"""Project registry: the nine pinned subject systems of the study.

Mirrors the reference corpus layout (/root/reference/src — SURVEY.md §2.3);
the corpus root is configurable so the pipeline mines any checkout.
"""
from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Dict, List, Optional


@dataclass(frozen=True)
class Project:
    name: str                 # taxonomy Repo spelling
    src_subdir: str           # path under the corpus root
    version: str
    languages: tuple          # of "python" | "cpp" | "ts"
    test_globs: tuple = ()    # project-specific test-dir hints


# versions per SURVEY.md §1 L0 / reference src/ layout
PROJECTS: Dict[str, Project] = {
    "apollo": Project("Apollo", "apollo/v6.0.0", "v6.0.0", ("cpp", "python")),
    "ray": Project("Ray", "ray/ray-1.1.0", "1.1.0", ("python", "cpp")),
    "DeepSpeech": Project("DeepSpeech2", "DeepSpeech/v0.9.3", "v0.9.3",
                          ("python", "cpp")),
    "nni": Project("nni", "nni/v2.0", "v2.0", ("python", "ts")),
    "nupic": Project("Nupic", "nupic/1.0.5", "1.0.5", ("python",)),
    "auto-sklearn": Project("auto_sklearn", "auto-sklearn/v0.12.0", "v0.12.0",
                            ("python",)),
    "autokeras": Project("autokeras", "autokeras/1.0.12", "1.0.12", ("python",)),
    "automl": Project("google_automl", "automl/1.1", "1.1", ("python",)),
    "tpot": Project("tpot", "tpot/v0.11.7", "v0.11.7", ("python",)),
}

DEFAULT_CORPUS_ROOT = "/root/reference/src"


def project_root(project: Project, corpus_root: Optional[str] = None) -> str:
    root = corpus_root or DEFAULT_CORPUS_ROOT
    return os.path.join(root, project.src_subdir)


def available_projects(corpus_root: Optional[str] = None) -> List[str]:
    return [k for k, p in PROJECTS.items()
            if os.path.isdir(project_root(p, corpus_root))]
