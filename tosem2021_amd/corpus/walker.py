"""Language-aware corpus walker + test-file detector.

Detection rules follow the conventions actually observed per subject
(SURVEY.md §4): pytest/unittest `test_*.py` / `*_test.py` / `*_tests.py`,
gtest `*_test.cc|cpp`, mocha/jest `*.test.ts`.
"""
from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Iterator, List, Optional, Sequence

PY_EXT = (".py",)
CPP_EXT = (".cc", ".cpp", ".cxx")
# headers are walked too, but is_test_file admits only tester headers in
# test dirs (openfst's include/fst/test/{algo_test,fst_test,weight-tester}.h
# back the study's biggest DeepSpeech components)
CPP_HDR_EXT = (".h", ".hh", ".hpp")
TS_EXT = (".ts", ".tsx")

# NOTE: third_party is NOT skipped — the study's taxonomy labels
# DeepSpeech's vendored openfst suite (components "FST Algo Test",
# "WeightTester", ... ~450 rows under native_client/ctcdecode/
# third_party/openfst-*/src/test/); DeepSpeech is the only corpus
# project with test files under a third_party dir, so including it is
# scope-neutral for the other eight.
SKIP_DIRS = {".git", "node_modules", "__pycache__", "build",
             "dist", ".tox", "external"}


@dataclass(frozen=True)
class SourceFile:
    path: str        # absolute
    rel: str         # relative to project root
    language: str    # python | cpp | ts
    is_test: bool


def classify_language(path: str) -> Optional[str]:
    if path.endswith(PY_EXT):
        return "python"
    if path.endswith(CPP_EXT) or path.endswith(CPP_HDR_EXT):
        return "cpp"
    if path.endswith(TS_EXT):
        return "ts"
    return None


def is_test_file(rel: str, language: str) -> bool:
    base = os.path.basename(rel)
    parts = rel.replace("\\", "/").split("/")
    in_test_dir = any(p in ("test", "tests", "testing", "unit_tests",
                            "integration_tests", "ut") for p in parts[:-1])
    if language == "python":
        name_hit = (base.startswith("test_") or base.endswith("_test.py")
                    or base.endswith("_tests.py") or base == "tests.py")
        return name_hit or (in_test_dir and base.endswith(".py")
                            and not base.startswith("__"))
    if language == "cpp":
        stem = base.rsplit(".", 1)[0]
        if base.endswith(CPP_HDR_EXT):
            # tester headers only, and only inside a test dir
            return in_test_dir and (stem.endswith("_test") or
                                    stem.endswith("-tester") or
                                    stem.endswith("_tester"))
        return stem.endswith("_test") or stem.endswith("_unittest") or \
            stem.startswith("test_") or in_test_dir
    if language == "ts":
        return ".test." in base or ".spec." in base
    return False


def walk(root: str, languages: Sequence[str] = ("python", "cpp", "ts"),
         tests_only: bool = False) -> Iterator[SourceFile]:
    root = os.path.abspath(root)
    for dirpath, dirnames, filenames in os.walk(root):
        dirnames[:] = sorted(d for d in dirnames if d not in SKIP_DIRS)
        for fn in sorted(filenames):
            path = os.path.join(dirpath, fn)
            lang = classify_language(fn)
            if lang is None or lang not in languages:
                continue
            rel = os.path.relpath(path, root)
            test = is_test_file(rel, lang)
            if tests_only and not test:
                continue
            yield SourceFile(path=path, rel=rel, language=lang, is_test=test)


def test_files(root: str, languages: Sequence[str] = ("python", "cpp", "ts")
               ) -> List[SourceFile]:
    return list(walk(root, languages, tests_only=True))
