"""Python test-case extractor (ast-based).

Lifts individual test cases and their assertions out of pytest/unittest
suites, producing rows shaped like the master taxonomy
(RQs/taxonomy_test2.csv — one row per test case plus one per assertion,
as the study's labelers did for e.g. auto-sklearn: taxonomy_test2.csv:2-5).
"""
from __future__ import annotations

import ast
from dataclasses import dataclass, field
from typing import List, Optional


@dataclass
class Assertion:
    """One oracle site inside a test: an assert / self.assert* / pytest.raises."""
    kind: str            # "assert" | "unittest" | "raises" | "warns" | "approx"
    call_name: str       # e.g. assertAlmostEqual, assertEqual, "", raises
    source: str          # source text of the assertion expression
    lineno: int
    exception: str = ""  # for raises: the exception name


@dataclass
class TestCase:
    name: str
    qualname: str        # Class.method or function name
    file_rel: str
    lineno: int
    end_lineno: int
    source: str          # full body source
    docstring: str = ""
    assertions: List[Assertion] = field(default_factory=list)
    decorators: List[str] = field(default_factory=list)
    markers: List[str] = field(default_factory=list)    # pytest marks
    uses_mock: bool = False
    is_parametrized: bool = False
    # gtest TEST_P/TYPED_TEST_P: number of INSTANTIATE_* macros naming this
    # suite (each runs the whole suite once); 1 for plain cases
    param_multiplicity: int = 1


UNITTEST_ASSERT_PREFIX = "assert"
RAISE_HELPERS = {"assertRaises", "assertRaisesRegex", "assertRaisesRegexp",
                 "raises", "assertWarns", "warns"}
MOCK_TOKENS = {"mock", "Mock", "MagicMock", "patch", "monkeypatch", "mocker"}


def _name_of(node: ast.AST) -> str:
    if isinstance(node, ast.Name):
        return node.id
    if isinstance(node, ast.Attribute):
        return node.attr
    if isinstance(node, ast.Call):
        return _name_of(node.func)
    return ""


def _dotted(node: ast.AST) -> str:
    if isinstance(node, ast.Attribute):
        base = _dotted(node.value)
        return f"{base}.{node.attr}" if base else node.attr
    if isinstance(node, ast.Name):
        return node.id
    return ""


class _AssertVisitor(ast.NodeVisitor):
    def __init__(self, src_lines: List[str]):
        self.src_lines = src_lines
        self.assertions: List[Assertion] = []
        self.uses_mock = False

    def _src(self, node: ast.AST) -> str:
        try:
            seg = ast.get_source_segment("\n".join(self.src_lines), node)
            return (seg or "").strip()
        except Exception:
            return ""

    def visit_Assert(self, node: ast.Assert):
        self.assertions.append(Assertion(
            kind="assert", call_name="assert", source=self._src(node),
            lineno=node.lineno))
        self.generic_visit(node)

    def visit_Call(self, node: ast.Call):
        name = _name_of(node.func)
        if name in MOCK_TOKENS or any(
                t in _dotted(node.func) for t in ("mock.", "patch")):
            self.uses_mock = True
        if name in RAISE_HELPERS:
            exc = ""
            if node.args:
                exc = _dotted(node.args[0]) or _name_of(node.args[0])
            self.assertions.append(Assertion(
                kind="raises" if "aise" in name or name == "raises" else "warns",
                call_name=name, source=self._src(node), lineno=node.lineno,
                exception=exc))
        elif name.startswith(UNITTEST_ASSERT_PREFIX) and name != "assert_":
            self.assertions.append(Assertion(
                kind="unittest", call_name=name, source=self._src(node),
                lineno=node.lineno))
        elif name in ("assert_allclose", "assert_array_equal",
                      "assert_array_almost_equal", "assert_almost_equal",
                      "assert_equal", "assert_frame_equal", "approx"):
            self.assertions.append(Assertion(
                kind="approx" if ("close" in name or "almost" in name
                                  or name == "approx") else "unittest",
                call_name=name, source=self._src(node), lineno=node.lineno))
        self.generic_visit(node)

    def visit_With(self, node: ast.With):
        for item in node.items:
            if isinstance(item.context_expr, ast.Call):
                name = _name_of(item.context_expr.func)
                if name in RAISE_HELPERS:
                    exc = ""
                    if item.context_expr.args:
                        exc = _dotted(item.context_expr.args[0])
                    self.assertions.append(Assertion(
                        kind="raises", call_name=name,
                        source=self._src(item.context_expr),
                        lineno=node.lineno, exception=exc))
        self.generic_visit(node)


def _is_test_func(name: str, in_test_class: bool) -> bool:
    return name.startswith("test") or (in_test_class and name.startswith("check"))


def _decorator_names(node) -> List[str]:
    return [_dotted(d) or _name_of(d) for d in node.decorator_list]


def extract_file(path: str, rel: Optional[str] = None) -> List[TestCase]:
    """Parse one Python file and return its test cases with assertions."""
    rel = rel or path
    try:
        with open(path, "rb") as f:
            raw = f.read()
        text = raw.decode("utf-8", errors="replace")
        tree = ast.parse(text)
    except (SyntaxError, ValueError):
        return []  # py2-only files in the corpus (e.g. nupic) — skip cleanly
    lines = text.splitlines()
    cases: List[TestCase] = []

    # one-level assertion-helper expansion (mirrors the gtest extractor's
    # local-macro handling): a non-test function/method in this file whose
    # body asserts (e.g. DeepSpeech tests/test_value_range.py
    # `_ending_tester` -> assertEqual) makes each of its call sites inside
    # a test case count as that helper's assertions.
    helper_asserts = {}

    def collect_helpers(body):
        for node in body:
            if isinstance(node, ast.ClassDef):
                collect_helpers(node.body)
            elif isinstance(node, (ast.FunctionDef, ast.AsyncFunctionDef)):
                if _is_test_func(node.name, True):
                    continue
                hv = _AssertVisitor(lines)
                hv.visit(node)
                if hv.assertions:
                    helper_asserts[node.name] = hv.assertions

    collect_helpers(tree.body)

    def helper_calls(node):
        out = []
        for sub in ast.walk(node):
            if isinstance(sub, ast.Call):
                fn = sub.func
                name = fn.attr if isinstance(fn, ast.Attribute) else \
                    (fn.id if isinstance(fn, ast.Name) else "")
                if name in helper_asserts:
                    for a in helper_asserts[name]:
                        out.append(Assertion(
                            kind=a.kind, call_name=a.call_name,
                            source=a.source, lineno=sub.lineno,
                            exception=a.exception))
        return out

    def visit_body(body, class_name: str, in_test_class: bool):
        for node in body:
            if isinstance(node, ast.ClassDef):
                is_tc = ("Test" in node.name or any(
                    "TestCase" in _dotted(b) for b in node.bases))
                visit_body(node.body, node.name, is_tc)
            elif isinstance(node, (ast.FunctionDef, ast.AsyncFunctionDef)):
                if not _is_test_func(node.name, in_test_class):
                    continue
                v = _AssertVisitor(lines)
                v.visit(node)
                v.assertions.extend(helper_calls(node))
                v.assertions.sort(key=lambda a: a.lineno)
                decs = _decorator_names(node)
                seg = ast.get_source_segment(text, node) or ""
                cases.append(TestCase(
                    name=node.name,
                    qualname=f"{class_name}.{node.name}" if class_name else node.name,
                    file_rel=rel,
                    lineno=node.lineno,
                    end_lineno=getattr(node, "end_lineno", node.lineno),
                    source=seg,
                    docstring=ast.get_docstring(node) or "",
                    assertions=v.assertions,
                    decorators=decs,
                    markers=[d.split(".")[-1] for d in decs
                             if "mark" in d or "parametrize" in d],
                    uses_mock=v.uses_mock,
                    is_parametrized=any("parametrize" in d for d in decs),
                ))

    visit_body(tree.body, "", False)
    return cases
