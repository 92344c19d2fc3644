"""TypeScript mocha/jest test extractor (regex + brace matching).

Covers the nni manager/webui convention (21 *.test.ts under
ts/nni_manager): it('...') / test('...') cases inside (nested)
describe('...') blocks, with chai `expect(...)` / `assert.*(...)`
assertions.

Round-2 hardening (VERDICT r1 missing item 6): case bodies are the
brace-matched extent of the it()/test() callback (round 1 sliced "until the
next it(", bleeding between-case assertions into the wrong case), and the
suite name is the full nested-describe path (brace-matched describe
extents), not just the latest describe seen.
"""
from __future__ import annotations

import re
from typing import List, Optional, Tuple

from tosem2021_amd.extract.gtest_extractor import (_extract_call,
                                                   _match_brace_block)
from tosem2021_amd.extract.python_extractor import Assertion, TestCase

RE_CASE = re.compile(r"\b(it|test)\s*\(\s*(['\"`])(.+?)\2", re.S)
RE_DESCRIBE = re.compile(r"\bdescribe\s*\(\s*(['\"`])(.+?)\1", re.S)
RE_ASSERT = re.compile(
    r"\b(expect\s*\(|assert\s*\.\s*\w+\s*\(|chai\.\w+\s*\()")


def _callback_body(text: str, match_end: int,
                   hard_stop: int) -> Tuple[int, int]:
    """(start, end) of the it()/describe() callback's brace block: the first
    '{' after the title argument, unless it lies beyond hard_stop (a
    braceless arrow body like `it('x', () => expect(a).eq(b));` — fall back
    to the statement up to hard_stop)."""
    brace = text.find("{", match_end)
    if brace < 0 or brace >= hard_stop:
        return match_end, hard_stop
    return brace, _match_brace_block(text, brace)


def extract_ts_file(path: str, rel: Optional[str] = None) -> List[TestCase]:
    rel = rel or path
    try:
        with open(path, "rb") as f:
            text = f.read().decode("utf-8", errors="replace")
    except OSError:
        return []
    # nested describe extents: (start, end, name)
    describes = []
    for m in RE_DESCRIBE.finditer(text):
        _b, end = _callback_body(text, m.end(), len(text))
        describes.append((m.start(), end, m.group(2)))
    case_marks = list(RE_CASE.finditer(text))
    cases: List[TestCase] = []
    for idx, m in enumerate(case_marks):
        start, name = m.start(), m.group(3)
        next_case = case_marks[idx + 1].start() if idx + 1 < len(case_marks) \
            else len(text)
        body_start, body_end = _callback_body(text, m.end(), next_case)
        body = text[body_start:body_end]
        lineno = text.count("\n", 0, start) + 1
        # suite = path of enclosing describes, outermost first
        suite = ".".join(dname for dstart, dend, dname in describes
                         if dstart < start <= dend)
        assertions: List[Assertion] = []
        for am in RE_ASSERT.finditer(body):
            open_paren = body.find("(", am.start())
            src = body[am.start():open_paren] + _extract_call(body, open_paren)
            # chains like expect(x).to.equal(y): extend to end of statement
            tail = body[am.start() + len(src):]
            stmt_end = tail.find(";")
            if 0 <= stmt_end <= 160:
                src = src + tail[:stmt_end]
            kind = "unittest"
            if re.search(r"\.to\.throw|rejectedWith", src):
                kind = "raises"
            elif re.search(r"closeTo|approximately", src):
                kind = "approx"
            assertions.append(Assertion(
                kind=kind, call_name=src.split("(")[0].strip(),
                source=src[:500],
                lineno=lineno + body.count("\n", 0, am.start())))
        cases.append(TestCase(
            name=name, qualname=f"{suite}.{name}" if suite else name,
            file_rel=rel, lineno=lineno,
            end_lineno=lineno + body.count("\n"),
            source=body[:4000], assertions=assertions,
            uses_mock=bool(re.search(r"\bsinon|jest\.mock|\bstub|\bspy\b",
                                     body)),
        ))
    return cases
