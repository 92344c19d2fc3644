"""TypeScript mocha/jest test extractor (regex-based).

Covers the nni manager/webui convention (21 *.test.ts under
ts/nni_manager): it('...') / test('...') cases inside describe('...') blocks,
with chai `expect(...)` / `assert.*(...)` assertions.
"""
from __future__ import annotations

import re
from typing import List, Optional

from tosem2021_amd.extract.python_extractor import Assertion, TestCase
from tosem2021_amd.extract.gtest_extractor import _extract_call

RE_CASE = re.compile(r"\b(it|test)\s*\(\s*(['\"`])(.+?)\2", re.S)
RE_DESCRIBE = re.compile(r"\bdescribe\s*\(\s*(['\"`])(.+?)\1", re.S)
RE_ASSERT = re.compile(
    r"\b(expect\s*\(|assert\s*\.\s*\w+\s*\(|chai\.\w+\s*\()")


def extract_ts_file(path: str, rel: Optional[str] = None) -> List[TestCase]:
    rel = rel or path
    try:
        with open(path, "rb") as f:
            text = f.read().decode("utf-8", errors="replace")
    except OSError:
        return []
    describes = [(m.start(), m.group(2)) for m in RE_DESCRIBE.finditer(text)]
    case_marks = [(m.start(), m.group(3)) for m in RE_CASE.finditer(text)]
    cases: List[TestCase] = []
    for idx, (start, name) in enumerate(case_marks):
        end = case_marks[idx + 1][0] if idx + 1 < len(case_marks) else len(text)
        body = text[start:end]
        lineno = text.count("\n", 0, start) + 1
        suite = ""
        for dstart, dname in describes:
            if dstart < start:
                suite = dname
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
