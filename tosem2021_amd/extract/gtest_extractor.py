"""C++ gtest extractor (regex + brace matching; no libclang in this image).

Covers the corpus' gtest conventions at Apollo/Ray/DeepSpeech scale
(665 apollo *_test.cc, 56 ray, 28 DeepSpeech):
TEST / TEST_F / TEST_P / TYPED_TEST / TYPED_TEST_P macros and the
EXPECT_* / ASSERT_* assertion families.
"""
from __future__ import annotations

import re
from typing import List, Optional

from tosem2021_amd.extract.python_extractor import Assertion, TestCase

RE_TEST_MACRO = re.compile(
    r"\b(TEST|TEST_F|TEST_P|TYPED_TEST|TYPED_TEST_P)"
    r"\s*\(\s*([A-Za-z_]\w*)\s*,\s*([A-Za-z_]\w*)\s*\)")
# INSTANTIATE_TEST_SUITE_P(Prefix, Suite, ...) and the pre-1.10 _CASE_P
# spelling, plus the typed variants: each one names a suite whose TEST_P /
# TYPED_TEST_P cases it multiplies (arg 2 = the suite).
RE_INSTANTIATE = re.compile(
    r"\bINSTANTIATE_(?:TYPED_)?TEST_(?:SUITE|CASE)_P"
    r"\s*\(\s*([A-Za-z_]\w*)\s*,\s*([A-Za-z_]\w*)")
# boost.test (kenlm, openfst in the DeepSpeech snapshot): one-arg case macro
RE_BOOST_CASE = re.compile(
    r"\b(BOOST_AUTO_TEST_CASE|BOOST_FIXTURE_TEST_CASE|BOOST_AUTO_TEST_CASE_TEMPLATE)"
    r"\s*\(\s*([A-Za-z_]\w*)")
RE_ASSERT = re.compile(
    r"\b((?:EXPECT|ASSERT)_[A-Z_0-9]+|BOOST_(?:CHECK|REQUIRE|WARN)(?:_[A-Z_0-9]+)?)\s*\(")
RE_DEATH = re.compile(r"DEATH|THROW", re.I)
# project-local assertion wrappers, e.g. kenlm's
#   #define SLOPPY_CHECK_CLOSE(ref, value, tol) BOOST_CHECK_CLOSE(...)
# (reference src/DeepSpeech/v0.9.3/native_client/kenlm/lm/model_test.cc:11):
# one-level expansion — a #define in the SAME file whose body contains a
# known assertion macro makes its invocations count as that assertion.
# glog/openfst-style CHECK assertions for FREESTANDING test files (no
# TEST/BOOST macro; e.g. the vendored openfst suite the study labeled as
# DeepSpeech components "FST Algo Test" / "WeightTester" — reference
# src/DeepSpeech/v0.9.3/native_client/ctcdecode/third_party/openfst-*/
# src/test/*.cc).  Only used on files where no framework macro matched,
# so gtest files keep their EXPECT/ASSERT-only accounting.
RE_CHECK = re.compile(r"\b((?:D|Q)?CHECK(?:_[A-Z_0-9]+)?)\s*\(")
# a top-level C++ function definition heading a freestanding test body
RE_FUNC_DEF = re.compile(
    r"^[ \t]*(?:template\s*<[^>]*>\s*)?"
    r"(?:[A-Za-z_][\w:<>,\s\*&]*?[\s\*&])"
    r"([A-Za-z_]\w*)\s*\([^;{)]*\)\s*(?:const\s*)?\{", re.M)

RE_LOCAL_MACRO = re.compile(
    r"^[ \t]*#[ \t]*define[ \t]+([A-Za-z_]\w*)[ \t]*\(",
    re.M)


def _match_brace_block(text: str, open_idx: int) -> int:
    """Index just past the matching '}' for the '{' at open_idx (best-effort:
    brace counting skipping string/char literals and comments)."""
    depth = 0
    i = open_idx
    n = len(text)
    while i < n:
        c = text[i]
        if c == '"' or c == "'":
            q = c
            i += 1
            while i < n and text[i] != q:
                if text[i] == "\\":
                    i += 1
                i += 1
        elif c == "/" and i + 1 < n and text[i + 1] == "/":
            i = text.find("\n", i)
            if i < 0:
                return n
        elif c == "/" and i + 1 < n and text[i + 1] == "*":
            i = text.find("*/", i)
            if i < 0:
                return n
            i += 1
        elif c == "{":
            depth += 1
        elif c == "}":
            depth -= 1
            if depth == 0:
                return i + 1
        i += 1
    return n


def _extract_call(text: str, start: int) -> str:
    """Source of a call starting at `start` through its closing paren."""
    depth = 0
    i = start
    n = len(text)
    while i < n:
        c = text[i]
        if c == '"':
            i += 1
            while i < n and text[i] != '"':
                if text[i] == "\\":
                    i += 1
                i += 1
        elif c == "(":
            depth += 1
        elif c == ")":
            depth -= 1
            if depth == 0:
                return text[start:i + 1]
        i += 1
    return text[start:min(start + 200, n)]


def _split_args(s: str) -> List[str]:
    """Split a C++ argument list at top-level commas (tracks (), {}, [],
    strings, and template <> with <</>>/comparison guards)."""
    out, depth, ang, i, last, n = [], 0, 0, 0, 0, len(s)
    while i < n:
        c = s[i]
        if c == '"':
            i += 1
            while i < n and s[i] != '"':
                i += 2 if s[i] == "\\" else 1
        elif c in "([{":
            depth += 1
        elif c in ")]}":
            depth -= 1
        elif c == "<":
            if i + 1 < n and s[i + 1] in "<=":
                i += 1
            else:
                ang += 1
        elif c == ">":
            if i + 1 < n and s[i + 1] in ">=":
                i += 1
            elif ang > 0 and not (i > 0 and s[i - 1] == "-"):
                ang -= 1
        elif c == "," and depth == 0 and ang == 0:
            out.append(s[last:i].strip())
            last = i + 1
        i += 1
    tail = s[last:].strip()
    if tail or out:
        out.append(tail)
    return out


_RE_GEN_HEAD = re.compile(
    r"^(?:::)?(?:testing::|::testing::)?(Values|ValuesIn|Bool|Range|"
    r"Combine|ConvertGenerator)\s*\(")


def _gen_cardinality(expr: str) -> Optional[int]:
    """Statically count the test cases an INSTANTIATE_* generator expands
    to: Values(...) = argc, Bool() = 2, Range(b, e[, s]) from int literals,
    Combine(...) = product.  None when unknowable (ValuesIn over a runtime
    container, function calls, ...)."""
    expr = expr.strip()
    m = _RE_GEN_HEAD.match(expr)
    if not m:
        return None
    head = m.group(1)
    inner = _extract_call(expr, m.end() - 1)
    args = _split_args(inner[1:-1]) if len(inner) >= 2 else []
    if head == "Bool":
        return 2
    if head == "Values":
        return len(args) or None
    if head == "Range":
        try:
            b, e = int(args[0], 0), int(args[1], 0)
            step = int(args[2], 0) if len(args) > 2 else 1
            return max((e - b + step - 1) // step, 0) or None
        except (ValueError, IndexError):
            return None
    if head == "Combine":
        total = 1
        for a in args:
            c = _gen_cardinality(a)
            if c is None:
                return None
            total *= c
        return total
    if head == "ConvertGenerator":
        return _gen_cardinality(args[0]) if args else None
    return None


def extract_gtest_file(path: str, rel: Optional[str] = None) -> List[TestCase]:
    rel = rel or path
    try:
        with open(path, "rb") as f:
            text = f.read().decode("utf-8", errors="replace")
    except OSError:
        return []
    cases: List[TestCase] = []
    # one-level local macro table: wrapper name -> underlying assertion macro
    local_asserts = {}
    for dm in RE_LOCAL_MACRO.finditer(text):
        mname = dm.group(1)
        # macro body: to end of line, following backslash continuations
        i = dm.end()
        body_lines = []
        while True:
            j = text.find("\n", i)
            if j < 0:
                j = len(text)
            line = text[i:j]
            body_lines.append(line)
            if line.rstrip().endswith("\\"):
                i = j + 1
            else:
                break
        mbody = " ".join(body_lines)
        am = RE_ASSERT.search(mbody)
        if am and mname not in ("EXPECT", "ASSERT"):
            local_asserts[mname] = am.group(1)
    re_local = (re.compile(r"\b(" + "|".join(map(re.escape, local_asserts))
                           + r")\s*\(")
                if local_asserts else None)
    # per-suite instantiation counts (TEST_P multiplicity, VERDICT r1
    # missing item 6): each INSTANTIATE_* runs the whole suite once more
    inst_counts: dict = {}
    for im in RE_INSTANTIATE.finditer(text):
        # static value-list cardinality (docs/ROADMAP.md completeness
        # item): Values/Bool/Range/Combine counted, else 1 per
        # instantiation (conservative)
        card = None
        lp = text.find("(", im.start())
        if lp >= 0:
            call_args = _split_args(_extract_call(text, lp)[1:-1])
            if len(call_args) >= 3:
                card = _gen_cardinality(call_args[2])
        inst_counts[im.group(2)] = \
            inst_counts.get(im.group(2), 0) + (card or 1)
    marks = []
    for m in RE_TEST_MACRO.finditer(text):
        marks.append((m, m.group(2), m.group(3)))
    for m in RE_BOOST_CASE.finditer(text):
        marks.append((m, "", m.group(2)))
    # one-level assertion-helper expansion (same idea as local macros):
    # a file-local function whose body asserts (kenlm model_test.cc
    # StartTest -> BOOST_CHECK/SLOPPY_CHECK_CLOSE) makes its call sites
    # inside a case count as that helper's assertions.
    helper_asserts = {}
    helper_bodies = {}
    case_names = {name for _, _, name in marks}
    for fm in RE_FUNC_DEF.finditer(text):
        fname = fm.group(1)
        if fname in case_names or fname in ("main", "if", "for", "while",
                                            "switch"):
            continue
        fb = text.rfind("{", fm.start(), fm.end())
        fbody = text[fm.start():_match_brace_block(text, fb)]
        found = [(am.group(1), am.end() - 1) for am in
                 RE_ASSERT.finditer(fbody)]
        if re_local is not None:
            found += [(local_asserts[am.group(1)], am.end() - 1) for am in
                      re_local.finditer(fbody)]
        helper_asserts[fname] = [
            (call, (call + _extract_call(fbody, pos))[:500])
            for call, pos in found]
        helper_bodies[fname] = fbody
    # propagate through helper->helper calls (kenlm: case -> Everything ->
    # Starters -> BOOST_CHECK), bounded depth, no self-recursion blowup
    for _ in range(3):
        changed = False
        for fname, fbody in helper_bodies.items():
            inherited = []
            for other, asserts in helper_asserts.items():
                if other != fname and asserts and \
                        re.search(r"\b" + re.escape(other) +
                                  r"\s*(?:<[^;{}()]*>)?\s*\(", fbody):
                    inherited.extend(asserts)
            merged = helper_asserts.get(fname, []) + [
                a for a in inherited
                if a not in helper_asserts.get(fname, [])]
            if len(merged) > len(helper_asserts.get(fname, [])) and \
                    len(merged) <= 200:
                helper_asserts[fname] = merged
                changed = True
        if not changed:
            break
    helper_asserts = {k: v for k, v in helper_asserts.items() if v}
    re_helper = (re.compile(
        r"\b(" + "|".join(map(re.escape, sorted(helper_asserts))) +
        r")\s*(?:<[^;{}()]*>)?\s*\(") if helper_asserts else None)
    for m, suite, name in sorted(marks, key=lambda t: t[0].start()):
        brace = text.find("{", m.end())
        if brace < 0:
            continue
        end = _match_brace_block(text, brace)
        body = text[m.start():end]
        lineno = text.count("\n", 0, m.start()) + 1
        assertions: List[Assertion] = []
        for am in RE_ASSERT.finditer(body):
            call = am.group(1)
            src = call + _extract_call(body, am.end() - 1)
            a_line = lineno + body.count("\n", 0, am.start())
            kind = "raises" if RE_DEATH.search(call) else "unittest"
            exc = ""
            if kind == "raises":
                inner = src[src.find("(") + 1:]
                parts = inner.rsplit(",", 1)
                exc = parts[-1].strip(" );") if len(parts) > 1 else ""
            if "NEAR" in call or "FLOAT_EQ" in call or "DOUBLE_EQ" in call \
                or "CLOSE" in call:
                kind = "approx"
            assertions.append(Assertion(
                kind=kind, call_name=call, source=src[:500], lineno=a_line,
                exception=exc))
        if re_local is not None:
            for am in re_local.finditer(body):
                wrapper = am.group(1)
                call = local_asserts[wrapper]
                src = wrapper + _extract_call(body, am.end() - 1)
                a_line = lineno + body.count("\n", 0, am.start())
                kind = "raises" if RE_DEATH.search(call) else "unittest"
                if "NEAR" in call or "FLOAT_EQ" in call \
                        or "DOUBLE_EQ" in call or "CLOSE" in call:
                    kind = "approx"
                assertions.append(Assertion(
                    kind=kind, call_name=call, source=src[:500],
                    lineno=a_line, exception=""))
            assertions.sort(key=lambda a: a.lineno)
        if re_helper is not None:
            body_off = text.find("{", m.end())
            inner = text[body_off:end] if body_off >= 0 else ""
            for hm in re_helper.finditer(inner):
                a_line = lineno + body[:len(body)].count(
                    "\n", 0, body_off - m.start() + hm.start())
                for call, src in helper_asserts[hm.group(1)]:
                    kind = "raises" if RE_DEATH.search(call) else "unittest"
                    if "NEAR" in call or "FLOAT_EQ" in call \
                            or "DOUBLE_EQ" in call or "CLOSE" in call:
                        kind = "approx"
                    assertions.append(Assertion(
                        kind=kind, call_name=call, source=src,
                        lineno=a_line, exception=""))
            assertions.sort(key=lambda a: a.lineno)
        macro = m.group(1) if m.re is RE_TEST_MACRO else ""
        parametrized = macro in ("TEST_P", "TYPED_TEST_P")
        cases.append(TestCase(
            name=name,
            qualname=f"{suite}.{name}",
            file_rel=rel,
            lineno=lineno,
            end_lineno=lineno + body.count("\n"),
            source=body[:4000],
            assertions=assertions,
            uses_mock=bool(re.search(r"\bMOCK_METHOD|NiceMock|StrictMock|gmock",
                                     body)),
            is_parametrized=parametrized,
            param_multiplicity=max(inst_counts.get(suite, 0), 1)
            if parametrized else 1,
        ))
    if not cases:
        cases = _extract_freestanding(text, rel)
    return cases


def _extract_freestanding(text: str, rel: str) -> List[TestCase]:
    """Fallback for hand-rolled C++ test files with no framework macro:
    one case per top-level function containing CHECK-family assertions
    (glog/openfst style; per-function granularity matches how the study
    labeled those suites as components like 'WeightTester')."""
    cases: List[TestCase] = []
    for m in RE_FUNC_DEF.finditer(text):
        brace = text.rfind("{", m.start(), m.end())
        end = _match_brace_block(text, brace)
        body = text[m.start():end]
        lineno = text.count("\n", 0, m.start()) + 1
        assertions: List[Assertion] = []
        for am in RE_CHECK.finditer(body):
            call = am.group(1)
            src = call + _extract_call(body, am.end() - 1)
            a_line = lineno + body.count("\n", 0, am.start())
            kind = "approx" if ("NEAR" in call or "CLOSE" in call) \
                else "unittest"
            assertions.append(Assertion(
                kind=kind, call_name=call, source=src[:500], lineno=a_line,
                exception=""))
        if not assertions:
            continue
        cases.append(TestCase(
            name=m.group(1),
            qualname=m.group(1),
            file_rel=rel,
            lineno=lineno,
            end_lineno=lineno + body.count("\n"),
            source=body[:4000],
            assertions=assertions,
            uses_mock=False,
            is_parametrized=False,
            param_multiplicity=1,
        ))
    return cases
