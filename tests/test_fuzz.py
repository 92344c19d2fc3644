"""Property-based fuzzing (hypothesis) over the parsing-adjacent surfaces.

The extractors are regex+brace-matching parsers over arbitrary repository
text — the classic crash/bleed surface.  These tests assert structural
invariants on randomized inputs rather than golden outputs.
"""
import string

from hypothesis import given, settings
from hypothesis import strategies as st

code_text = st.text(
    alphabet=string.ascii_letters + string.digits + " \n\t(){}[];,'\"=+-_<>.",
    max_size=400)


@settings(max_examples=80, deadline=None)
@given(code_text)
def test_brace_matcher_bounds(src):
    from tosem2021_amd.extract.gtest_extractor import _match_brace_block
    for idx, ch in enumerate(src[:50]):
        if ch == "{":
            end = _match_brace_block(src, idx)
            assert idx < end <= len(src)


@settings(max_examples=80, deadline=None)
@given(code_text)
def test_call_extractor_bounds(src):
    from tosem2021_amd.extract.gtest_extractor import _extract_call
    i = src.find("(")
    if i >= 0:
        out = _extract_call(src, i)
        assert isinstance(out, str)
        assert len(out) <= len(src) - i or len(out) <= 200


@settings(max_examples=60, deadline=None)
@given(code_text)
def test_ts_extractor_never_crashes(tmp_path_factory, src):
    from tosem2021_amd.extract.ts_extractor import extract_ts_file
    p = tmp_path_factory.mktemp("fz") / "f.test.ts"
    p.write_text("describe('s', () => {\n" + src + "\n});\n")
    cases = extract_ts_file(str(p))
    for c in cases:
        assert c.lineno >= 1
        assert c.end_lineno >= c.lineno
        for a in c.assertions:
            assert a.lineno >= c.lineno


@settings(max_examples=60, deadline=None)
@given(code_text)
def test_gtest_extractor_never_crashes(tmp_path_factory, src):
    from tosem2021_amd.extract.gtest_extractor import extract_gtest_file
    p = tmp_path_factory.mktemp("fz") / "f_test.cc"
    p.write_text("TEST(S, A) {\n" + src + "\n}\n")
    cases = extract_gtest_file(str(p))
    for c in cases:
        assert c.lineno >= 1
        assert c.param_multiplicity >= 1


@settings(max_examples=80, deadline=None)
@given(st.text(max_size=200))
def test_rules_classifier_total(src):
    """classify_text is total over arbitrary text and always yields a
    schema-valid row."""
    from tosem2021_amd.classify.rules import classify_text
    from tosem2021_amd.extract.schema import (METHODS, PROPERTIES,
                                              STRATEGIES)
    row = classify_text(src)
    assert row.method in METHODS
    for s in row.strategies():
        assert s in STRATEGIES
    for p in row.properties():
        assert p in PROPERTIES
    assert len(row.to_csv_row()) == 41


@settings(max_examples=40, deadline=None)
@given(st.text(max_size=120))
def test_tokenizer_ids_in_vocab(src):
    from tosem2021_amd.models.tokenizer import CodeTokenizer
    tok = CodeTokenizer(1024)
    ids = tok.encode(src, 64)
    assert len(ids) <= 64
    assert all(0 <= i < 1024 for i in ids)
