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


def _mini_xlsx(tmp, cells_xml, shared_xml=None):
    import zipfile
    p = str(tmp / "f.xlsx")
    with zipfile.ZipFile(p, "w") as z:
        z.writestr("[Content_Types].xml", "<Types/>")
        z.writestr(
            "xl/workbook.xml",
            '<workbook xmlns="http://schemas.openxmlformats.org/'
            'spreadsheetml/2006/main" xmlns:r="http://schemas.'
            'openxmlformats.org/officeDocument/2006/relationships">'
            '<sheets><sheet name="S" sheetId="1" r:id="rId1"/></sheets>'
            '</workbook>')
        z.writestr(
            "xl/_rels/workbook.xml.rels",
            '<Relationships xmlns="http://schemas.openxmlformats.org/'
            'package/2006/relationships"><Relationship Id="rId1" '
            'Type="x" Target="worksheets/sheet1.xml"/></Relationships>')
        if shared_xml is not None:
            z.writestr("xl/sharedStrings.xml", shared_xml)
        z.writestr(
            "xl/worksheets/sheet1.xml",
            '<worksheet xmlns="http://schemas.openxmlformats.org/'
            'spreadsheetml/2006/main"><sheetData>%s</sheetData>'
            '</worksheet>' % cells_xml)
    return p


# XML 1.0 cannot carry most control chars even escaped, and \r
# normalizes to \n per spec — keep the alphabet XML-transparent
_XML_SAFE = (string.ascii_letters + string.digits +
             " \t\n!#$%&'()*+,-./:;<=>?@[]^_`{|}~\"")


@settings(max_examples=40, deadline=None)
@given(st.text(alphabet=_XML_SAFE, max_size=60),
       st.integers(min_value=0, max_value=40))
def test_xlsx_reader_fuzz(tmp_path_factory, text, col):
    """read_xlsx is total over arbitrary cell text and sparse columns,
    and round-trips inline-string content."""
    from xml.sax.saxutils import escape

    from tosem2021_amd.utils.xlsx import read_xlsx
    tmp = tmp_path_factory.mktemp("xl")
    letters = ""
    c = col
    while True:
        letters = chr(ord("A") + c % 26) + letters
        c = c // 26 - 1
        if c < 0:
            break
    cells = ('<row r="1"><c r="%s1" t="inlineStr"><is><t>%s</t></is></c>'
             '</row>' % (letters, escape(text)))
    rows = read_xlsx(_mini_xlsx(tmp, cells))["S"]
    assert len(rows) == 1 and len(rows[0]) == col + 1
    assert rows[0][col] == text
    for k in range(col):
        assert rows[0][k] == ""


def test_xlsx_reader_hostile_inputs(tmp_path):
    """Malformed parts degrade, never crash: truncated zip, bad XML,
    out-of-range shared-string index."""
    import zipfile

    import pytest as _pytest

    from tosem2021_amd.utils.xlsx import read_xlsx
    bad = tmp_path / "bad.xlsx"
    bad.write_bytes(b"PK\x03\x04 not a zip really")
    with _pytest.raises((zipfile.BadZipFile, KeyError, OSError)):
        read_xlsx(str(bad))
    # shared-string index beyond the table
    p = _mini_xlsx(tmp_path,
                   '<row r="1"><c r="A1" t="s"><v>99</v></c></row>',
                   '<sst xmlns="http://schemas.openxmlformats.org/'
                   'spreadsheetml/2006/main"><si><t>only</t></si></sst>')
    rows = read_xlsx(p)["S"]
    assert len(rows) == 1   # degrades to empty/raw, no IndexError


@settings(max_examples=60, deadline=None)
@given(st.lists(st.tuples(
    st.sampled_from(["Apollo", "Ray", "nni", "tpot"]),
    st.floats(min_value=0.01, max_value=99.99)), min_size=0, max_size=4,
    unique_by=lambda t: t[0]))
def test_mirror_cell_encoding_roundtrip(pairs):
    """The `repo:(x%)` correlation-cell encoding written by the mirror is
    parsed back losslessly by golden_mirror's encoded-cell extractor."""
    from tosem2021_amd.analyze.golden_mirror import _encoded_cells
    cell = ",".join(f"{r}:({v:.2f}%)" for r, v in pairs)
    rows = [["strategy", "col"], ["negative_test", cell]]
    out = _encoded_cells(rows)
    assert len(out) == len(pairs)
    for r, v in pairs:
        got = out[("negative_test", 0, r)]
        assert abs(round(v, 2) - got) < 0.005
