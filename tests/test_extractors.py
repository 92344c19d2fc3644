"""Unit tests: extractors on synthetic sources (SURVEY.md §4 test model —
unit layer: parsers/classifiers on synthetic test files)."""
import os
import textwrap

from tosem2021_amd.corpus import walker
from tosem2021_amd.extract.gtest_extractor import extract_gtest_file
from tosem2021_amd.extract.python_extractor import extract_file
from tosem2021_amd.extract.ts_extractor import extract_ts_file

PY_SRC = textwrap.dedent('''
    import pytest
    from unittest import mock

    class TestThing:
        def test_accuracy(self):
            """checks model accuracy"""
            score = 0.97
            self.assertAlmostEqual(0.96, score, places=2)
            assert score > 0.8

        def test_raises(self):
            with pytest.raises(ValueError):
                parse("bad")

    @pytest.mark.parametrize("x", [1, 2])
    def test_mocked(x, monkeypatch):
        m = mock.MagicMock()
        assert isinstance(m, object)

    def helper():
        assert False  # not a test
''')

CC_SRC = textwrap.dedent('''
    #include "gtest/gtest.h"
    TEST(MathTest, HandlesZero) {
      EXPECT_EQ(0, Add(0, 0));
      EXPECT_NEAR(1.0, Compute(), 1e-6);
    }
    TEST_F(PipelineFixture, Throws) {
      EXPECT_THROW(Run("bad"), std::invalid_argument);
      if (x) { EXPECT_TRUE(ok()); }
    }
''')

TS_SRC = textwrap.dedent('''
    describe('manager', () => {
        it('starts an experiment', async () => {
            const r = await mgr.start();
            expect(r.status).to.equal('RUNNING');
            assert.strictEqual(r.id, 1);
        });
        it('rejects bad config', () => {
            expect(() => mgr.load('x')).to.throw();
        });
    });
''')


def _write(tmp_path, name, content):
    p = tmp_path / name
    p.write_text(content)
    return str(p)


def test_python_extractor(tmp_path):
    cases = extract_file(_write(tmp_path, "test_thing.py", PY_SRC))
    names = {c.qualname for c in cases}
    assert names == {"TestThing.test_accuracy", "TestThing.test_raises",
                     "test_mocked"}
    acc = next(c for c in cases if c.name == "test_accuracy")
    assert acc.docstring == "checks model accuracy"
    kinds = {a.kind for a in acc.assertions}
    assert "unittest" in kinds and "assert" in kinds
    raises = next(c for c in cases if c.name == "test_raises")
    assert any(a.kind == "raises" and a.exception == "ValueError"
               for a in raises.assertions)
    mocked = next(c for c in cases if c.name == "test_mocked")
    assert mocked.uses_mock and mocked.is_parametrized


def test_python_extractor_handles_syntax_error(tmp_path):
    assert extract_file(_write(tmp_path, "test_py2.py",
                               "print 'hello'\ndef test_x(): pass")) == []


def test_gtest_extractor(tmp_path):
    cases = extract_gtest_file(_write(tmp_path, "math_test.cc", CC_SRC))
    assert {c.qualname for c in cases} == {"MathTest.HandlesZero",
                                           "PipelineFixture.Throws"}
    zero = next(c for c in cases if c.name == "HandlesZero")
    assert len(zero.assertions) == 2
    assert any(a.kind == "approx" for a in zero.assertions)
    throws = next(c for c in cases if c.name == "Throws")
    kinds = [a.kind for a in throws.assertions]
    assert "raises" in kinds and "unittest" in kinds
    exc = next(a for a in throws.assertions if a.kind == "raises")
    assert "invalid_argument" in exc.exception


def test_ts_extractor(tmp_path):
    cases = extract_ts_file(_write(tmp_path, "manager.test.ts", TS_SRC))
    assert len(cases) == 2
    first = cases[0]
    assert first.qualname == "manager.starts an experiment"
    assert len(first.assertions) == 2
    assert any(a.kind == "raises" for a in cases[1].assertions)


def test_walker_detection(tmp_path):
    (tmp_path / "pkg").mkdir()
    (tmp_path / "tests").mkdir()
    files = {
        "pkg/module.py": "x = 1",
        "pkg/module_test.py": "def test_a(): pass",
        "tests/test_b.py": "def test_b(): pass",
        "tests/helper.py": "pass",
        "pkg/engine_test.cc": "TEST(A,B){}",
        "pkg/engine.cc": "",
        "pkg/ui.test.ts": "it('x',()=>{});",
        "pkg/ui.ts": "",
    }
    for rel, content in files.items():
        p = tmp_path / rel
        p.write_text(content)
    found = {f.rel for f in walker.test_files(str(tmp_path))}
    assert "pkg/module_test.py" in found
    assert "tests/test_b.py" in found
    assert "tests/helper.py" in found          # in a tests/ dir
    assert "pkg/engine_test.cc" in found
    assert "pkg/ui.test.ts" in found
    assert "pkg/module.py" not in found
    assert "pkg/engine.cc" not in found
    assert "pkg/ui.ts" not in found


def test_gtest_local_macro_expansion(tmp_path):
    """One-level project-local assertion wrappers (kenlm SLOPPY_CHECK_* style,
    reference src/DeepSpeech .../lm/model_test.cc:11) count as assertions."""
    from tosem2021_amd.extract.gtest_extractor import extract_gtest_file
    src = r"""
#define SLOPPY_CHECK_CLOSE(ref, value, tol) BOOST_CHECK_CLOSE( \
    static_cast<double>(ref), static_cast<double>(value), tol);
#define MY_OK(x) EXPECT_TRUE(x)

TEST(Wrapped, UsesLocalMacros) {
  SLOPPY_CHECK_CLOSE(1.0, compute(), 0.001);
  MY_OK(flag());
  EXPECT_EQ(1, one());
}
"""
    p = tmp_path / "wrap_test.cc"
    p.write_text(src)
    cases = extract_gtest_file(str(p))
    assert len(cases) == 1
    kinds = sorted(a.kind for a in cases[0].assertions)
    names = sorted(a.call_name for a in cases[0].assertions)
    assert "approx" in kinds          # SLOPPY_CHECK_CLOSE -> BOOST_CHECK_CLOSE
    assert "BOOST_CHECK_CLOSE" in names
    assert "EXPECT_TRUE" in names     # MY_OK
    assert len(cases[0].assertions) == 3


def test_ts_nested_describe_and_body_extent(tmp_path):
    """Round-2 hardening: brace-matched it() bodies (no assertion bleed
    between cases) and full nested-describe suite paths."""
    from tosem2021_amd.extract.ts_extractor import extract_ts_file
    src = """
describe('outer', () => {
    describe('inner', () => {
        it('first', () => {
            expect(a).to.equal(1);
        });
        // helper between cases — must not attach to 'first'
        const check = () => { assert.ok(stray()); };
        it('second', async () => {
            expect(b).to.equal(2);
            expect(c).to.throw();
        });
    });
    it('outer-level', () => {
        assert.strictEqual(d, 4);
    });
});
"""
    p = tmp_path / "x.test.ts"
    p.write_text(src)
    cases = extract_ts_file(str(p))
    byname = {c.name: c for c in cases}
    assert set(byname) == {"first", "second", "outer-level"}
    assert byname["first"].qualname == "outer.inner.first"
    assert byname["second"].qualname == "outer.inner.second"
    assert byname["outer-level"].qualname == "outer.outer-level"
    # 'first' must hold exactly its own assertion; the stray helper assert
    # between the cases belongs to neither
    assert len(byname["first"].assertions) == 1
    assert len(byname["second"].assertions) == 2
    assert len(byname["outer-level"].assertions) == 1
    assert "stray" not in byname["first"].source


def test_ts_braceless_arrow_case(tmp_path):
    from tosem2021_amd.extract.ts_extractor import extract_ts_file
    src = """
it('compact', () => expect(x).to.equal(1));
it('next', () => { expect(y).to.equal(2); });
"""
    p = tmp_path / "y.test.ts"
    p.write_text(src)
    cases = extract_ts_file(str(p))
    assert len(cases) == 2
    assert len(cases[0].assertions) == 1
    assert len(cases[1].assertions) == 1


def test_gtest_instantiate_multiplicity(tmp_path):
    """TEST_P cases carry the summed static cardinality of their
    INSTANTIATE_* value lists as param multiplicity (Values(1,2) +
    Values(3) = 3 runs of each case)."""
    from tosem2021_amd.extract.gtest_extractor import extract_gtest_file
    src = """
class ParamSuite : public ::testing::TestWithParam<int> {};

TEST_P(ParamSuite, Works) {
  EXPECT_GT(GetParam(), 0);
}
TEST_P(ParamSuite, AlsoWorks) {
  EXPECT_LT(GetParam(), 100);
}
TEST(PlainSuite, One) { EXPECT_TRUE(ok()); }

INSTANTIATE_TEST_SUITE_P(Small, ParamSuite, ::testing::Values(1, 2));
INSTANTIATE_TEST_CASE_P(Legacy, ParamSuite, ::testing::Values(3));
"""
    p = tmp_path / "p_test.cc"
    p.write_text(src)
    cases = {c.name: c for c in extract_gtest_file(str(p))}
    assert set(cases) == {"Works", "AlsoWorks", "One"}
    assert cases["Works"].is_parametrized
    assert cases["Works"].param_multiplicity == 3
    assert cases["AlsoWorks"].param_multiplicity == 3
    assert not cases["One"].is_parametrized
    assert cases["One"].param_multiplicity == 1
    # classify_case folds the multiplicity into the Cases column
    from tosem2021_amd.classify.rules import classify_case
    rows = classify_case(cases["Works"], repo="Apollo", file_id=1)
    assert rows[0].cases == 3     # 1 assertion x 3 parameter values


def test_gtest_instantiate_value_cardinality(tmp_path):
    """INSTANTIATE_* generators are statically counted: Values argc,
    Bool 2, Range span, Combine product; unknown generators fall back to
    1 per instantiation."""
    src = """
TEST_P(MySuite, Works) { EXPECT_EQ(GetParam(), GetParam()); }
INSTANTIATE_TEST_SUITE_P(Three, MySuite, testing::Values(1, 2, 3));
INSTANTIATE_TEST_SUITE_P(Four, MySuite,
    testing::Combine(testing::Values("a", "b"), testing::Bool()));

TEST_P(Opaque, Runs) { EXPECT_TRUE(GetParam()); }
INSTANTIATE_TEST_SUITE_P(Dyn, Opaque, testing::ValuesIn(kRuntimeVec));
"""
    p = tmp_path / "card_test.cc"
    p.write_text(src)
    from tosem2021_amd.extract.gtest_extractor import extract_gtest_file
    cases = {c.name: c for c in extract_gtest_file(str(p))}
    assert cases["Works"].param_multiplicity == 3 + 4
    assert cases["Runs"].param_multiplicity == 1


def test_freestanding_check_fallback(tmp_path):
    """C++ test files with no framework macro (openfst style) fall back to
    per-function cases over glog CHECK-family assertions."""
    src = """
#include <fst/float-weight.h>
namespace {
void TestWeightCopy(int w) {
  CHECK_EQ(w, w);
  CHECK(w >= 0);
}
template <class W>
void TestNear(W a, W b) {
  CHECK_NEAR(a, b, 1e-5);
}
void Helper(int x) { use(x); }   // no assertions -> no case
}
int main() { TestWeightCopy(3); return 0; }
"""
    p = tmp_path / "weight_test.cc"
    p.write_text(src)
    from tosem2021_amd.extract.gtest_extractor import extract_gtest_file
    cases = {c.name: c for c in extract_gtest_file(str(p))}
    assert "TestWeightCopy" in cases and "TestNear" in cases
    assert "Helper" not in cases
    assert len(cases["TestWeightCopy"].assertions) == 2
    assert cases["TestNear"].assertions[0].kind == "approx"


def test_walker_admits_tester_headers():
    """Tester headers inside test dirs are walked (the study labeled
    openfst's algo_test.h / weight-tester.h as DeepSpeech components);
    ordinary headers are not."""
    from tosem2021_amd.corpus.walker import classify_language, is_test_file
    assert classify_language("a/b/weight-tester.h") == "cpp"
    assert is_test_file("src/include/fst/test/weight-tester.h", "cpp")
    assert is_test_file("src/include/fst/test/algo_test.h", "cpp")
    assert not is_test_file("src/include/fst/float-weight.h", "cpp")
    assert not is_test_file("src/lib/weight-tester.h", "cpp")   # not in test dir
    # third_party is mined (study scope); node_modules still skipped
    from tosem2021_amd.corpus.walker import SKIP_DIRS
    assert "third_party" not in SKIP_DIRS
    assert "node_modules" in SKIP_DIRS


def test_python_assertion_helper_expansion(tmp_path):
    """Assertions inside file-local helper methods count at their call
    sites in test cases (DeepSpeech test_value_range.py pattern)."""
    src = '''
import unittest

class T(unittest.TestCase):
    def _ending_tester(self, v, e):
        r = compute(v)
        self.assertEqual(r, e)

    def test_scalar(self):
        self._ending_tester(1, 1)
        self._ending_tester(2, 2)

    def test_direct(self):
        self.assertTrue(ok())
'''
    p = tmp_path / "test_vr.py"
    p.write_text(src)
    from tosem2021_amd.extract.python_extractor import extract_file
    cases = {c.name: c for c in extract_file(str(p))}
    assert len(cases["test_scalar"].assertions) == 2
    assert cases["test_scalar"].assertions[0].call_name == "assertEqual"
    assert len(cases["test_direct"].assertions) == 1


def test_gtest_helper_function_expansion(tmp_path):
    """C++ helper functions holding the assertions (kenlm model_test.cc
    pattern, incl. templated calls and helper->helper nesting) are
    expanded transitively into the calling case."""
    src = """
template <class M> void Starters(const M &m) {
  BOOST_CHECK_EQUAL(m.Size(), 1);
  BOOST_CHECK(m.Ok());
}
template <class M> void Everything(const M &m) {
  Starters<M>(m);
}
BOOST_AUTO_TEST_CASE(probing) {
  ProbingModel m("f.arpa");
  Everything<ProbingModel>(m);
}
BOOST_AUTO_TEST_CASE(direct) {
  BOOST_REQUIRE(true);
}
"""
    p = tmp_path / "model_test.cc"
    p.write_text(src)
    from tosem2021_amd.extract.gtest_extractor import extract_gtest_file
    cases = {c.name: c for c in extract_gtest_file(str(p))}
    assert len(cases["probing"].assertions) == 2
    assert len(cases["direct"].assertions) == 1
