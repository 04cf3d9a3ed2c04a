"""Golden tests for the text model formats (payload parity)."""

import math

import pytest

from flink_ms_amd.utils import textio as t


# Expected strings verified against Java Double.toString semantics.
JAVA_DOUBLE_CASES = [
    (1.0, "1.0"), (0.5, "0.5"), (100.0, "100.0"), (1e7, "1.0E7"),
    (9999999.0, "9999999.0"), (0.001, "0.001"), (0.0009, "9.0E-4"),
    (1e-4, "1.0E-4"), (-2.5, "-2.5"), (0.0, "0.0"), (-0.0, "-0.0"),
    (123456789.0, "1.23456789E8"), (0.1, "0.1"),
    (1 / 3, "0.3333333333333333"), (1.5e-10, "1.5E-10"),
    (1.7976931348623157e308, "1.7976931348623157E308"),
    (12345.678, "12345.678"), (2e7, "2.0E7"), (-1e-3, "-0.001"),
    (42.0, "42.0"), (3.14159, "3.14159"), (-7.25e-5, "-7.25E-5"),
]


@pytest.mark.parametrize("x,expected", JAVA_DOUBLE_CASES)
def test_java_double_format(x, expected):
    assert t.java_double_to_string(x) == expected


def test_java_double_specials():
    assert t.java_double_to_string(float("nan")) == "NaN"
    assert t.java_double_to_string(float("inf")) == "Infinity"
    assert t.java_double_to_string(float("-inf")) == "-Infinity"


def test_java_double_roundtrip():
    import random
    rng = random.Random(42)
    for _ in range(2000):
        x = rng.uniform(-1e9, 1e9) * 10 ** rng.randint(-12, 12)
        s = t.java_double_to_string(x)
        assert float(s.replace("E", "e")) == x, (x, s)


def test_als_row_roundtrip():
    row = t.als_factor_row(42, "U", [1.0, 0.5, -0.25])
    assert row == "42,U,1.0;0.5;-0.25"
    rid, kind, facs = t.parse_als_row(row)
    assert (rid, kind, facs) == ("42", "U", [1.0, 0.5, -0.25])


def test_mean_row():
    row = t.als_factor_row(t.MEAN_ID, "I", [2.0])
    assert row == "MEAN,I,2.0"
    assert t.als_state_key("MEAN", "I") == "MEAN-I"
    assert t.als_state_key(7, "U") == "7-U"


def test_svm_rows():
    assert t.svm_flat_row(1, 0.5) == "1,0.5"
    assert t.parse_svm_flat_row("17,-2.0") == (17, -2.0)
    # bucket = 1-based index / range (SVMImpl.scala:42)
    assert t.svm_bucket_of(999, 1000) == 0
    assert t.svm_bucket_of(1000, 1000) == 1
    row = t.svm_range_row(2, [(2000, 1.5), (2001, -0.125)])
    assert row == "2,2000:1.5;2001:-0.125"
    b, pairs = t.parse_svm_range_row(row)
    assert b == 2 and pairs == [(2000, 1.5), (2001, -0.125)]


def test_latency_csv_rows():
    assert t.als_latency_csv_row(1, 2, 3.5, 12) == "1,2,3.5,12"
    assert t.svm_latency_csv_row(0, 5, -1.0, 3) == "0,5,-1.0,3"


# ---- property tests (hypothesis): the codecs round-trip arbitrary doubles

from hypothesis import given, settings, strategies as st


@settings(max_examples=500, deadline=None)
@given(st.floats(allow_nan=False, allow_infinity=False))
def test_java_double_roundtrip_property(x):
    s = t.java_double_to_string(x)
    assert float(s.replace("E", "e")) == x
    # Java surface rules: decimal form iff 1e-3 <= |x| < 1e7 (or zero)
    if x != 0 and 1e-3 <= abs(x) < 1e7:
        assert "E" not in s and "." in s
    elif x != 0:
        assert "E" in s


@settings(max_examples=200, deadline=None)
@given(st.lists(st.floats(allow_nan=False, allow_infinity=False,
                          width=32), min_size=1, max_size=8),
       st.integers(min_value=0, max_value=2**31 - 1),
       st.sampled_from(["U", "I"]))
def test_als_row_roundtrip_property(factors, rid, kind):
    row = t.als_factor_row(rid, kind, factors)
    pid, pkind, pfactors = t.parse_als_row(row)
    assert pid == str(rid) and pkind == kind
    assert pfactors == [float(f) for f in factors]


@settings(max_examples=200, deadline=None)
@given(st.lists(st.tuples(st.integers(min_value=1, max_value=10**6),
                          st.floats(allow_nan=False, allow_infinity=False)),
                min_size=1, max_size=20))
def test_svm_range_row_roundtrip_property(pairs):
    row = t.svm_range_row(7, pairs)
    b, parsed = t.parse_svm_range_row(row)
    assert b == 7
    assert parsed == [(i, float(w)) for i, w in pairs]


def test_params_negative_number_values():
    from flink_ms_amd.utils.params import Params

    p = Params.from_args(["--lambda", "-0.5", "--seed", "-7", "--flag",
                          "--name", "x"])
    assert p.get_float("lambda") == -0.5
    assert p.get_int("seed") == -7
    assert p.get_bool("flag") is True
    assert p.get("name") == "x"


# ---------------------------------------------------- property tests

from hypothesis import given, settings
from hypothesis import strategies as st


@settings(max_examples=500, deadline=None)
@given(st.floats(allow_nan=False, allow_infinity=False))
def test_java_double_round_trips(x):
    """Shortest-round-trip property: parsing the Java-formatted string
    recovers the exact double (JLS Double.toString contract)."""
    s = t.java_double_to_string(x)
    assert float(s) == x
    # surface-shape invariants
    assert "e" not in s or "E" in s  # exponents are uppercase E
    if 1e-3 <= abs(x) < 1e7 or x == 0.0:
        assert "E" not in s          # plain decimal in the JLS window


@settings(max_examples=200, deadline=None)
@given(st.integers(min_value=0, max_value=2**40),
       st.sampled_from(["U", "I"]),
       st.lists(st.floats(allow_nan=False, allow_infinity=False,
                          width=32), min_size=1, max_size=16))
def test_als_row_round_trip_property(eid, kind, vec):
    row = t.als_factor_row(eid, kind, vec)
    pid, pkind, pvec = t.parse_als_row(row)
    assert pid == str(eid) and pkind == kind
    assert pvec == pytest.approx(vec, rel=0, abs=0)  # exact round trip


@settings(max_examples=100, deadline=None)
@given(st.lists(st.tuples(
    st.integers(min_value=0, max_value=10**6),
    st.sampled_from(["U", "I"]),
    st.lists(st.floats(allow_nan=False, allow_infinity=False, width=32),
             min_size=3, max_size=3)), min_size=1, max_size=20))
def test_native_parser_matches_python_parser(rows):
    """The threaded C++ block parser agrees with the scalar Python codec
    on arbitrary well-formed rows (fp32-exact values)."""
    _hip_ops = pytest.importorskip("flink_ms_amd._hip_ops")
    text = "\n".join(t.als_factor_row(e, k, v) for e, k, v in rows)
    ids, kinds, facs, offs, lens, bad = _hip_ops.parse_als_block(
        text.encode(), 3)
    assert int(bad) == 0
    for r, (eid, kind, vec) in enumerate(rows):
        assert int(ids[r]) == eid
        assert ("U" if int(kinds[r]) == 0 else "I") == kind
        for a, b in zip(facs[r].tolist(), vec):
            assert a == b, (a, b)
