"""Property-based differential test: the native JSON5 parser agrees with
Python's json module on the strict-JSON subset, including round-trips."""

import json
import math

from hypothesis import given, settings, strategies as st

from containerpilot_amd import native

json_scalars = st.one_of(
    st.none(),
    st.booleans(),
    st.integers(min_value=-(2**53), max_value=2**53),
    st.floats(allow_nan=False, allow_infinity=False, width=64),
    st.text(max_size=40),
)

json_values = st.recursive(
    json_scalars,
    lambda children: st.one_of(
        st.lists(children, max_size=6),
        st.dictionaries(st.text(max_size=12), children, max_size=6),
    ),
    max_leaves=25,
)


def normalize(v):
    """ints that arrive as floats compare equal; collapse for comparison"""
    if isinstance(v, float) and math.isfinite(v) and v == int(v) \
            and abs(v) < 2**53:
        return float(v)
    if isinstance(v, dict):
        return {k: normalize(x) for k, x in v.items()}
    if isinstance(v, list):
        return [normalize(x) for x in v]
    return v


@settings(max_examples=300, deadline=None)
@given(json_values)
def test_parser_agrees_with_python_json(value):
    text = json.dumps(value)
    parsed = json.loads(native.json5_to_json(text))

    def eq(a, b):
        if isinstance(a, float) or isinstance(b, float):
            if isinstance(a, bool) != isinstance(b, bool):
                return False
            return math.isclose(float(a), float(b),
                                rel_tol=1e-15, abs_tol=1e-300)
        if isinstance(a, dict) and isinstance(b, dict):
            return a.keys() == b.keys() and all(eq(a[k], b[k]) for k in a)
        if isinstance(a, list) and isinstance(b, list):
            return len(a) == len(b) and all(eq(x, y) for x, y in zip(a, b))
        return a == b

    assert eq(parsed, value), (text, parsed)


@settings(max_examples=200, deadline=None)
@given(st.text(max_size=60))
def test_parser_never_crashes_on_garbage(text):
    try:
        native.json5_to_json(text)
    except ValueError:
        pass  # rejecting is fine; crashing is not


def test_parser_rejects_bracket_bombs():
    for bomb in ("[" * 200000, "{" * 100000,
                 "[{" * 50000, '{"a":' * 50000):
        try:
            native.json5_to_json(bomb)
        except ValueError:
            pass


@settings(max_examples=200, deadline=None)
@given(st.text(alphabet=st.sampled_from(list("{}().|$\"ab \n-")), max_size=50))
def test_template_never_crashes_on_garbage(text):
    """The template engine either renders or raises ValueError — never
    crashes the process."""
    try:
        native.render_template(text)
    except ValueError:
        pass
