"""Property-based fuzz of the native JSON parser (csrc/dispatcher/json.cpp).

It parses every untrusted request body on the ingress path (model
extraction, admin bodies, backend responses).  Properties:
  1. Arbitrary bytes never crash — parse returns a dump or None.
  2. Anything Python's json emits, our parser accepts, and the dump
     re-parses in Python to the SAME value (semantic round-trip through
     both parsers).
  3. Model extraction agrees with Python on generated request bodies.
"""
import json
import math

import hypothesis.strategies as st
from hypothesis import given, settings

from ollamamq_amd.dispatch import load

d = load()

fuzz_st = st.text(alphabet="{}[]\":,0123456789.-+eE truefalsn\\/ \t\n\x00á",
                  min_size=0, max_size=200)

json_val = st.recursive(
    st.none() | st.booleans()
    | st.integers(min_value=-(10 ** 12), max_value=10 ** 12)
    | st.floats(allow_nan=False, allow_infinity=False, width=32)
    | st.text(alphabet=st.characters(codec="utf-8",
                                     exclude_categories=("Cs",)),
              max_size=20),
    lambda children: st.lists(children, max_size=4)
    | st.dictionaries(st.text(max_size=8), children, max_size=4),
    max_leaves=12)


def _eq(a, b):
    if isinstance(a, float) or isinstance(b, float):
        return math.isclose(float(a), float(b), rel_tol=1e-6, abs_tol=1e-9)
    if isinstance(a, list):
        return isinstance(b, list) and len(a) == len(b) and \
            all(_eq(x, y) for x, y in zip(a, b))
    if isinstance(a, dict):
        return isinstance(b, dict) and a.keys() == b.keys() and \
            all(_eq(v, b[k]) for k, v in a.items())
    return a == b


@settings(derandomize=True, max_examples=400, deadline=None)
@given(fuzz_st)
def test_fuzz_never_crashes(text):
    out = d.json_roundtrip(text)
    assert out is None or isinstance(out, str)
    d.json_get_model(text)  # must not crash either


@settings(derandomize=True, max_examples=200, deadline=None)
@given(json_val)
def test_semantic_roundtrip_vs_python(val):
    src = json.dumps(val)
    out = d.json_roundtrip(src)
    assert out is not None, f"rejected valid JSON: {src!r}"
    assert _eq(json.loads(out), val), (src, out)


@settings(derandomize=True, max_examples=100, deadline=None)
@given(model=st.text(max_size=16), extra=json_val)
def test_model_extraction_agrees(model, extra):
    body = json.dumps({"model": model, "options": extra})
    assert d.json_get_model(body) == model
