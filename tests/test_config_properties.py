"""Property-based fuzz of the native appconf.yaml parser (C2).

The parser is hand-written C++ (csrc/dispatcher/config.cpp) that hard-refuses
YAML constructs outside the reference's subset (reference src/config.rs uses
serde_yaml).  Two properties:
  1. Arbitrary text NEVER crashes: every input either parses to a config
     dict or raises RuntimeError with a message (a segfault would kill the
     pytest process, which is the crash detector here).
  2. Structured valid configs round-trip exactly.
"""
import os

import hypothesis.strategies as st
from hypothesis import given, settings

from ollamamq_amd.dispatch import load

d = load()

# includes every metachar the refusal matrix cares about (anchors, flow,
# block scalars, tabs, merge keys) plus structure chars and junk
fuzz_text = st.text(
    alphabet="abcdefghijklmnopqrstuvwxyz0123456789"
             " :#-_.\"'&*{}[]|><%@!\t\n",
    min_size=0, max_size=300)

name_st = st.text(alphabet="abcdefghijklmnopqrstuvwxyz0123456789.-:",
                  min_size=1, max_size=16).filter(
    lambda s: s[0].isalnum() and s[-1] != ":")
url_st = st.builds(lambda h, p: f"http://{h}:{p}",
                   st.text(alphabet="abcdefghij.", min_size=1, max_size=10)
                   .filter(lambda s: "." != s[0] and "." != s[-1]),
                   st.integers(min_value=1, max_value=65535))


def _write(tmp, text):
    p = os.path.join(tmp, "appconf.yaml")
    with open(p, "w") as f:
        f.write(text)
    return p


@settings(derandomize=True, max_examples=250, deadline=None)
@given(fuzz_text)
def test_fuzz_never_crashes(tmp_path_factory, text):
    p = _write(str(tmp_path_factory.mktemp("cfg")), text)
    try:
        cfg = d.load_config(p)
        # on success the shape contract holds
        assert isinstance(cfg["backends"], list)
        assert cfg["settings"]["port"] >= 0
        assert isinstance(cfg["models"], list)
    except RuntimeError as e:
        assert str(e)  # refusals carry a message (with line number)


@settings(derandomize=True, max_examples=60, deadline=None)
@given(backends=st.lists(url_st, max_size=4),
       port=st.integers(min_value=1, max_value=65535),
       timeout=st.integers(min_value=1, max_value=86400),
       models=st.lists(
           st.tuples(name_st,
                     st.integers(min_value=512, max_value=200000),
                     st.integers(min_value=-1, max_value=86400)),
           max_size=3))
def test_valid_config_roundtrip(tmp_path_factory, backends, port, timeout,
                                models):
    y = ""
    if backends:
        y += "backends:\n" + "".join(f"  - {u}\n" for u in backends)
    y += f"settings:\n  port: {port}\n  timeout: {timeout}\n"
    if models:
        y += "models:\n"
        for n, ctx, ka in models:
            y += (f"  - name: \"{n}\"\n    max_ctx: {ctx}\n"
                  f"    keep_alive: {ka}\n")
    p = _write(str(tmp_path_factory.mktemp("cfg")), y)
    cfg = d.load_config(p)
    assert cfg["backends"] == backends
    assert cfg["settings"]["port"] == port
    assert cfg["settings"]["timeout"] == timeout
    assert [(m["name"], m["max_ctx"], m["keep_alive"])
            for m in cfg["models"]] == models
