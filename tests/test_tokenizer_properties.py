"""Property tests for the byte-level synthetic tokenizer (engine/tokenizer).

It feeds every prompt into the engine and renders every generated id back
onto the wire, so its invariants gate the whole serving path:
  - encode() ids are always in-vocab and non-empty
  - printable ASCII round-trips exactly through encode+decode
  - decode() never raises for ANY id the sampler can emit (0..vocab-1,
    including the stop token and out-of-byte-range ids)
"""
import hypothesis.strategies as st
from hypothesis import given, settings

from ollamamq_amd.engine.tokenizer import ByteTokenizer


@settings(derandomize=True, max_examples=200, deadline=None)
@given(text=st.text(max_size=200),
       vocab=st.integers(min_value=1, max_value=200000))
def test_encode_in_vocab_nonempty(text, vocab):
    tok = ByteTokenizer(vocab)
    ids = tok.encode(text)
    assert ids, "encode must never return an empty prompt"
    assert all(0 <= t < vocab for t in ids)


@settings(derandomize=True, max_examples=100, deadline=None)
@given(text=st.text(alphabet=st.characters(min_codepoint=32,
                                           max_codepoint=126),
                    max_size=100))
def test_printable_ascii_roundtrip(text):
    tok = ByteTokenizer(512)
    assert tok.decode(tok.encode(text)) == (text or tok.decode([0]))


@settings(derandomize=True, max_examples=200, deadline=None)
@given(ids=st.lists(st.integers(min_value=0, max_value=200000 - 1),
                    max_size=64))
def test_decode_total(ids):
    tok = ByteTokenizer(200000)
    out = tok.decode(ids)
    assert isinstance(out, str)


def test_stop_token_reserved_only_when_roomy():
    assert ByteTokenizer(512).stop_token is None
    big = ByteTokenizer(128256)
    assert big.stop_token == 128255
    # the stop token must never collide with an encodable byte id
    assert big.stop_token >= 256
