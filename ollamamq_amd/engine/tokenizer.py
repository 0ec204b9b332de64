"""Byte-level synthetic tokenizer.

There is no network to fetch real tokenizer vocabularies (BASELINE.json:
synthetic prompts, random-init weights), so workers tokenize bytes 1:1
(token id = byte value) and render generated ids deterministically:
ids < 256 decode to their byte, larger ids to a printable ⟨id⟩ marker.
This keeps the wire contract exercised end-to-end (prompt in, streamed
text out) with real model compute in between.
"""
from __future__ import annotations

from typing import List


class ByteTokenizer:
    def __init__(self, vocab_size: int):
        self.vocab_size = vocab_size
        # reserve the last id as a conventional stop token when roomy
        self.stop_token = vocab_size - 1 if vocab_size > 512 else None

    def encode(self, text: str) -> List[int]:
        data = text.encode("utf-8", errors="replace")
        cap = min(self.vocab_size, 256)
        return [b % cap for b in data] or [0]

    def decode_one(self, tok: int) -> str:
        if 32 <= tok < 127:
            return chr(tok)
        if tok < 256:
            return f"\\x{tok:02x}"
        return f"⟨{tok}⟩"

    def decode(self, toks: List[int]) -> str:
        return "".join(self.decode_one(t) for t in toks)
