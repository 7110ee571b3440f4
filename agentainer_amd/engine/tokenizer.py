"""Byte-level tokenizer — self-contained (no network, no tokenizer files).

Deterministic byte mapping: token = byte + 3, with PAD=0, BOS=1, EOS=2.
Real checkpoints would bring their own tokenizer; for synthetic/random-init
models (BASELINE.json: "synthetic prompts, random-init weights") a byte
tokenizer gives reproducible prompt/response round-trips.
"""

from __future__ import annotations

from typing import List

PAD, BOS, EOS = 0, 1, 2
OFFSET = 3
MIN_VOCAB = 256 + OFFSET


class ByteTokenizer:
    def __init__(self, vocab_size: int):
        assert vocab_size >= MIN_VOCAB, f"vocab must be >= {MIN_VOCAB}"
        self.vocab_size = vocab_size
        self.eos_id = EOS
        self.bos_id = BOS

    def encode(self, text: str, bos: bool = False) -> List[int]:
        ids = [b + OFFSET for b in text.encode("utf-8")]
        return ([BOS] if bos else []) + ids

    def decode(self, ids: List[int]) -> str:
        bs = bytes(i - OFFSET for i in ids if OFFSET <= i < OFFSET + 256)
        return bs.decode("utf-8", "replace")

    def id_bytes(self, i: int) -> bytes:
        """Raw bytes of one token (b'' for specials) — feeds the streaming
        path's incremental UTF-8 decoder so multi-byte sequences are only
        emitted once complete."""
        return bytes([i - OFFSET]) if OFFSET <= i < OFFSET + 256 else b""
