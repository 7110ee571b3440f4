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

    def stream_decoder(self) -> "_ByteStreamDecoder":
        return _ByteStreamDecoder(self)


class _ByteStreamDecoder:
    """Incremental token->text for the SSE path: byte tokens through a
    stateful UTF-8 decoder, so multi-byte chars surface exactly once."""

    def __init__(self, tok: ByteTokenizer):
        import codecs
        self.tok = tok
        self.dec = codecs.getincrementaldecoder("utf-8")("replace")

    def feed(self, token_id: int) -> str:
        return self.dec.decode(self.tok.id_bytes(token_id))

    def flush(self) -> str:
        return self.dec.decode(b"", True)


class HFTokenizer:
    """Real-checkpoint tokenizer (tokenizer.json via the `tokenizers`
    library) for weights-path deploys. Same surface as ByteTokenizer:
    encode/decode/eos_id/stream_decoder."""

    def __init__(self, path: str, eos_id: int = -1):
        from tokenizers import Tokenizer
        self.tok = Tokenizer.from_file(path)
        self.vocab_size = self.tok.get_vocab_size()
        self.eos_id = eos_id
        self.bos_id = -1

    @classmethod
    def from_checkpoint(cls, model_dir: str) -> "HFTokenizer":
        """Load <dir>/tokenizer.json; eos id from config.json /
        generation_config.json (first entry when a list)."""
        import json
        import os
        eos = -1
        for cfg_name in ("generation_config.json", "config.json"):
            p = os.path.join(model_dir, cfg_name)
            if os.path.exists(p):
                with open(p) as f:
                    v = json.load(f).get("eos_token_id")
                if isinstance(v, list) and v:
                    eos = int(v[0])
                    break
                if isinstance(v, int):
                    eos = v
                    break
        return cls(os.path.join(model_dir, "tokenizer.json"), eos_id=eos)

    def encode(self, text: str, bos: bool = False) -> List[int]:
        return self.tok.encode(text).ids

    def decode(self, ids: List[int]) -> str:
        return self.tok.decode(ids)

    def stream_decoder(self) -> "_HFStreamDecoder":
        return _HFStreamDecoder(self)


class _HFStreamDecoder:
    """Prefix-delta incremental decode for BPE/unigram vocabularies: holds
    back a trailing replacement char (incomplete byte-level sequence)
    until the next token completes it."""

    def __init__(self, tok: HFTokenizer):
        self.tok = tok
        self.ids: List[int] = []
        self.emitted = ""

    def _stable(self) -> str:
        full = self.tok.decode(self.ids)
        return full[:-1] if full.endswith("�") else full

    def feed(self, token_id: int) -> str:
        self.ids.append(token_id)
        stable = self._stable()
        if stable.startswith(self.emitted):
            delta = stable[len(self.emitted):]
            self.emitted = stable
            return delta
        return ""  # prefix revised (rare); resync at flush

    def flush(self) -> str:
        full = self.tok.decode(self.ids)
        if full.startswith(self.emitted):
            return full[len(self.emitted):]
        return ""
