"""Byte-level tokenizer used when no trained tokenizer is available (no
network for tokenizer downloads; synthetic/random-init models don't have one).
Token ids: 0=pad/eos, 1=bos, 2..257 = bytes 0..255. Bijective for vocab>=258
(all real configs)."""

from __future__ import annotations

BOS = 1
EOS = 0
BYTE_OFFSET = 2


def encode(text: str, vocab_size: int, add_bos: bool = True) -> list[int]:
    toks = [BYTE_OFFSET + (b % min(256, vocab_size - BYTE_OFFSET - 1))
            for b in text.encode("utf-8")]
    return ([BOS] if add_bos else []) + toks


def decode(tokens: list[int]) -> str:
    bs = bytes(t - BYTE_OFFSET for t in tokens
               if BYTE_OFFSET <= t < BYTE_OFFSET + 256)
    return bs.decode("utf-8", errors="replace")
