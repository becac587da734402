"""Tokenizers for the model server.

Two implementations behind one interface (encode/decode/eos/bos):
  * HFTokenizer — a real `tokenizer.json` from the model artifacts dir,
    loaded with the `tokenizers` library (the reference's external server
    images used the model's HF tokenizer; reference docs/container-contract
    mounts the artifacts at /content/model).
  * ByteTokenizer — byte-level fallback when no trained tokenizer exists
    (no network for tokenizer downloads; synthetic/random-init models).
    Token ids: 0=pad/eos, 1=bos, 2..257 = bytes 0..255.

The model loader (loader/model_loader.py) exports a byte-level BPE
tokenizer.json (specials at 0=</s>, 1=<s>), so loader-produced artifacts
exercise the real-tokenizer path end-to-end even offline.
"""

from __future__ import annotations

import os
from typing import Optional

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


class ByteTokenizer:
    bos = BOS
    eos = EOS

    def __init__(self, vocab_size: int):
        self.vocab_size = vocab_size

    def encode(self, text: str, add_bos: bool = True) -> list[int]:
        return encode(text, self.vocab_size, add_bos)

    def decode(self, tokens: list[int]) -> str:
        return decode(tokens)


class HFTokenizer:
    def __init__(self, path: str):
        from tokenizers import Tokenizer
        self.t = Tokenizer.from_file(path)
        self.eos = self._special("</s>", "<|endoftext|>", "<|end_of_text|>",
                                 default=EOS)
        self.bos = self._special("<s>", "<|begin_of_text|>", default=None)

    def _special(self, *names: str, default):
        for n in names:
            tid = self.t.token_to_id(n)
            if tid is not None:
                return tid
        return default

    def encode(self, text: str, add_bos: bool = True) -> list[int]:
        ids = self.t.encode(text, add_special_tokens=False).ids
        if add_bos and self.bos is not None:
            return [self.bos] + ids
        return ids

    def decode(self, tokens: list[int]) -> str:
        specials = {self.bos, self.eos}
        return self.t.decode([t for t in tokens if t not in specials],
                             skip_special_tokens=True)


class StreamDecoder:
    """Incremental detokenizer for SSE streaming: decoding each token id
    independently emits U+FFFD when a multi-byte UTF-8 character spans
    token boundaries (byte-level BPE splits them freely). This decodes the
    CUMULATIVE id list and emits the delta, holding back a trailing
    replacement char until the next token resolves (or flush() at
    end-of-stream emits it as-is for genuinely invalid bytes)."""

    def __init__(self, tok):
        self.tok = tok
        self.ids: list[int] = []
        self.emitted = 0            # chars of the cumulative text sent

    def push(self, token_id: int) -> str:
        self.ids.append(token_id)
        text = self.tok.decode(self.ids)
        safe = len(text)
        if text.endswith("�"):
            safe -= 1               # maybe a partial sequence: hold back
        if safe <= self.emitted:
            return ""
        out = text[self.emitted:safe]
        self.emitted = safe
        return out

    def flush(self) -> str:
        text = self.tok.decode(self.ids)
        out = text[self.emitted:]
        self.emitted = len(text)
        return out


def load(model_dir: Optional[str], vocab_size: int):
    """The server's tokenizer: artifacts tokenizer.json if present, else the
    byte-level fallback."""
    if model_dir:
        p = os.path.join(model_dir, "tokenizer.json")
        if os.path.exists(p):
            try:
                return HFTokenizer(p)
            except Exception:
                pass  # corrupt file: fall back rather than fail serving
    return ByteTokenizer(vocab_size)


def write_byte_level_tokenizer(path: str) -> None:
    """Export a byte-level BPE tokenizer.json (0=eos </s>, 1=bos <s>,
    2..257 = the 256 byte-level symbols in alphabet order)."""
    from tokenizers import Tokenizer, decoders, models, pre_tokenizers
    alphabet = sorted(pre_tokenizers.ByteLevel.alphabet())
    assert len(alphabet) == 256
    vocab = {"</s>": 0, "<s>": 1}
    for i, ch in enumerate(alphabet):
        vocab[ch] = BYTE_OFFSET + i
    tk = Tokenizer(models.BPE(vocab=vocab, merges=[]))
    tk.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
    tk.decoder = decoders.ByteLevel()
    tk.save(path)
