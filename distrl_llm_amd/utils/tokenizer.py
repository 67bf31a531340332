"""Tokenizers.

The reference delegates tokenization to Unsloth/HF
(reference train_distributed.py:5,46; distributed_actor.py:217-229). Here:
a local HF tokenizer is used when a tokenizer directory is available;
otherwise (offline / synthetic-bench environments, BASELINE.json) a
deterministic byte-level tokenizer with the model's vocab size stands in.

Encoding: when the vocab is large enough, text is encoded two UTF-8 bytes
per token (id = 512 + b0 + 256*b1; a trailing odd byte maps to ids 0..255)
— ~2 chars/token, comparable to a real BPE's density on math text, so
prompt budgets behave like the reference's. Small-vocab test models fall
back to one byte per token. Out-of-range ids (sampled from random-init
models) decode to a printable escape so reward regexes operate on real
strings.
"""

from __future__ import annotations

from typing import List

_PAIR_BASE = 512
_PAIR_END = _PAIR_BASE + 256 * 256


class ByteTokenizer:
    """Deterministic, dependency-free byte-pair-of-two tokenizer."""

    def __init__(self, vocab_size: int = 152064):
        if vocab_size < 512:
            raise ValueError("vocab_size too small for ByteTokenizer")
        self.vocab_size = vocab_size
        self.pad_token_id = 256
        self.eos_token_id = 257
        self.bos_token_id = 258
        self.im_start_id = 259
        self.im_end_id = 260
        self.chat_template = None
        self.pair_mode = vocab_size >= _PAIR_END

    def encode(self, text: str, add_special_tokens: bool = False) -> List[int]:
        bs = text.encode("utf-8")
        if self.pair_mode:
            ids = [(_PAIR_BASE + bs[i] + 256 * bs[i + 1])
                   for i in range(0, len(bs) - 1, 2)]
            if len(bs) % 2:
                ids.append(bs[-1])
        else:
            ids = list(bs)
        if add_special_tokens:
            ids = [self.bos_token_id] + ids
        return ids

    def decode(self, ids, skip_special_tokens: bool = True) -> str:
        out = bytearray()
        for t in ids:
            t = int(t)
            if t < 256:
                out.append(t)
            elif t <= 260:
                if not skip_special_tokens:
                    out.extend(f"<|{t}|>".encode())
            elif self.pair_mode and _PAIR_BASE <= t < _PAIR_END:
                v = t - _PAIR_BASE
                out.append(v & 0xFF)
                out.append(v >> 8)
            else:
                # out-of-range id from a random-init model: printable escape
                out.extend(f"\\u{t:05x}".encode())
        return out.decode("utf-8", errors="replace")

    def __call__(self, text: str):
        return {"input_ids": self.encode(text)}


def load_tokenizer(model_name_or_path: str, vocab_size: int = 152064):
    """HF tokenizer if ``model_name_or_path`` is a local directory with
    tokenizer files, else ByteTokenizer (no network in this environment)."""
    import os
    if os.path.isdir(model_name_or_path):
        try:
            from transformers import AutoTokenizer
            return AutoTokenizer.from_pretrained(model_name_or_path)
        except Exception:
            pass
    return ByteTokenizer(vocab_size=vocab_size)
