"""Tokenizers.

The reference delegates tokenization to Unsloth/HF
(reference train_distributed.py:5,46; distributed_actor.py:217-229). Here:
a local HF tokenizer is used when a tokenizer directory is available;
otherwise (offline / synthetic-bench environments, BASELINE.json) a
deterministic byte-level tokenizer with the model's vocab size stands in —
UTF-8 bytes map to ids 0..255, specials sit at the top of the byte range,
and out-of-byte-range ids (sampled from a random-init model) decode to a
printable escape so reward regexes operate on real strings.
"""

from __future__ import annotations

from typing import List


class ByteTokenizer:
    """Deterministic, dependency-free byte-level tokenizer."""

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

    def encode(self, text: str, add_special_tokens: bool = False) -> List[int]:
        ids = list(text.encode("utf-8"))
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
            else:
                # out-of-byte-range id from a random-init model: printable escape
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
