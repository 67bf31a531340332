"""Dataset preparation, prompting and synthetic data.

Mirrors the reference's data layer (reference helper.py:3-23,
train_distributed.py:38-48): R1-style system prompt, chat templating with a
generation prompt, and the MATH answer->solution column remap. Because this
environment has no network, a deterministic synthetic MATH-shaped prompt
generator is provided for benches and tests (BASELINE.json: synthetic
prompts / random-init weights).
"""

from __future__ import annotations

import random
from typing import Dict, List, Optional, Sequence

r1_preprompt = (
    "A conversation between User and Assistant. The user asks a question, and the Assistant solves it.\n"
    "The assistant first thinks about the reasoning process and then provides the user with the answer.\n"
    "The response must follow this format:\n"
    "<think> reasoning process here </think>\n"
    "<answer> answer here </answer>\n"
)


def default_chat_template(messages: List[Dict[str, str]], add_generation_prompt: bool = True) -> str:
    """ChatML-style template (the Qwen2 chat format) used when no tokenizer
    template is available (offline / synthetic runs)."""
    out = []
    for m in messages:
        out.append(f"<|im_start|>{m['role']}\n{m['content']}<|im_end|>\n")
    if add_generation_prompt:
        out.append("<|im_start|>assistant\n")
    return "".join(out)


def apply_template(tokenizer, messages: List[Dict[str, str]]) -> str:
    if tokenizer is not None and getattr(tokenizer, "chat_template", None):
        return tokenizer.apply_chat_template(messages, add_generation_prompt=True, tokenize=False)
    return default_chat_template(messages, add_generation_prompt=True)


def process_dataset(tokenizer, dataset, preprompt: str = "", postprompt: str = ""):
    """Chat-template every problem (reference helper.py:11-23).

    ``dataset`` is either an HF dataset (has .map) or a list of dicts with
    "problem" / "solution" keys; returns the same kind of object with
    "problem" replaced by the templated prompt string.
    """
    def to_messages(problem: str):
        return [
            {"role": "system", "content": preprompt},
            {"role": "user", "content": problem + " " + postprompt},
        ]

    if hasattr(dataset, "map"):
        def generate_messages(examples):
            return {"problem": [apply_template(tokenizer, to_messages(p))
                                for p in examples["problem"]]}
        return dataset.map(generate_messages, batched=True)

    out = []
    for row in dataset:
        row = dict(row)
        row["problem"] = apply_template(tokenizer, to_messages(row["problem"]))
        out.append(row)
    return out


_TEMPLATES = [
    ("Let $x = {a}$ and $y = {b}$. Compute $x + y \\cdot {c}$.", lambda a, b, c: a + b * c),
    ("What is the remainder when ${a}^{{{b}}}$ is divided by ${c}$?", lambda a, b, c: pow(a, b, c)),
    ("Evaluate $\\gcd({a}, {b}) + {c}$.", None),
    ("A sequence satisfies $a_1 = {a}$ and $a_{{n+1}} = a_n + {b}$. Find $a_{{{c}}}$.",
     lambda a, b, c: a + b * (c - 1)),
    ("Compute the sum of the first ${a}$ positive multiples of ${b}$, minus ${c}$.",
     lambda a, b, c: b * a * (a + 1) // 2 - c),
]


def synthetic_math_dataset(n: int, seed: int = 0, pad_words: int = 0) -> List[Dict[str, str]]:
    """Deterministic synthetic MATH-500-shaped rows: {"problem", "solution"}.

    Shapes match the reference dataset statistics (prompt mean ~144 tokens,
    max 865 — reference train_distributed.py:18 comment); ``pad_words``
    appends filler context to lengthen prompts.
    """
    import math
    rng = random.Random(seed)
    rows = []
    for i in range(n):
        tmpl, fn = _TEMPLATES[i % len(_TEMPLATES)]
        a, b, c = rng.randint(2, 50), rng.randint(2, 9), rng.randint(2, 97)
        problem = tmpl.format(a=a, b=b, c=c)
        if fn is None:
            sol = math.gcd(a, b) + c
        else:
            sol = fn(a, b, c)
        if pad_words:
            filler = " ".join(f"w{rng.randint(0, 999)}" for _ in range(pad_words))
            problem = problem + " Context: " + filler
        rows.append({"problem": problem, "solution": str(sol)})
    return rows


def load_local_rows(path: str) -> List[Dict[str, str]]:
    """Load a local dataset file into [{"problem", "solution"}, ...] rows
    — the air-gapped counterpart of the reference's HF `load_dataset`
    (reference train_distributed.py:38-44). Accepts .json (list of
    objects), .jsonl, or .parquet; a directory is scanned for the first
    such file. MATH-style "answer" columns are remapped to "solution"
    (same remap as the reference)."""
    import json as _json
    import os as _os

    if _os.path.isdir(path):
        for f in sorted(_os.listdir(path)):
            if f.endswith((".json", ".jsonl", ".parquet")):
                return load_local_rows(_os.path.join(path, f))
        raise FileNotFoundError(f"no .json/.jsonl/.parquet file in {path}")

    if path.endswith(".jsonl"):
        with open(path) as fh:
            raw = [_json.loads(l) for l in fh if l.strip()]
    elif path.endswith(".json"):
        with open(path) as fh:
            raw = _json.load(fh)
        if isinstance(raw, dict):  # {"data": [...]} wrappers
            raw = next(v for v in raw.values() if isinstance(v, list))
    elif path.endswith(".parquet"):
        import pandas as pd
        raw = pd.read_parquet(path).to_dict("records")
    else:
        raise ValueError(f"unsupported dataset file type: {path}")

    rows = []
    for r in raw:
        sol = r.get("solution", r.get("answer"))
        if r.get("problem") is None or sol is None:
            raise ValueError(
                f"{path}: rows need 'problem' and 'solution' (or 'answer') "
                f"columns, got {sorted(r)}")
        rows.append({"problem": str(r["problem"]), "solution": str(sol)})
    return rows


class ListDataset:
    """Minimal stand-in for the HF dataset surface the Trainer uses
    (reference distributed_trainer.py:245-246,386: .shuffle() and
    .iter(batch_size) yielding dicts of lists)."""

    def __init__(self, rows: Sequence[Dict], seed: Optional[int] = None):
        self.rows = list(rows)
        self._rng = random.Random(seed)

    def __len__(self):
        return len(self.rows)

    def shuffle(self, seed: Optional[int] = None):
        rng = random.Random(seed) if seed is not None else self._rng
        rows = list(self.rows)
        rng.shuffle(rows)
        return ListDataset(rows)

    def iter(self, batch_size: int):
        for start in range(0, len(self.rows), batch_size):
            chunk = self.rows[start:start + batch_size]
            keys = chunk[0].keys()
            yield {k: [row[k] for row in chunk] for k in keys}
