"""Reward functions for MATH-style <think>/<answer> completions.

Semantics match the reference's reward stack (reference reward_functions.py:
extract 4-7, correctness 9-11, soft format 20-24, xml tag count 26-41,
column-stacked reward_function 44-49): column 0 = format reward
(soft-format 0.1 + xml tag counting), column 1 = accuracy (exact string
match on the extracted <answer> block). ``strict_format_reward_func`` exists
in the reference but is never wired into ``reward_function``
(reference reward_functions.py:14-18); we keep it available for parity.
"""

from __future__ import annotations

import re
from typing import Sequence

import numpy as np

_SOFT_FORMAT_RE = re.compile(r"<think>.*?</think>\s*<answer>.*?</answer>", re.DOTALL)
_STRICT_FORMAT_RE = re.compile(r"^<think>\n.*?\n</think>\n<answer>\n.*?\n</answer>\n$", re.DOTALL)


def extract_xml_answer(text: str) -> str:
    """Pull the contents of the last <answer>...</answer> region, stripped."""
    answer = text.split("<answer>")[-1]
    answer = answer.split("</answer>")[0]
    return answer.strip()


def correctness_reward(completions: Sequence[str], solutions: Sequence[str]) -> np.ndarray:
    extracted = [extract_xml_answer(c) for c in completions]
    return np.array([1.0 if r == s else 0.0 for r, s in zip(extracted, solutions)])


def soft_format_reward(completions: Sequence[str]) -> np.ndarray:
    # Note: anchored at the start like the reference (re.match, not re.search).
    return np.array([0.1 if _SOFT_FORMAT_RE.match(c) else 0.0 for c in completions])


def strict_format_reward(completions: Sequence[str]) -> np.ndarray:
    return np.array([0.1 if _STRICT_FORMAT_RE.match(c) else 0.0 for c in completions])


def count_xml(text: str) -> float:
    """Per-tag partial credit with trailing-text penalty."""
    count = 0.0
    if text.count("<think>\n") == 1:
        count += 0.05
    if text.count("\n</think>\n") == 1:
        count += 0.05
    if text.count("\n<answer>\n") == 1:
        count += 0.05
        count -= len(text.split("\n</answer>\n")[-1]) * 0.001
    if text.count("\n</answer>") == 1:
        count += 0.05
        count -= (len(text.split("\n</answer>")[-1]) - 1) * 0.001
    return count


def xmlcount_reward(completions: Sequence[str]) -> np.ndarray:
    return np.array([count_xml(c) for c in completions])


def reward_function(completions: Sequence[str], solutions: Sequence[str]) -> np.ndarray:
    """Returns an (N, 2) array: column 0 = format (+xml) reward, column 1 = accuracy."""
    acc = correctness_reward(completions, solutions)
    form = soft_format_reward(completions) + xmlcount_reward(completions)
    return np.column_stack((form, acc))
