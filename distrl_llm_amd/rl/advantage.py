"""Baseline / advantage / top-k math on candidate groups.

Numerics match the reference's per-prompt-group reward processing
(reference distributed_trainer.py:262-294): the per-group baseline is the
mean of summed (format + accuracy) rewards, the GRPO advantage is the
whitened summed reward ``(R - mean) / (std + 1e-8)`` over all n candidates
of a group, computed BEFORE top-k filtering, and the top-k filter keeps the
``topk`` highest-total-reward candidates per group via argsort.

Deviation (documented): the reference's multi-learner PG
path drops the baseline when flattening (merge_candidates loses the
"baselines" key, distributed_trainer.py:221-230), training PG on raw summed
rewards. Here the PG baseline is subtracted before any flattening, so
single- and multi-learner PG optimize the same objective.
"""

from __future__ import annotations

from typing import Dict, List, Tuple

import numpy as np


def group_baseline(group_rewards: np.ndarray) -> float:
    """Mean summed reward over a (n, 2) group reward array."""
    return float(np.mean(group_rewards.sum(axis=1)))


def group_advantages(group_rewards: np.ndarray) -> np.ndarray:
    """Whitened summed rewards over one prompt group (GRPO advantages)."""
    summed = group_rewards.sum(axis=1)
    return (summed - np.mean(summed)) / (np.std(summed) + 1e-8)


def topk_indices(scores: np.ndarray, topk: int) -> np.ndarray:
    """Indices of the top-k scores, in the reference's argsort order
    (ascending, last k). Ties are resolved by a stable sort here (lowest
    index first within a tie); the reference's np.argsort default is an
    unstable quicksort, so tied-reward candidates may differ from it —
    equivalent in expectation, deterministic here (an improvement, not
    exact tie parity). Reference: distributed_trainer.py:287."""
    return np.argsort(scores, kind="stable")[-topk:]


def process_candidates(candidates: List[Dict], learner_type: str, topk: int) -> Tuple[List[Dict], Dict]:
    """Convert raw per-worker candidate dicts (with "rewards": list of (n,2)
    arrays) into training candidates + round metrics.

    After this call each candidate dict's "rewards" holds per-group 1-D
    arrays of training scalars (GRPO: whitened advantages; PG: summed reward
    minus the group baseline), top-k filtered.
    Returns (candidates, metrics) where metrics carries the reference's
    logging statistics (distributed_trainer.py:262-274,348-366).
    """
    mean_acc, mean_form, mean_tok, max_acc, min_acc = [], [], [], [], []

    for cand in candidates:
        new_rewards = []
        for group_rewards, group_tokens in zip(cand["rewards"], cand["token_lengths"]):
            group_rewards = np.asarray(group_rewards)
            mean_acc.append(np.mean(group_rewards[:, 1]))
            max_acc.append(np.max(group_rewards[:, 1]))
            min_acc.append(np.min(group_rewards[:, 1]))
            mean_form.append(np.mean(group_rewards[:, 0]))
            mean_tok.append(np.mean(group_tokens))
            if learner_type == "grpo":
                new_rewards.append(group_advantages(group_rewards))
            else:
                summed = group_rewards.sum(axis=1)
                new_rewards.append(summed - group_baseline(group_rewards))
        cand["rewards"] = new_rewards

        # top-k filter per group (reference distributed_trainer.py:282-294)
        f_answers, f_rewards, f_problems = [], [], []
        for j, scores in enumerate(cand["rewards"]):
            idx = topk_indices(scores, topk)
            f_answers.append([cand["answers"][j][i] for i in idx])
            f_rewards.append(scores[idx])
            f_problems.append(cand["problem"][j][:topk])
        cand["answers"] = f_answers
        cand["rewards"] = f_rewards
        cand["problem"] = f_problems

    metrics = {
        "mean_format_reward": float(np.mean(mean_form)) if mean_form else 0.0,
        "mean_accuracy_reward": float(np.mean(mean_acc)) if mean_acc else 0.0,
        "min_accuracy_reward": float(np.mean(min_acc)) if min_acc else 0.0,
        "max_accuracy_reward": float(np.mean(max_acc)) if max_acc else 0.0,
        "mean_token_length": float(np.mean(mean_tok)) if mean_tok else 0.0,
    }
    return candidates, metrics


def merge_candidates(candidates: List[Dict]) -> Tuple[list, list, list]:
    """Flatten candidate groups into parallel (problems, answers, rewards)
    lists (reference distributed_trainer.py:221-230). Rewards here are
    already the per-sample training scalars (advantages / baselined)."""
    problems, answers, rewards = [], [], []
    for cand in candidates:
        for a, p, r in zip(cand["answers"], cand["problem"], cand["rewards"]):
            problems.extend(p)
            answers.extend(a)
            rewards.extend(list(r))
    return problems, answers, rewards


def even_chunk_sizes(total: int, parts: int) -> List[int]:
    """Even split with remainder +1 to the first chunks
    (reference distributed_trainer.py:312-322)."""
    base, extra = divmod(total, parts)
    return [base + 1 if i < extra else base for i in range(parts)]
