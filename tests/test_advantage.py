import numpy as np

from distrl_llm_amd.rl.advantage import (even_chunk_sizes, group_advantages,
                                         group_baseline, merge_candidates,
                                         process_candidates, topk_indices)


def _group(n=4):
    # (n, 2): col 0 format, col 1 accuracy
    return np.array([[0.1, 0.0], [0.2, 1.0], [0.0, 0.0], [0.3, 1.0]])


def test_baseline_and_advantage():
    g = _group()
    summed = g.sum(axis=1)
    assert abs(group_baseline(g) - summed.mean()) < 1e-12
    adv = group_advantages(g)
    expected = (summed - summed.mean()) / (summed.std() + 1e-8)
    np.testing.assert_allclose(adv, expected)
    assert abs(adv.mean()) < 1e-9


def test_topk():
    scores = np.array([0.5, 2.0, 1.0, -1.0])
    idx = topk_indices(scores, 2)
    assert set(idx.tolist()) == {1, 2}
    # topk == n is a no-op selection (all kept)
    assert len(topk_indices(scores, 4)) == 4


def _candidates():
    return [{
        "problem": [["p0"] * 4, ["p1"] * 4],
        "answers": [["a", "b", "c", "d"], ["e", "f", "g", "h"]],
        "solution": [["s0"] * 4, ["s1"] * 4],
        "token_lengths": [[10, 20, 30, 40], [5, 5, 5, 5]],
        "rewards": [_group(), _group() * 0.5],
    }]


def test_process_candidates_grpo():
    cands, metrics = process_candidates(_candidates(), "grpo", topk=4)
    r = cands[0]["rewards"][0]
    assert len(r) == 4
    # whitened: mean ~0 (sorted by topk ordering but complete)
    assert abs(np.sum(r)) < 1e-6
    assert metrics["mean_token_length"] == (25.0 + 5.0) / 2
    # mean over groups of per-group max: (1.0 + 0.5) / 2
    assert metrics["max_accuracy_reward"] == 0.75


def test_process_candidates_pg_baseline_subtracted():
    cands, _ = process_candidates(_candidates(), "pg", topk=4)
    r = cands[0]["rewards"][0]
    # PG rewards are baselined before merging (deviation fix)
    assert abs(np.sum(r)) < 1e-9


def test_process_candidates_topk_filters():
    cands, _ = process_candidates(_candidates(), "grpo", topk=2)
    assert all(len(g) == 2 for g in cands[0]["answers"])
    assert all(len(g) == 2 for g in cands[0]["rewards"])
    assert all(len(g) == 2 for g in cands[0]["problem"])
    # kept answers are the top-reward candidates (indices 1, 3 of each group)
    assert set(cands[0]["answers"][0]) == {"b", "d"}


def test_merge_candidates():
    cands, _ = process_candidates(_candidates(), "grpo", topk=4)
    p, a, r = merge_candidates(cands)
    assert len(p) == len(a) == len(r) == 8


def test_even_chunk_sizes():
    assert even_chunk_sizes(10, 3) == [4, 3, 3]
    assert even_chunk_sizes(2, 4) == [1, 1, 0, 0]
