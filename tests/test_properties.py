"""Property-based tests (hypothesis) for the pure control-plane math."""

import numpy as np
from hypothesis import given, settings
from hypothesis import strategies as st

from distrl_llm_amd.rl.advantage import (even_chunk_sizes, group_advantages,
                                         topk_indices)
from distrl_llm_amd.rl.sched import split_dict_lists, worker_chunk_sizes
from distrl_llm_amd.utils.tokenizer import ByteTokenizer


@given(batch=st.integers(1, 500), actors=st.integers(0, 16),
       learners=st.integers(1, 16), lcs=st.integers(1, 64))
@settings(max_examples=300, deadline=None)
def test_worker_chunks_invariants(batch, actors, learners, lcs):
    sizes = worker_chunk_sizes(batch, actors, learners, lcs)
    assert len(sizes) == actors + learners
    assert sum(sizes) == batch          # every prompt dispatched exactly once
    assert all(s >= 0 for s in sizes)
    # actors are prioritized: if any actor idles, learners got nothing
    if actors and min(sizes[:actors]) == 0:
        assert all(s == 0 for s in sizes[actors:])


@given(total=st.integers(0, 1000), parts=st.integers(1, 32))
@settings(max_examples=200, deadline=None)
def test_even_chunks(total, parts):
    sizes = even_chunk_sizes(total, parts)
    assert sum(sizes) == total and len(sizes) == parts
    assert max(sizes) - min(sizes) <= 1


@given(st.lists(st.floats(-10, 10, allow_nan=False), min_size=2, max_size=64))
@settings(max_examples=200, deadline=None)
def test_advantages_whitened(rs):
    g = np.array([[r, 0.0] for r in rs])
    adv = group_advantages(g)
    assert abs(adv.mean()) < 1e-6 or np.allclose(rs, rs[0])
    assert np.isfinite(adv).all()


@given(st.lists(st.floats(-5, 5, allow_nan=False), min_size=1, max_size=32),
       st.integers(1, 32))
@settings(max_examples=200, deadline=None)
def test_topk_selects_best(scores, k):
    s = np.array(scores)
    k = min(k, len(s))
    idx = topk_indices(s, k)
    assert len(idx) == k
    kept = sorted(s[idx])
    dropped = sorted(np.delete(s, idx))
    if dropped and kept:
        assert kept[0] >= dropped[-1] - 1e-12


@given(st.text(max_size=300))
@settings(max_examples=200, deadline=None)
def test_tokenizer_roundtrip(text):
    for vocab in (152064, 2048):
        tok = ByteTokenizer(vocab)
        assert tok.decode(tok.encode(text)) == text


@given(n=st.integers(1, 40), sizes=st.lists(st.integers(0, 20), min_size=1,
                                            max_size=8))
@settings(max_examples=100, deadline=None)
def test_split_dict_lists_partition(n, sizes):
    total = sum(sizes)
    data = {"a": list(range(total)), "b": [str(i) for i in range(total)]}
    if total == 0:
        return
    chunks = split_dict_lists(data, sizes)
    flat = [x for c in chunks for x in c["a"]]
    assert flat == data["a"]


@given(
    st.lists(st.integers(min_value=1, max_value=4), min_size=1, max_size=4),
    st.integers(min_value=2, max_value=8),
    st.integers(min_value=1, max_value=8),
    st.integers(min_value=1, max_value=4),
    st.integers(min_value=0, max_value=2 ** 31 - 1),
)
@settings(max_examples=40, deadline=None)
def test_pipeline_sample_conservation(group_sizes, n, topk, learners, seed):
    """End-to-end data-path invariant: process_candidates (advantages +
    top-k) -> merge_candidates -> even_chunk_sizes must train every kept
    sample exactly once — min(topk, n) samples per prompt, partitioned
    across learners with no loss or duplication."""
    import numpy as np

    from distrl_llm_amd.rl.advantage import (even_chunk_sizes,
                                             merge_candidates,
                                             process_candidates)
    rng = np.random.default_rng(seed)
    cands = []
    uid = 0
    total_prompts = 0
    for g in group_sizes:
        total_prompts += g
        cand = {"problem": [], "answers": [], "solution": [],
                "token_lengths": [], "rewards": []}
        for _ in range(g):
            cand["problem"].append([f"p{uid}"] * n)
            cand["answers"].append([f"a{uid}_{j}" for j in range(n)])
            cand["solution"].append([f"s{uid}"] * n)
            cand["token_lengths"].append([int(x) for x in
                                          rng.integers(1, 50, n)])
            cand["rewards"].append(rng.normal(size=(n, 2)))
            uid += 1
        cands.append(cand)

    kept, _stats = process_candidates(cands, "grpo", topk)
    problems, answers, rewards = merge_candidates(kept)
    expect = total_prompts * min(topk, n)
    assert len(problems) == len(answers) == len(rewards) == expect
    # answers are globally unique -> no duplicated training sample
    assert len(set(answers)) == expect

    sizes = even_chunk_sizes(len(problems), learners)
    assert sum(sizes) == expect and len(sizes) == learners
    assert max(sizes) - min(sizes) <= 1
