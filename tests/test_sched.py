import pytest

from distrl_llm_amd.rl.sched import (calculate_chunk_sizes, split_dict_lists,
                                     worker_chunk_sizes)


def test_standard_split():
    # reference default: batch 30, 2 actors, 1 learner, learner_chunk 8
    assert calculate_chunk_sizes(30, 2, 1, 8) == [11, 11, 8]
    assert worker_chunk_sizes(30, 2, 1, 8) == [11, 11, 8]


def test_remainder_to_first_actors():
    # actor_size 23 over 3 actors -> 8, 8, 7
    assert calculate_chunk_sizes(31, 3, 1, 8) == [8, 8, 7, 8]


def test_learners_only():
    assert calculate_chunk_sizes(16, 0, 2, 8) == [8, 8]
    assert worker_chunk_sizes(16, 0, 2, 8) == [8, 8]


def test_degraded_learner_shrinks():
    # batch 5 < 4 actors + 2*8: actors prioritized, learner chunk shrinks
    assert worker_chunk_sizes(5, 4, 2, 8) == [1, 1, 1, 1, 1, 0]
    assert sum(worker_chunk_sizes(5, 4, 2, 8)) == 5


def test_degraded_learners_dropped():
    # batch == num_actors: learners idle, no double dispatch
    w = worker_chunk_sizes(2, 2, 1, 8)
    assert w == [1, 1, 0]
    assert calculate_chunk_sizes(2, 2, 1, 8) == [1, 1]


def test_degraded_too_few_for_actors():
    w = worker_chunk_sizes(2, 3, 1, 8)
    assert w == [1, 1, 0, 0]


def test_invalid_inputs():
    with pytest.raises(ValueError):
        calculate_chunk_sizes(0, 2, 1, 8)
    with pytest.raises(ValueError):
        calculate_chunk_sizes(10, -1, 1, 8)
    with pytest.raises(ValueError):
        calculate_chunk_sizes(10, 2, 0, 8)


def test_split_dict_lists():
    data = {"problem": list(range(10)), "solution": list("abcdefghij")}
    chunks = split_dict_lists(data, [4, 3, 3])
    assert [len(c["problem"]) for c in chunks] == [4, 3, 3]
    assert chunks[0]["problem"] == [0, 1, 2, 3]
    assert chunks[2]["solution"] == list("hij")
    with pytest.raises(ValueError):
        split_dict_lists(data, [4, 4])
    with pytest.raises(ValueError):
        split_dict_lists({"a": [1, 2], "b": [1]}, [2])
