"""Batch chunking across the heterogeneous actor/learner pool.

Reproduces the sizing rules of the reference Trainer
(reference distributed_trainer.py:77-169): learners each take a fixed
``learner_chunk_size`` prompts, actors evenly split the remainder with +1
going to the first ``extra`` actors, and when the batch is too small actors
are prioritized (each gets >=1 prompt) while learner chunks shrink or drop
to zero.

Deviation (documented): in the degraded small-batch regime the reference
hands ``chunked_batch[-num_learners:]`` to learners even when no learner
chunks exist, so the last actor chunks are generated twice
(reference distributed_trainer.py:194-197). Here ``worker_chunk_sizes``
always returns one entry per worker (zero where a worker idles), so no
prompt is dispatched twice.
"""

from __future__ import annotations

from typing import Dict, List, Sequence


def _degraded_plan(batch_size: int, num_actors: int, num_learners: int,
                   learner_chunk_size: int):
    """Apply the small-batch degradation rules; returns the effective
    (num_actors, num_learners, learner_chunk_size)."""
    if batch_size >= num_actors:
        remaining = batch_size - num_actors
        if remaining > 0 and num_learners > 0:
            learner_chunk_size = max(1, remaining // num_learners)
            num_learners = min(num_learners, remaining // learner_chunk_size)
        else:
            num_learners = 0
    else:
        num_actors = batch_size
        num_learners = 0
    return num_actors, num_learners, learner_chunk_size


def calculate_chunk_sizes(batch_size: int, num_actors: int, num_learners: int = 1,
                          learner_chunk_size: int = 1) -> List[int]:
    """Chunk sizes for actors followed by learners (compacted: degraded
    workers are omitted, matching the reference's return shape)."""
    if batch_size <= 0 or num_learners <= 0 or num_actors < 0:
        raise ValueError(
            "Batch size, number of learners and number of actors must be positive")

    eff_actors, eff_learners, eff_lcs = num_actors, num_learners, learner_chunk_size
    if batch_size < num_actors + learner_chunk_size * num_learners:
        eff_actors, eff_learners, eff_lcs = _degraded_plan(
            batch_size, num_actors, num_learners, learner_chunk_size)

    total_learner = eff_lcs * eff_learners
    actor_size = batch_size - total_learner

    chunks: List[int] = []
    if eff_actors > 0:
        base, extra = divmod(actor_size, eff_actors)
        chunks = [base + 1 if i < extra else base for i in range(eff_actors)]
    if eff_learners > 0:
        chunks.extend([eff_lcs] * eff_learners)
    return chunks


def worker_chunk_sizes(batch_size: int, num_actors: int, num_learners: int = 1,
                       learner_chunk_size: int = 1) -> List[int]:
    """Per-worker chunk sizes, always length ``num_actors + num_learners``
    (zeros for idled workers). Sum equals ``batch_size``."""
    if batch_size <= 0 or num_learners <= 0 or num_actors < 0:
        raise ValueError(
            "Batch size, number of learners and number of actors must be positive")

    eff_actors, eff_learners, eff_lcs = num_actors, num_learners, learner_chunk_size
    if batch_size < num_actors + learner_chunk_size * num_learners:
        eff_actors, eff_learners, eff_lcs = _degraded_plan(
            batch_size, num_actors, num_learners, learner_chunk_size)

    total_learner = eff_lcs * eff_learners
    actor_size = batch_size - total_learner

    actor_chunks = [0] * num_actors
    if eff_actors > 0:
        base, extra = divmod(actor_size, eff_actors)
        for i in range(eff_actors):
            actor_chunks[i] = base + 1 if i < extra else base
    learner_chunks = [0] * num_learners
    for i in range(eff_learners):
        learner_chunks[i] = eff_lcs

    # With no actors to absorb it, the remainder (batch not divisible by
    # the learner chunks — e.g. a dataset's last batch) goes +1 to the
    # first learners. The reference drops it and then crashes in
    # split_dict_lists (sum mismatch) — found by property test.
    leftover = batch_size - sum(actor_chunks) - sum(learner_chunks)
    i = 0
    while leftover > 0 and num_learners > 0:
        learner_chunks[i % num_learners] += 1
        leftover -= 1
        i += 1

    out = actor_chunks + learner_chunks
    assert sum(out) == batch_size, (out, batch_size)
    return out


def split_dict_lists(data: Dict[str, Sequence], chunk_sizes) -> List[Dict[str, list]]:
    """Split a dict of parallel lists into per-worker dicts.

    Same contract as the reference (distributed_trainer.py:142-169): all
    values must be equal-length lists and chunk sizes must sum to that
    length.
    """
    if isinstance(chunk_sizes, int):
        chunk_sizes = [chunk_sizes]

    list_length = len(next(iter(data.values())))
    if not all(len(v) == list_length for v in data.values()):
        raise ValueError("All lists in the dictionary must have the same length")
    if sum(chunk_sizes) != list_length:
        raise ValueError(
            f"Sum of chunk sizes ({sum(chunk_sizes)}) must equal the length of lists ({list_length})")

    chunks = []
    start = 0
    for size in chunk_sizes:
        end = start + size
        chunks.append({k: list(v[start:end]) for k, v in data.items()})
        start = end
    return chunks
