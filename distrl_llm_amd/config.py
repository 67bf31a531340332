"""Config dataclasses.

Native replacements for the external config surfaces the reference imports:
``vllm.SamplingParams`` (reference distributed_actor.py:13,43-48) and
``transformers.GenerationConfig`` (reference distributed_trainer.py:8,22-28).
Only the fields the reference actually uses are first-class; everything else
is rejected loudly rather than silently ignored.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional


@dataclass
class SamplingParams:
    """Sampling parameters for the generation engine.

    Field-compatible with the subset of vLLM ``SamplingParams`` the reference
    constructs (max_tokens, temperature, n, top_p — distributed_actor.py:43-48
    and distributed_trainer.py:53-58).
    """

    max_tokens: int = 16
    temperature: float = 1.0
    n: int = 1
    top_p: float = 1.0
    top_k: int = 0  # 0 = disabled
    seed: Optional[int] = None
    # optional realistic-length mode: each candidate draws an exponential
    # output cap with this mean (clamped to [1, max_tokens]) — used by the
    # EOS-realistic bench so in-wave retirement is measured, not assumed
    geom_len_mean: Optional[float] = None

    def __post_init__(self) -> None:
        if self.max_tokens <= 0:
            raise ValueError(f"max_tokens must be positive, got {self.max_tokens}")
        if self.n <= 0:
            raise ValueError(f"n must be positive, got {self.n}")
        if not (0.0 < self.top_p <= 1.0):
            raise ValueError(f"top_p must be in (0, 1], got {self.top_p}")
        if self.temperature < 0.0:
            raise ValueError(f"temperature must be >= 0, got {self.temperature}")


@dataclass
class GenerationConfig:
    """Generation config, mirroring the fields the reference Trainer builds
    (distributed_trainer.py:22-28)."""

    max_new_tokens: int = 16
    temperature: float = 1.0
    num_return_sequences: int = 1
    do_sample: bool = True
    use_cache: bool = True

    def to_sampling_params(self, top_p: float = 0.95) -> SamplingParams:
        """Translate to engine SamplingParams the way BaseActor does
        (reference distributed_actor.py:41-48: top_p hardcoded 0.95)."""
        return SamplingParams(
            max_tokens=self.max_new_tokens,
            temperature=self.temperature if self.do_sample else 0.0,
            n=self.num_return_sequences,
            top_p=top_p,
        )


@dataclass
class EngineConfig:
    """Per-GPU generation-engine configuration.

    ``gpu_memory_utilization`` mirrors the reference's per-role GPU fractions
    (train_distributed.py:34-35, distributed_actor.py:65) but is re-derived
    for 288 GB HBM3E: it bounds the KV block pool, not the whole process.
    """

    max_seq_length: int = 2048
    kv_block_size: int = 16
    gpu_memory_utilization: float = 0.9
    max_num_seqs: int = 4096
    # When >0, cap on number of KV blocks (used by CPU tests); 0 = derive
    # from free memory * gpu_memory_utilization.
    num_kv_blocks: int = 0
    enforce_eager: bool = False  # True disables hipGraph capture of decode
    # Automatic cross-request prefix caching (vLLM APC analogue): full
    # prompt blocks are kept in the pool keyed by their token prefix and
    # reused by later requests sharing the prefix (classic win: a long
    # system prompt + short user tails). Cache-hit tails prefill through
    # the decode step (existing kernels only); LRU-evicted under pool
    # pressure. Off by default: RL rollouts use unique prompts.
    enable_prefix_caching: bool = False
