"""Automatic cross-request prefix caching (EngineConfig.enable_prefix_caching
— the vLLM APC analogue, absent in the reference but part of the vLLM
surface it runs on). Full prompt blocks persist in the pool under an LRU
keyed by their token prefix; later requests sharing the prefix reuse the
blocks and prefill only the tail THROUGH THE DECODE STEP (existing
kernels only — no new attention surface). Correctness contract: outputs
are identical to an uncached engine, block accounting is exact, and the
cache yields to pool pressure (LRU eviction)."""

import os

import pytest
import torch

from distrl_llm_amd.config import EngineConfig, SamplingParams
from distrl_llm_amd.engine import Engine
from distrl_llm_amd.models import CausalLM, get_spec


@pytest.fixture(scope="module")
def model():
    m = CausalLM(get_spec("tiny-qwen2"), lora_r=4, lora_alpha=8,
                 dtype=torch.float32)
    m.random_init(seed=51)
    return m


def _engine(model, cache=True, blocks=256, **over):
    cfg = EngineConfig(max_seq_length=96, kv_block_size=8,
                       num_kv_blocks=blocks, max_num_seqs=32,
                       enable_prefix_caching=cache)
    for k, v in over.items():
        setattr(cfg, k, v)
    return Engine(model, cfg, device=torch.device("cpu"), seed=0)


SYS = list(range(100, 125))  # 25-token "system prompt" (3 full blocks @ 8)


def test_cached_outputs_identical_and_blocks_reused(model):
    """Second call sharing a long prefix: same greedy outputs as an
    uncached engine, with the shared blocks actually reused."""
    prompts_a = [SYS + [7, 8, 9], SYS + [1, 2]]
    prompts_b = [SYS + [4, 4, 4, 4], SYS + [9]]
    sp = SamplingParams(max_tokens=6, temperature=0.0, n=2)

    plain = _engine(model, cache=False)
    exp_a = plain.generate(prompts_a, sp, eos_token_id=None)
    exp_b = plain.generate(prompts_b, sp, eos_token_id=None)

    eng = _engine(model, cache=True)
    got_a = eng.generate(prompts_a, sp, eos_token_id=None)
    assert got_a == exp_a
    assert len(eng._prefix_cache) > 0
    hits0 = eng._prefix_hits
    got_b = eng.generate(prompts_b, sp, eos_token_id=None)
    assert got_b == exp_b
    # prompts_b reused the 3 SYS blocks per prompt (cap leaves >=1 tail tok)
    assert eng._prefix_hits - hits0 >= 3 * len(prompts_b)
    # all non-cache blocks returned; cache holds exactly its entries
    assert (eng.pool.allocator.num_free
            == eng.pool.num_blocks - len(eng._prefix_cache))
    eng.clear_prefix_cache()
    assert eng.pool.allocator.num_free == eng.pool.num_blocks


def test_session_path_with_cache(model, monkeypatch):
    """Same equality through the session/graph state machine."""
    monkeypatch.setenv("DISTRL_FORCE_SESSION", "1")
    prompts = [SYS + [3, 1], SYS + [5, 5, 5]]
    sp = SamplingParams(max_tokens=5, temperature=0.0, n=1)
    plain = _engine(model, cache=False)
    exp = plain.generate(prompts, sp, eos_token_id=None)
    eng = _engine(model, cache=True)
    eng.generate([SYS + [9, 9]], sp, eos_token_id=None)  # warm the cache
    assert eng.generate(prompts, sp, eos_token_id=None) == exp


def test_fully_cached_prompt_keeps_one_tail_token(model):
    """A prompt whose length is an exact multiple of the block size and
    fully cached still recomputes >= 1 tail token (logits source)."""
    p = SYS[:24]  # exactly 3 blocks
    sp = SamplingParams(max_tokens=4, temperature=0.0, n=1)
    plain = _engine(model, cache=False)
    exp = plain.generate([p], sp, eos_token_id=None)
    eng = _engine(model, cache=True)
    eng.generate([p], sp, eos_token_id=None)
    got = eng.generate([p], sp, eos_token_id=None)  # now a cache hit
    assert got == exp
    # at most 2 of the 3 blocks may be reused ((L-1)//bs cap)
    assert eng._prefix_hits <= 2


def test_eviction_under_pool_pressure(model):
    """A tiny pool: the cache yields blocks via LRU eviction instead of
    starving admission, and outputs stay correct."""
    sp = SamplingParams(max_tokens=4, temperature=0.0, n=1)
    plain = _engine(model, cache=False, blocks=16)
    eng = _engine(model, cache=True, blocks=16)
    rounds = []
    for s in range(10):  # distinct long prompts fill + churn the cache
        p = [200 + s] * 20 + [s, s + 1]
        rounds.append((p, plain.generate([p], sp, eos_token_id=None)))
        assert eng.generate([p], sp, eos_token_id=None) == rounds[-1][1]
    assert eng._prefix_evicts > 0
    # replay the first round (likely evicted): still correct
    p0, exp0 = rounds[0]
    assert eng.generate([p0], sp, eos_token_id=None) == exp0
    eng.clear_prefix_cache()
    assert eng.pool.allocator.num_free == eng.pool.num_blocks


def test_cache_with_eos_and_limits(model):
    """Cache interacts with EOS and per-candidate caps like a plain
    engine (same generate semantics, just faster prefill)."""
    prompts = [SYS + [2, 3], SYS + [8]]
    sp = SamplingParams(max_tokens=6, temperature=0.0, n=2)
    plain = _engine(model, cache=False)
    limits = [[2, 5], [6, 1]]
    exp = plain.generate(prompts, sp, eos_token_id=None,
                         token_limits=limits)
    eng = _engine(model, cache=True)
    eng.generate([SYS + [1]], sp, eos_token_id=None)  # warm
    got = eng.generate(prompts, sp, eos_token_id=None, token_limits=limits)
    assert got == exp
