"""Property-based engine fuzzing (CPU): the paged-KV continuous-batching
engine must agree with naive full-recompute decoding for ANY prompt set,
block size and pool size, and must always return every block to the pool.
Complements the targeted cases in test_engine.py."""

import pytest
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from distrl_llm_amd.config import EngineConfig, SamplingParams
from distrl_llm_amd.engine import Engine
from distrl_llm_amd.models import CausalLM, get_spec


@pytest.fixture(scope="module")
def model():
    spec = get_spec("tiny-qwen2")
    m = CausalLM(spec, lora_r=4, lora_alpha=8, dtype=torch.float32)
    m.random_init(seed=42)
    return m


def _naive_greedy(model, prompt_ids, steps, eos=None):
    ids = list(prompt_ids)
    out = []
    for _ in range(steps):
        logits = model(torch.tensor([ids]))[0, -1]
        t = int(logits.argmax())
        out.append(t)
        ids.append(t)
        if eos is not None and t == eos:
            break
    return out


@settings(max_examples=15, deadline=None)
@given(
    prompts=st.lists(
        st.lists(st.integers(min_value=1, max_value=500), min_size=1,
                 max_size=24),
        min_size=1, max_size=5),
    block_size=st.sampled_from([4, 8, 16]),
    num_blocks=st.integers(min_value=30, max_value=200),
    max_tokens=st.integers(min_value=1, max_value=8),
    n=st.integers(min_value=1, max_value=3),
)
def test_greedy_equals_naive_any_shape(model, prompts, block_size, num_blocks,
                                       max_tokens, n):
    cfg = EngineConfig(max_seq_length=64, kv_block_size=block_size,
                       num_kv_blocks=num_blocks, max_num_seqs=64)
    engine = Engine(model, cfg, device=torch.device("cpu"), seed=0)
    sp = SamplingParams(max_tokens=max_tokens, temperature=0.0, n=n)
    results = engine.generate(prompts, sp, eos_token_id=None)
    assert engine.pool.allocator.num_free == num_blocks
    for p, res in zip(prompts, results):
        steps = min(max_tokens, cfg.max_seq_length - len(p))
        expected = _naive_greedy(model, p, steps)
        assert len(res) == n
        for ids in res:
            assert ids == expected, (p, ids, expected)


@settings(max_examples=10, deadline=None)
@given(
    seed=st.integers(min_value=0, max_value=2**31 - 1),
    temperature=st.floats(min_value=0.3, max_value=2.0),
    top_p=st.floats(min_value=0.3, max_value=1.0),
)
def test_sampling_reproducible_across_engines(model, seed, temperature, top_p):
    """Same engine seed => identical sampled streams (the checkpoint
    bit-identity guarantee rests on this)."""
    prompts = [[3, 1, 4, 1, 5], [9, 2, 6]]
    sp = SamplingParams(max_tokens=6, temperature=temperature, top_p=top_p,
                       n=2)
    outs = []
    for _ in range(2):
        cfg = EngineConfig(max_seq_length=64, kv_block_size=8,
                           num_kv_blocks=64, max_num_seqs=16)
        engine = Engine(model, cfg, device=torch.device("cpu"), seed=seed)
        outs.append(engine.generate(prompts, sp, eos_token_id=None))
    assert outs[0] == outs[1]


@settings(max_examples=10, deadline=None)
@given(eos_prompt=st.lists(st.integers(min_value=1, max_value=500),
                           min_size=1, max_size=10))
def test_eos_mid_batch_frees_blocks(model, eos_prompt):
    """A sequence hitting EOS while others continue must terminate with
    EOS included and the pool must fully drain afterwards."""
    first = _naive_greedy(model, eos_prompt, 1)[0]
    cfg = EngineConfig(max_seq_length=64, kv_block_size=8, num_kv_blocks=64,
                       max_num_seqs=16)
    engine = Engine(model, cfg, device=torch.device("cpu"), seed=0)
    sp = SamplingParams(max_tokens=6, temperature=0.0, n=1)
    res = engine.generate([eos_prompt, [7, 7, 7, 7]], sp, eos_token_id=first)
    assert res[0][0][-1] == first  # terminated by EOS (possibly step 1)
    assert len(res[0][0]) <= 6
    assert engine.pool.allocator.num_free == 64


@settings(max_examples=10, deadline=None)
@given(
    prompts=st.lists(
        st.lists(st.integers(min_value=1, max_value=500), min_size=1,
                 max_size=20),
        min_size=1, max_size=4),
    max_tokens=st.integers(min_value=1, max_value=6),
)
def test_llama_family_greedy_equals_naive(prompts, max_tokens):
    """The Llama architecture branch (no qkv bias, untied lm_head,
    different rope theta) through the same engine paths."""
    m = _llama_model()
    cfg = EngineConfig(max_seq_length=64, kv_block_size=8, num_kv_blocks=128,
                       max_num_seqs=32)
    engine = Engine(m, cfg, device=torch.device("cpu"), seed=0)
    sp = SamplingParams(max_tokens=max_tokens, temperature=0.0, n=2)
    results = engine.generate(prompts, sp, eos_token_id=None)
    for p, res in zip(prompts, results):
        expected = _naive_greedy(m, p, max_tokens)
        for ids in res:
            assert ids == expected
    assert engine.pool.allocator.num_free == 128


_LLAMA = {}


def _llama_model():
    if "m" not in _LLAMA:
        m = CausalLM(get_spec("tiny-llama"), lora_r=4, lora_alpha=8,
                     dtype=torch.float32)
        m.random_init(seed=77)
        assert m.lm_head is not None  # untied head actually exercised
        _LLAMA["m"] = m
    return _LLAMA["m"]


@settings(max_examples=8, deadline=None)
@given(
    prompts=st.lists(
        st.lists(st.integers(min_value=1, max_value=500), min_size=1,
                 max_size=20),
        min_size=1, max_size=4),
    max_tokens=st.integers(min_value=1, max_value=6),
)
def test_mistral_family_greedy_equals_naive(prompts, max_tokens):
    """The Mistral (v0.3-style) architecture branch — same stack as
    Llama but its own spec/name resolution and HF round-trip — through
    the same engine paths."""
    if "m" not in _MISTRAL:
        m = CausalLM(get_spec("tiny-mistral"), lora_r=4, lora_alpha=8,
                     dtype=torch.float32)
        m.random_init(seed=78)
        _MISTRAL["m"] = m
    m = _MISTRAL["m"]
    cfg = EngineConfig(max_seq_length=64, kv_block_size=8, num_kv_blocks=128,
                       max_num_seqs=32)
    engine = Engine(m, cfg, device=torch.device("cpu"), seed=0)
    sp = SamplingParams(max_tokens=max_tokens, temperature=0.0, n=2)
    results = engine.generate(prompts, sp, eos_token_id=None)
    for p, res in zip(prompts, results):
        expected = _naive_greedy(m, p, max_tokens)
        for ids in res:
            assert ids == expected
    assert engine.pool.allocator.num_free == 128


_MISTRAL = {}


@settings(max_examples=8, deadline=None)
@given(
    prompts=st.lists(
        st.lists(st.integers(min_value=1, max_value=500), min_size=1,
                 max_size=20),
        min_size=1, max_size=4),
    max_tokens=st.integers(min_value=1, max_value=6),
)
def test_qwen3_family_greedy_equals_naive(prompts, max_tokens):
    """The Qwen3 architecture branch (per-head q/k RMSNorm before RoPE,
    q_size != hidden) through the same engine paths — including the
    session state machine (DISTRL_FORCE_SESSION flips inside the run)."""
    import os
    if "m" not in _QWEN3:
        m = CausalLM(get_spec("tiny-qwen3"), lora_r=4, lora_alpha=8,
                     dtype=torch.float32)
        m.random_init(seed=79)
        with torch.no_grad():  # non-trivial norms
            for layer in m.model.layers:
                at = layer.self_attn
                at.q_norm.weight.add_(
                    torch.rand(at.q_norm.weight.shape) * 0.5 - 0.25)
                at.k_norm.weight.add_(
                    torch.rand(at.k_norm.weight.shape) * 0.5 - 0.25)
        _QWEN3["m"] = m
    m = _QWEN3["m"]
    cfg = EngineConfig(max_seq_length=64, kv_block_size=8, num_kv_blocks=128,
                       max_num_seqs=32)
    for force in ("0", "1"):
        os.environ["DISTRL_FORCE_SESSION"] = force
        try:
            engine = Engine(m, cfg, device=torch.device("cpu"), seed=0)
            sp = SamplingParams(max_tokens=max_tokens, temperature=0.0, n=2)
            results = engine.generate(prompts, sp, eos_token_id=None)
            for p, res in zip(prompts, results):
                expected = _naive_greedy(m, p, max_tokens)
                for ids in res:
                    assert ids == expected, f"force={force}"
            assert engine.pool.allocator.num_free == 128
        finally:
            os.environ.pop("DISTRL_FORCE_SESSION", None)


_QWEN3 = {}
