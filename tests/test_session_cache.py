"""Decode-session graph cache (DISTRL_GRAPH_CACHE=1, default off —
docs/ROADMAP.md #4): cached state buffers + wave padding must decode
identically to the per-wave session path, reuse state across waves, and
never leak pool blocks (including the padding lanes' scratch block).
On GPU the same machinery also reuses the captured hipGraph; that
enablement is a round-2 validation item — here the full state machine
runs on CPU via DISTRL_FORCE_SESSION."""

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


@pytest.fixture
def cached_session(monkeypatch):
    monkeypatch.setenv("DISTRL_FORCE_SESSION", "1")
    monkeypatch.setenv("DISTRL_GRAPH_CACHE", "1")


def _naive(model, prompt, steps):
    ids = list(prompt)
    out = []
    for _ in range(steps):
        t = int(model(torch.tensor([ids]))[0, -1].argmax())
        out.append(t)
        ids.append(t)
    return out


def _engine(model, seed=0, blocks=256):
    cfg = EngineConfig(max_seq_length=64, kv_block_size=8,
                       num_kv_blocks=blocks, max_num_seqs=32)
    return Engine(model, cfg, device=torch.device("cpu"), seed=seed)


def test_cached_greedy_matches_eager(model, cached_session, monkeypatch):
    prompts = [[1, 5, 9], [2, 2, 2, 2], [7] * 11]
    sp = SamplingParams(max_tokens=6, temperature=0.0, n=2)
    e = _engine(model)
    cached = e.generate(prompts, sp, eos_token_id=None)
    assert e.pool.allocator.num_free == 256  # scratch returned too

    monkeypatch.delenv("DISTRL_FORCE_SESSION")
    monkeypatch.delenv("DISTRL_GRAPH_CACHE")
    eager = _engine(model).generate(prompts, sp, eos_token_id=None)
    assert cached == eager


def test_cache_reuses_state_across_waves(model, cached_session):
    e = _engine(model)
    sp = SamplingParams(max_tokens=4, temperature=0.0, n=1)
    r1 = e.generate([[1, 2, 3]], sp, eos_token_id=None)
    r2 = e.generate([[4, 5, 6, 7]], sp, eos_token_id=None)
    cache = e._session_cache
    assert cache is not None and len(cache._cache) == 1  # same key reused
    # different sampling params -> new cache entry
    e.generate([[1, 2]], SamplingParams(max_tokens=2, temperature=0.0, n=1),
               eos_token_id=None)
    assert len(cache._cache) == 2
    assert e.pool.allocator.num_free == 256
    # results still correct on the reused buffers
    assert r1[0][0] == _naive(model, [1, 2, 3], 4)
    assert r2[0][0] == _naive(model, [4, 5, 6, 7], 4)


def test_cached_streaming_and_eos(model, cached_session):
    e = _engine(model)
    sp_probe = SamplingParams(max_tokens=1, temperature=0.0, n=1)
    first = e.generate([[4, 4, 4]], sp_probe, eos_token_id=None)[0][0][0]
    streamed = {}
    sp = SamplingParams(max_tokens=6, temperature=0.0, n=1)
    res = e.generate([[4, 4, 4], [9, 1, 2]], sp, eos_token_id=first,
                     stream_cb=lambda pi, ci, t:
                     streamed.setdefault((pi, ci), []).extend(t))
    assert res[0][0][-1] == first
    for (pi, ci), toks in streamed.items():
        assert toks == res[pi][ci]
    assert e.pool.allocator.num_free == 256


def test_cached_sampling_reproducible(model, cached_session):
    prompts = [[3, 1, 4], [1, 5, 9, 2]]
    sp = SamplingParams(max_tokens=5, temperature=0.8, top_p=0.9, n=2)
    a = _engine(model, seed=9).generate(prompts, sp, eos_token_id=None)
    b = _engine(model, seed=9).generate(prompts, sp, eos_token_id=None)
    assert a == b


def test_cached_tight_pool_falls_back(model, cached_session, monkeypatch):
    """When the cache can't get its buffers (pool too tight for the
    padding scratch block), generation falls back to an exact-size
    session and still succeeds."""
    from distrl_llm_amd.engine import decode_session as ds
    monkeypatch.setattr(
        ds.SessionCache, "acquire",
        lambda self, seqs, sp, eos: (_ for _ in ()).throw(MemoryError()))
    e = _engine(model)
    sp = SamplingParams(max_tokens=4, temperature=0.0, n=1)
    res = e.generate([[1, 2, 3]], sp, eos_token_id=None)
    assert res[0][0] == _naive(model, [1, 2, 3], 4)
    assert e.pool.allocator.num_free == 256
