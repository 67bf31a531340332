"""CPU coverage of the DecodeSession state machine — the exact step body
that runs inside the hipGraph on GPU (device-resident positions/finished/
out_buf, frozen-sequence clamping, worst-case block prealloc, extraction).
DISTRL_FORCE_SESSION=1 routes CPU generation through the session path so
CI exercises this logic without a GPU; outputs must match the eager
continuous-batching loop exactly."""

import os

import pytest
import torch

from distrl_llm_amd.config import EngineConfig, SamplingParams
from distrl_llm_amd.engine import Engine
from distrl_llm_amd.models import CausalLM, get_spec


@pytest.fixture(scope="module")
def model():
    spec = get_spec("tiny-qwen2")
    m = CausalLM(spec, lora_r=4, lora_alpha=8, dtype=torch.float32)
    m.random_init(seed=21)
    return m


@pytest.fixture
def force_session(monkeypatch):
    monkeypatch.setenv("DISTRL_FORCE_SESSION", "1")


def _engine(model, seed=0, **over):
    cfg = EngineConfig(max_seq_length=64, kv_block_size=8, num_kv_blocks=256,
                       max_num_seqs=32)
    for k, v in over.items():
        setattr(cfg, k, v)
    return Engine(model, cfg, device=torch.device("cpu"), seed=seed)


def test_session_greedy_matches_eager(model, force_session):
    prompts = [[1, 5, 9, 2, 7], [3, 3, 8], [11] * 17]
    sp = SamplingParams(max_tokens=6, temperature=0.0, n=2)
    session_out = _engine(model).generate(prompts, sp, eos_token_id=None)

    os.environ.pop("DISTRL_FORCE_SESSION", None)
    eager_out = _engine(model).generate(prompts, sp, eos_token_id=None)
    assert session_out == eager_out


def test_session_eos_and_mixed_lengths(model, force_session):
    """EOS freezes one lane while others continue; frozen lanes must not
    advance positions or emit further tokens."""
    prompt = [4, 4, 4]
    e = _engine(model)
    sp_probe = SamplingParams(max_tokens=1, temperature=0.0, n=1)
    first = e.generate([prompt], sp_probe, eos_token_id=None)[0][0][0]

    e2 = _engine(model)
    sp = SamplingParams(max_tokens=7, temperature=0.0, n=1)
    res = e2.generate([prompt, [9, 1, 2, 6]], sp, eos_token_id=first)
    assert res[0][0][-1] == first and len(res[0][0]) <= 7
    assert len(res[1][0]) <= 7
    assert e2.pool.allocator.num_free == 256


def test_session_sampling_seed_deterministic(model, force_session):
    prompts = [[2, 4, 6, 8], [1, 3, 5]]
    sp = SamplingParams(max_tokens=5, temperature=0.9, top_p=0.9, n=2)
    a = _engine(model, seed=77).generate(prompts, sp, eos_token_id=None)
    b = _engine(model, seed=77).generate(prompts, sp, eos_token_id=None)
    c = _engine(model, seed=78).generate(prompts, sp, eos_token_id=None)
    assert a == b
    assert a != c  # different engine seed gives a different stream


def test_session_streaming_deltas_match_final(model, force_session):
    """stream_cb deltas (chunk-granular on the session path) concatenate
    exactly to the returned sequences, including the prefill-sampled
    first token and EOS-frozen lanes."""
    e = _engine(model)
    prompts = [[1, 5, 9, 2, 7], [3, 3, 8]]
    sp = SamplingParams(max_tokens=6, temperature=0.0, n=2)
    streamed = {}  # (prompt, cand) -> [tokens]

    def cb(pi, ci, toks):
        streamed.setdefault((pi, ci), []).extend(toks)

    res = e.generate(prompts, sp, eos_token_id=None, stream_cb=cb)
    for pi, per_prompt in enumerate(res):
        for ci, ids in enumerate(per_prompt):
            assert streamed[(pi, ci)] == ids, (pi, ci)


def test_session_max_seq_len_clamp(model, force_session):
    """A prompt near max_seq_length stops at the limit, not past it."""
    e = _engine(model)
    prompt = list(range(1, 61))  # 60 tokens, limit 64
    sp = SamplingParams(max_tokens=10, temperature=0.0, n=1)
    res = e.generate([prompt], sp, eos_token_id=None)
    assert len(res[0][0]) == 4  # 64 - 60
    assert e.pool.allocator.num_free == 256


def test_in_wave_retirement_and_per_seq_limits(tiny_engine_factory=None):
    """Session path with in-wave retirement (retire_at) + per-candidate
    token limits: outputs must equal the eager path's (reference parity:
    vLLM continuous batching, distributed_actor.py:147-172)."""
    import os
    import torch
    from distrl_llm_amd.config import EngineConfig, SamplingParams
    from distrl_llm_amd.engine import Engine
    from distrl_llm_amd.models import CausalLM, get_spec
    spec = get_spec("tiny-qwen2")
    model = CausalLM(spec, lora_r=2, lora_alpha=4, dtype=torch.float32,
                     device=torch.device("cpu"))
    model.random_init(seed=11)
    prompts = [[3, 5, 7], [11, 13], [2, 4, 6, 8], [9]]
    limits = [[3, 9], [1, 5], [7, 2], [4, 4]]
    sp = SamplingParams(max_tokens=9, temperature=0.0, n=2)
    outs = {}
    for mode, env in (("eager", {}),
                      ("session", {"DISTRL_FORCE_SESSION": "1",
                                   "DISTRL_RETIRE_UNIT": "2"}),
                      ("session_cache", {"DISTRL_FORCE_SESSION": "1",
                                         "DISTRL_GRAPH_CACHE": "1",
                                         "DISTRL_RETIRE_UNIT": "2"})):
        for k, v in env.items():
            os.environ[k] = v
        try:
            cfg = EngineConfig(max_seq_length=32, kv_block_size=4,
                               num_kv_blocks=256, max_num_seqs=8)
            eng = Engine(model, cfg, device=torch.device("cpu"), seed=5)
            outs[mode] = eng.generate(prompts, sp, eos_token_id=None,
                                      token_limits=limits)
            assert eng.pool.allocator.num_free == 256
        finally:
            for k in env:
                os.environ.pop(k, None)
    assert outs["eager"] == outs["session"], (outs["eager"], outs["session"])
    assert outs["eager"] == outs["session_cache"]
    for pi, per_prompt in enumerate(outs["eager"]):
        for ci, ids in enumerate(per_prompt):
            assert len(ids) == limits[pi][ci]  # greedy, no EOS -> exact cap


def test_cancel_check_aborts_mid_generation(model, force_session):
    """Request-abort support (vLLM abort_request analogue): a cancelled
    prompt retires with partial output within one decode chunk and frees
    its KV blocks; co-batched prompts are unaffected."""
    eng = _engine(model)
    sp = SamplingParams(max_tokens=40, temperature=0.0, n=1)
    prompts = [[3, 1, 4], [2, 7, 2]]
    full = eng.generate(prompts, sp, eos_token_id=None)

    # cancel prompt 0 once it is in flight (first token streamed), as a
    # disconnecting client would — an immediately-set cancel is purged
    # pre-admission instead (covered below)
    seen = []
    out = eng.generate(prompts, sp, eos_token_id=None,
                       stream_cb=lambda pi, ci, t: seen.append(pi),
                       cancel_check=lambda pi: pi == 0 and 0 in seen)
    # prompt 0 aborted: partial (prefill token + at most one chunk of 16)
    assert 1 <= len(out[0][0]) <= 17 < 40
    assert out[0][0] == full[0][0][:len(out[0][0])]  # prefix of greedy
    # prompt 1 ran to its cap untouched
    assert out[1][0] == full[1][0]
    assert eng.pool.allocator.num_free == eng.pool.num_blocks


def test_cancel_before_admission_returns_empty(model, force_session):
    eng = _engine(model)
    sp = SamplingParams(max_tokens=5, temperature=0.0, n=2)
    out = eng.generate([[5, 6], [7, 8]], sp, eos_token_id=None,
                       cancel_check=lambda pi: True)
    assert out == [[[], []], [[], []]]
    assert eng.pool.allocator.num_free == eng.pool.num_blocks


def test_cancel_check_eager_path(model, monkeypatch):
    monkeypatch.setenv("DISTRL_FORCE_SESSION", "0")
    eng = _engine(model)
    sp = SamplingParams(max_tokens=30, temperature=0.0, n=1)
    prompts = [[3, 1, 4], [2, 7, 2]]
    full = eng.generate(prompts, sp, eos_token_id=None)
    seen = []
    out = eng.generate(prompts, sp, eos_token_id=None,
                       stream_cb=lambda pi, ci, t: seen.append(pi),
                       cancel_check=lambda pi: pi == 0 and seen.count(0) >= 3)
    assert 3 <= len(out[0][0]) < 30
    assert out[0][0] == full[0][0][:len(out[0][0])]
    assert out[1][0] == full[1][0]
    assert eng.pool.allocator.num_free == eng.pool.num_blocks
