"""GPU engine + learner integration: the full native path (HIP kernels for
norm/rope/cache/attention/sampling/loss) on a small bf16 model."""

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def setup():
    from distrl_llm_amd.config import EngineConfig
    from distrl_llm_amd.engine import Engine
    from distrl_llm_amd.models import CausalLM, get_spec
    spec = get_spec("small-qwen2")
    model = CausalLM(spec, lora_r=8, lora_alpha=16, dtype=torch.bfloat16,
                     device=torch.device("cuda:0"))
    model.random_init(seed=7)
    cfg = EngineConfig(max_seq_length=512, kv_block_size=16,
                       num_kv_blocks=2048, max_num_seqs=256)
    return model, Engine(model, cfg, device=torch.device("cuda:0"), seed=0)


def _naive_greedy(model, prompt_ids, steps):
    ids = list(prompt_ids)
    out = []
    for _ in range(steps):
        with torch.no_grad():
            logits = model(torch.tensor([ids], device="cuda:0"))[0, -1]
        t = int(logits.argmax())
        out.append(t)
        ids.append(t)
    return out


def test_gpu_greedy_matches_naive(setup):
    model, engine = setup
    from distrl_llm_amd.config import SamplingParams
    prompts = [[1, 5, 9, 2, 7, 11, 200, 3000], [3, 3, 8]]
    sp = SamplingParams(max_tokens=8, temperature=0.0, n=1)
    results = engine.generate(prompts, sp, eos_token_id=None)
    for p, res in zip(prompts, results):
        expected = _naive_greedy(model, p, 8)
        # bf16 paged path vs bf16 recompute path: tiny numeric divergence
        # can flip argmax; require near-total agreement
        agree = sum(a == b for a, b in zip(res[0], expected))
        assert agree >= 7, (res[0], expected)


def test_gpu_fanout_and_long_context(setup):
    model, engine = setup
    from distrl_llm_amd.config import SamplingParams
    prompts = [list(torch.randint(0, 4000, (200,)).tolist()) for _ in range(3)]
    sp = SamplingParams(max_tokens=32, temperature=1.0, n=4, top_p=0.95)
    before = engine.pool.allocator.num_free
    results = engine.generate(prompts, sp, eos_token_id=None)
    assert [len(r) for r in results] == [4, 4, 4]
    for r in results:
        for ids in r:
            assert len(ids) == 32
            assert all(0 <= t < model.spec.vocab_size for t in ids)
    assert engine.pool.allocator.num_free == before


def test_gpu_learner_round(setup):
    model, engine = setup
    from distrl_llm_amd.train.learner import Learner
    from distrl_llm_amd.utils.tokenizer import ByteTokenizer
    tok = ByteTokenizer(vocab_size=model.spec.vocab_size)
    learner = Learner(model, tok, lr=1e-3, max_prompt_tokens=32,
                      max_new_tokens=32, train_batch_size=2)
    problems = ["what is 1+1?", "compute 2*3", "evaluate 5-2", "sum 1..4"]
    answers = ["<answer>2</answer>", "<answer>6</answer>",
               "<answer>3</answer>", "<answer>10</answer>"]
    lora_before = model.model.layers[0].self_attn.q_proj.lora_B.clone()
    loss = learner.accumulate_gradients(problems, answers, [1.0, -0.5, 0.3, 0.1])
    assert torch.isfinite(torch.tensor(loss))
    learner.step()
    assert not torch.equal(lora_before,
                           model.model.layers[0].self_attn.q_proj.lora_B)


def test_native_extension_is_loaded():
    """The HIP path must be the one that runs on GPU (no silent eager
    fallback)."""
    from distrl_llm_amd.ops.build import get_extension
    ext = get_extension()
    assert ext is not None, "gfx950 extension must load on the GPU box"
    assert "_build" in ext.__file__  # in-tree .so


def test_llama3_8b_bf16_path():
    """BASELINE config 4 readiness: Llama-3-8B bf16 (non-quantized) decode
    + PG learner micro-step on the same engine machinery (no qkv bias,
    GQA 4:1, vocab 128256)."""
    from distrl_llm_amd.config import EngineConfig, SamplingParams
    from distrl_llm_amd.engine import Engine
    from distrl_llm_amd.models import CausalLM, get_spec
    from distrl_llm_amd.train.learner import Learner
    from distrl_llm_amd.utils.tokenizer import ByteTokenizer

    dev = torch.device("cuda:0")
    spec = get_spec("meta-llama/Meta-Llama-3-8B")
    model = CausalLM(spec, lora_r=32, lora_alpha=16, dtype=torch.bfloat16,
                     device=dev).random_init(3407)
    tok = ByteTokenizer(vocab_size=spec.vocab_size)
    engine = Engine(model, EngineConfig(max_seq_length=512,
                                        gpu_memory_utilization=0.2),
                    device=dev, seed=0)
    prompts = [tok.encode("Compute 6*7."), tok.encode("What is 10-3?")]
    sp = SamplingParams(max_tokens=12, temperature=1.2, n=2, top_p=0.95)
    outs = engine.generate(prompts, sp, eos_token_id=tok.eos_token_id)
    assert len(outs) == 2 and all(len(o) == 2 for o in outs)
    for o in outs:
        for ids in o:
            assert 1 <= len(ids) <= 12

    learner = Learner(model, tok, lr=2e-5, max_prompt_tokens=64,
                      max_new_tokens=32, train_batch_size=2)
    answers = [tok.decode(ids) for o in outs for ids in o]
    probs = ["Compute 6*7."] * 2 + ["What is 10-3?"] * 2
    loss = learner.accumulate_gradients(probs, answers, [1.0, -0.2, 0.4, -0.6])
    learner.step()
    assert torch.isfinite(torch.tensor(loss))


def test_graph_vs_eager_decode_identical(setup):
    """hipGraph-replayed decode must produce exactly the same greedy tokens
    as the eager per-step loop (same kernels, same state machine)."""
    model, _ = setup
    from distrl_llm_amd.config import EngineConfig, SamplingParams
    from distrl_llm_amd.engine import Engine
    cfg_g = EngineConfig(max_seq_length=256, kv_block_size=16,
                         num_kv_blocks=512, max_num_seqs=64)
    cfg_e = EngineConfig(max_seq_length=256, kv_block_size=16,
                         num_kv_blocks=512, max_num_seqs=64,
                         enforce_eager=True)
    prompts = [list(range(3, 40)), [7, 11, 13, 17, 19]]
    sp = SamplingParams(max_tokens=12, temperature=0.0, n=1)
    out_g = Engine(model, cfg_g, device=torch.device("cuda:0"),
                   seed=3).generate(prompts, sp, eos_token_id=None)
    out_e = Engine(model, cfg_e, device=torch.device("cuda:0"),
                   seed=3).generate(prompts, sp, eos_token_id=None)
    assert out_g == out_e, (out_g, out_e)


def test_session_cache_vs_fresh_identical(setup):
    """CachedDecodeSession (DISTRL_GRAPH_CACHE=1: padded lanes, reused
    buffers, hipGraph reused across waves) must produce exactly the same
    greedy tokens as per-wave fresh sessions (docs/ROADMAP.md #4)."""
    import os
    model, _ = setup
    from distrl_llm_amd.config import EngineConfig, SamplingParams
    from distrl_llm_amd.engine import Engine
    cfg = dict(max_seq_length=256, kv_block_size=16, num_kv_blocks=512,
               max_num_seqs=64)
    prompts_a = [list(range(3, 40)), [7, 11, 13, 17, 19], [2, 4]]
    prompts_b = [[5, 6, 7, 300, 9], list(range(100, 120))]
    sp = SamplingParams(max_tokens=12, temperature=0.0, n=2)
    outs = {}
    for cache in ("0", "1"):
        os.environ["DISTRL_GRAPH_CACHE"] = cache
        try:
            eng = Engine(model, EngineConfig(**cfg),
                         device=torch.device("cuda:0"), seed=3)
            # two waves: the second exercises buffer/graph REUSE
            outs[cache] = (eng.generate(prompts_a, sp, eos_token_id=None),
                           eng.generate(prompts_b, sp, eos_token_id=None))
        finally:
            os.environ.pop("DISTRL_GRAPH_CACHE", None)
    assert outs["0"] == outs["1"], (outs["0"], outs["1"])


def test_long_prompt_prefill_no_cap(setup):
    """Prompts beyond the varlen prefill kernel's ~2200-token score-tile
    cap route through the first-party flash kernel (no SDPA/aotriton, no
    length cap — docs/ROADMAP.md #9)."""
    model, _ = setup
    from distrl_llm_amd.config import EngineConfig, SamplingParams
    from distrl_llm_amd.engine import Engine
    cfg = EngineConfig(max_seq_length=3072, kv_block_size=16,
                       num_kv_blocks=1024, max_num_seqs=8)
    eng = Engine(model, cfg, device=torch.device("cuda:0"), seed=0)
    prompt = [(i * 7 + 3) % 900 + 1 for i in range(2500)]
    sp = SamplingParams(max_tokens=4, temperature=0.0, n=1)
    res = eng.generate([prompt], sp, eos_token_id=None)
    expected = _naive_greedy(model, prompt, 4)
    agree = sum(a == b for a, b in zip(res[0][0], expected))
    assert agree >= 3, (res[0][0], expected)


def test_nf4_freed_base_learner_parity():
    """free_base_to_nf4_ (bf16 image dropped, fused nf4 GEMM forward +
    on-the-fly dequant dX): forward and LoRA grads must match the
    materialized nf4-image path (SURVEY.md §2.4-B learner nf4 GEMM)."""
    from distrl_llm_amd.models import CausalLM, get_spec
    spec = get_spec("small-qwen2")
    dev = torch.device("cuda:0")
    outs = {}
    for freed in (False, True):
        torch.manual_seed(0)
        model = CausalLM(spec, lora_r=8, lora_alpha=16,
                         dtype=torch.bfloat16, device=dev)
        model.random_init(seed=3)
        model.quantize_nf4_(keep_bf16=not freed)
        ids = torch.randint(1, 500, (2, 48), device=dev)
        logits = model(ids)
        loss = logits.float().log_softmax(-1).mean()
        loss.backward()
        g = [m.lora_A.grad.clone() for m in model.modules()
             if hasattr(m, "lora_A") and m.lora_A is not None
             and m.lora_A.grad is not None]
        outs[freed] = (logits.detach().float(), g)
    lo, go = outs[False]
    lf, gf = outs[True]
    assert torch.allclose(lo, lf, atol=2e-2, rtol=2e-2), (lo - lf).abs().max()
    for a, b in zip(go, gf):
        err = (a - b).abs().max()
        ref = a.abs().max().clamp_min(1e-6)
        assert err <= 0.05 * ref + 1e-5, (err, ref)


def test_graph_cache_stable_across_weight_refresh():
    """Cached hipGraphs must survive weight syncs: every tensor a captured
    graph references (adapter fragment packs, the LoRA-u pool) must keep
    ONE storage for the cache's lifetime. Regression test for the
    multi-wave batch-30 silent-NaN bug (profiles/r02_summary.md): run a
    generate, perturb the adapter (forces a refresh), run again with the
    cache, and compare against a FRESH engine with identical weights."""
    import copy
    from distrl_llm_amd.config import EngineConfig, SamplingParams
    from distrl_llm_amd.engine import Engine
    from distrl_llm_amd.models import CausalLM, get_spec
    spec = get_spec("small-qwen2")
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    model = CausalLM(spec, lora_r=8, lora_alpha=16, dtype=torch.bfloat16,
                     device=dev)
    model.random_init(seed=21)
    cfg = dict(max_seq_length=256, kv_block_size=16, num_kv_blocks=512,
               max_num_seqs=64)
    prompts = [[5, 9, 2, 7], list(range(30, 60))]
    sp = SamplingParams(max_tokens=10, temperature=0.0, n=2)

    eng = Engine(model, EngineConfig(**cfg), device=dev, seed=3)
    eng.generate(prompts, sp, eos_token_id=None)    # captures + caches
    with torch.no_grad():                           # a "weight sync"
        for m in model.modules():
            if hasattr(m, "lora_B") and m.lora_B is not None:
                m.lora_B.add_(torch.randn_like(m.lora_B) * 0.05)
    out_cached = eng.generate(prompts, sp, eos_token_id=None)  # reuses graphs

    fresh = Engine(model, EngineConfig(**cfg), device=dev, seed=3)
    out_fresh = fresh.generate(prompts, sp, eos_token_id=None)
    assert out_cached == out_fresh, (out_cached, out_fresh)


@pytest.mark.gpu
def test_graph_cache_stable_across_refresh_nf4_hybrid():
    """Same graph-stability contract through the QUANTIZED decode paths:
    with nf4 sidecars attached, the 7B-class default runs merged bf16 +
    the fused nf4 down-projection (hybrid_down), whose adapter fragment
    packs are rebuilt per weight sync — the exact tensors whose per-round
    reallocation caused the multi-wave batch-30 silent-NaN bug. The
    in-place refresh (weights.py::_refresh_nf4_adapters) must keep the
    captured graphs valid across syncs."""
    from distrl_llm_amd.config import EngineConfig, SamplingParams
    from distrl_llm_amd.engine import Engine
    from distrl_llm_amd.models import CausalLM, get_spec
    spec = get_spec("small-qwen2")
    dev = torch.device("cuda:0")
    model = CausalLM(spec, lora_r=8, lora_alpha=16, dtype=torch.bfloat16,
                     device=dev)
    model.random_init(seed=22)
    model.quantize_nf4_()   # sidecars on -> hybrid_down active
    cfg = dict(max_seq_length=256, kv_block_size=16, num_kv_blocks=512,
               max_num_seqs=64)
    prompts = [[5, 9, 2, 7], list(range(30, 60)), [1000, 2000, 3000]]
    sp = SamplingParams(max_tokens=10, temperature=0.0, n=2)

    eng = Engine(model, EngineConfig(**cfg), device=dev, seed=3)
    assert getattr(eng.fused, "hybrid_down", False), \
        "test requires the hybrid nf4 down-proj decode path"
    eng.generate(prompts, sp, eos_token_id=None)    # captures + caches
    with torch.no_grad():                           # a "weight sync"
        for m in model.modules():
            if hasattr(m, "lora_B") and m.lora_B is not None:
                m.lora_B.add_(torch.randn_like(m.lora_B) * 0.05)
    out_cached = eng.generate(prompts, sp, eos_token_id=None)

    fresh = Engine(model, EngineConfig(**cfg), device=dev, seed=3)
    out_fresh = fresh.generate(prompts, sp, eos_token_id=None)
    assert out_cached == out_fresh, (out_cached, out_fresh)


@pytest.mark.gpu
def test_cancel_check_aborts_on_gpu(setup):
    """Request abort over the real hipGraph session path: stop_check is
    polled between replay chunks, the cancelled prompt retires with a
    greedy-prefix partial output, co-batched prompts are unaffected and
    no KV blocks leak (CPU counterpart:
    test_decode_session_cpu.py::test_cancel_check_aborts_mid_generation)."""
    from distrl_llm_amd.config import SamplingParams
    model, engine = setup
    sp = SamplingParams(max_tokens=40, temperature=0.0, n=1)
    prompts = [[3, 1, 4], [2, 7, 2]]
    free0 = engine.pool.allocator.num_free
    full = engine.generate(prompts, sp, eos_token_id=None)

    seen = []
    out = engine.generate(prompts, sp, eos_token_id=None,
                          stream_cb=lambda pi, ci, t: seen.append(pi),
                          cancel_check=lambda pi: pi == 0 and 0 in seen)
    assert 1 <= len(out[0][0]) <= 17 < 40
    assert out[0][0] == full[0][0][:len(out[0][0])]
    assert out[1][0] == full[1][0]
    assert engine.pool.allocator.num_free == free0


@pytest.mark.gpu
def test_qwen3_qk_norm_gpu_decode():
    """Qwen3 family on the real GPU decode path: the per-head q/k norms
    run as in-place torch ops on the packed qkv buffer inside the
    hipGraph-captured step (engine.py qk_norm_packed). Greedy engine
    output must match the naive layer-by-layer forward (which applies
    the norms through models/model.py — a different code path)."""
    from distrl_llm_amd.config import EngineConfig, SamplingParams
    from distrl_llm_amd.engine import Engine
    from distrl_llm_amd.models import CausalLM, get_spec
    dev = torch.device("cuda:0")
    model = CausalLM(get_spec("small-qwen3"), lora_r=8, lora_alpha=16,
                     dtype=torch.bfloat16, device=dev)
    model.random_init(seed=33)
    with torch.no_grad():  # non-trivial norms
        for layer in model.model.layers:
            at = layer.self_attn
            at.q_norm.weight.add_(
                (torch.rand_like(at.q_norm.weight.float()) * 0.5 - 0.25)
                .to(at.q_norm.weight.dtype))
            at.k_norm.weight.add_(
                (torch.rand_like(at.k_norm.weight.float()) * 0.5 - 0.25)
                .to(at.k_norm.weight.dtype))
    eng = Engine(model, EngineConfig(max_seq_length=256, kv_block_size=16,
                                     num_kv_blocks=512, max_num_seqs=64),
                 device=dev, seed=0)
    prompts = [[5, 9, 2, 7], list(range(30, 50))]
    sp = SamplingParams(max_tokens=8, temperature=0.0, n=2)
    out = eng.generate(prompts, sp, eos_token_id=None)
    for p, res in zip(prompts, out):
        exp = _naive_greedy(model, p, 8)
        for ids in res:
            assert ids == exp


@pytest.mark.gpu
def test_prefix_caching_gpu_equality(setup):
    """Automatic prefix caching on the GPU: cache-hit tails prefill
    through the fused decode step; outputs must equal the uncached
    engine's, and clear_prefix_cache must return every block."""
    from distrl_llm_amd.config import EngineConfig, SamplingParams
    from distrl_llm_amd.engine import Engine
    model, _ = setup
    dev = torch.device("cuda:0")
    cfg = dict(max_seq_length=256, kv_block_size=16, num_kv_blocks=512,
               max_num_seqs=64)
    sys_p = list(range(700, 740))  # 40-token shared prefix (2 full blocks)
    prompts_a = [sys_p + [7, 8, 9], sys_p + [1]]
    prompts_b = [sys_p + [4, 4], sys_p + [2, 3, 5]]
    sp = SamplingParams(max_tokens=8, temperature=0.0, n=2)

    plain = Engine(model, EngineConfig(**cfg), device=dev, seed=3)
    exp_a = plain.generate(prompts_a, sp, eos_token_id=None)
    exp_b = plain.generate(prompts_b, sp, eos_token_id=None)

    eng = Engine(model, EngineConfig(enable_prefix_caching=True, **cfg),
                 device=dev, seed=3)
    assert eng.generate(prompts_a, sp, eos_token_id=None) == exp_a
    hits0 = eng._prefix_hits
    assert eng.generate(prompts_b, sp, eos_token_id=None) == exp_b
    assert eng._prefix_hits > hits0  # the shared prefix was reused
    eng.clear_prefix_cache()
    assert eng.pool.allocator.num_free == eng.pool.num_blocks
