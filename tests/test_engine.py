import pytest
import torch

from distrl_llm_amd.config import EngineConfig, SamplingParams
from distrl_llm_amd.engine import Engine
from distrl_llm_amd.models import CausalLM, get_spec


@pytest.fixture(scope="module")
def setup():
    spec = get_spec("tiny-qwen2")
    model = CausalLM(spec, lora_r=4, lora_alpha=8, dtype=torch.float32)
    model.random_init(seed=7)
    cfg = EngineConfig(max_seq_length=128, kv_block_size=8, num_kv_blocks=256,
                       max_num_seqs=64)
    return model, Engine(model, cfg, device=torch.device("cpu"), seed=0)


def _naive_greedy(model, prompt_ids, steps):
    """Full-recompute greedy decoding as ground truth."""
    ids = list(prompt_ids)
    out = []
    for _ in range(steps):
        logits = model(torch.tensor([ids]))[0, -1]
        t = int(logits.argmax())
        out.append(t)
        ids.append(t)
    return out


def test_greedy_matches_naive(setup):
    model, engine = setup
    prompts = [[1, 5, 9, 2, 7], [3, 3, 8]]
    sp = SamplingParams(max_tokens=6, temperature=0.0, n=1)
    results = engine.generate(prompts, sp, eos_token_id=None)
    for p, res in zip(prompts, results):
        assert len(res) == 1
        expected = _naive_greedy(model, p, 6)
        assert res[0] == expected, (res[0], expected)


def test_fanout_counts_and_pool_freed(setup):
    model, engine = setup
    prompts = [list(range(1, 12)), [2, 4, 6]]
    sp = SamplingParams(max_tokens=5, temperature=1.0, n=4, top_p=0.9)
    before_free = engine.pool.allocator.num_free
    results = engine.generate(prompts, sp, eos_token_id=None)
    assert [len(r) for r in results] == [4, 4]
    for r in results:
        for ids in r:
            assert len(ids) == 5
    # all KV blocks returned to the pool
    assert engine.pool.allocator.num_free == before_free


def test_eos_termination(setup):
    model, engine = setup
    # find the greedy-argmax token after a short prompt and declare it EOS:
    prompt = [1, 2, 3]
    first = _naive_greedy(model, prompt, 1)[0]
    sp = SamplingParams(max_tokens=8, temperature=0.0, n=1)
    res = engine.generate([prompt], sp, eos_token_id=first)
    assert res[0][0] == [first]  # stopped immediately, EOS included


def test_shared_prefill_isolation(setup):
    """n candidates of one prompt must match n separate single runs in
    greedy mode (shared prompt blocks don't corrupt each other)."""
    model, engine = setup
    prompt = list(range(1, 18))  # spans 2+ blocks of 8 with a partial tail
    sp = SamplingParams(max_tokens=4, temperature=0.0, n=3)
    res = engine.generate([prompt], sp, eos_token_id=None)
    expected = _naive_greedy(model, prompt, 4)
    for ids in res[0]:
        assert ids == expected


def test_continuous_batching_admission(setup):
    """More prompts than the pool can hold at once must still all finish."""
    model, engine = setup
    spec = model.spec
    cfg = EngineConfig(max_seq_length=64, kv_block_size=8, num_kv_blocks=40,
                       max_num_seqs=64)
    small = Engine(model, cfg, device=torch.device("cpu"), seed=0)
    prompts = [[i + 1, i + 2, i + 3, i + 4] for i in range(10)]
    sp = SamplingParams(max_tokens=6, temperature=0.0, n=2)
    results = small.generate(prompts, sp, eos_token_id=None)
    assert [len(r) for r in results] == [2] * 10
    expected0 = _naive_greedy(model, prompts[0], 6)
    assert results[0][0] == expected0
    assert small.pool.allocator.num_free == 40


def test_prefill_token_budget_chunks(setup):
    """Admission splits prefill into batches bounded by the token budget
    without losing prompts."""
    model, _ = setup
    from distrl_llm_amd.config import EngineConfig, SamplingParams
    from distrl_llm_amd.engine import Engine
    cfg = EngineConfig(max_seq_length=128, kv_block_size=8, num_kv_blocks=512,
                       max_num_seqs=64)
    engine = Engine(model, cfg, device=torch.device("cpu"), seed=0)
    prompts = [[i + 1] * 20 for i in range(6)]
    sp = SamplingParams(max_tokens=4, temperature=0.0, n=1)
    results = engine.generate(prompts, sp, eos_token_id=None,
                              prefill_token_budget=25)  # forces 1/batch
    assert [len(r) for r in results] == [1] * 6
    for p, r in zip(prompts, results):
        assert r[0] == _naive_greedy(model, p, 4)


def test_sampling_params_validation():
    import pytest as _pytest
    from distrl_llm_amd.config import SamplingParams
    with _pytest.raises(ValueError):
        SamplingParams(max_tokens=0)
    with _pytest.raises(ValueError):
        SamplingParams(top_p=0.0)
    with _pytest.raises(ValueError):
        SamplingParams(temperature=-1.0)
    with _pytest.raises(ValueError):
        SamplingParams(n=0)


def test_nf4_quantized_model_generates():
    """A 4-bit (nf4) quantized model runs the same engine paths: the
    decode math operates on the quantized weight image (identical to the
    merged nf4(W)+BA semantics the GPU path uses)."""
    from distrl_llm_amd.models import CausalLM, get_spec
    model = CausalLM(get_spec("tiny-qwen2"), lora_r=4, lora_alpha=8,
                     dtype=torch.float32)
    model.random_init(seed=13)
    model.quantize_nf4_()
    cfg = EngineConfig(max_seq_length=64, kv_block_size=8, num_kv_blocks=128,
                       max_num_seqs=16)
    engine = Engine(model, cfg, device=torch.device("cpu"), seed=0)
    prompts = [[1, 2, 3, 4], [9, 8]]
    sp = SamplingParams(max_tokens=5, temperature=0.0, n=2)
    res = engine.generate(prompts, sp, eos_token_id=None)
    for p, r in zip(prompts, res):
        expected = _naive_greedy(model, p, 5)
        for ids in r:
            assert ids == expected
    assert engine.pool.allocator.num_free == 128


def test_n_exceeding_max_num_seqs_clear_error(setup):
    """n-fan-out is admitted atomically per prompt; an impossible n must
    fail with the real reason, not a pool-exhaustion error."""
    model, _ = setup
    cfg = EngineConfig(max_seq_length=64, kv_block_size=8, num_kv_blocks=256,
                       max_num_seqs=8)
    engine = Engine(model, cfg, device=torch.device("cpu"), seed=0)
    with pytest.raises(ValueError, match="max_num_seqs"):
        engine.generate([[1, 2, 3]],
                        SamplingParams(max_tokens=2, temperature=0.0, n=20),
                        eos_token_id=None)
    # error path leaves the pool fully drained
    assert engine.pool.allocator.num_free == 256


def test_duplicate_prompts_independent(setup):
    """Identical prompts in one batch are independent sequences (each
    prefills its own blocks) and produce identical greedy outputs."""
    model, engine = setup
    p = [5, 5, 6, 7]
    res = engine.generate([p, p, p],
                          SamplingParams(max_tokens=4, temperature=0.0, n=1),
                          eos_token_id=None)
    exp = _naive_greedy(model, p, 4)
    assert [r[0] for r in res] == [exp, exp, exp]


def test_derive_num_blocks_cpu_sizing(setup):
    """CPU pool sizing: 64 MB budget, floor of 16 blocks, capped at
    max_num_seqs * blocks_for(max_seq_length)."""
    model, _ = setup
    cfg = EngineConfig(max_seq_length=64, kv_block_size=8, num_kv_blocks=0,
                       max_num_seqs=4)
    e = Engine(model, cfg, device=torch.device("cpu"), seed=0)
    # cap: 4 seqs x 8 blocks each
    assert e.pool.num_blocks == 4 * 8
    cfg2 = EngineConfig(max_seq_length=64, kv_block_size=8, num_kv_blocks=0,
                        max_num_seqs=100000)
    e2 = Engine(model, cfg2, device=torch.device("cpu"), seed=0)
    # budget-bound: 64 MB / bytes-per-block, at least the floor
    spec = model.spec
    per_block = (2 * spec.num_layers * 8 * spec.num_kv_heads * spec.head_dim
                 * 4)  # fp32 on CPU
    assert e2.pool.num_blocks == max(64 * 1024 * 1024 // per_block, 16)


def test_geom_len_mean_draws_deterministic_limits(setup):
    """sp.geom_len_mean: per-candidate exponential caps drawn from the
    engine's seeded generator — deterministic per (engine seed, call)."""
    model, _ = setup
    cfg = EngineConfig(max_seq_length=64, kv_block_size=8, num_kv_blocks=256,
                       max_num_seqs=16)
    sp = SamplingParams(max_tokens=20, temperature=0.0, n=3,
                        geom_len_mean=6.0)
    outs = []
    for _ in range(2):
        eng = Engine(model, cfg, device=torch.device("cpu"), seed=9)
        outs.append(eng.generate([[3, 5, 7], [2, 4]], sp, eos_token_id=None))
    assert outs[0] == outs[1]
    lens = [len(ids) for per in outs[0] for ids in per]
    assert all(1 <= L <= 20 for L in lens)
    assert len(set(lens)) > 1  # the exponential actually varies the caps
