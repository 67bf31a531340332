"""Statistical checks of the CPU reference sampler (the stream serving
and CPU engines draw from; the GPU kernel is tested against the same
reference in test_ops_gpu)."""

import torch

from distrl_llm_amd.ops import reference as R


def _counts(logits, temperature, top_p, top_k, n, seed):
    g = torch.Generator().manual_seed(seed)
    c = torch.zeros(logits.shape[-1], dtype=torch.long)
    for _ in range(n):
        t = R.sample_tokens(logits.unsqueeze(0), temperature, top_p, top_k,
                            generator=g)
        c[int(t)] += 1
    return c


def test_temperature_sharpens_distribution():
    logits = torch.tensor([2.0, 1.0, 0.0, -1.0])
    hot = _counts(logits, 2.0, 1.0, 0, 2000, 0)
    cold = _counts(logits, 0.25, 1.0, 0, 2000, 0)
    # colder temperature concentrates mass on the argmax
    assert cold[0] > hot[0]
    assert cold[0] > 1900  # p(argmax) at T=0.25 is ~0.98
    # hot roughly matches softmax(logits/2): p0 ~ 0.45
    assert 700 < hot[0] < 1100


def test_top_p_truncates_tail():
    logits = torch.log(torch.tensor([0.5, 0.3, 0.15, 0.05]))
    c = _counts(logits, 1.0, 0.8, 0, 3000, 1)
    # nucleus at p=0.8 keeps {0, 1} (cumsum 0.5, 0.8); tail never sampled
    assert c[2] == 0 and c[3] == 0
    assert c[0] > c[1] > 0
    # renormalized ratio ~ 0.5/0.3
    ratio = c[0].item() / max(c[1].item(), 1)
    assert 1.2 < ratio < 2.3


def test_top_k_truncates():
    logits = torch.tensor([1.0, 0.9, 0.8, 0.7, 0.6])
    c = _counts(logits, 1.0, 1.0, 2, 2000, 2)
    assert c[2] == 0 and c[3] == 0 and c[4] == 0
    assert c[0] > 0 and c[1] > 0


def test_greedy_is_argmax():
    logits = torch.randn(32)
    t = R.sample_tokens(logits.unsqueeze(0), 0.0, 1.0, 0)
    assert int(t) == int(logits.argmax())


def test_generation_config_translation():
    """GenerationConfig -> SamplingParams the way the reference's
    BaseActor translates HF GenerationConfig to vLLM params (reference
    distributed_actor.py:41-48: top_p hardcoded 0.95)."""
    from distrl_llm_amd.config import GenerationConfig
    gc = GenerationConfig(max_new_tokens=77, temperature=1.2,
                          num_return_sequences=16, do_sample=True)
    sp = gc.to_sampling_params()
    assert (sp.max_tokens, sp.temperature, sp.n, sp.top_p) == (77, 1.2, 16, 0.95)
    # do_sample=False means greedy regardless of temperature
    sp2 = GenerationConfig(temperature=0.9, do_sample=False).to_sampling_params()
    assert sp2.temperature == 0.0


def test_byte_tokenizer_specials_rendering():
    """skip_special_tokens=False renders specials visibly; True drops
    them (the learner/serve paths rely on this split)."""
    from distrl_llm_amd.utils.tokenizer import ByteTokenizer
    tok = ByteTokenizer(vocab_size=152064)
    ids = tok.encode("hi") + [tok.eos_token_id]
    assert tok.decode(ids, skip_special_tokens=True) == "hi"
    assert "<|257|>" in tok.decode(ids, skip_special_tokens=False)
