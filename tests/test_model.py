import os

import pytest
import torch

from distrl_llm_amd.models import CausalLM, get_spec
from distrl_llm_amd.models.lora import (load_adapter, lora_state_dict,
                                        save_adapter)
from distrl_llm_amd.ops import reference as R


@pytest.fixture(scope="module")
def tiny_model():
    spec = get_spec("tiny-qwen2")
    m = CausalLM(spec, lora_r=4, lora_alpha=8, dtype=torch.float32)
    m.random_init(seed=0)
    return m


def test_forward_shapes(tiny_model):
    B, T = 2, 12
    ids = torch.randint(0, tiny_model.spec.vocab_size, (B, T))
    logits = tiny_model(ids)
    assert logits.shape == (B, T, tiny_model.spec.vocab_size)
    assert torch.isfinite(logits).all()


def test_fresh_lora_is_noop(tiny_model):
    """B zeros ==> adapter contributes nothing (reference first-round
    behavior)."""
    ids = torch.randint(0, tiny_model.spec.vocab_size, (1, 8))
    logits = tiny_model(ids)
    # disable lora by zeroing A as well; with B=0 output must be unchanged
    with torch.no_grad():
        saved = [p.clone() for p in tiny_model.parameters() if p.requires_grad]
        for mod in tiny_model.modules():
            if hasattr(mod, "lora_A") and mod.lora_A is not None:
                mod.lora_A.zero_()
    logits2 = tiny_model(ids)
    torch.testing.assert_close(logits, logits2)
    with torch.no_grad():
        for p, s in zip([p for p in tiny_model.parameters() if p.requires_grad], saved):
            p.copy_(s)


def test_left_pad_invariance(tiny_model):
    """Left-padding must not change the logits of real tokens."""
    ids = torch.randint(0, tiny_model.spec.vocab_size, (1, 6))
    logits = tiny_model(ids)
    padded = torch.cat([torch.zeros(1, 3, dtype=torch.long), ids], dim=1)
    mask = torch.cat([torch.zeros(1, 3, dtype=torch.long),
                      torch.ones(1, 6, dtype=torch.long)], dim=1)
    logits_p = tiny_model(padded, mask)
    torch.testing.assert_close(logits, logits_p[:, 3:], rtol=1e-4, atol=1e-4)


def test_grads_only_on_lora(tiny_model):
    ids = torch.randint(0, tiny_model.spec.vocab_size, (2, 8))
    loss = tiny_model(ids).float().pow(2).mean()
    loss.backward()
    for name, p in tiny_model.named_parameters():
        if "lora_" in name:
            assert p.requires_grad
        else:
            assert not p.requires_grad and p.grad is None
    tiny_model.zero_grad(set_to_none=True)


def test_adapter_roundtrip(tmp_path, tiny_model):
    with torch.no_grad():
        for p in tiny_model.parameters():
            if p.requires_grad:
                p.add_(torch.randn_like(p) * 0.01)
    path = str(tmp_path / "adapter")
    save_adapter(tiny_model, path, "tiny-qwen2", r=4, alpha=8)
    assert os.path.exists(os.path.join(path, "adapter_config.json"))
    assert os.path.exists(os.path.join(path, "adapter_model.safetensors"))
    before = {k: v.clone() for k, v in lora_state_dict(tiny_model).items()}

    spec = get_spec("tiny-qwen2")
    m2 = CausalLM(spec, lora_r=4, lora_alpha=8, dtype=torch.float32)
    m2.random_init(seed=0)
    n = load_adapter(m2, path)
    assert n == 7 * spec.num_layers
    after = lora_state_dict(m2)
    for k in before:
        torch.testing.assert_close(before[k], after[k])


def test_peft_key_format(tiny_model):
    keys = list(lora_state_dict(tiny_model))
    assert all(k.startswith("base_model.model.model.layers.") for k in keys)
    assert any(k.endswith(".self_attn.q_proj.lora_A.weight") for k in keys)
    assert any(k.endswith(".mlp.down_proj.lora_B.weight") for k in keys)


def test_nf4_roundtrip():
    w = torch.randn(128, 64)
    packed, absmax = R.quantize_nf4(w, block_size=64)
    assert packed.dtype == torch.uint8 and packed.numel() == w.numel() // 2
    deq = R.dequantize_nf4(packed, absmax, w.shape, 64)
    # nf4 quantization error is bounded by half the largest code gap
    # (0.304/2 = 0.152) x absmax
    err = (w - deq).abs()
    bound = absmax.repeat_interleave(64).view(w.shape) * 0.1521 + 1e-6
    assert (err <= bound).all()
    # exact codebook values round-trip exactly
    w2 = torch.tensor([R.NF4_CODE.tolist() * 4]) * 3.0
    p2, a2 = R.quantize_nf4(w2, 64)
    torch.testing.assert_close(R.dequantize_nf4(p2, a2, w2.shape, 64), w2)


def test_quantize_model_attaches_sidecar():
    spec = get_spec("tiny-qwen2")
    m = CausalLM(spec, lora_r=0, dtype=torch.float32).random_init(1)
    before = m.model.layers[0].self_attn.q_proj.weight.clone()
    m.quantize_nf4_()
    mod = m.model.layers[0].self_attn.q_proj
    assert mod.weight_nf4 is not None and mod.weight_absmax is not None
    # weight replaced by its quantized image: close but not identical
    assert not torch.equal(before, mod.weight)
    assert (before - mod.weight).abs().max() < 0.1


def test_cli_warm_start_adapter(tmp_path):
    """--load_adapter loads a saved PEFT adapter into a fresh worker's
    model (warm-start extension; the reference has no resume path,
    SURVEY §5.4)."""
    from distrl_llm_amd.models.lora import load_adapter, lora_state_dict, save_adapter
    spec = get_spec("tiny-qwen2")
    m1 = CausalLM(spec, lora_r=4, lora_alpha=8, dtype=torch.float32).random_init(0)
    with torch.no_grad():
        for p in m1.parameters():
            if p.requires_grad:
                p.add_(torch.randn_like(p) * 0.05)
    path = str(tmp_path / "warm")
    save_adapter(m1, path, "tiny-qwen2", r=4, alpha=8)

    m2 = CausalLM(spec, lora_r=4, lora_alpha=8, dtype=torch.float32).random_init(1)
    load_adapter(m2, path)
    a, b = lora_state_dict(m1), lora_state_dict(m2)
    for k in a:
        torch.testing.assert_close(a[k], b[k])


def test_lora_dropout_train_eval_semantics():
    """--lora_dropout > 0: stochastic adapter path in train mode,
    deterministic (no-drop) in eval mode (PEFT semantics; reference
    helper.py:25-46 passes lora_dropout through)."""
    import torch

    from distrl_llm_amd.models import CausalLM, get_spec
    m = CausalLM(get_spec("tiny-qwen2"), lora_r=4, lora_alpha=8,
                 lora_dropout=0.5, dtype=torch.float32).random_init(2)
    with torch.no_grad():  # nonzero B so the adapter contributes
        for name, p in m.named_parameters():
            if "lora_B" in name:
                p.add_(torch.randn_like(p))
    ids = torch.randint(0, 2048, (1, 6))
    m.train()
    torch.manual_seed(0)
    a = m(ids)
    torch.manual_seed(1)
    b = m(ids)
    assert not torch.equal(a, b)  # dropout is live in train mode
    m.eval()
    with torch.no_grad():
        c, d = m(ids), m(ids)
    assert torch.equal(c, d)  # eval: deterministic, no dropping


def test_spec_registry_resolution_and_shapes():
    """Registry name resolution for every supported family/size, and
    internal dimension consistency (published architecture shapes)."""
    from distrl_llm_amd.models.spec import get_spec
    cases = {
        "unsloth/Qwen2.5-7B-Instruct-bnb-4bit": (3584, 28, 28, 4),
        "Qwen/Qwen2.5-0.5B-Instruct": (896, 24, 14, 2),
        "qwen2.5-1.5b": (1536, 28, 12, 2),
        "Qwen/Qwen2.5-14B-Instruct": (5120, 48, 40, 8),
        "unsloth/Qwen2.5-32B-Instruct-bnb-4bit": (5120, 64, 40, 8),
        "Qwen/Qwen2.5-72B-Instruct": (8192, 80, 64, 8),
        "meta-llama/Meta-Llama-3-8B": (4096, 32, 32, 8),
        "meta-llama/Meta-Llama-3-70B-Instruct": (8192, 80, 64, 8),
        "mistralai/Mistral-7B-Instruct-v0.3": (4096, 32, 32, 8),
        "Qwen/Qwen3-8B": (4096, 36, 32, 8),
        "Qwen/Qwen3-4B-Instruct-2507": (2560, 36, 32, 8),
        "Qwen/Qwen3-32B": (5120, 64, 64, 8),
    }
    for name, (h, L, q, kv) in cases.items():
        s = get_spec(name)
        assert (s.hidden_size, s.num_layers, s.num_heads,
                s.num_kv_heads) == (h, L, q, kv), name
        assert s.q_size == s.num_heads * s.head_dim
        assert s.num_heads % s.num_kv_heads == 0
    import pytest as _p
    with _p.raises(ValueError):
        get_spec("mystery-model-99b")
