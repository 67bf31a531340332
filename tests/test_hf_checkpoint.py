"""HF safetensors checkpoint interop: save/load roundtrip, ModelSpec
derivation from config.json, sharded checkpoints, strictness, and the
build_worker local-directory path (the reference's from_pretrained mode,
reference distributed_actor.py:58-66 — exercised offline by writing the
checkpoint ourselves)."""

import json
import os

import pytest
import torch

from distrl_llm_amd.models import CausalLM, get_spec
from distrl_llm_amd.models.hf_io import (is_hf_checkpoint_dir,
                                         load_hf_checkpoint, resolve_spec,
                                         save_hf_checkpoint,
                                         spec_from_hf_config)


@pytest.fixture(scope="module")
def ckpt_dir(tmp_path_factory):
    d = str(tmp_path_factory.mktemp("hf_ckpt"))
    m = CausalLM(get_spec("tiny-qwen2"), lora_r=0,
                 dtype=torch.float32).random_init(31)
    save_hf_checkpoint(m, d)
    return d, m


def test_roundtrip_logits_identical(ckpt_dir):
    d, m = ckpt_dir
    assert is_hf_checkpoint_dir(d)
    m2 = CausalLM(get_spec("tiny-qwen2"), lora_r=0,
                  dtype=torch.float32).random_init(99)  # different init
    n = load_hf_checkpoint(m2, d)
    assert n > 0
    ids = torch.randint(0, 2048, (2, 9))
    torch.testing.assert_close(m(ids), m2(ids), rtol=0, atol=0)


def test_spec_from_config_json(ckpt_dir):
    d, m = ckpt_dir
    spec = spec_from_hf_config(d)
    ref = get_spec("tiny-qwen2")
    for f in ("vocab_size", "hidden_size", "intermediate_size", "num_layers",
              "num_heads", "num_kv_heads", "head_dim", "rope_theta",
              "rms_norm_eps", "tie_word_embeddings", "qkv_bias"):
        assert getattr(spec, f) == getattr(ref, f), f
    assert resolve_spec(d).hidden_size == ref.hidden_size
    # names still resolve through the registry
    assert resolve_spec("unsloth/Qwen2.5-7B-Instruct-bnb-4bit").hidden_size == 3584


def test_sharded_checkpoint_load(ckpt_dir, tmp_path):
    """Multi-shard layout with model.safetensors.index.json loads too."""
    from safetensors.torch import load_file, save_file
    d, m = ckpt_dir
    full = load_file(os.path.join(d, "model.safetensors"))
    keys = sorted(full)
    half = len(keys) // 2
    sd = str(tmp_path / "sharded")
    os.makedirs(sd)
    save_file({k: full[k] for k in keys[:half]},
              os.path.join(sd, "model-00001-of-00002.safetensors"))
    save_file({k: full[k] for k in keys[half:]},
              os.path.join(sd, "model-00002-of-00002.safetensors"))
    wmap = {k: ("model-00001-of-00002.safetensors" if i < half
                else "model-00002-of-00002.safetensors")
            for i, k in enumerate(keys)}
    with open(os.path.join(sd, "model.safetensors.index.json"), "w") as f:
        json.dump({"weight_map": wmap}, f)
    with open(os.path.join(d, "config.json")) as f:
        cfg = f.read()
    with open(os.path.join(sd, "config.json"), "w") as f:
        f.write(cfg)

    m2 = CausalLM(get_spec("tiny-qwen2"), lora_r=0,
                  dtype=torch.float32).random_init(7)
    load_hf_checkpoint(m2, sd)
    ids = torch.randint(0, 2048, (1, 6))
    torch.testing.assert_close(m(ids), m2(ids), rtol=0, atol=0)


def test_strict_missing_and_unexpected(ckpt_dir, tmp_path):
    from safetensors.torch import load_file, save_file
    d, _ = ckpt_dir
    full = load_file(os.path.join(d, "model.safetensors"))

    bad = str(tmp_path / "missing")
    os.makedirs(bad)
    drop = {k: v for k, v in full.items() if "q_proj" not in k}
    save_file(drop, os.path.join(bad, "model.safetensors"))
    m = CausalLM(get_spec("tiny-qwen2"), lora_r=0, dtype=torch.float32)
    with pytest.raises(ValueError, match="missing"):
        load_hf_checkpoint(m, bad)

    bad2 = str(tmp_path / "unexpected")
    os.makedirs(bad2)
    extra = dict(full)
    extra["model.layers.0.self_attn.weird.weight"] = torch.zeros(2)
    save_file(extra, os.path.join(bad2, "model.safetensors"))
    with pytest.raises(ValueError, match="unexpected"):
        load_hf_checkpoint(m, bad2)
    # ignorable legacy keys pass
    ok = dict(full)
    ok["model.layers.0.self_attn.rotary_emb.inv_freq"] = torch.zeros(8)
    okd = str(tmp_path / "legacy")
    os.makedirs(okd)
    save_file(ok, os.path.join(okd, "model.safetensors"))
    load_hf_checkpoint(m, okd)


def test_transformers_can_load_our_checkpoint(ckpt_dir):
    """Cross-check the layout against transformers itself (offline,
    local files only)."""
    d, m = ckpt_dir
    from transformers import AutoModelForCausalLM
    hf = AutoModelForCausalLM.from_pretrained(d, torch_dtype=torch.float32)
    ids = torch.randint(0, 2048, (1, 8))
    ours = m(ids)
    theirs = hf(ids).logits
    torch.testing.assert_close(ours, theirs, rtol=2e-4, atol=2e-4)


def test_transformers_cross_check_llama(tmp_path):
    """Same cross-validation for the Llama family (no qkv bias, untied
    embeddings): transformers' LlamaForCausalLM must reproduce our
    logits from our own checkpoint."""
    d = str(tmp_path / "llama_ckpt")
    m = CausalLM(get_spec("tiny-llama"), lora_r=0,
                 dtype=torch.float32).random_init(17)
    save_hf_checkpoint(m, d)
    from transformers import AutoModelForCausalLM
    hf = AutoModelForCausalLM.from_pretrained(d, torch_dtype=torch.float32)
    ids = torch.randint(0, 2048, (2, 7))
    torch.testing.assert_close(m(ids), hf(ids).logits, rtol=2e-4, atol=2e-4)


def test_engine_generation_matches_transformers(ckpt_dir):
    """End-to-end decode cross-validation: our paged-KV engine's greedy
    continuations equal transformers' KV-cached ``generate`` from the
    same checkpoint — an independent implementation agreeing token for
    token."""
    from transformers import AutoModelForCausalLM

    from distrl_llm_amd.config import EngineConfig, SamplingParams
    from distrl_llm_amd.engine import Engine

    d, m = ckpt_dir
    hf = AutoModelForCausalLM.from_pretrained(d, dtype=torch.float32)
    eng = Engine(m, EngineConfig(max_seq_length=128, kv_block_size=8,
                                 num_kv_blocks=256, max_num_seqs=16),
                 device=torch.device("cpu"), seed=0)
    prompts = [[1, 5, 9, 2, 7], [3, 3, 8, 4], [17] * 12,
               list(range(2, 25))]  # crosses multiple KV blocks
    ours = eng.generate(prompts, SamplingParams(max_tokens=8,
                                                temperature=0.0, n=1),
                        eos_token_id=None)
    for p, o in zip(prompts, ours):
        with torch.no_grad():
            out = hf.generate(torch.tensor([p]), max_new_tokens=8,
                              do_sample=False, use_cache=True)
        assert o[0] == out[0, len(p):].tolist()


def test_merged_checkpoint_equals_adapter_forward(tmp_path):
    """save_merged_checkpoint: a LoRA-less model loading the merged
    export must match the adapter-bearing model's logits (W + scale*B@A
    folded in)."""
    from distrl_llm_amd.models.hf_io import save_merged_checkpoint
    spec = get_spec("tiny-qwen2")
    m = CausalLM(spec, lora_r=4, lora_alpha=8,
                 dtype=torch.float32).random_init(3)
    with torch.no_grad():  # give the adapter real content
        for name, p in m.named_parameters():
            if "lora_" in name:
                p.add_(torch.randn_like(p) * 0.05)
    d = str(tmp_path / "merged")
    n = save_merged_checkpoint(m, d)
    assert n == 7 * spec.num_layers

    flat = CausalLM(spec, lora_r=0, dtype=torch.float32).random_init(99)
    load_hf_checkpoint(flat, d)
    ids = torch.randint(0, 2048, (2, 10))
    torch.testing.assert_close(flat(ids), m(ids), rtol=1e-4, atol=1e-5)
    # and transformers agrees with the merged export too
    from transformers import AutoModelForCausalLM
    hf = AutoModelForCausalLM.from_pretrained(d, dtype=torch.float32)
    torch.testing.assert_close(hf(ids).logits, m(ids), rtol=2e-4, atol=2e-4)


def test_worker_loads_local_checkpoint_dir(ckpt_dir, tmp_path):
    """build_worker with --model <local dir> trains from the checkpoint's
    weights, not random init (single-rank gloo world)."""
    import torch.multiprocessing as mp
    d, _ = ckpt_dir
    port = 25500 + os.getpid() % 500
    ok_file = str(tmp_path / "ok")
    mp.spawn(_worker_main, nprocs=1, args=(d, port, ok_file), join=True)
    assert os.path.exists(ok_file)


def _worker_main(rank, ckpt, port, ok_file):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    from safetensors.torch import load_file

    from distrl_llm_amd.parallel.worker import build_worker
    config = {
        "run_name": "hfload", "lora_save_path": ok_file + ".adapter",
        "lr": 1e-3, "max_prompt_tokens": 32, "max_new_tokens": 8,
        "episodes": 1, "num_candidates": 2, "batch_size": 2,
        "train_batch_size": 2, "temperature": 1.0, "save_every": 0,
        "eval_every": 0, "model": ckpt, "dataset": "synthetic",
        "number_of_actors": 0, "number_of_learners": 1, "learner": "grpo",
        "max_lora_rank": 4, "topk": 2, "learner_chunk_size": 2,
        "actor_gpu_usage": 0.9, "learner_gpu_usage": 0.35,
        "lora_alpha": 8, "lora_dropout": 0.0, "seed": 55,
    }
    trainer = build_worker(rank, 1, config, device=torch.device("cpu"),
                           engine_overrides={"num_kv_blocks": 256,
                                             "kv_block_size": 8,
                                             "max_seq_length": 256})
    want = load_file(os.path.join(ckpt, "model.safetensors"))
    got = dict(trainer.engine.model.named_parameters())
    key = "model.layers.0.self_attn.q_proj.weight"
    assert torch.equal(got[key], want[key])
    trainer.fabric.close()
    open(ok_file, "w").write("ok")


def test_transformers_cross_check_mistral(tmp_path):
    """Mistral family (v0.3-style: no qkv bias, untied embeddings, no
    sliding window): transformers' MistralForCausalLM must reproduce our
    logits from our own checkpoint, and windowed (pre-v0.3) configs must
    be rejected loudly instead of silently running full attention."""
    d = str(tmp_path / "mistral_ckpt")
    m = CausalLM(get_spec("tiny-mistral"), lora_r=0,
                 dtype=torch.float32).random_init(31)
    save_hf_checkpoint(m, d)
    import json
    with open(os.path.join(d, "config.json")) as f:
        cfg = json.load(f)
    assert cfg["architectures"] == ["MistralForCausalLM"]
    assert cfg["sliding_window"] is None
    from distrl_llm_amd.models.hf_io import spec_from_hf_config
    s = spec_from_hf_config(d)
    assert (s.hidden_size, s.num_kv_heads, s.qkv_bias) == (64, 2, False)
    from transformers import AutoModelForCausalLM
    hf = AutoModelForCausalLM.from_pretrained(d, torch_dtype=torch.float32)
    assert type(hf).__name__ == "MistralForCausalLM"
    ids = torch.randint(0, 2048, (2, 9))
    torch.testing.assert_close(m(ids), hf(ids).logits, rtol=2e-4, atol=2e-4)
    # pre-v0.3 windowed checkpoint (window < max_position, i.e. the
    # window actually binds): rejected
    cfg["sliding_window"] = 128
    with open(os.path.join(d, "config.json"), "w") as f:
        json.dump(cfg, f)
    with pytest.raises(ValueError, match="sliding_window"):
        spec_from_hf_config(d)


def test_transformers_cross_check_qwen3(tmp_path):
    """Qwen3 family: per-head q/k RMSNorm before RoPE, no qkv bias, and
    q_size != hidden_size (heads*head_dim decoupled from hidden).
    transformers' Qwen3ForCausalLM must reproduce our logits from our
    own checkpoint — with q/k norm weights perturbed away from 1 so the
    extra norms are exercised non-trivially."""
    d = str(tmp_path / "qwen3_ckpt")
    m = CausalLM(get_spec("tiny-qwen3"), lora_r=0,
                 dtype=torch.float32).random_init(41)
    assert m.spec.q_size != m.spec.hidden_size  # 4*24=96 vs 64
    g = torch.Generator().manual_seed(7)
    with torch.no_grad():
        for layer in m.model.layers:
            at = layer.self_attn
            at.q_norm.weight.add_(torch.rand(at.q_norm.weight.shape,
                                             generator=g) - 0.5)
            at.k_norm.weight.add_(torch.rand(at.k_norm.weight.shape,
                                             generator=g) - 0.5)
    save_hf_checkpoint(m, d)
    import json
    with open(os.path.join(d, "config.json")) as f:
        cfg = json.load(f)
    assert cfg["architectures"] == ["Qwen3ForCausalLM"]
    from distrl_llm_amd.models.hf_io import spec_from_hf_config
    s = spec_from_hf_config(d)
    assert s.qk_norm and not s.qkv_bias and s.head_dim == 24
    from transformers import AutoModelForCausalLM
    hf = AutoModelForCausalLM.from_pretrained(d, torch_dtype=torch.float32)
    assert type(hf).__name__ == "Qwen3ForCausalLM"
    ids = torch.randint(0, 2048, (2, 11))
    torch.testing.assert_close(m(ids), hf(ids).logits, rtol=2e-4, atol=2e-4)


def test_engine_generation_matches_transformers_qwen3(tmp_path):
    """End-to-end Qwen3 decode cross-validation: the engine's prefill +
    paged decode (with the per-head q/k norms applied before RoPE) must
    agree token-for-token with transformers' KV-cached generate."""
    from transformers import AutoModelForCausalLM

    from distrl_llm_amd.config import EngineConfig, SamplingParams
    from distrl_llm_amd.engine import Engine

    d = str(tmp_path / "qwen3_gen")
    m = CausalLM(get_spec("tiny-qwen3"), lora_r=0,
                 dtype=torch.float32).random_init(43)
    g = torch.Generator().manual_seed(9)
    with torch.no_grad():
        for layer in m.model.layers:
            at = layer.self_attn
            at.q_norm.weight.add_(torch.rand(at.q_norm.weight.shape,
                                             generator=g) - 0.5)
            at.k_norm.weight.add_(torch.rand(at.k_norm.weight.shape,
                                             generator=g) - 0.5)
    save_hf_checkpoint(m, d)
    hf = AutoModelForCausalLM.from_pretrained(d, dtype=torch.float32)
    eng = Engine(m, EngineConfig(max_seq_length=128, kv_block_size=8,
                                 num_kv_blocks=256, max_num_seqs=16),
                 device=torch.device("cpu"), seed=0)
    prompts = [[1, 5, 9, 2, 7], [17] * 12, list(range(2, 25))]
    ours = eng.generate(prompts, SamplingParams(max_tokens=8,
                                                temperature=0.0, n=1),
                        eos_token_id=None)
    for p, o in zip(prompts, ours):
        with torch.no_grad():
            out = hf.generate(torch.tensor([p]), max_new_tokens=8,
                              do_sample=False, use_cache=True)
        assert o[0] == out[0, len(p):].tolist()
