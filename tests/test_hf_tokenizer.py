"""Real-tokenizer path: ``load_tokenizer`` must pick up a local HF
tokenizer directory (the reference's deployment mode — it downloads the
model repo's tokenizer, reference distributed_actor.py:217-229) and the
data/trainer plumbing must work with it. No network: the tokenizer is
trained on the spot with the ``tokenizers`` library and saved locally.
"""

import json
import os

import pytest


@pytest.fixture(scope="module")
def hf_tokenizer_dir(tmp_path_factory):
    from tokenizers import Tokenizer
    from tokenizers.decoders import ByteLevel as ByteLevelDecoder
    from tokenizers.models import BPE
    from tokenizers.pre_tokenizers import ByteLevel
    from tokenizers.trainers import BpeTrainer

    d = tmp_path_factory.mktemp("hf_tok")
    corpus = [
        "A conversation between User and Assistant.",
        "<think> reasoning process here </think>",
        "<answer> 42 </answer>",
        "Let $x = 3$ and $y = 5$. Compute $x + y \\cdot 7$.",
    ] * 50
    tok = Tokenizer(BPE(unk_token=None))
    tok.pre_tokenizer = ByteLevel(add_prefix_space=False)
    tok.decoder = ByteLevelDecoder()
    trainer = BpeTrainer(vocab_size=600,
                         special_tokens=["<|endoftext|>", "<|im_start|>",
                                         "<|im_end|>"])
    tok.train_from_iterator(corpus, trainer)
    tok.save(str(d / "tokenizer.json"))
    # ChatML template (the Qwen2 chat format the reference models use)
    template = (
        "{% for message in messages %}"
        "{{ '<|im_start|>' + message['role'] + '\n' + message['content'] + '<|im_end|>' + '\n' }}"
        "{% endfor %}"
        "{% if add_generation_prompt %}{{ '<|im_start|>assistant\n' }}{% endif %}")
    with open(d / "tokenizer_config.json", "w") as f:
        json.dump({"tokenizer_class": "PreTrainedTokenizerFast",
                   "eos_token": "<|endoftext|>",
                   "pad_token": "<|endoftext|>",
                   "chat_template": template}, f)
    return str(d)


def test_load_tokenizer_prefers_local_hf_dir(hf_tokenizer_dir):
    from distrl_llm_amd.utils.tokenizer import ByteTokenizer, load_tokenizer
    tok = load_tokenizer(hf_tokenizer_dir, vocab_size=152064)
    assert not isinstance(tok, ByteTokenizer)
    text = "Compute $x + y \\cdot 7$."
    ids = tok.encode(text)
    assert isinstance(ids, list) and all(isinstance(i, int) for i in ids)
    assert tok.decode(ids, skip_special_tokens=True) == text
    assert tok.eos_token_id is not None


def test_load_tokenizer_falls_back_offline(tmp_path):
    from distrl_llm_amd.utils.tokenizer import ByteTokenizer, load_tokenizer
    # model *names* (no local dir) always get the offline tokenizer
    assert isinstance(load_tokenizer("unsloth/Qwen2.5-7B-Instruct-bnb-4bit",
                                     152064), ByteTokenizer)
    # an empty directory isn't a tokenizer dir either
    assert isinstance(load_tokenizer(str(tmp_path), 152064), ByteTokenizer)


def test_process_dataset_uses_hf_chat_template(hf_tokenizer_dir):
    from distrl_llm_amd.rl.data import (process_dataset, r1_preprompt,
                                        synthetic_math_dataset)
    from distrl_llm_amd.utils.tokenizer import load_tokenizer
    tok = load_tokenizer(hf_tokenizer_dir)
    rows = process_dataset(tok, synthetic_math_dataset(3, seed=0),
                           r1_preprompt)
    for r in rows:
        assert r["problem"].startswith("<|im_start|>system")
        assert r["problem"].endswith("<|im_start|>assistant\n")


def test_engine_roundtrip_with_hf_tokenizer(hf_tokenizer_dir):
    """The generate path (encode -> engine -> decode) works with an HF
    tokenizer whose vocab is smaller than the model's."""
    import torch
    from distrl_llm_amd.config import EngineConfig, SamplingParams
    from distrl_llm_amd.engine import Engine
    from distrl_llm_amd.models import CausalLM, get_spec
    from distrl_llm_amd.utils.tokenizer import load_tokenizer

    tok = load_tokenizer(hf_tokenizer_dir)
    spec = get_spec("tiny-qwen2")
    model = CausalLM(spec, lora_r=4, lora_alpha=8, dtype=torch.float32)
    model.random_init(seed=3)
    engine = Engine(model, EngineConfig(max_seq_length=128, kv_block_size=8,
                                        num_kv_blocks=128, max_num_seqs=16),
                    device=torch.device("cpu"), seed=0)
    ids = tok.encode("Compute $3 + 5$.")
    assert max(ids) < spec.vocab_size
    outs = engine.generate([ids], SamplingParams(max_tokens=4, temperature=0.0,
                                                 n=1),
                           eos_token_id=tok.eos_token_id)
    text = tok.decode([t for t in outs[0][0] if t < tok.vocab_size],
                      skip_special_tokens=True)
    assert isinstance(text, str)
