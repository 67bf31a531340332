"""Learner-math cross-validation against transformers: the packed
teacher-forced log-prob loss (train/learner.py) must equal the loss
computed with HF's own model implementation on the same sequences —
an independent check of the packing, position/causal semantics and the
fused loss (reference learner math: distributed_actor.py:215-261,375)."""

import pytest
import torch

from distrl_llm_amd.models import CausalLM, get_spec
from distrl_llm_amd.models.hf_io import save_hf_checkpoint
from distrl_llm_amd.train.learner import Learner
from distrl_llm_amd.utils.tokenizer import ByteTokenizer


@pytest.fixture(scope="module")
def setup(tmp_path_factory):
    d = str(tmp_path_factory.mktemp("lckpt"))
    spec = get_spec("tiny-qwen2")
    # lora_r=4 with fresh zero-B adapters: logits identical to base
    m = CausalLM(spec, lora_r=4, lora_alpha=8,
                 dtype=torch.float32).random_init(23)
    save_hf_checkpoint(m, d)
    from transformers import AutoModelForCausalLM
    hf = AutoModelForCausalLM.from_pretrained(d, dtype=torch.float32)
    tok = ByteTokenizer(vocab_size=spec.vocab_size)
    learner = Learner(m, tok, lr=1e-3, max_prompt_tokens=24,
                      max_new_tokens=24, train_batch_size=8,
                      use_8bit_adam=False)
    return learner, hf, tok


def _hf_loss(learner, hf, problems, answers, rewards):
    """Independent computation: HF forward on the learner's own packed
    encoding, per-token log-softmax gather at the predicting positions,
    masked mean, PG weighting."""
    ids, attn, targets, ans_mask, gidx = learner._encode_batch(problems,
                                                               answers)
    with torch.no_grad():
        logits = hf(ids, attention_mask=attn).logits
    H = logits.shape[-1]
    ans_logits = logits.gather(1, gidx.unsqueeze(-1).expand(-1, -1, H))
    logp = torch.log_softmax(ans_logits.float(), -1)
    tok_logp = logp.gather(-1, targets.unsqueeze(-1)).squeeze(-1)
    m = ans_mask.float()
    per_seq = (tok_logp * m).sum(-1) / m.sum(-1).clamp_min(1.0)
    return -(per_seq * rewards).mean()


def test_learner_loss_matches_transformers(setup):
    learner, hf, tok = setup
    problems = ["What is 2+2?", "Compute the sum of 3 and 4 now.",
                "Short?"]
    answers = ["<think>easy</think><answer>4</answer>",
               "<answer>7</answer>",
               "<think>a much longer reasoning chain here</think>"
               "<answer>0</answer>"]
    rewards = torch.tensor([0.8, -1.1, 0.25])
    ours = learner._micro_loss(problems, answers, rewards, loss_scale=1.0)
    theirs = _hf_loss(learner, hf, problems, answers, rewards)
    torch.testing.assert_close(ours.float(), theirs, rtol=2e-4, atol=2e-4)


def test_learner_loss_matches_transformers_with_truncation(setup):
    """Over-long prompts/answers hit the reference's truncation rules
    (first max_prompt / first max_new tokens) — the HF recomputation uses
    the same encoding, so equality checks the model+loss, while the
    explicit length assert checks the truncation."""
    learner, hf, tok = setup
    problems = ["x" * 400]   # >> max_prompt_tokens
    answers = ["y" * 400]    # >> max_new_tokens
    ids, attn, targets, ans_mask, gidx = learner._encode_batch(problems,
                                                               answers)
    assert attn.sum() == 24 + 24  # both truncated to their budgets
    rewards = torch.tensor([1.0])
    ours = learner._micro_loss(problems, answers, rewards, loss_scale=0.5)
    theirs = 0.5 * _hf_loss(learner, hf, problems, answers, rewards)
    torch.testing.assert_close(ours.float(), theirs, rtol=2e-4, atol=2e-4)
