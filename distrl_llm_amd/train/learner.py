"""Learner: teacher-forced log-prob training of the LoRA adapter.

Native replacement for the reference learner stack (reference
distributed_actor.py:196-333,336-514): fixed-shape batching (prompts
left-padded/truncated to max_prompt_tokens, answers right-padded/truncated
to max_new_tokens — distributed_actor.py:215-239), micro-batched gradient
accumulation over train_batch_size, bf16 autocast on GPU, the PG loss
``-mean((sum logp*mask / sum mask) * R)`` (distributed_actor.py:375) — which
is also the exact GRPO gradient since the reference's one-step on-policy
surrogate ``exp(logp - logp.detach())`` is identically 1 (SURVEY.md
§2.6-6) — and 8-bit Adam.

Deviations (documented):
- the degenerate-batch skip implements the reference's INTENT (skip when
  all advantages are zero), not its buggy ``batch_rewards.all() == 0``
  form;
- log-probs + loss run through the fused HIP kernel emitting dlogits
  directly instead of the reference's per-row log_softmax+gather loop
  (distributed_actor.py:253-260);
- the LM head is evaluated only on the answer region (the prompt region's
  logits are never formed).
"""

from __future__ import annotations

from typing import List, Optional, Sequence

import torch

from ..models.lora import trainable_parameters
from ..models.model import CausalLM
from ..ops import functional as OF
from .optim import make_optimizer


class Learner:
    def __init__(self, model: CausalLM, tokenizer, lr: float,
                 max_prompt_tokens: int, max_new_tokens: int,
                 train_batch_size: int, use_8bit_adam: bool = True,
                 pad_token_id: Optional[int] = None):
        self.model = model
        self.tokenizer = tokenizer
        self.max_prompt_tokens = max_prompt_tokens
        self.max_new_tokens = max_new_tokens
        self.train_batch_size = train_batch_size
        self.device = model.device
        self.params = trainable_parameters(model)
        self.optimizer = make_optimizer(self.params, lr, use_8bit_adam)
        self.pad_token_id = (pad_token_id if pad_token_id is not None
                             else getattr(tokenizer, "pad_token_id", 0) or 0)

    # -------------------------------------------------- checkpoint state

    def state_dict(self):
        return {"optimizer": self.optimizer.state_dict()}

    def load_state_dict(self, sd):
        self.optimizer.load_state_dict(sd["optimizer"])

    # ------------------------------------------------------- tokenization

    def _encode_batch(self, problems: Sequence[str], answers: Sequence[str]):
        """Packed right-padded layout: [prompt(<=max_prompt) | answer
        (<=max_new) | pad]. Keeps the reference's truncation semantics
        (prompt keeps its first max_prompt_tokens, answers their first
        max_new_tokens — distributed_actor.py:217-229) but packs instead of
        left-padding, so causal attention needs no mask (flash-capable) and
        trailing pad compute shrinks to the batch max.
        """
        P, A = self.max_prompt_tokens, self.max_new_tokens
        B = len(problems)
        enc = [(self.tokenizer.encode(p)[:P], self.tokenizer.encode(a)[:A])
               for p, a in zip(problems, answers)]
        T = max(len(pi) + len(ai) for pi, ai in enc)
        T = min((T + 7) // 8 * 8, P + A)  # pad batch length to multiple of 8
        A_max = max(max(len(ai) for _, ai in enc), 1)
        input_ids = torch.full((B, T), self.pad_token_id, dtype=torch.long)
        attn = torch.zeros(B, T, dtype=torch.long)
        targets = torch.full((B, A_max), self.pad_token_id, dtype=torch.long)
        ans_mask = torch.zeros(B, A_max, dtype=torch.long)
        gather_idx = torch.zeros(B, A_max, dtype=torch.long)
        for i, (pi, ai) in enumerate(enc):
            L = len(pi) + len(ai)
            input_ids[i, :L] = torch.tensor(pi + ai)
            attn[i, :L] = 1
            if ai:
                targets[i, :len(ai)] = torch.tensor(ai)
                ans_mask[i, :len(ai)] = 1
            # hidden positions predicting the answer tokens:
            # len(pi)-1 .. len(pi)+len(ai)-2 (clamped into range)
            start = max(len(pi) - 1, 0)
            gather_idx[i] = torch.arange(start, start + A_max).clamp_max(T - 1)
        dev = self.device
        return (input_ids.to(dev), attn.to(dev), targets.to(dev),
                ans_mask.to(dev), gather_idx.to(dev))

    # -------------------------------------------------------------- loss

    def _micro_loss(self, problems, answers, rewards: torch.Tensor,
                    loss_scale: float) -> torch.Tensor:
        input_ids, attn, targets, ans_mask, gidx = self._encode_batch(
            problems, answers)
        hidden = self.model.forward_hidden(input_ids, attn)
        H = hidden.shape[-1]
        ans_hidden = hidden.gather(
            1, gidx.unsqueeze(-1).expand(-1, -1, H))
        ans_logits = self.model.logits(ans_hidden)
        return OF.logprob_loss(ans_logits, targets, ans_mask, rewards, loss_scale)

    def accumulate_gradients(self, problems: List[str], answers: List[str],
                             rewards: List[float]) -> float:
        """Zero grads, run gradient accumulation over micro-batches of
        train_batch_size; returns the (unscaled) total loss. Does NOT step
        the optimizer — the caller all-reduces across learners first."""
        self.model.train()
        self.optimizer.zero_grad(set_to_none=True)
        rewards_t = torch.tensor(rewards, dtype=torch.float32, device=self.device)
        B = len(problems)
        mb = self.train_batch_size
        num_batches = (B + mb - 1) // mb
        total_loss = 0.0
        for i in range(num_batches):
            s, e = i * mb, min((i + 1) * mb, B)
            r = rewards_t[s:e]
            if bool((r == 0).all()):
                # degenerate group: every advantage zero -> no gradient signal
                continue
            ctx = (torch.autocast("cuda", dtype=torch.bfloat16)
                   if self.device.type == "cuda" else torch.autocast("cpu", enabled=False))
            with ctx:
                loss = self._micro_loss(problems[s:e], answers[s:e], r,
                                        loss_scale=1.0 / num_batches)
            loss.backward()
            total_loss += float(loss.item()) * num_batches
        return total_loss

    def step(self) -> None:
        self.optimizer.step()
        self.optimizer.zero_grad(set_to_none=True)
