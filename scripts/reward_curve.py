#!/usr/bin/env python3
"""Learning demonstration: a rising reward curve from the FULL RL stack.

BASELINE.json's headline metric is samples/sec PLUS the 2-hour reward
curve (reference README:73-85). There is no network for MATH-500 or
pretrained weights, so this is the offline-learnable equivalent: a real
BPE tokenizer (trained on the spot) whose vocabulary contains the
reference reward's four XML tag strings as single tokens, a random-init
small model, and the UNMODIFIED reference reward stack
(rl/rewards.py == reference reward_functions.py semantics). The policy
must discover — purely from GRPO advantages — that emitting the
<think>/<answer> skeleton earns count_xml/soft-format reward and that
placing the prompt's number inside <answer> earns accuracy reward.
Every component is the production stack: engine generation, rewards,
whitened group advantages, fused loss, LoRA backward, Adam8bit.

Run (GPU): python scripts/reward_curve.py --steps 150
Outputs: profiles/reward_curve.jsonl (reference metric keys per step)
and a windowed summary on stdout.
"""
import argparse
import json
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def build_tokenizer_dir(d: str) -> str:
    from tokenizers import Tokenizer
    from tokenizers.decoders import ByteLevel as ByteLevelDecoder
    from tokenizers.models import BPE
    from tokenizers.pre_tokenizers import ByteLevel
    from tokenizers.trainers import BpeTrainer

    os.makedirs(d, exist_ok=True)
    corpus = []
    for n in range(10):
        corpus.append(f"The display shows the number {n}. "
                      f"What number is on the display?")
    corpus += [
        "A conversation between User and Assistant.",
        "respond in the following format",
        "<think>\nreasoning\n</think>\n<answer>\nanswer\n</answer>",
    ]
    corpus = corpus * 40
    tok = Tokenizer(BPE(unk_token=None))
    tok.pre_tokenizer = ByteLevel(add_prefix_space=False)
    tok.decoder = ByteLevelDecoder()
    trainer = BpeTrainer(vocab_size=572, special_tokens=[
        "<|endoftext|>", "<|im_start|>", "<|im_end|>"])
    tok.train_from_iterator(corpus, trainer)
    # the four count_xml tag patterns are single REGULAR tokens (added,
    # not special — decode(skip_special_tokens=True) must keep them or
    # the reward never sees them): the tokenizer DESIGN choice that makes
    # the format reward discoverable by exploration (each tag is one
    # sampling event, not a 5-token coincidence)
    tok.add_tokens(["<think>\n", "\n</think>\n", "\n<answer>\n",
                    "\n</answer>"])
    tok.save(os.path.join(d, "tokenizer.json"))
    template = (
        "{% for message in messages %}"
        "{{ '<|im_start|>' + message['role'] + '\n' + message['content']"
        " + '<|im_end|>' + '\n' }}"
        "{% endfor %}"
        "{% if add_generation_prompt %}{{ '<|im_start|>assistant\n' }}"
        "{% endif %}")
    with open(os.path.join(d, "tokenizer_config.json"), "w") as f:
        json.dump({"tokenizer_class": "PreTrainedTokenizerFast",
                   "eos_token": "<|endoftext|>",
                   "pad_token": "<|endoftext|>",
                   "chat_template": template}, f)
    # model config: small-qwen2-like dims, vocab covering the tokenizer
    with open(os.path.join(d, "config.json"), "w") as f:
        json.dump({"architectures": ["Qwen2ForCausalLM"],
                   "hidden_size": 512, "intermediate_size": 1024,
                   "num_hidden_layers": 4, "num_attention_heads": 8,
                   "num_key_value_heads": 4, "head_dim": 64,
                   "vocab_size": 640, "rope_theta": 1e5,
                   "rms_norm_eps": 1e-6, "tie_word_embeddings": False,
                   "max_position_embeddings": 2048,
                   "attention_bias": True}, f)
    return d


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=150)
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--cands", type=int, default=8)
    ap.add_argument("--lr", type=float, default=5e-4)
    ap.add_argument("--out", type=str, default="profiles/reward_curve.jsonl")
    args = ap.parse_args()

    tok_dir = build_tokenizer_dir("/tmp/reward_curve_tok")

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29611")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")

    from distrl_llm_amd.parallel.worker import build_worker
    from distrl_llm_amd.rl.data import ListDataset, process_dataset
    from distrl_llm_amd.rl.rewards import reward_function
    from distrl_llm_amd.utils.tokenizer import load_tokenizer

    tokenizer = load_tokenizer(tok_dir, 640)
    rows = []
    import random
    rng = random.Random(7)
    for i in range(args.batch * (args.steps + 2)):
        n = rng.randint(0, 9)
        rows.append({
            "problem": f"The display shows the number {n}. "
                       f"What number is on the display?",
            "solution": str(n),
        })
    preprompt = ("Respond in the following format: "
                 "<think>\n...\n</think>\n<answer>\n...\n</answer>")
    rows = process_dataset(tokenizer, rows, preprompt)

    config = {
        "run_name": "reward_curve", "project_name": "curve",
        "lora_save_path": "/tmp/reward_curve_adapter",
        "lr": args.lr, "max_new_tokens": 24, "max_prompt_tokens": 96,
        "num_candidates": args.cands, "episodes": 1,
        "batch_size": args.batch, "train_batch_size": 16,
        "temperature": 1.5, "save_every": 10**9, "eval_every": 0,
        "model": tok_dir, "dataset": "synthetic",
        "number_of_actors": 0, "number_of_learners": 1,
        "learner": "grpo", "use_vllm": True, "max_lora_rank": 32,
        "topk": args.cands, "learner_chunk_size": args.batch,
        "actor_gpu_usage": 0.3, "learner_gpu_usage": 0.3,
        "lora_alpha": 32, "lora_dropout": 0.0, "seed": 3407,
        "use_8bit_adam": True,
    }
    train = ListDataset(rows, seed=7)
    overrides = {}
    if not torch.cuda.is_available():
        overrides = {"num_kv_blocks": 2048, "kv_block_size": 8,
                     "max_seq_length": 128}
    trainer = build_worker(0, 1, config, train_dataset=train,
                           test_dataset=ListDataset(rows[:2]),
                           reward_function=reward_function,
                           engine_overrides=overrides)
    sp_dict = dict(trainer.sampling_params.__dict__)

    os.makedirs(os.path.dirname(args.out), exist_ok=True)
    batches = list(train.iter(batch_size=args.batch))
    hist = []
    with open(args.out, "w") as f:
        for step in range(args.steps):
            stats = trainer.rl_round(batches[step % len(batches)], sp_dict)
            rec = {k: v for k, v in stats.items()
                   if isinstance(v, (int, float))}
            rec["step"] = step
            f.write(json.dumps(rec) + "\n")
            f.flush()
            hist.append(rec)
            if (step + 1) % 10 == 0:
                w = hist[-10:]
                mf = sum(r.get("mean_format_reward", 0) for r in w) / len(w)
                ma = sum(r.get("mean_accuracy_reward", 0) for r in w) / len(w)
                print(f"step {step+1:4d}: mean_format_reward(10) {mf:+.4f} "
                      f"mean_accuracy_reward(10) {ma:.4f} "
                      f"loss {hist[-1].get('loss', float('nan')):+.4f}",
                      flush=True)
    first = hist[:10]
    last = hist[-10:]
    f0 = sum(r.get("mean_format_reward", 0) for r in first) / len(first)
    f1 = sum(r.get("mean_format_reward", 0) for r in last) / len(last)
    a0 = sum(r.get("mean_accuracy_reward", 0) for r in first) / len(first)
    a1 = sum(r.get("mean_accuracy_reward", 0) for r in last) / len(last)
    print(f"CURVE format {f0:+.4f} -> {f1:+.4f} | "
          f"accuracy {a0:.4f} -> {a1:.4f}")


if __name__ == "__main__":
    main()
