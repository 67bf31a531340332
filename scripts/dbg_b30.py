#!/usr/bin/env python3
"""Diagnose the batch-30 update skip."""
import os, sys, time
sys.path.insert(0, ".")
os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29741")
os.environ.setdefault("RANK", "0")
os.environ.setdefault("WORLD_SIZE", "1")
import numpy as np
from distrl_llm_amd.parallel.worker import build_worker
from distrl_llm_amd.rl.data import ListDataset, process_dataset, r1_preprompt, synthetic_math_dataset
from distrl_llm_amd.rl.rewards import reward_function as rf
import zlib

def reward_function(completions, solutions):
    r = rf(completions, solutions)
    jit = np.array([(zlib.crc32(c[-256:].encode()) % 1000) / 1000.0 * 0.1 for c in completions])
    r[:, 0] += jit
    return r

B = int(os.environ.get("DBG_B", "30"))
rows = process_dataset(None, synthetic_math_dataset(B * 2, seed=17), r1_preprompt)
config = {
    "run_name": "dbg", "project_name": "dbg", "lora_save_path": "/tmp/dbg_adapter",
    "lr": 2e-5, "max_new_tokens": int(os.environ.get("DBG_MAXNEW", "64")), "max_prompt_tokens": 350,
    "num_candidates": 16, "episodes": 1, "batch_size": B, "train_batch_size": 16,
    "temperature": 1.2, "save_every": 10**9, "eval_every": 0,
    "model": "unsloth/Qwen2.5-7B-Instruct-bnb-4bit", "dataset": "synthetic",
    "number_of_actors": 0, "number_of_learners": 1, "learner": "grpo",
    "use_vllm": True, "max_lora_rank": 32, "topk": 16,
    "learner_chunk_size": B, "actor_gpu_usage": 0.5, "learner_gpu_usage": 0.2,
    "lora_alpha": 16, "lora_dropout": 0.0, "seed": 3407, "use_8bit_adam": True,
}
train = ListDataset(rows, seed=17)
ov = {}
if os.environ.get("DBG_POOL"):
    ov = {"num_kv_blocks": int(os.environ["DBG_POOL"])}
tr = build_worker(0, 1, config, train_dataset=train, test_dataset=ListDataset(rows[:2]),
                  reward_function=reward_function, engine_overrides=ov)
batch = next(iter(train.iter(batch_size=B)))
# instrument: capture the generate output before the update
sp_dict = dict(tr.sampling_params.__dict__)
cands, gen_dur = tr._cmd("generate", (batch, sp_dict))
n_empty = sum(1 for c in cands for grp in c["answers"] for a in grp if len(a) == 0)
tls = [t for c in cands for grp in c["token_lengths"] for t in grp]
div = []
for c in cands:
    for grp in c["answers"]:
        div.append(len(set(grp)))
print("gen_dur", round(gen_dur, 1), "n_answers", len(tls), "n_empty", n_empty,
      "tok_len min/max", min(tls), max(tls), "distinct-per-group", sorted(div)[:5], "...", sorted(div)[-3:])
# second generation (cache-reuse path) diversity
cands_b, gen_b = tr._cmd("generate", (batch, sp_dict))
div_b = [len(set(grp)) for c in cands_b for grp in c["answers"]]
print("gen2", round(gen_b, 1), "distinct-per-group", sorted(div_b)[:6], "...", sorted(div_b)[-3:])
# analyze CALL-2 candidates (the cache-reuse path that fails)
from distrl_llm_amd.rl.advantage import process_candidates, merge_candidates
import numpy as _np
rd = tr._compute_rewards(cands_b)
raw_stds = []
for c in cands_b:
    for g in c["rewards"]:
        raw_stds.append(float(_np.asarray(g).sum(axis=1).std()))
print("call2 raw reward group stds: min", min(raw_stds), "max", max(raw_stds),
      "zeros", sum(1 for x in raw_stds if x == 0.0))
g0 = cands_b[0]["answers"][0]
print("g0 ans lens", [len(a) for a in g0[:6]])
print("g0 a0 tail", repr(g0[0][-48:]))
print("g0 a1 tail", repr(g0[1][-48:]))
cands2, st2 = process_candidates(cands_b, tr.learner_type, tr.topk)
probs, ans, rews = merge_candidates(cands2)
ra = _np.array(rews)
print("call2 advantages: nonzero", int((ra != 0).sum()), "std", float(ra.std()))
stats = {"skip": 1}

# candidate diversity
cands, _ = None, None
print("sample answer repr:", repr(stats.get("sample_answer", ""))[:120])
