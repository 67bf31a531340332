"""Unit test of the eval metrics math (pass@1(meanN) = mean over prompts
of the candidate-mean accuracy; BoN = mean of per-prompt max accuracy —
reference distributed_trainer.py:384-416) with a stubbed generate fan-out,
so the exact numbers are asserted, not just key presence."""

from types import SimpleNamespace

import numpy as np

from distrl_llm_amd.rl.data import ListDataset
from distrl_llm_amd.rl.rewards import reward_function
from distrl_llm_amd.rl.trainer import Trainer


class _CaptureLogger:
    def __init__(self):
        self.records = []

    def log(self, d, step=None):
        self.records.append((dict(d), step))


def _make_trainer(candidates):
    config = {
        "run_name": "eval_test", "lora_save_path": "x", "batch_size": 2,
        "num_candidates": 8, "learner_chunk_size": 1, "topk": 8,
        "learner": "grpo", "episodes": 1, "save_every": 0, "eval_every": 1,
        "max_new_tokens": 8, "temperature": 1.0,
    }
    rows = [{"problem": "p1", "solution": "42"},
            {"problem": "p2", "solution": "7"}]
    t = Trainer(fabric=SimpleNamespace(rank=0), config=config, engine=None,
                learner=None, tokenizer=SimpleNamespace(eos_token_id=None),
                test_dataset=ListDataset(rows), reward_function=reward_function,
                logger=_CaptureLogger())
    t._cmd = lambda name, payload=None: (candidates, 0.01)
    return t


def test_eval_pass_at_1_and_bon_exact():
    ok = "<think>x</think><answer>42</answer>"
    ok2 = "<think>x</think><answer>7</answer>"
    bad = "<think>x</think><answer>0</answer>"
    # prompt 1: 3 of 8 correct -> mean 0.375, max 1.0
    # prompt 2: 0 of 8 correct -> mean 0.0, max 0.0
    candidates = [{
        "problem": [["p1"] * 8, ["p2"] * 8],
        "solution": [["42"] * 8, ["7"] * 8],
        "answers": [[ok] * 3 + [bad] * 5, [bad] * 8],
        "token_lengths": [[10] * 8, [20] * 8],
    }]
    t = _make_trainer(candidates)
    t.evaluate(total_steps=5)
    rec, step = t.logger.records[-1]
    assert step == 5
    assert rec["eval/pass@1(mean8)"] == (0.375 + 0.0) / 2
    assert rec["eval/BoN(8)"] == (1.0 + 0.0) / 2
    assert rec["eval/mean_token_length"] == 15.0
    # sanity: ok2 would flip prompt 2
    assert reward_function([ok2], ["7"])[0, 1] == 1.0


def test_eval_multiple_worker_chunks_aggregate():
    """Two worker-chunk dicts (the gathered fan-out shape) aggregate the
    same as one."""
    ok = "<think>x</think><answer>42</answer>"
    bad = "<answer>no</answer>"
    candidates = [
        {"problem": [["p1"] * 8], "solution": [["42"] * 8],
         "answers": [[ok] * 4 + [bad] * 4], "token_lengths": [[4] * 8]},
        {"problem": [["p2"] * 8], "solution": [["42"] * 8],
         "answers": [[ok] * 8], "token_lengths": [[6] * 8]},
    ]
    t = _make_trainer(candidates)
    t.evaluate(total_steps=1)
    rec, _ = t.logger.records[-1]
    assert rec["eval/pass@1(mean8)"] == (0.5 + 1.0) / 2
    assert rec["eval/BoN(8)"] == 1.0
    assert rec["eval/mean_token_length"] == 5.0
