"""Data-prep units: chat templating fallback, the CLI's dataset loading
(90/10 split — reference train_distributed.py:44 — synthetic fallback,
local files) without spawning workers."""

import json
from types import SimpleNamespace

from distrl_llm_amd.rl.data import (apply_template, default_chat_template,
                                    process_dataset, r1_preprompt,
                                    synthetic_math_dataset)


def test_chatml_fallback_template():
    msgs = [{"role": "system", "content": "sys"},
            {"role": "user", "content": "q?"}]
    out = apply_template(None, msgs)
    assert out == default_chat_template(msgs)
    assert out.startswith("<|im_start|>system\nsys<|im_end|>\n")
    assert out.endswith("<|im_start|>assistant\n")


def test_process_dataset_wraps_preprompt():
    rows = synthetic_math_dataset(2, seed=0)
    out = process_dataset(None, rows, r1_preprompt, postprompt="P!")
    for r in out:
        assert r1_preprompt in r["problem"]
        assert "P!" in r["problem"]
        assert r["solution"]  # untouched


def _args(**over):
    base = dict(synthetic_dataset=0, dataset="HuggingFaceH4/MATH-500",
                seed=3407)
    base.update(over)
    return SimpleNamespace(**base)


def test_load_datasets_split_and_fallback(tmp_path, capsys):
    from train_distributed import load_datasets

    # offline: HF load fails -> synthetic fallback, 90/10 split of 500
    train, test = load_datasets(_args(), tokenizer=None)
    assert len(train) == 450 and len(test) == 50
    assert "falling back to synthetic" in capsys.readouterr().out

    # explicit synthetic size
    train, test = load_datasets(_args(synthetic_dataset=20), tokenizer=None)
    assert len(train) == 18 and len(test) == 2

    # local file
    p = tmp_path / "d.jsonl"
    rows = [{"problem": f"q{i}", "solution": str(i)} for i in range(30)]
    p.write_text("\n".join(json.dumps(r) for r in rows))
    train, test = load_datasets(_args(dataset=str(p)), tokenizer=None)
    assert len(train) == 27 and len(test) == 3
    assert r1_preprompt in train.rows[0]["problem"]
