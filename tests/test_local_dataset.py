"""Local dataset files for air-gapped runs with real data: json / jsonl /
parquet with problem+solution (or MATH-style answer) columns, via
``--dataset <path>`` (the offline counterpart of the reference's
HF load_dataset, reference train_distributed.py:38-44)."""

import json
import os
import subprocess
import sys

import pytest

from distrl_llm_amd.rl.data import load_local_rows

ROWS = [{"problem": "What is 1+1?", "solution": "2"},
        {"problem": "What is 2*3?", "solution": "6"},
        {"problem": "What is 9-4?", "solution": "5"}]


def test_jsonl(tmp_path):
    p = tmp_path / "d.jsonl"
    p.write_text("\n".join(json.dumps(r) for r in ROWS))
    assert load_local_rows(str(p)) == ROWS


def test_json_list_and_answer_remap(tmp_path):
    p = tmp_path / "d.json"
    remapped = [{"problem": r["problem"], "answer": r["solution"]}
                for r in ROWS]
    p.write_text(json.dumps(remapped))
    assert load_local_rows(str(p)) == ROWS


def test_parquet_and_directory_scan(tmp_path):
    import pandas as pd
    pd.DataFrame(ROWS).to_parquet(tmp_path / "d.parquet")
    assert load_local_rows(str(tmp_path)) == ROWS  # directory scan


def test_missing_columns_error(tmp_path):
    p = tmp_path / "bad.jsonl"
    p.write_text(json.dumps({"question": "x"}))
    with pytest.raises(ValueError, match="problem"):
        load_local_rows(str(p))


@pytest.mark.timeout(600)
def test_cli_trains_on_local_dataset_file(tmp_path):
    data = tmp_path / "math.jsonl"
    # 10 rows so the 90/10 split leaves 9 train / 1 test
    data.write_text("\n".join(json.dumps(ROWS[i % 3]) for i in range(10)))
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    env.setdefault("MASTER_ADDR", "127.0.0.1")
    out = subprocess.run(
        [sys.executable, os.path.join(repo, "train_distributed.py"),
         "--model", "tiny-qwen2", "--dataset", str(data),
         "--batch_size", "4", "--num_candidates", "2", "--topk", "2",
         "--max_new_tokens", "8", "--max_prompt_tokens", "48",
         "--number_of_actors", "1", "--number_of_learners", "1",
         "--learner_chunk_size", "2", "--train_batch_size", "2",
         "--max_lora_rank", "4", "--episodes", "1", "--eval_every", "0",
         "--save_every", "100", "--backend_device", "cpu",
         "--run_name", "localds"],
        capture_output=True, text=True, timeout=540, env=env,
        cwd=str(tmp_path))
    assert out.returncode == 0, out.stderr[-3000:]
    assert "Number of train samples: 9" in out.stdout
    assert os.path.exists(tmp_path / "run_localds")
