"""Small-surface units: the JSONL metrics logger (reference key set,
wandb-off path — reference distributed_trainer.py:237-239,348-366) and
the trace no-op guarantee on CPU-only hosts."""

import json
import os

import torch


def test_metrics_logger_jsonl(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    monkeypatch.delenv("DISTRL_WANDB", raising=False)
    from distrl_llm_amd.utils.logging import MetricsLogger
    lg = MetricsLogger("unit", "proj", {"lr": 1e-5, "model": "tiny"})
    lg.log({"loss": 0.5, "mean_format_reward": 0.1}, step=1)
    lg.log({"eval/BoN(8)": 0.25}, step=2)
    path = os.path.join(str(tmp_path), "metrics_unit.jsonl")
    assert os.path.exists(path)
    recs = [json.loads(l) for l in open(path)]
    assert recs[0]["loss"] == 0.5 and recs[0]["_step"] == 1
    assert recs[1]["eval/BoN(8)"] == 0.25


def test_trace_range_noop_without_gpu():
    from distrl_llm_amd.utils import trace
    if torch.cuda.is_available():  # CPU-only CI guarantee
        return
    executed = []
    with trace.trace_range("x"):
        executed.append(1)
    trace.trace_mark("y")
    assert executed == [1]


def test_trace_range_propagates_exceptions():
    from distrl_llm_amd.utils.trace import trace_range
    import pytest
    with pytest.raises(KeyError):
        with trace_range("x"):
            raise KeyError("boom")
