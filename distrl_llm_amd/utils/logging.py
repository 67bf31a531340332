"""Metrics logging: wandb when importable, JSONL fallback with the same keys.

The key set matches the reference exactly (reference
distributed_trainer.py:348-366,412-415): loss, mean_format_reward,
mean_accuracy_reward, min/max_accuracy_reward, mean_token_length, episode,
total_batch_steps, total_samples_processed, timing/update_duration,
timing/reward_duration, timing/generation_duration, eval/pass@1(meanN),
eval/BoN(N), eval/mean_token_length, timing/eval_duration.
"""

from __future__ import annotations

import json
import os
import time
from typing import Dict, Optional


class MetricsLogger:
    def __init__(self, run_name: Optional[str], project_name: str, config: Dict,
                 out_dir: str = "."):
        self.run_name = run_name or f"run-{int(time.time())}"
        self.project_name = project_name
        self._wandb = None
        self._fh = None
        try:
            import wandb  # type: ignore
            if os.environ.get("WANDB_MODE", "") != "disabled" and os.environ.get("DISTRL_WANDB", "0") == "1":
                self._wandb = wandb.init(name=run_name, config=config, project=project_name)
        except Exception:
            self._wandb = None
        if self._wandb is None:
            os.makedirs(out_dir, exist_ok=True)
            path = os.path.join(out_dir, f"metrics_{self.run_name}.jsonl")
            self._fh = open(path, "a")
            self.path = path

    def log(self, metrics: Dict, step: Optional[int] = None) -> None:
        if self._wandb is not None:
            self._wandb.log(metrics, step=step)
        if self._fh is not None:
            rec = dict(metrics)
            rec["_step"] = step
            rec["_ts"] = time.time()
            self._fh.write(json.dumps(rec) + "\n")
            self._fh.flush()

    def finish(self) -> None:
        if self._wandb is not None:
            self._wandb.finish()
        if self._fh is not None:
            self._fh.close()
            self._fh = None
