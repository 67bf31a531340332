"""SPMD RL Trainer: the reference Trainer's orchestration, re-built on
collectives instead of Ray RPC.

Every rank runs the same program (one process per GPU). Rank 0 drives the
control flow by broadcasting (command, payload) tuples over the gloo
control plane; all ranks execute each command's collective sequence in
lockstep. This replaces the reference's Ray remote-call surface
(SURVEY.md §1 L4->L3: generate / train / compute_gradients /
apply_merged_gradients / save_adapter / save_checkpoint).

Round structure matches reference distributed_trainer.py:232-382:
generate fan-out across ALL workers (learners generate too —
reference README:19), rank-0 reward + advantage + top-k math, learner
update (gradient all-reduce over the learner subgroup instead of CPU
gather to learner 0), weight sync (RCCL broadcast instead of adapter disk
round-trip, plus the adapter save kept for checkpoint parity), metrics
with the reference's exact key set, periodic eval
(pass@1 / BoN — distributed_trainer.py:384-416) and checkpoint cadence.
"""

from __future__ import annotations

import os
import time
from typing import Dict, List, Optional

import numpy as np

from ..config import SamplingParams
from ..models import lora as lora_io
from ..parallel.fabric import Fabric
from ..utils.trace import trace_range
from .advantage import even_chunk_sizes, merge_candidates, process_candidates
from .sched import split_dict_lists, worker_chunk_sizes


class Trainer:
    def __init__(self, fabric: Fabric, config: Dict, engine, learner,
                 tokenizer, train_dataset=None, test_dataset=None,
                 reward_function=None, logger=None):
        self.fabric = fabric
        self.config = config
        self.engine = engine
        self.learner = learner  # None on pure actor ranks
        self.tokenizer = tokenizer
        self.train_dataset = train_dataset  # rank 0 only needs these
        self.test_dataset = test_dataset
        self.reward_function = reward_function
        self.logger = logger

        c = config
        self.batch_size = c["batch_size"]
        self.num_candidates = c["num_candidates"]
        self.learner_chunk_size = c["learner_chunk_size"]
        self.topk = c["topk"]
        self.learner_type = c["learner"]
        self.episodes = c["episodes"]
        self.save_every = c["save_every"]
        self.eval_every = c["eval_every"]
        self.lora_save_path = c["lora_save_path"]
        self.run_directory = f"run_{c.get('run_name') or 'default'}"
        self.sampling_params = SamplingParams(
            max_tokens=c["max_new_tokens"], temperature=c["temperature"],
            n=c["num_candidates"], top_p=0.95)
        # eval params mirror reference distributed_trainer.py:53-58
        self.eval_sampling_params = SamplingParams(
            max_tokens=c["max_new_tokens"], temperature=0.6, top_p=0.95, n=8)
        self.eos_token_id = getattr(tokenizer, "eos_token_id", None)
        # --resume <dir>: rank 0 restores counters + dataset RNG; the
        # adapter/optimizer/engine states are restored in build_worker
        self._resume_state = None
        if config.get("resume") and fabric.rank == 0:
            import torch
            st = os.path.join(config["resume"], "trainer_state.pt")
            if os.path.exists(st):
                self._resume_state = torch.load(st, weights_only=False)

    # ----------------------------------------------------------- plumbing

    def _cmd(self, name: str, payload=None):
        """Rank 0: broadcast a command then execute it locally. Each
        command carries a monotonically increasing sequence id — the
        explicit round handshake SURVEY.md §5.2 asks for in place of the
        reference's implicit adapter-file ordering."""
        assert self.fabric.rank == 0
        self._cmd_seq = getattr(self, "_cmd_seq", 0) + 1
        self.fabric.broadcast_obj((self._cmd_seq, name, payload), src=0)
        return self._handle(name, payload)

    def follower_loop(self):
        """Ranks != 0: execute the command stream until stop."""
        expected = 0
        while True:
            seq, name, payload = self.fabric.broadcast_obj(src=0)
            expected += 1
            if seq != expected:
                raise RuntimeError(
                    f"rank {self.fabric.rank}: command stream out of sync "
                    f"(got seq {seq}, expected {expected}, cmd {name!r})")
            if name == "stop":
                break
            self._handle(name, payload)

    def _handle(self, name: str, payload):
        if name == "generate":
            return self._generate_handler(payload)
        if name == "update":
            return self._update_handler(payload)
        if name == "sync_weights":
            params = self._lora_params()
            with trace_range("rl/sync_weights"):
                self.fabric.broadcast_lora(params)
            return None
        if name == "barrier":
            import torch
            if self.fabric.device.type == "cuda":
                torch.cuda.synchronize()
            self.fabric.barrier()
            if self.fabric.device.type == "cuda":
                torch.cuda.synchronize()
            return None
        if name == "save_adapter":
            if self.fabric.rank == 0:
                self._save_adapter(payload)
            return None
        if name == "save_checkpoint":
            return self._save_checkpoint_handler(payload)
        raise ValueError(f"unknown command {name!r}")

    def _save_checkpoint_handler(self, payload):
        """Full resumable checkpoint (extension over the reference's
        adapter-only save, distributed_trainer.py:377-380 / SURVEY §5.4):
        PEFT adapter + rank-0 trainer counters/dataset-RNG + lead-learner
        optimizer state + per-rank engine RNG stream. With the same world
        size, ``--resume <dir>`` continues the run bit-identically."""
        import torch
        path, meta = (payload if isinstance(payload, (tuple, list))
                      else (payload, None))
        os.makedirs(path, exist_ok=True)
        if meta is not None:
            torch.save({"generator": self.engine.generator.get_state(),
                        "seq_counter": self.engine._seq_counter},
                       os.path.join(path,
                                    f"engine_state_rank{self.fabric.rank}.pt"))
            if (self.fabric.is_learner
                    and self.fabric.rank == self.fabric.learner_ranks[0]):
                torch.save(self.learner.state_dict(),
                           os.path.join(path, "optimizer_state.pt"))
        if self.fabric.rank == 0:
            self._save_adapter(path)
            if meta is not None:
                torch.save(meta, os.path.join(path, "trainer_state.pt"))
        return None

    def _lora_params(self):
        model = self.engine.model
        return [p for p in model.parameters() if p.requires_grad]

    def _save_adapter(self, path: str):
        c = self.config
        lora_io.save_adapter(self.engine.model, path, c["model"],
                             c["max_lora_rank"], c["lora_alpha"],
                             c["lora_dropout"])

    # ----------------------------------------------------------- generate

    def _generate_handler(self, payload):
        """All ranks: generate this rank's chunk, gather to rank 0.

        payload: (batch_dict, sp_dict). Returns (candidates, duration) on
        rank 0, None elsewhere. Candidate dicts use the reference wire
        format: answers / token_lengths /
        solution / problem each replicated n-per-prompt.
        """
        batch, sp_dict = payload
        sp = SamplingParams(**sp_dict)
        t0 = time.time()
        bsz = len(batch["problem"])
        sizes = worker_chunk_sizes(bsz, self.fabric.num_actors,
                                   self.fabric.num_learners,
                                   self.learner_chunk_size)
        chunks = split_dict_lists(batch, sizes)
        my_task = chunks[self.fabric.rank]
        result = None
        if len(my_task["problem"]) > 0:
            with trace_range("rl/generate"):
                result = self._generate_task(my_task, sp)
        gathered = self.fabric.gather_obj(result, dst=0)
        if self.fabric.rank != 0:
            return None
        candidates = [g for g in gathered if g is not None]
        return candidates, time.time() - t0

    def _generate_task(self, task: Dict, sp: SamplingParams) -> Dict:
        """Run the engine on one task dict — the native vllm_generate
        (reference distributed_actor.py:147-172)."""
        prompt_ids = [self.tokenizer.encode(p) for p in task["problem"]]
        outs = self.engine.generate(prompt_ids, sp,
                                    eos_token_id=self.eos_token_id)
        task = dict(task)
        task["answers"] = [[self.tokenizer.decode(ids, skip_special_tokens=True)
                            for ids in per_prompt] for per_prompt in outs]
        task["token_lengths"] = [[len(ids) for ids in per_prompt]
                                 for per_prompt in outs]
        task["solution"] = [[s] * sp.n for s in task["solution"]]
        task["problem"] = [[p] * sp.n for p in task["problem"]]
        return task

    def _compute_rewards(self, candidates: List[Dict]) -> float:
        """Rank 0: per-group reward arrays (reference
        distributed_trainer.py:205-219)."""
        t0 = time.time()
        with trace_range("rl/reward"):
            for cand in candidates:
                rewards = []
                for answers, solutions in zip(cand["answers"], cand["solution"]):
                    rewards.append(self.reward_function(answers, solutions))
                cand["rewards"] = rewards
        return time.time() - t0

    # ------------------------------------------------------------- update

    def _update_handler(self, payload):
        """Learners: accumulate grads on their chunk, all-reduce, step.
        Returns this rank's loss (gathered by rank 0)."""
        chunks = payload  # list of (problems, answers, rewards) per learner
        loss = None
        if self.fabric.is_learner:
            problems, answers, rewards = chunks[self.fabric.learner_index]
            with trace_range("rl/update"):
                if len(problems) > 0:
                    loss = self.learner.accumulate_gradients(problems, answers,
                                                             rewards)
                else:
                    loss = 0.0
                self.fabric.allreduce_mean_grads(self.learner.params)
                self.learner.step()
            loss = self.fabric.allreduce_mean_scalar(loss)
        gathered = self.fabric.gather_obj(loss, dst=0)
        if self.fabric.rank != 0:
            return None
        losses = [x for x in gathered if x is not None]
        return float(np.mean(losses)) if losses else 0.0

    # -------------------------------------------------------------- train

    def train(self):
        """Entry point on every rank."""
        if self.fabric.rank != 0:
            return self.follower_loop()
        try:
            self._train_rank0()
        finally:
            self._cmd_seq = getattr(self, "_cmd_seq", 0) + 1
            self.fabric.broadcast_obj((self._cmd_seq, "stop", None), src=0)

    def rl_round(self, batch: Dict, sp_dict: Optional[Dict] = None) -> Dict:
        """Rank 0: one full RL round — generate fan-out, rewards,
        advantage/top-k, learner update, weight sync. Returns the round's
        metrics dict (also the unit of work bench.py times)."""
        sp_dict = sp_dict or self.sampling_params.__dict__
        candidates, gen_dur = self._cmd("generate", (batch, sp_dict))
        reward_dur = self._compute_rewards(candidates)
        candidates, stats = process_candidates(candidates, self.learner_type,
                                               self.topk)

        # split flattened training samples across learners evenly
        # (reference distributed_trainer.py:311-322)
        t0 = time.time()
        problems, answers, rewards = merge_candidates(candidates)
        sizes = even_chunk_sizes(len(problems), self.fabric.num_learners)
        chunks, start = [], 0
        for sz in sizes:
            chunks.append((problems[start:start + sz],
                           answers[start:start + sz],
                           rewards[start:start + sz]))
            start += sz
        loss = self._cmd("update", chunks)
        self._cmd("sync_weights")
        update_dur = time.time() - t0

        gen_tokens = sum(
            int(t) for cand in candidates for group in cand["token_lengths"]
            for t in group)
        sample = {
            "sample_problem": candidates[0]["problem"][0][0] if candidates else "",
            "sample_answer": candidates[0]["answers"][0][0] if candidates else "",
            "sample_reward": (float(candidates[0]["rewards"][0][0])
                              if candidates else 0.0),
        }
        stats = dict(stats)
        stats.update(sample)
        stats.update({
            "loss": loss,
            "num_samples": len(problems),
            "timing/update_duration": update_dur,
            "timing/reward_duration": reward_dur,
            "timing/generation_duration": gen_dur,
            "timing/generation_tokens_per_sec":
                gen_tokens / gen_dur if gen_dur > 0 else 0.0,
            "timing/samples_per_sec": (
                len(problems) / (gen_dur + reward_dur + update_dur)),
        })
        return stats

    def _train_rank0(self):
        if self.config.get("eval_only"):
            # standalone evaluation (--eval_only, optionally with
            # --load_adapter): pass@1/BoN over the test split, no training
            self.evaluate(0)
            return
        rs = self._resume_state
        total_batch_steps = rs["total_batch_steps"] if rs else 0
        total_samples = rs["total_samples"] if rs else 0
        start_episode = rs["episode"] if rs else 0
        skip_batches = rs["batch_in_episode"] if rs else 0
        if rs is not None:
            # restore the dataset RNG to the start of the checkpointed
            # episode so the re-shuffle reproduces the same batch order
            self.train_dataset._rng.setstate(rs["dataset_rng"])

        if self.eval_every > 0 and rs is None:
            self.evaluate(total_batch_steps)

        for episode in range(start_episode, self.episodes):
            ep_rng_state = self.train_dataset._rng.getstate()
            dataset = self.train_dataset.shuffle()
            for bi, batch in enumerate(dataset.iter(batch_size=self.batch_size)):
                if bi < skip_batches:
                    continue  # replayed prefix of a mid-episode resume
                total_batch_steps += 1
                total_samples += len(batch["problem"])

                stats = self.rl_round(batch)
                # per-round sample dump (reference
                # distributed_trainer.py:297-299)
                print(f"Sample problem: {stats['sample_problem'][:200]!r}")
                print(f"Sample answer: {stats['sample_answer'][:200]!r}")
                print(f"Sample reward: {stats['sample_reward']}")
                self._cmd("save_adapter", self.lora_save_path)

                if self.logger is not None:
                    self.logger.log({
                        "loss": stats["loss"],
                        "mean_format_reward": stats["mean_format_reward"],
                        "mean_accuracy_reward": stats["mean_accuracy_reward"],
                        "min_accuracy_reward": stats["min_accuracy_reward"],
                        "max_accuracy_reward": stats["max_accuracy_reward"],
                        "mean_token_length": stats["mean_token_length"],
                        "episode": episode,
                        "total_batch_steps": total_batch_steps,
                        "total_samples_processed": total_samples,
                        "timing/update_duration": stats["timing/update_duration"],
                        "timing/reward_duration": stats["timing/reward_duration"],
                        "timing/generation_duration": stats["timing/generation_duration"],
                        "timing/generation_tokens_per_sec":
                            stats["timing/generation_tokens_per_sec"],
                        "timing/samples_per_sec": stats["timing/samples_per_sec"],
                    }, step=total_batch_steps)

                if self.eval_every > 0 and total_batch_steps % self.eval_every == 0:
                    self.evaluate(total_batch_steps)
                if self.save_every > 0 and total_batch_steps % self.save_every == 0:
                    meta = {"episode": episode, "batch_in_episode": bi + 1,
                            "total_batch_steps": total_batch_steps,
                            "total_samples": total_samples,
                            "dataset_rng": ep_rng_state}
                    self._cmd("save_checkpoint",
                              (os.path.join(self.run_directory,
                                            f"model_{total_batch_steps}"),
                               meta))
            skip_batches = 0
            meta = {"episode": episode + 1, "batch_in_episode": 0,
                    "total_batch_steps": total_batch_steps,
                    "total_samples": total_samples,
                    "dataset_rng": self.train_dataset._rng.getstate()}
            self._cmd("save_checkpoint",
                      (os.path.join(self.run_directory,
                                    f"model_{total_batch_steps}"), meta))

    # --------------------------------------------------------------- eval

    def evaluate(self, total_steps: int):
        """Rank 0 drives; same fan-out machinery with eval sampling params
        (reference distributed_trainer.py:384-416)."""
        t0 = time.time()
        sp_dict = self.eval_sampling_params.__dict__
        total_passed = 0.0
        total_max = 0.0
        total_problems = 0
        token_lengths = []
        for batch in self.test_dataset.iter(batch_size=self.batch_size):
            candidates, _dur = self._cmd("generate", (batch, sp_dict))
            self._compute_rewards(candidates)
            for cand in candidates:
                for r, toks in zip(cand["rewards"], cand["token_lengths"]):
                    token_lengths.append(float(np.mean(toks)))
                    total_passed += float(np.mean(r[:, 1]))
                    total_max += float(np.max(r[:, 1]))
                    total_problems += 1
        n = self.eval_sampling_params.n
        if self.logger is not None and total_problems > 0:
            self.logger.log({
                f"eval/pass@1(mean{n})": total_passed / total_problems,
                f"eval/BoN({n})": total_max / total_problems,
                "eval/mean_token_length": float(np.mean(token_lengths)),
                "timing/eval_duration": time.time() - t0,
            }, step=total_steps)
