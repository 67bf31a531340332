"""Process fabric: one process per GPU, torch.distributed over RCCL/xGMI.

Native replacement for the reference's Ray transport (Ray
object store + shared-filesystem adapter sync, zero collectives). Here:

- one ``torch.distributed`` world spans all actor + learner ranks
  (backend "cpu:gloo,cuda:nccl" — nccl IS RCCL on ROCm; gloo carries the
  CPU control plane: prompt dicts, generation strings, rewards);
- a learner subgroup carries the gradient all-reduce (replacing the
  reference's CPU gather-to-learner-0 staging, distributed_trainer.py:
  329-342);
- LoRA weight sync is a single flat-bucket broadcast from the first
  learner rank to the world each round (replacing the reference's
  save_lora/load_lora disk round-trip, distributed_actor.py:84-86,150).
  Payloads are tiny (~40-80 MB), so single-shot latency-bound collectives
  are the right shape for xGMI's 7x153 GB/s point-to-point links — not
  multi-hop rings.

Rank layout matches the reference's GPU assignment (distributed_actor.py:
535-537): ranks [0, num_actors) are actors, [num_actors, world) learners.
"""

from __future__ import annotations

import datetime
import os
from typing import List, Optional

import torch
import torch.distributed as dist


class Fabric:
    def __init__(self, rank: int, world_size: int, num_actors: int,
                 num_learners: int, device: torch.device,
                 timeout_s: float = 240.0):
        """``timeout_s`` mirrors the reference's 240 s ray.get timeouts
        (distributed_trainer.py:200,333) — a stuck collective crashes the
        run with a diagnostic rather than hanging."""
        assert world_size == num_actors + num_learners
        assert 0 <= rank < world_size
        self.rank = rank
        self.world_size = world_size
        self.num_actors = num_actors
        self.num_learners = num_learners
        self.device = device
        self.is_actor = rank < num_actors
        self.is_learner = not self.is_actor
        self.learner_index = rank - num_actors if self.is_learner else -1
        self.learner_ranks = list(range(num_actors, world_size))
        self.lead_learner_rank = num_actors  # "learner 0"

        if not dist.is_initialized():
            backend = "cpu:gloo,cuda:nccl" if device.type == "cuda" else "gloo"
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29531")
            dist.init_process_group(
                backend=backend, rank=rank, world_size=world_size,
                timeout=datetime.timedelta(seconds=timeout_s),
                device_id=device if device.type == "cuda" else None)
        self._owns_pg = True

        # explicit gloo group for the CPU control plane: object collectives
        # (prompt dicts, generations) and barriers are pinned here so their
        # routing never depends on the combined backend's default device
        if device.type == "cuda":
            self.cpu_group = dist.new_group(
                backend="gloo",
                timeout=datetime.timedelta(seconds=timeout_s))
        else:
            self.cpu_group = None  # default group already is gloo

        self.learner_group = dist.new_group(self.learner_ranks,
                                            timeout=datetime.timedelta(seconds=timeout_s))

    # ------------------------------------------------------ control plane

    def broadcast_obj(self, obj=None, src: int = 0):
        box = [obj]
        dist.broadcast_object_list(box, src=src, group=self.cpu_group,
                                   device=torch.device("cpu"))
        return box[0]

    def gather_obj(self, obj, dst: int = 0) -> Optional[List]:
        out = [None] * self.world_size if self.rank == dst else None
        dist.gather_object(obj, out, dst=dst, group=self.cpu_group)
        return out

    def barrier(self):
        dist.barrier(group=self.cpu_group)

    # ------------------------------------------------------- data plane

    def allreduce_mean_grads(self, params: List[torch.nn.Parameter]) -> None:
        """Average gradients across the learner subgroup as one flat bucket
        (replaces the reference's CPU gradient staging)."""
        if self.num_learners <= 1 or not self.is_learner:
            return
        grads = [p.grad if p.grad is not None else torch.zeros_like(p)
                 for p in params]
        flat = torch.cat([g.reshape(-1).float() for g in grads])
        dist.all_reduce(flat, op=dist.ReduceOp.SUM, group=self.learner_group)
        flat /= self.num_learners
        off = 0
        for p, g in zip(params, grads):
            n = g.numel()
            p.grad = flat[off:off + n].view_as(p).to(p.dtype)
            off += n

    def allreduce_mean_scalar(self, value: float) -> float:
        """Mean of a python scalar over the learner subgroup (losses)."""
        if self.num_learners <= 1 or not self.is_learner:
            return value
        t = torch.tensor([value], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.SUM, group=self.learner_group)
        return float(t.item() / self.num_learners)

    def broadcast_lora(self, params: List[torch.nn.Parameter]) -> None:
        """Flat-bucket broadcast of the LoRA tensors from the lead learner
        to every rank — the weight-sync point of each round (replaces the
        reference's adapter disk round-trip)."""
        if self.world_size <= 1:
            return
        flat = torch.cat([p.detach().reshape(-1) for p in params])
        if self.device.type == "cuda":
            flat = flat.to(self.device)
        dist.broadcast(flat, src=self.lead_learner_rank)
        if self.rank != self.lead_learner_rank:
            off = 0
            with torch.no_grad():
                for p in params:
                    n = p.numel()
                    p.copy_(flat[off:off + n].view_as(p).to(p.device, p.dtype))
                    off += n

    def close(self):
        if dist.is_initialized():
            dist.destroy_process_group()
