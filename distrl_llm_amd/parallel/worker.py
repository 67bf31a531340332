"""Per-rank worker construction shared by the CLI (spawn mode) and
bench.py / torchrun (SPMD mode).

Replaces the reference's actor factory (reference distributed_actor.py:
517-585): GPU discovery, actor-first/learner-last placement, per-role GPU
memory fractions, learner-type selection — as deterministic per-rank
construction instead of Ray placement groups.
"""

from __future__ import annotations

import os
from typing import Dict, Optional

import torch

from ..config import EngineConfig
from ..engine.engine import Engine
from ..models.model import CausalLM
from ..models.spec import is_4bit_model_name
from ..parallel.fabric import Fabric
from ..rl.trainer import Trainer
from ..train.learner import Learner
from ..utils.logging import MetricsLogger
from ..utils.tokenizer import load_tokenizer


def build_worker(rank: int, world_size: int, config: Dict,
                 train_dataset=None, test_dataset=None, reward_function=None,
                 device: Optional[torch.device] = None,
                 engine_overrides: Optional[Dict] = None) -> Trainer:
    """Construct model + engine + learner + fabric + trainer for one rank.

    Rank layout: [0, num_actors) = actors, [num_actors, world) = learners
    (reference distributed_actor.py:535-537).
    """
    num_actors = config["number_of_actors"]
    num_learners = config["number_of_learners"]
    assert world_size == num_actors + num_learners

    if device is None:
        if torch.cuda.is_available():
            local = int(os.environ.get("LOCAL_RANK", rank))
            if torch.cuda.device_count() < world_size:
                raise RuntimeError(
                    f"Not enough GPUs available. Available: "
                    f"{torch.cuda.device_count()}, Required: {world_size}")
            device = torch.device(f"cuda:{local}")
            torch.cuda.set_device(device)
        else:
            device = torch.device("cpu")

    from ..models.hf_io import (is_hf_checkpoint_dir, load_hf_checkpoint,
                                resolve_spec)
    spec = resolve_spec(config["model"])
    dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
    seed = int(config.get("seed", 3407))

    model = CausalLM(spec, lora_r=config["max_lora_rank"],
                     lora_alpha=config["lora_alpha"],
                     lora_dropout=config["lora_dropout"],
                     dtype=dtype, device=device)
    model.random_init(seed)  # identical on every rank (replicated base + LoRA)
    if is_hf_checkpoint_dir(config["model"]):
        # local pretrained checkpoint (the reference's from_pretrained
        # path, reference distributed_actor.py:58-66) — loaded before any
        # nf4 quantization
        load_hf_checkpoint(model, config["model"])
    load_4bit = config.get("load_in_4bit")
    if load_4bit is None:
        load_4bit = is_4bit_model_name(config["model"])
    if load_4bit:
        model.quantize_nf4_()

    if config.get("load_adapter") or config.get("resume"):
        # warm-start from a PEFT adapter directory (every rank loads the
        # same file, so replicas stay identical); --resume implies loading
        # the checkpoint's adapter
        from ..models.lora import adapter_hyperparams, load_adapter
        adir = config.get("resume") or config["load_adapter"]
        ar, aa, _ad = adapter_hyperparams(adir)
        if ar != config["max_lora_rank"] or aa != config["lora_alpha"]:
            raise ValueError(
                f"adapter at {adir} was trained with r={ar}, alpha={aa} "
                f"but the run is configured with "
                f"--max_lora_rank {config['max_lora_rank']} "
                f"--lora_alpha {config['lora_alpha']} — pass the matching "
                f"values")
        load_adapter(model, adir)

    tokenizer = load_tokenizer(config["model"], spec.vocab_size)

    is_learner = rank >= num_actors
    max_seq = config["max_prompt_tokens"] + config["max_new_tokens"]
    gpu_usage = (config.get("learner_gpu_usage", 0.35) if is_learner
                 else config.get("actor_gpu_usage", 0.91))
    eng_cfg = EngineConfig(max_seq_length=max_seq,
                           gpu_memory_utilization=gpu_usage)
    for k, v in (engine_overrides or {}).items():
        setattr(eng_cfg, k, v)
    engine = Engine(model, eng_cfg, device=device, seed=seed + 1000 * rank)

    resume = config.get("resume")
    if resume:
        # restore this rank's sampling-RNG stream (bit-identical resume
        # when the world size matches the checkpointing run; absent file
        # = topology changed, continue with the fresh seeded stream)
        es = os.path.join(resume, f"engine_state_rank{rank}.pt")
        if os.path.exists(es):
            st = torch.load(es, map_location="cpu", weights_only=True)
            engine.generator.set_state(st["generator"])
            engine._seq_counter = int(st["seq_counter"])

    learner = None
    if is_learner:
        learner = Learner(model, tokenizer, lr=config["lr"],
                          max_prompt_tokens=config["max_prompt_tokens"],
                          max_new_tokens=config["max_new_tokens"],
                          train_batch_size=config["train_batch_size"],
                          use_8bit_adam=config.get("use_8bit_adam", True))
        if resume:
            opt_path = os.path.join(resume, "optimizer_state.pt")
            if os.path.exists(opt_path):
                learner.load_state_dict(
                    torch.load(opt_path, map_location="cpu",
                               weights_only=True))

    fabric = Fabric(rank, world_size, num_actors, num_learners, device,
                    timeout_s=float(config.get("fabric_timeout_s", 240.0)))

    logger = None
    if rank == 0:
        logger = MetricsLogger(config.get("run_name"),
                               config.get("project_name", "math-reasoning"),
                               config)

    return Trainer(fabric, config, engine, learner, tokenizer,
                   train_dataset=train_dataset, test_dataset=test_dataset,
                   reward_function=reward_function, logger=logger)
