"""distrl_llm_amd — MI355X-native distributed RL fine-tuning engine for LLMs.

A from-scratch CDNA4 (gfx950) implementation of the capabilities of
BY571/DistRL-LLM: heterogeneous actor/learner GPU pools, batched paged-KV
rollout generation, 4-bit (nf4) LoRA training with 8-bit Adam, PG and GRPO
learners, top-k candidate subselection, PEFT-format LoRA checkpoints and the
same ``train_distributed.py`` CLI — built on PyTorch-ROCm + hand-written
HIP/CDNA4 kernels + RCCL over xGMI (no Ray, no vLLM, no Unsloth, no Triton).
"""

__version__ = "0.1.0"

from .config import EngineConfig, GenerationConfig, SamplingParams  # noqa: F401


def __getattr__(name):
    # lazy re-exports so `import distrl_llm_amd` stays light (torch model
    # code only loads when the public classes are actually used)
    if name == "Engine":
        from .engine import Engine
        return Engine
    if name == "CausalLM":
        from .models import CausalLM
        return CausalLM
    if name == "get_spec":
        from .models import get_spec
        return get_spec
    raise AttributeError(f"module {__name__!r} has no attribute {name!r}")
