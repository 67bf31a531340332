"""Model architecture registry.

Covers the model families the reference targets (BASELINE.json configs:
Qwen2.5-{0.5B,7B,32B}, Llama-3-8B) plus tiny variants for CPU tests. Specs
are resolved from the model-name string the CLI passes (the reference
resolves names through HF/Unsloth; offline we map known names to
architectures and random-init weights — BASELINE.json: synthetic prompts /
random-init weights).
"""

from __future__ import annotations

from dataclasses import dataclass


@dataclass(frozen=True)
class ModelSpec:
    name: str
    vocab_size: int
    hidden_size: int
    intermediate_size: int
    num_layers: int
    num_heads: int
    num_kv_heads: int
    head_dim: int
    rope_theta: float
    rms_norm_eps: float
    tie_word_embeddings: bool
    qkv_bias: bool  # Qwen2 uses biases on q/k/v projections
    max_position: int = 32768
    # Qwen3: per-head RMSNorm on q/k (over head_dim, before RoPE)
    qk_norm: bool = False

    @property
    def q_size(self) -> int:
        return self.num_heads * self.head_dim

    @property
    def kv_size(self) -> int:
        return self.num_kv_heads * self.head_dim


_REGISTRY = {
    "qwen2.5-0.5b": ModelSpec("qwen2.5-0.5b", 151936, 896, 4864, 24, 14, 2, 64,
                              1e6, 1e-6, True, True),
    "qwen2.5-1.5b": ModelSpec("qwen2.5-1.5b", 151936, 1536, 8960, 28, 12, 2, 128,
                              1e6, 1e-6, True, True),
    "qwen2.5-7b": ModelSpec("qwen2.5-7b", 152064, 3584, 18944, 28, 28, 4, 128,
                            1e6, 1e-6, False, True),
    "qwen2.5-14b": ModelSpec("qwen2.5-14b", 152064, 5120, 13824, 48, 40, 8, 128,
                             1e6, 1e-5, False, True),
    "qwen2.5-32b": ModelSpec("qwen2.5-32b", 152064, 5120, 27648, 64, 40, 8, 128,
                             1e6, 1e-5, False, True),
    "qwen2.5-72b": ModelSpec("qwen2.5-72b", 152064, 8192, 29568, 80, 64, 8, 128,
                             1e6, 1e-5, False, True),
    "llama-3-8b": ModelSpec("llama-3-8b", 128256, 4096, 14336, 32, 32, 8, 128,
                            5e5, 1e-5, False, False),
    "llama-3-70b": ModelSpec("llama-3-70b", 128256, 8192, 28672, 80, 64, 8, 128,
                             5e5, 1e-5, False, False),
    # Mistral-7B v0.3 (sliding_window null => plain causal attention,
    # which is the attention this stack implements; pre-v0.3 windowed
    # checkpoints are rejected by hf_io.spec_from_hf_config)
    "mistral-7b": ModelSpec("mistral-7b", 32768, 4096, 14336, 32, 32, 8, 128,
                            1e6, 1e-5, False, False),
    # Qwen3 dense (per-head q/k RMSNorm before RoPE, no qkv bias;
    # note q_size != hidden for 4B: heads*head_dim = 32*128 = 4096)
    "qwen3-4b": ModelSpec("qwen3-4b", 151936, 2560, 9728, 36, 32, 8, 128,
                          1e6, 1e-6, True, False, qk_norm=True),
    "qwen3-8b": ModelSpec("qwen3-8b", 151936, 4096, 12288, 36, 32, 8, 128,
                          1e6, 1e-6, False, False, qk_norm=True),
    "qwen3-32b": ModelSpec("qwen3-32b", 151936, 5120, 25600, 64, 64, 8, 128,
                           1e6, 1e-6, False, False, qk_norm=True),
    # tiny models for CPU tests / the gloo plumbing config
    "tiny-qwen2": ModelSpec("tiny-qwen2", 2048, 64, 128, 2, 4, 2, 16,
                            1e4, 1e-6, True, True, max_position=512),
    # small models for GPU kernel/engine tests (head_dim 64 = kernel-supported)
    "small-qwen2": ModelSpec("small-qwen2", 4096, 512, 1024, 2, 8, 2, 64,
                             1e5, 1e-6, True, True, max_position=2048),
    "small-qwen3": ModelSpec("small-qwen3", 4096, 512, 1024, 2, 8, 2, 64,
                             1e5, 1e-6, True, False, max_position=2048,
                             qk_norm=True),
    "tiny-llama": ModelSpec("tiny-llama", 2048, 64, 128, 2, 4, 2, 16,
                            1e4, 1e-5, False, False, max_position=512),
    "tiny-mistral": ModelSpec("tiny-mistral", 2048, 64, 128, 2, 4, 2, 16,
                              1e4, 1e-5, False, False, max_position=512),
    # tiny Qwen3 exercises qk_norm + q_size != hidden (4 heads x 24 = 96)
    "tiny-qwen3": ModelSpec("tiny-qwen3", 2048, 64, 128, 2, 4, 2, 24,
                            1e4, 1e-6, False, False, max_position=512,
                            qk_norm=True),
}


def get_spec(model_name: str) -> ModelSpec:
    """Resolve a model-name string (e.g. the reference's default
    'unsloth/Qwen2.5-7B-Instruct-bnb-4bit') to an architecture spec."""
    low = model_name.lower()
    for key in ("tiny-qwen2", "tiny-llama", "tiny-mistral", "tiny-qwen3",
                "small-qwen2", "small-qwen3"):
        if key in low:
            return _REGISTRY[key]
    if "qwen2.5-0.5b" in low or "qwen2-0.5b" in low:
        return _REGISTRY["qwen2.5-0.5b"]
    if "qwen2.5-1.5b" in low:
        return _REGISTRY["qwen2.5-1.5b"]
    if "qwen2.5-7b" in low:
        return _REGISTRY["qwen2.5-7b"]
    if "qwen2.5-14b" in low:
        return _REGISTRY["qwen2.5-14b"]
    if "qwen2.5-32b" in low:
        return _REGISTRY["qwen2.5-32b"]
    if "qwen2.5-72b" in low:
        return _REGISTRY["qwen2.5-72b"]
    if "llama-3" in low and "8b" in low:
        return _REGISTRY["llama-3-8b"]
    if "llama-3" in low and "70b" in low:
        return _REGISTRY["llama-3-70b"]
    if "mistral" in low and "7b" in low:
        return _REGISTRY["mistral-7b"]
    if "qwen3" in low:
        for size in ("4b", "8b", "32b"):
            if size in low:
                return _REGISTRY[f"qwen3-{size}"]
    raise ValueError(f"Unknown model architecture for name: {model_name!r}; "
                     f"known: {sorted(_REGISTRY)}")


def is_4bit_model_name(model_name: str) -> bool:
    """The reference selects 4-bit via the '-bnb-4bit' model suffix and the
    LOAD_IN_4BIT constant (reference distributed_actor.py:17)."""
    return "4bit" in model_name.lower()
