"""Model architecture presets (random-init; no network for checkpoints).

Shapes match the public architectures named by BASELINE.json's configs
(Llama-3-8B / Llama-3-70B / Mistral-7B), plus tiny presets for CPU tests.
"""

from __future__ import annotations

from dataclasses import dataclass, replace


@dataclass(frozen=True)
class ModelConfig:
    name: str
    hidden_size: int
    intermediate_size: int
    num_layers: int
    num_heads: int
    num_kv_heads: int
    vocab_size: int
    head_dim: int = 128
    rope_theta: float = 500000.0
    rms_eps: float = 1e-5
    max_positions: int = 8192
    tie_embeddings: bool = False

    @property
    def q_size(self) -> int:
        return self.num_heads * self.head_dim

    @property
    def kv_size(self) -> int:
        return self.num_kv_heads * self.head_dim

    def scaled_for_tp(self, tp: int) -> "ModelConfig":
        """Per-rank shard shapes for tensor parallelism over xGMI."""
        assert self.num_heads % tp == 0 and self.num_kv_heads % tp == 0
        assert self.intermediate_size % tp == 0
        return replace(
            self,
            num_heads=self.num_heads // tp,
            num_kv_heads=self.num_kv_heads // tp,
            intermediate_size=self.intermediate_size // tp,
        )


MODEL_PRESETS = {
    "llama-3-8b": ModelConfig(
        name="llama-3-8b",
        hidden_size=4096,
        intermediate_size=14336,
        num_layers=32,
        num_heads=32,
        num_kv_heads=8,
        vocab_size=128256,
        rope_theta=500000.0,
    ),
    "llama-3-70b": ModelConfig(
        name="llama-3-70b",
        hidden_size=8192,
        intermediate_size=28672,
        num_layers=80,
        num_heads=64,
        num_kv_heads=8,
        vocab_size=128256,
        rope_theta=500000.0,
    ),
    # MHA family (GQA group = 1)
    "llama-2-7b": ModelConfig(
        name="llama-2-7b",
        hidden_size=4096,
        intermediate_size=11008,
        num_layers=32,
        num_heads=32,
        num_kv_heads=32,
        vocab_size=32000,
        rope_theta=10000.0,
        max_positions=4096,
    ),
    # odd GQA group (G = 7) exercises the templated decode kernel
    "qwen2-7b": ModelConfig(
        name="qwen2-7b",
        hidden_size=3584,
        intermediate_size=18944,
        num_layers=28,
        num_heads=28,
        num_kv_heads=4,
        vocab_size=152064,
        rope_theta=1000000.0,
    ),
    "mistral-7b": ModelConfig(
        name="mistral-7b",
        hidden_size=4096,
        intermediate_size=14336,
        num_layers=32,
        num_heads=32,
        num_kv_heads=8,
        vocab_size=32000,
        rope_theta=10000.0,
    ),
    # tiny shapes for CPU tests / smoke
    "tiny-llama": ModelConfig(
        name="tiny-llama",
        hidden_size=256,
        intermediate_size=512,
        num_layers=2,
        num_heads=4,
        num_kv_heads=2,
        vocab_size=512,
        head_dim=64,
        rope_theta=10000.0,
        max_positions=512,
    ),
    # tiny shape divisible for world-4 TP lockstep tests
    "tiny-llama-tp4": ModelConfig(
        name="tiny-llama-tp4",
        hidden_size=256,
        intermediate_size=512,
        num_layers=2,
        num_heads=8,
        num_kv_heads=4,
        vocab_size=512,
        head_dim=32,
        rope_theta=10000.0,
        max_positions=512,
    ),
    # small-but-real shape for 1-GPU kernel shakedown
    "llama-1b": ModelConfig(
        name="llama-1b",
        hidden_size=2048,
        intermediate_size=8192,
        num_layers=16,
        num_heads=32,
        num_kv_heads=8,
        vocab_size=128256,
        head_dim=64,
        rope_theta=500000.0,
    ),
}


def get_model_config(name: str) -> ModelConfig:
    key = name.lower()
    if key not in MODEL_PRESETS:
        raise KeyError(f"Unknown model preset '{name}'. Known: {sorted(MODEL_PRESETS)}")
    return MODEL_PRESETS[key]
