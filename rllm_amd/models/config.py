"""Model architecture configs for the Qwen2 family (the reference's north-star
models: BASELINE.json configs — Qwen2.5-0.5B/1.5B/7B, DeepSeek-R1-Distill-
Qwen-1.5B, 14B)."""

from __future__ import annotations

from dataclasses import dataclass


@dataclass
class ModelConfig:
    name: str
    hidden_size: int
    intermediate_size: int
    num_layers: int
    num_heads: int
    num_kv_heads: int
    head_dim: int
    vocab_size: int
    rope_theta: float = 10000.0
    rms_eps: float = 1e-6
    tie_word_embeddings: bool = False
    max_position_embeddings: int = 131072
    qkv_bias: bool = True  # Qwen2 family uses q/k/v bias

    @property
    def q_size(self) -> int:
        return self.num_heads * self.head_dim

    @property
    def kv_size(self) -> int:
        return self.num_kv_heads * self.head_dim

    def param_count(self) -> int:
        H, I, L, V = self.hidden_size, self.intermediate_size, self.num_layers, self.vocab_size
        qkv = H * (self.q_size + 2 * self.kv_size) + (self.q_size + 2 * self.kv_size)
        o = self.q_size * H
        mlp = 3 * H * I
        norms = 2 * H * L + H
        emb = V * H * (1 if self.tie_word_embeddings else 2)
        return L * (qkv + o + mlp) + norms + emb


MODEL_CONFIGS: dict[str, ModelConfig] = {
    # DeepSeek-R1-Distill-Qwen-1.5B == Qwen2.5-Math-1.5B architecture
    "r1-distill-qwen-1.5b": ModelConfig(
        name="r1-distill-qwen-1.5b", hidden_size=1536, intermediate_size=8960,
        num_layers=28, num_heads=12, num_kv_heads=2, head_dim=128,
        vocab_size=151936, rope_theta=10000.0, tie_word_embeddings=False,
    ),
    "qwen2.5-0.5b": ModelConfig(
        name="qwen2.5-0.5b", hidden_size=896, intermediate_size=4864,
        num_layers=24, num_heads=14, num_kv_heads=2, head_dim=64,
        vocab_size=151936, rope_theta=1000000.0, tie_word_embeddings=True,
    ),
    "qwen2.5-1.5b": ModelConfig(
        name="qwen2.5-1.5b", hidden_size=1536, intermediate_size=8960,
        num_layers=28, num_heads=12, num_kv_heads=2, head_dim=128,
        vocab_size=151936, rope_theta=1000000.0, tie_word_embeddings=True,
    ),
    "qwen2.5-7b": ModelConfig(
        name="qwen2.5-7b", hidden_size=3584, intermediate_size=18944,
        num_layers=28, num_heads=28, num_kv_heads=4, head_dim=128,
        vocab_size=152064, rope_theta=1000000.0, tie_word_embeddings=False,
    ),
    "qwen2.5-14b": ModelConfig(
        name="qwen2.5-14b", hidden_size=5120, intermediate_size=13824,
        num_layers=48, num_heads=40, num_kv_heads=8, head_dim=128,
        vocab_size=152064, rope_theta=1000000.0, tie_word_embeddings=False,
    ),
    # tiny config for CPU plumbing tests / smoke
    "tiny": ModelConfig(
        name="tiny", hidden_size=256, intermediate_size=512,
        num_layers=2, num_heads=4, num_kv_heads=2, head_dim=128,
        vocab_size=1024, rope_theta=10000.0, tie_word_embeddings=True,
    ),
}


def get_model_config(name: str) -> ModelConfig:
    key = name.lower()
    if key not in MODEL_CONFIGS:
        raise KeyError(f"unknown model config {name!r}; known: {sorted(MODEL_CONFIGS)}")
    return MODEL_CONFIGS[key]
