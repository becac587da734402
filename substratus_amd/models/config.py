"""Model architecture configs for the data plane.

The reference orchestrates external HF images for concrete models named in its
examples (reference: examples/llama2-7b/*, examples/falcon-7b-instruct/*,
examples/facebook-opt-125m/*, examples/llama2-13b-chat-gguf/*,
examples/llama2-70b/server.yaml). We implement those model families natively.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional


@dataclass
class ModelConfig:
    name: str
    family: str  # "llama" | "falcon" | "opt"
    hidden_size: int
    intermediate_size: int
    num_layers: int
    num_heads: int
    num_kv_heads: int
    vocab_size: int
    max_position_embeddings: int
    head_dim: int = 0
    rope_theta: float = 10000.0
    rms_norm_eps: float = 1e-5
    layer_norm_eps: float = 1e-5
    tie_word_embeddings: bool = False
    # falcon-style: attention and MLP in parallel off one layernorm
    parallel_attn: bool = False
    # falcon-40b "new decoder architecture": parallel attn+MLP but with TWO
    # norms of the residual (ln_attn feeds attention, ln_mlp feeds the MLP)
    parallel_norms: bool = False
    # opt-style: learned positional embeddings + LayerNorm + gelu
    learned_pos_emb: bool = False
    activation: str = "silu"  # "silu" (gated) | "gelu"

    def __post_init__(self) -> None:
        if self.head_dim == 0:
            self.head_dim = self.hidden_size // self.num_heads

    @property
    def kv_dim(self) -> int:
        return self.num_kv_heads * self.head_dim

    def param_count(self) -> int:
        h, f, L, v = self.hidden_size, self.intermediate_size, self.num_layers, self.vocab_size
        qkv = h * (self.num_heads + 2 * self.num_kv_heads) * self.head_dim
        o = self.num_heads * self.head_dim * h
        if self.activation == "silu":
            mlp = 3 * h * f
        else:
            mlp = 2 * h * f
        emb = v * h * (1 if self.tie_word_embeddings else 2)
        return L * (qkv + o + mlp) + emb


_REGISTRY: dict[str, ModelConfig] = {}


def register(cfg: ModelConfig) -> ModelConfig:
    _REGISTRY[cfg.name] = cfg
    return cfg


def get_config(name: str) -> ModelConfig:
    key = name.lower().split("/")[-1]
    if key in _REGISTRY:
        return _REGISTRY[key]
    raise KeyError(f"unknown model config: {name!r}; known: {sorted(_REGISTRY)}")


# Llama-2 family (reference: examples/llama2-7b, examples/llama2-13b-chat-gguf,
# examples/llama2-70b).
register(ModelConfig(
    name="llama2-7b", family="llama", hidden_size=4096, intermediate_size=11008,
    num_layers=32, num_heads=32, num_kv_heads=32, vocab_size=32000,
    max_position_embeddings=4096, rms_norm_eps=1e-5))
register(ModelConfig(
    name="llama2-13b", family="llama", hidden_size=5120, intermediate_size=13824,
    num_layers=40, num_heads=40, num_kv_heads=40, vocab_size=32000,
    max_position_embeddings=4096, rms_norm_eps=1e-5))
register(ModelConfig(
    name="llama2-70b", family="llama", hidden_size=8192, intermediate_size=28672,
    num_layers=80, num_heads=64, num_kv_heads=8, vocab_size=32000,
    max_position_embeddings=4096, rms_norm_eps=1e-5))
# Llama-3 family (beyond reference parity — same llama weight layout, so
# HF checkpoints map via the existing name translation; GQA groups 4 and 8
# ride the MFMA decode kernel, rope theta 500k, 128k vocab).
register(ModelConfig(
    name="llama3-8b", family="llama", hidden_size=4096,
    intermediate_size=14336, num_layers=32, num_heads=32, num_kv_heads=8,
    vocab_size=128256, max_position_embeddings=8192, rope_theta=500000.0,
    rms_norm_eps=1e-5))
register(ModelConfig(
    name="llama3-70b", family="llama", hidden_size=8192,
    intermediate_size=28672, num_layers=80, num_heads=64, num_kv_heads=8,
    vocab_size=128256, max_position_embeddings=8192, rope_theta=500000.0,
    rms_norm_eps=1e-5))

# A tiny llama for CPU tests and CI.
# Tiny llama for CPU tests/CI and GPU numerics tests (head_dim=64, GQA=2).
register(ModelConfig(
    name="llama-tiny", family="llama", hidden_size=256, intermediate_size=704,
    num_layers=4, num_heads=4, num_kv_heads=2, vocab_size=512,
    max_position_embeddings=2048, rms_norm_eps=1e-5))

# Falcon-7b-instruct (reference: examples/falcon-7b-instruct/finetuned-model.yaml).
# 71 query heads with multi-query attention (1 kv head), parallel attn+mlp.
register(ModelConfig(
    name="falcon-7b-instruct", family="falcon", hidden_size=4544,
    intermediate_size=4 * 4544, num_layers=32, num_heads=71, num_kv_heads=1,
    vocab_size=65024, max_position_embeddings=2048, layer_norm_eps=1e-5,
    parallel_attn=True, activation="gelu", tie_word_embeddings=True))
register(ModelConfig(
    name="falcon-tiny", family="falcon", hidden_size=256,
    intermediate_size=1024, num_layers=2, num_heads=4, num_kv_heads=1,
    vocab_size=512, max_position_embeddings=2048, parallel_attn=True,
    activation="gelu", tie_word_embeddings=True))

# Falcon-40b (reference: examples/falcon-40b/*.yaml). GQA (128 q / 8 kv
# heads) with the "new decoder architecture": two parallel layernorms per
# block (ln_attn -> attention, ln_mlp -> MLP).
register(ModelConfig(
    name="falcon-40b", family="falcon", hidden_size=8192,
    intermediate_size=4 * 8192, num_layers=60, num_heads=128, num_kv_heads=8,
    vocab_size=65024, max_position_embeddings=2048, layer_norm_eps=1e-5,
    parallel_attn=True, parallel_norms=True, activation="gelu",
    tie_word_embeddings=True))
register(ModelConfig(
    name="falcon40-tiny", family="falcon", hidden_size=256,
    intermediate_size=1024, num_layers=2, num_heads=4, num_kv_heads=2,
    vocab_size=512, max_position_embeddings=2048, parallel_attn=True,
    parallel_norms=True, activation="gelu", tie_word_embeddings=True))

# facebook/opt-125m (reference: examples/facebook-opt-125m/base-model.yaml —
# the reference's CPU system-test model, test/system.sh).
register(ModelConfig(
    name="opt-125m", family="opt", hidden_size=768, intermediate_size=3072,
    num_layers=12, num_heads=12, num_kv_heads=12, vocab_size=50272,
    max_position_embeddings=2048, learned_pos_emb=True, activation="gelu",
    tie_word_embeddings=True))
