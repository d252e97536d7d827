"""Model and engine configuration.

Model presets mirror the BASELINE.json configs: Llama-3-8B (headline),
Llama-3-70B (TP=8) and Mixtral 8x7B (MoE).  All weights are random-init
bf16 (no network; the benchmark contract is synthetic data / random weights).
"""
from __future__ import annotations

from dataclasses import dataclass


@dataclass
class ModelConfig:
    name: str = "llama-3-8b"
    hidden_size: int = 4096
    num_layers: int = 32
    num_q_heads: int = 32
    num_kv_heads: int = 8
    head_dim: int = 128
    intermediate_size: int = 14336
    vocab_size: int = 128256
    rope_theta: float = 500000.0
    rms_eps: float = 1e-5
    max_position: int = 16384
    # MoE (0 experts = dense)
    num_experts: int = 0
    top_k_experts: int = 2

    @property
    def is_moe(self) -> bool:
        return self.num_experts > 0

    @property
    def qkv_dim(self) -> int:
        return (self.num_q_heads + 2 * self.num_kv_heads) * self.head_dim


def llama3_8b() -> ModelConfig:
    return ModelConfig()


def llama3_70b() -> ModelConfig:
    return ModelConfig(name="llama-3-70b", hidden_size=8192, num_layers=80,
                       num_q_heads=64, num_kv_heads=8,
                       intermediate_size=28672)


def mixtral_8x7b() -> ModelConfig:
    return ModelConfig(name="mixtral-8x7b", hidden_size=4096, num_layers=32,
                       num_q_heads=32, num_kv_heads=8,
                       intermediate_size=14336, vocab_size=32000,
                       rope_theta=1e6, num_experts=8, top_k_experts=2)


def tiny_mixtral() -> ModelConfig:
    return ModelConfig(name="tiny-mixtral", hidden_size=256, num_layers=2,
                       num_q_heads=2, num_kv_heads=2, head_dim=128,
                       intermediate_size=256, vocab_size=512,
                       max_position=512, num_experts=4, top_k_experts=2)


def tiny_llama() -> ModelConfig:
    """CPU-test sized model (same head_dim=128 the kernels require)."""
    return ModelConfig(name="tiny-llama", hidden_size=256, num_layers=2,
                       num_q_heads=2, num_kv_heads=2, head_dim=128,
                       intermediate_size=512, vocab_size=512,
                       max_position=512)


MODEL_PRESETS = {
    "llama-3-8b": llama3_8b,
    "llama-3-70b": llama3_70b,
    "mixtral-8x7b": mixtral_8x7b,
    "tiny-llama": tiny_llama,
    "tiny-mixtral": tiny_mixtral,
}


@dataclass
class EngineConfig:
    block_size: int = 16
    max_model_len: int = 4096
    max_sessions: int = 256
    # KV blocks: explicit count, or derived from free HBM at init
    num_kv_blocks: int = 0
    kv_mem_fraction: float = 0.90
    max_prefill_tokens: int = 16384
    use_graphs: bool = True
    graph_buckets: tuple = (8, 16, 32, 64, 96, 128, 192, 256)
    decode_splits: int = 1          # split-KV factor for small-batch decode
    decode_microbatch: int = 32     # decode steps per host sync (self-
                                    # advancing graph replay train)
    kv_dtype: str = "bf16"          # "bf16" | "fp8" (OCP e4m3 cache: half
                                    # the KV bandwidth, 2x the capacity)
    seed: int = 1234
    tp_size: int = 1

    @property
    def max_blocks_per_seq(self) -> int:
        return (self.max_model_len + self.block_size - 1) // self.block_size


@dataclass
class SamplingParams:
    temperature: float = 0.7
    top_k: int = 50
    top_p: float = 0.9
    max_new_tokens: int = 128
    # generation ends when one of these ids is sampled (the id itself is
    # the turn's last token); agent protocols stop at eos/tool markers
    # long before max_new_tokens
    stop_token_ids: tuple = ()
