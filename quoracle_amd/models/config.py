"""Model architecture configs (Llama-family + MoE) for the local pool.

The reference delegates model choice to remote providers (reference:
lib/quoracle/models/model_query.ex); here each pool entry names a locally
hosted architecture.  A model key is "<preset>#<instance>" — instances share
the architecture but get different random-init seeds, giving the consensus
pool genuinely decorrelated voters.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict


@dataclass(frozen=True)
class ModelConfig:
    name: str
    vocab_size: int
    n_layers: int
    hidden: int
    n_heads: int
    n_kv_heads: int
    head_dim: int
    intermediate: int
    rope_theta: float = 500000.0
    rmsnorm_eps: float = 1e-5
    max_context: int = 8192
    max_output: int = 4096
    tie_embeddings: bool = False
    # MoE (0 experts = dense)
    n_experts: int = 0
    top_k_experts: int = 2

    @property
    def q_dim(self) -> int:
        return self.n_heads * self.head_dim

    @property
    def kv_dim(self) -> int:
        return self.n_kv_heads * self.head_dim

    @property
    def qkv_dim(self) -> int:
        return self.q_dim + 2 * self.kv_dim

    @property
    def is_moe(self) -> bool:
        return self.n_experts > 0

    def kv_bytes_per_token(self) -> int:
        """bf16 paged-KV footprint of one token across all layers."""
        return 2 * 2 * self.n_layers * self.kv_dim

    def param_count(self) -> int:
        embed = self.vocab_size * self.hidden
        lm_head = 0 if self.tie_embeddings else self.vocab_size * self.hidden
        per_layer = (self.hidden * self.qkv_dim + self.q_dim * self.hidden
                     + 2 * self.hidden)
        if self.is_moe:
            per_layer += self.n_experts * 3 * self.hidden * self.intermediate
            per_layer += self.hidden * self.n_experts  # router
        else:
            per_layer += 3 * self.hidden * self.intermediate
        return embed + lm_head + self.n_layers * per_layer + self.hidden


PRESETS: Dict[str, ModelConfig] = {
    # CPU-testable toy model (head_dim multiple of 8, <=128 for the kernels)
    "tiny": ModelConfig("tiny", vocab_size=512, n_layers=2, hidden=64,
                        n_heads=4, n_kv_heads=2, head_dim=16,
                        intermediate=128, rope_theta=10000.0,
                        max_context=32768, max_output=1024,
                        tie_embeddings=True),
    # CPU-testable MoE toy (engine-level Mixtral-path tests)
    "tiny-moe": ModelConfig("tiny-moe", vocab_size=512, n_layers=2, hidden=64,
                            n_heads=4, n_kv_heads=2, head_dim=16,
                            intermediate=128, rope_theta=10000.0,
                            max_context=32768, max_output=1024,
                            tie_embeddings=True, n_experts=4,
                            top_k_experts=2),
    # GPT-2-small scale, llama-style blocks (BASELINE config 1)
    "gpt2s": ModelConfig("gpt2s", vocab_size=50304, n_layers=12, hidden=768,
                         n_heads=12, n_kv_heads=12, head_dim=64,
                         intermediate=3072, rope_theta=10000.0,
                         max_context=32768, max_output=2048,
                         tie_embeddings=True),
    # Llama-3-8B architecture (BASELINE configs 2-4)
    "llama3-8b": ModelConfig("llama3-8b", vocab_size=128256, n_layers=32,
                             hidden=4096, n_heads=32, n_kv_heads=8,
                             head_dim=128, intermediate=14336,
                             rope_theta=500000.0, max_context=32768,
                             max_output=4096),
    # Llama-3-70B (TP=4, BASELINE config 5)
    "llama3-70b": ModelConfig("llama3-70b", vocab_size=128256, n_layers=80,
                              hidden=8192, n_heads=64, n_kv_heads=8,
                              head_dim=128, intermediate=28672,
                              rope_theta=500000.0, max_context=32768,
                              max_output=4096),
    # Mixtral-8x7B MoE (TP=4, BASELINE config 5)
    "mixtral-8x7b": ModelConfig("mixtral-8x7b", vocab_size=32000,
                                n_layers=32, hidden=4096, n_heads=32,
                                n_kv_heads=8, head_dim=128,
                                intermediate=14336, rope_theta=1000000.0,
                                max_context=32768, max_output=4096,
                                n_experts=8, top_k_experts=2),
    # small bidirectional-ish embedder for the consensus cosine vote
    "embed-small": ModelConfig("embed-small", vocab_size=50304, n_layers=2,
                               hidden=256, n_heads=4, n_kv_heads=4,
                               head_dim=64, intermediate=512,
                               rope_theta=10000.0, max_context=4096,
                               max_output=1, tie_embeddings=True),
}


def get_config(key: str) -> ModelConfig:
    """Resolve "<preset>#<instance>" or "<preset>" to its architecture."""
    preset = key.split("#", 1)[0]
    if preset not in PRESETS:
        raise KeyError(f"unknown model preset {preset!r} "
                       f"(known: {sorted(PRESETS)})")
    return PRESETS[preset]


def instance_seed(key: str) -> int:
    """Deterministic per-instance weight seed from the model key."""
    import zlib
    return zlib.crc32(key.encode()) & 0x7FFFFFFF
