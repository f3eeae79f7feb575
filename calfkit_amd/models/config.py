"""Model-family configurations for the LocalModelClient engine.

BASELINE.json names Llama-3-8B (config 2/3), Llama-3-70B TP=8 (config 4) and
Mixtral 8x7B (config 5); ``TINY``/``SMOKE`` are test-lane configs that keep
head_dim=128 (the kernel contract) while being CPU-fast.
"""

from __future__ import annotations

from dataclasses import dataclass, replace

__all__ = ["ModelConfig", "LLAMA3_8B", "LLAMA3_70B", "MIXTRAL_8X7B", "TINY",
           "SMOKE", "get_config", "MAX_RUN_OVERSHOOT"]

#: Multi-step decode runs may blind-run a finishing sequence up to this many
#: tokens past its budget (engine RUN_STEPS[0] <= this, asserted there); the
#: RoPE table and the engine's page-table width carry this much headroom so
#: an overshooting position can never index out of bounds.
MAX_RUN_OVERSHOOT = 8


@dataclass(frozen=True)
class ModelConfig:
    name: str
    vocab_size: int
    hidden: int
    layers: int
    heads: int
    kv_heads: int
    head_dim: int
    ffn: int
    rope_theta: float = 500000.0
    norm_eps: float = 1e-5
    max_seq: int = 8192
    #: end-of-sequence token id the engine stops on (SyntheticTokenizer.EOS
    #: for the synthetic configs; a real llama-3 checkpoint would override
    #: with 128001/128009)
    eos_id: int = 2
    # MoE (Mixtral): 0 experts = dense
    num_experts: int = 0
    experts_per_token: int = 0
    tie_embeddings: bool = False

    @property
    def is_moe(self) -> bool:
        return self.num_experts > 0

    @property
    def q_dim(self) -> int:
        return self.heads * self.head_dim

    @property
    def kv_dim(self) -> int:
        return self.kv_heads * self.head_dim

    def param_count(self) -> int:
        """Approximate parameter count (dense part + experts)."""
        d, f = self.hidden, self.ffn
        attn = d * (self.q_dim + 2 * self.kv_dim) + self.q_dim * d
        mlp = 3 * d * f * max(1, self.num_experts)
        router = d * self.num_experts
        per_layer = attn + mlp + router + 2 * d
        return self.layers * per_layer + 2 * self.vocab_size * d + d


LLAMA3_8B = ModelConfig(
    name="llama-3-8b", vocab_size=128256, hidden=4096, layers=32, heads=32,
    kv_heads=8, head_dim=128, ffn=14336, rope_theta=500000.0)

LLAMA3_70B = ModelConfig(
    name="llama-3-70b", vocab_size=128256, hidden=8192, layers=80, heads=64,
    kv_heads=8, head_dim=128, ffn=28672, rope_theta=500000.0)

MIXTRAL_8X7B = ModelConfig(
    name="mixtral-8x7b", vocab_size=32000, hidden=4096, layers=32, heads=32,
    kv_heads=8, head_dim=128, ffn=14336, rope_theta=1e6,
    num_experts=8, experts_per_token=2)

#: CPU-fast config for unit tests (keeps head_dim=128, G=2)
TINY = ModelConfig(
    name="tiny", vocab_size=512, hidden=256, layers=2, heads=2, kv_heads=1,
    head_dim=128, ffn=512, rope_theta=10000.0, max_seq=1024)

#: GPU smoke config — big enough to exercise every kernel shape, small enough
#: to build in seconds (heads match llama GQA ratio)
SMOKE = ModelConfig(
    name="smoke", vocab_size=4096, hidden=1024, layers=4, heads=8, kv_heads=2,
    head_dim=128, ffn=2816, rope_theta=500000.0, max_seq=2048)

_BY_NAME = {c.name: c for c in (LLAMA3_8B, LLAMA3_70B, MIXTRAL_8X7B, TINY, SMOKE)}


def get_config(name: str, **overrides) -> ModelConfig:
    cfg = _BY_NAME[name]
    return replace(cfg, **overrides) if overrides else cfg
