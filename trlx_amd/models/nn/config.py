"""Native transformer configuration.

One config covers the decoder families the reference supports through HF
wrappers + per-arch branches (GPT2/OPT/Bloom/Llama/GPTBigCode,
GPT-J/NeoX via examples): the architectural axes are factored into flags
instead of per-arch classes, because the compute path is ours (HIP kernels),
not HF's.
"""

from dataclasses import dataclass, field
from typing import Any, Dict, Optional


@dataclass
class TransformerConfig:
    vocab_size: int = 50257
    hidden_size: int = 768
    num_layers: int = 12
    num_heads: int = 12
    num_kv_heads: Optional[int] = None  # GQA/MQA; None -> num_heads
    head_dim: Optional[int] = None  # None -> hidden_size // num_heads
    intermediate_size: Optional[int] = None  # None -> 4*hidden
    max_position_embeddings: int = 1024

    # normalization: "layernorm" | "rmsnorm"
    norm: str = "layernorm"
    norm_eps: float = 1e-5

    # position encoding: "learned" | "rope" | "alibi"
    position_encoding: str = "learned"
    rope_base: float = 10000.0
    rope_interleaved: bool = False  # True for GPT-J/NeoX-style pairs
    rope_pct: float = 1.0  # fraction of head_dim rotated (NeoX rotary_pct)

    # MLP activation: "gelu" | "gelu_new" | "relu" | "silu"
    activation: str = "gelu_new"
    swiglu: bool = False  # Llama-style gated MLP

    # residual topology: False = sequential (GPT-2/Llama), True = parallel
    # attention+MLP off one norm (GPT-J/NeoX parallel_residual)
    parallel_residual: bool = False

    attn_bias: bool = True
    mlp_bias: bool = True
    tie_word_embeddings: bool = True
    lm_head_bias: bool = False

    attn_scale: Optional[float] = None  # None -> 1/sqrt(head_dim)

    embd_pdrop: float = 0.0
    resid_pdrop: float = 0.0
    attn_pdrop: float = 0.0

    # bookkeeping for HF interop
    arch_name: str = "gpt2"
    extra: Dict[str, Any] = field(default_factory=dict)

    def __post_init__(self):
        if self.num_kv_heads is None:
            self.num_kv_heads = self.num_heads
        if self.head_dim is None:
            self.head_dim = self.hidden_size // self.num_heads
        if self.intermediate_size is None:
            self.intermediate_size = 4 * self.hidden_size

    @property
    def qkv_out(self) -> int:
        return (self.num_heads + 2 * self.num_kv_heads) * self.head_dim

    def to_dict(self) -> Dict[str, Any]:
        return dict(self.__dict__)

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "TransformerConfig":
        return cls(**d)


# ---------------------------------------------------------------------------
# canonical small configs (used by tests and the synthetic benchmarks; shapes
# match the HF checkpoints of the same name)
# ---------------------------------------------------------------------------

PRESETS = {
    "gpt2": dict(
        vocab_size=50257, hidden_size=768, num_layers=12, num_heads=12,
        max_position_embeddings=1024, norm="layernorm", position_encoding="learned",
        activation="gelu_new", attn_bias=True, mlp_bias=True, tie_word_embeddings=True,
        arch_name="gpt2",
    ),
    "gpt2-medium": dict(
        vocab_size=50257, hidden_size=1024, num_layers=24, num_heads=16,
        max_position_embeddings=1024, norm="layernorm", position_encoding="learned",
        activation="gelu_new", arch_name="gpt2",
    ),
    "gpt2-large": dict(
        vocab_size=50257, hidden_size=1280, num_layers=36, num_heads=20,
        max_position_embeddings=1024, norm="layernorm", position_encoding="learned",
        activation="gelu_new", arch_name="gpt2",
    ),
    "gpt2-xl": dict(
        vocab_size=50257, hidden_size=1600, num_layers=48, num_heads=25,
        max_position_embeddings=1024, norm="layernorm", position_encoding="learned",
        activation="gelu_new", arch_name="gpt2",
    ),
    "gptj-6b": dict(
        vocab_size=50400, hidden_size=4096, num_layers=28, num_heads=16,
        max_position_embeddings=2048, norm="layernorm", position_encoding="rope",
        rope_interleaved=True, rope_pct=64 / 256, activation="gelu_new",
        parallel_residual=True, attn_bias=False, mlp_bias=True,
        tie_word_embeddings=False, lm_head_bias=True, arch_name="gptj",
    ),
    "llama2-7b": dict(
        vocab_size=32000, hidden_size=4096, num_layers=32, num_heads=32,
        intermediate_size=11008, max_position_embeddings=4096, norm="rmsnorm",
        norm_eps=1e-5, position_encoding="rope", activation="silu", swiglu=True,
        attn_bias=False, mlp_bias=False, tie_word_embeddings=False, arch_name="llama",
    ),
    "pythia-160m": dict(
        vocab_size=50304, hidden_size=768, num_layers=12, num_heads=12,
        max_position_embeddings=2048, norm="layernorm", position_encoding="rope",
        rope_interleaved=False, rope_pct=0.25, activation="gelu",
        parallel_residual=True, tie_word_embeddings=False, arch_name="gpt_neox",
    ),
    "gpt-neox-20b": dict(
        vocab_size=50432, hidden_size=6144, num_layers=44, num_heads=64,
        max_position_embeddings=2048, norm="layernorm", position_encoding="rope",
        rope_interleaved=False, rope_pct=0.25, activation="gelu",
        parallel_residual=True, tie_word_embeddings=False, arch_name="gpt_neox",
    ),
    "bloom-560m": dict(
        vocab_size=250880, hidden_size=1024, num_layers=24, num_heads=16,
        max_position_embeddings=2048, norm="layernorm", position_encoding="alibi",
        activation="gelu", arch_name="bloom", tie_word_embeddings=True,
        extra={"pre_embed_norm": True},
    ),
    "gpt_bigcode-santacoder": dict(
        vocab_size=49280, hidden_size=2048, num_layers=24, num_heads=16,
        num_kv_heads=1, max_position_embeddings=2048, norm="layernorm",
        position_encoding="learned", activation="gelu_new", arch_name="gpt_bigcode",
    ),
    "opt-125m": dict(
        vocab_size=50272, hidden_size=768, num_layers=12, num_heads=12,
        max_position_embeddings=2048, norm="layernorm", position_encoding="learned",
        activation="relu", arch_name="opt", tie_word_embeddings=True,
        extra={"position_offset": 2},
    ),
}


def preset(name: str, **overrides) -> TransformerConfig:
    base = dict(PRESETS[name])
    base.update(overrides)
    return TransformerConfig(**base)
