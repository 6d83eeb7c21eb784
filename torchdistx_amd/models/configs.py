# Model configurations for the benchmark / test model zoo. These are the
# configs BASELINE.json names (GPT-2-XL, Llama-3-8B, Llama-3-70B,
# Mixtral-8x22B) plus tiny variants for CPU tests. Random-init weights,
# synthetic data; no checkpoints are ever loaded.

from dataclasses import dataclass
from typing import Optional


@dataclass
class TransformerConfig:
    name: str
    vocab_size: int
    dim: int
    n_layers: int
    n_heads: int
    n_kv_heads: int
    ffn_hidden: int
    max_seq_len: int = 8192
    norm: str = "rmsnorm"  # "rmsnorm" (llama-style) or "layernorm" (gpt2)
    act: str = "swiglu"  # "swiglu" or "gelu"
    rope: bool = True  # rotary embeddings; False -> learned positional
    tie_embeddings: bool = False
    # MoE (Mixtral-style); None -> dense FFN
    moe_num_experts: Optional[int] = None
    moe_top_k: int = 2
    # "fast": one normal_ per weight, zeros_ bias (llama-style, keeps each
    # tape segment one RNG op long). "stock": PyTorch's default
    # nn.Linear/nn.Embedding resets (kaiming_uniform_ weights + uniform_
    # bias), which records the empty -> uniform_ tape shape of unmodified
    # user models.
    init: str = "fast"

    @property
    def n_params(self) -> int:
        """Approximate parameter count (exact for the modules we build)."""
        p = self.vocab_size * self.dim  # token embedding
        if not self.rope:
            p += self.max_seq_len * self.dim
        head_dim = self.dim // self.n_heads
        kv_dim = head_dim * self.n_kv_heads
        norm_size = self.dim * (2 if self.norm == "layernorm" else 1)
        per_layer = (
            self.dim * self.dim  # wq
            + self.dim * kv_dim * 2  # wk, wv
            + self.dim * self.dim  # wo
            + norm_size * 2  # attn + ffn norms
        )
        if self.act == "swiglu":
            ffn = 3 * self.dim * self.ffn_hidden
        else:
            ffn = 2 * self.dim * self.ffn_hidden + self.ffn_hidden + self.dim
        if self.moe_num_experts:
            per_layer += self.moe_num_experts * ffn + self.dim * self.moe_num_experts
        else:
            per_layer += ffn
        p += self.n_layers * per_layer
        p += self.dim * (2 if self.norm == "layernorm" else 1)  # final norm
        if not self.tie_embeddings:
            p += self.vocab_size * self.dim  # lm head
        return p


TINY = TransformerConfig(
    name="tiny",
    vocab_size=128,
    dim=64,
    n_layers=2,
    n_heads=4,
    n_kv_heads=2,
    ffn_hidden=128,
    max_seq_len=64,
)

TINY_GPT2 = TransformerConfig(
    name="tiny-gpt2",
    vocab_size=128,
    dim=64,
    n_layers=2,
    n_heads=4,
    n_kv_heads=4,
    ffn_hidden=256,
    max_seq_len=64,
    norm="layernorm",
    act="gelu",
    rope=False,
    tie_embeddings=True,
)

TINY_MOE = TransformerConfig(
    name="tiny-moe",
    vocab_size=128,
    dim=64,
    n_layers=2,
    n_heads=4,
    n_kv_heads=2,
    ffn_hidden=128,
    max_seq_len=64,
    moe_num_experts=4,
    moe_top_k=2,
)

GPT2_XL = TransformerConfig(
    name="gpt2-xl",
    vocab_size=50257,
    dim=1600,
    n_layers=48,
    n_heads=25,
    n_kv_heads=25,
    ffn_hidden=6400,
    max_seq_len=1024,
    norm="layernorm",
    act="gelu",
    rope=False,
    tie_embeddings=True,
)

LLAMA3_8B = TransformerConfig(
    name="llama3-8b",
    vocab_size=128256,
    dim=4096,
    n_layers=32,
    n_heads=32,
    n_kv_heads=8,
    ffn_hidden=14336,
    max_seq_len=8192,
)

LLAMA3_70B = TransformerConfig(
    name="llama3-70b",
    vocab_size=128256,
    dim=8192,
    n_layers=80,
    n_heads=64,
    n_kv_heads=8,
    ffn_hidden=28672,
    max_seq_len=8192,
)

# Too large for ANY single device (812 GB bf16): exists to exercise the
# fake/deferred layer at beyond-hardware scale (the reference's qualitative
# "model too big for the machine" capability, docs/src/fake_tensor.rst:65)
# and the sharded materialization planners.
LLAMA3_405B = TransformerConfig(
    name="llama3-405b",
    vocab_size=128256,
    dim=16384,
    n_layers=126,
    n_heads=128,
    n_kv_heads=8,
    ffn_hidden=53248,
    max_seq_len=8192,
)

MIXTRAL_8X22B = TransformerConfig(
    name="mixtral-8x22b",
    vocab_size=32768,
    dim=6144,
    n_layers=56,
    n_heads=48,
    n_kv_heads=8,
    ffn_hidden=16384,
    max_seq_len=8192,
    moe_num_experts=8,
    moe_top_k=2,
)

CONFIGS = {
    c.name: c
    for c in [
        TINY,
        TINY_GPT2,
        TINY_MOE,
        GPT2_XL,
        LLAMA3_8B,
        LLAMA3_70B,
        LLAMA3_405B,
        MIXTRAL_8X22B,
    ]
}
