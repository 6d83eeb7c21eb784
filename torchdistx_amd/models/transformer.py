# Transformer model zoo for the benchmark configs (GPT-2-XL / Llama-3 /
# Mixtral shapes). These modules exist so deferred_init -> materialize can
# be exercised and measured on real model layouts; forward/backward runs on
# stock PyTorch-ROCm ops (scaled_dot_product_attention lowers to an
# MI355X-tuned kernel via MIOpen/aotriton).
#
# The reference has no model zoo (it is a library); these definitions are
# plain nn.Modules whose construction consists of exactly the op set the
# deferred-init tape records (empty / uniform_ / normal_ / zero_ / fill_ /
# copy_), see SURVEY.md section 2.7.

from typing import Optional

import torch
from torch import Tensor, nn

from torchdistx_amd.models.configs import TransformerConfig



_INIT_STD = 0.02


class InitLinear(nn.Linear):
    """Linear with a single-RNG-op init (N(0, 0.02) weight, zero bias) run
    once at construction. Under deferred_init this keeps every parameter's
    tape segment contiguous and at most one RNG op long, so per-tensor
    replay consumes the generator stream in exactly the eager order and the
    materialized weights are bitwise-equal to eager construction — and the
    default kaiming pass (a full extra write over the weights at
    materialization time) is never recorded at all."""

    def reset_parameters(self) -> None:
        nn.init.normal_(self.weight, mean=0.0, std=_INIT_STD)
        if self.bias is not None:
            nn.init.zeros_(self.bias)


class InitEmbedding(nn.Embedding):
    def reset_parameters(self) -> None:
        nn.init.normal_(self.weight, mean=0.0, std=_INIT_STD)
        self._fill_padding_idx_with_zero()


import threading

# Init mode in effect during model construction (set by build_model from
# cfg.init). "stock" builds unmodified nn.Linear / nn.Embedding, whose
# default resets record the empty -> kaiming(uniform_) tape shape of
# ordinary user models; "fast" uses the single-normal_ variants above.
_init_mode = threading.local()


def _stock_init() -> bool:
    return getattr(_init_mode, "mode", "fast") == "stock"


def _linear(d_in: int, d_out: int, bias: bool = False) -> nn.Linear:
    if _stock_init():
        return nn.Linear(d_in, d_out, bias=bias)
    return InitLinear(d_in, d_out, bias=bias)


def _embedding(num: int, dim: int) -> nn.Embedding:
    if _stock_init():
        return nn.Embedding(num, dim)
    return InitEmbedding(num, dim)


class RMSNorm(nn.Module):
    def __init__(self, dim: int, eps: float = 1e-5):
        super().__init__()
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(dim))

    def forward(self, x: Tensor) -> Tensor:
        dt = x.dtype
        x = x.float()
        x = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + self.eps)
        return (x * self.weight.float()).to(dt)


def _rope_cache(head_dim: int, max_seq_len: int, base: float = 500000.0):
    inv_freq = 1.0 / (
        base ** (torch.arange(0, head_dim, 2, dtype=torch.float32) / head_dim)
    )
    t = torch.arange(max_seq_len, dtype=torch.float32)
    freqs = torch.outer(t, inv_freq)
    return torch.cos(freqs), torch.sin(freqs)


def _apply_rope(x: Tensor, cos: Tensor, sin: Tensor) -> Tensor:
    # x: [B, H, S, D]
    s = x.shape[-2]
    if s > cos.shape[0]:
        raise ValueError(
            f"sequence length {s} exceeds the RoPE cache "
            f"(max_seq_len={cos.shape[0]})"
        )
    cos = cos[:s].to(x.dtype)
    sin = sin[:s].to(x.dtype)
    x1, x2 = x.chunk(2, dim=-1)
    return torch.cat((x1 * cos - x2 * sin, x2 * cos + x1 * sin), dim=-1)


class Attention(nn.Module):
    def __init__(self, cfg: TransformerConfig):
        super().__init__()
        self.n_heads = cfg.n_heads
        self.n_kv_heads = cfg.n_kv_heads
        self.head_dim = cfg.dim // cfg.n_heads
        kv_dim = self.head_dim * cfg.n_kv_heads
        self.wq = _linear(cfg.dim, cfg.dim)
        self.wk = _linear(cfg.dim, kv_dim)
        self.wv = _linear(cfg.dim, kv_dim)
        self.wo = _linear(cfg.dim, cfg.dim)

    def forward(self, x: Tensor, cos: Optional[Tensor], sin: Optional[Tensor]):
        b, s, _ = x.shape
        q = self.wq(x).view(b, s, self.n_heads, self.head_dim).transpose(1, 2)
        k = self.wk(x).view(b, s, self.n_kv_heads, self.head_dim).transpose(1, 2)
        v = self.wv(x).view(b, s, self.n_kv_heads, self.head_dim).transpose(1, 2)
        if cos is not None:
            q = _apply_rope(q, cos, sin)
            k = _apply_rope(k, cos, sin)
        out = torch.nn.functional.scaled_dot_product_attention(
            q, k, v, is_causal=True, enable_gqa=self.n_kv_heads != self.n_heads
        )
        out = out.transpose(1, 2).reshape(b, s, -1)
        return self.wo(out)


class SwiGLU(nn.Module):
    def __init__(self, dim: int, hidden: int):
        super().__init__()
        self.w1 = _linear(dim, hidden)  # gate
        self.w3 = _linear(dim, hidden)  # up
        self.w2 = _linear(hidden, dim)  # down

    def forward(self, x: Tensor) -> Tensor:
        return self.w2(torch.nn.functional.silu(self.w1(x)) * self.w3(x))


class GeluMLP(nn.Module):
    def __init__(self, dim: int, hidden: int):
        super().__init__()
        self.fc_in = _linear(dim, hidden, bias=True)
        self.fc_out = _linear(hidden, dim, bias=True)

    def forward(self, x: Tensor) -> Tensor:
        return self.fc_out(torch.nn.functional.gelu(self.fc_in(x)))


class MoELayer(nn.Module):
    """Mixtral-style top-k sparse MoE over SwiGLU experts."""

    def __init__(self, cfg: TransformerConfig):
        super().__init__()
        self.top_k = cfg.moe_top_k
        self.gate = _linear(cfg.dim, cfg.moe_num_experts)
        self.experts = nn.ModuleList(
            SwiGLU(cfg.dim, cfg.ffn_hidden) for _ in range(cfg.moe_num_experts)
        )

    def forward(self, x: Tensor) -> Tensor:
        b, s, d = x.shape
        flat = x.view(-1, d)
        logits = self.gate(flat)
        weights, selected = torch.topk(logits, self.top_k, dim=-1)
        weights = torch.softmax(weights, dim=-1, dtype=torch.float32).to(x.dtype)
        out = torch.zeros_like(flat)
        for e, expert in enumerate(self.experts):
            token_idx, slot = (selected == e).nonzero(as_tuple=True)
            if token_idx.numel() == 0:
                continue
            out.index_add_(
                0,
                token_idx,
                expert(flat[token_idx]) * weights[token_idx, slot, None],
            )
        return out.view(b, s, d)


class Block(nn.Module):
    def __init__(self, cfg: TransformerConfig):
        super().__init__()
        norm = RMSNorm if cfg.norm == "rmsnorm" else nn.LayerNorm
        self.attn_norm = norm(cfg.dim)
        self.attn = Attention(cfg)
        self.ffn_norm = norm(cfg.dim)
        if cfg.moe_num_experts:
            self.ffn = MoELayer(cfg)
        elif cfg.act == "swiglu":
            self.ffn = SwiGLU(cfg.dim, cfg.ffn_hidden)
        else:
            self.ffn = GeluMLP(cfg.dim, cfg.ffn_hidden)

    def forward(self, x, cos, sin):
        x = x + self.attn(self.attn_norm(x), cos, sin)
        return x + self.ffn(self.ffn_norm(x))


class TransformerLM(nn.Module):
    def __init__(self, cfg: TransformerConfig):
        super().__init__()
        self.cfg = cfg
        self.tok_emb = _embedding(cfg.vocab_size, cfg.dim)
        if cfg.rope:
            self.pos_emb = None
            # RoPE rotates half the head dim; cache covers head_dim/2 freqs.
            cos, sin = _rope_cache(cfg.dim // cfg.n_heads, cfg.max_seq_len)
            self.register_buffer("rope_cos", cos, persistent=False)
            self.register_buffer("rope_sin", sin, persistent=False)
        else:
            self.pos_emb = _embedding(cfg.max_seq_len, cfg.dim)
            self.rope_cos = None
            self.rope_sin = None
        self.blocks = nn.ModuleList(Block(cfg) for _ in range(cfg.n_layers))
        norm = RMSNorm if cfg.norm == "rmsnorm" else nn.LayerNorm
        self.final_norm = norm(cfg.dim)
        if cfg.tie_embeddings:
            self.lm_head = None
        else:
            self.lm_head = _linear(cfg.dim, cfg.vocab_size)

    def forward(self, tokens: Tensor) -> Tensor:
        x = self.tok_emb(tokens)
        if self.pos_emb is not None:
            pos = torch.arange(tokens.shape[1], device=tokens.device)
            x = x + self.pos_emb(pos)
        cos, sin = self.rope_cos, self.rope_sin
        for block in self.blocks:
            x = block(x, cos, sin)
        x = self.final_norm(x)
        if self.lm_head is not None:
            return self.lm_head(x)
        return torch.nn.functional.linear(x, self.tok_emb.weight)

    def loss(self, tokens: Tensor) -> Tensor:
        logits = self.forward(tokens[:, :-1])
        return torch.nn.functional.cross_entropy(
            logits.reshape(-1, logits.shape[-1]).float(),
            tokens[:, 1:].reshape(-1),
        )


def build_model(cfg: TransformerConfig, device=None, dtype=None) -> TransformerLM:
    """Builds the model with the given default device/dtype; suitable for
    use as the `module_fn` of deferred_init. cfg.init selects between the
    single-normal_ fast init and PyTorch's stock kaiming resets."""
    _init_mode.mode = cfg.init
    try:
        if dtype is None and device is None:
            return TransformerLM(cfg)
        ctx_dtype = dtype if dtype is not None else torch.get_default_dtype()
        prev_dtype = torch.get_default_dtype()
        torch.set_default_dtype(ctx_dtype)
        try:
            if device is not None:
                with torch.device(device):
                    return TransformerLM(cfg)
            return TransformerLM(cfg)
        finally:
            torch.set_default_dtype(prev_dtype)
    finally:
        _init_mode.mode = "fast"


__all__ = ["TransformerLM", "build_model", "RMSNorm", "MoELayer"]
