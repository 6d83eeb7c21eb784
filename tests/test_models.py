# Model-zoo tests: every tiny config builds under deferred_init,
# materializes bitwise-equal to eager construction (the replay-exactness
# property the zoo's single-RNG-op-per-parameter init guarantees), and
# trains a step.

import pytest
import torch

from torchdistx_amd import deferred_init, is_deferred, materialize_module
from torchdistx_amd.models import (
    CONFIGS,
    TINY,
    TINY_GPT2,
    TINY_MOE,
    build_model,
)


@pytest.mark.parametrize("cfg", [TINY, TINY_GPT2, TINY_MOE], ids=lambda c: c.name)
def test_deferred_matches_eager_bitwise(cfg) -> None:
    torch.manual_seed(0)
    deferred = deferred_init(build_model, cfg)
    assert is_deferred(deferred)
    materialize_module(deferred)

    torch.manual_seed(0)
    eager = build_model(cfg)

    for (dn, dp), (en, ep) in zip(
        deferred.named_parameters(), eager.named_parameters()
    ):
        assert dn == en
        assert torch.equal(dp, ep), dn
    for (dn, db), (en, eb) in zip(
        deferred.named_buffers(), eager.named_buffers()
    ):
        assert dn == en
        assert torch.equal(db, eb), dn


@pytest.mark.parametrize("cfg", [TINY, TINY_GPT2, TINY_MOE], ids=lambda c: c.name)
def test_forward_backward(cfg) -> None:
    torch.manual_seed(1)
    m = deferred_init(build_model, cfg)
    materialize_module(m)
    tokens = torch.randint(0, cfg.vocab_size, (2, 16))
    loss = m.loss(tokens)
    loss.backward()
    assert loss.isfinite().item()
    assert all(
        p.grad is not None for p in m.parameters() if p.requires_grad
    )


def test_param_count_formula_is_exact() -> None:
    for name in ("tiny", "tiny-gpt2", "tiny-moe"):
        cfg = CONFIGS[name]
        actual = sum(p.numel() for p in build_model(cfg).parameters())
        assert cfg.n_params == actual, name


def test_rope_cache_overflow_raises() -> None:
    m = build_model(TINY)
    with pytest.raises(ValueError, match="RoPE"):
        m(torch.randint(0, TINY.vocab_size, (1, TINY.max_seq_len + 1)))


def test_bf16_dtype_context() -> None:
    m = deferred_init(build_model, TINY, device="cpu", dtype=torch.bfloat16)
    materialize_module(m)
    assert m.tok_emb.weight.dtype == torch.bfloat16
    assert torch.get_default_dtype() == torch.float32


def test_stock_init_matches_eager_bitwise() -> None:
    # cfg.init="stock" builds unmodified nn.Linear/nn.Embedding: the tape
    # records PyTorch's kaiming resets (empty -> uniform_ chains) and CPU
    # replay reproduces eager construction exactly.
    import dataclasses

    from torchdistx_amd import deferred_init
    from torchdistx_amd.deferred_init import materialize_module
    from torchdistx_amd.models import TINY, build_model

    cfg = dataclasses.replace(TINY, init="stock")
    torch.manual_seed(21)
    m = deferred_init(build_model, cfg)
    materialize_module(m)

    torch.manual_seed(21)
    e = build_model(cfg)
    for (n1, p1), (n2, p2) in zip(m.named_parameters(), e.named_parameters()):
        assert n1 == n2 and torch.equal(p1, p2), n1
    # Stock init must actually produce plain nn.Linear modules.
    from torchdistx_amd.models.transformer import InitLinear

    assert not any(isinstance(mod, InitLinear) for mod in m.modules())
