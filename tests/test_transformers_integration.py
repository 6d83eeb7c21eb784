# Third-party ecosystem check: HuggingFace transformers models build under
# deferred_init and materialize/train normally (the reference's qualitative
# Blenderbot capability, docs/src/fake_tensor.rst:65). Exact bitwise
# equality with eager construction does not hold for HF models because
# their two-pass init (constructor defaults + post_init re-init) consumes
# the RNG stream in an order per-tensor replay cannot reproduce — the same
# property the reference's replay has; distributions are identical.

import pytest
import torch

transformers = pytest.importorskip("transformers")

from torchdistx_amd import deferred_init, is_deferred, materialize_module


def test_hf_gpt2_deferred_materialize_train() -> None:
    cfg = transformers.GPT2Config(
        n_embd=64, n_layer=2, n_head=4, vocab_size=256, n_positions=64
    )
    torch.manual_seed(0)
    m = deferred_init(transformers.GPT2LMHeadModel, cfg)
    assert is_deferred(m)
    assert sum(p.numel() for p in m.parameters()) > 100_000

    materialize_module(m)
    assert not is_deferred(m)

    ids = torch.randint(0, 256, (2, 16))
    out = m(input_ids=ids, labels=ids)
    out.loss.backward()
    assert out.loss.isfinite().item()
    # init statistics: embedding weights ~ N(0, initializer_range)
    std = m.transformer.wte.weight.detach().std().item()
    assert std == pytest.approx(cfg.initializer_range, rel=0.15)


def test_hf_llama_deferred() -> None:
    cfg = transformers.LlamaConfig(
        hidden_size=64,
        intermediate_size=128,
        num_hidden_layers=2,
        num_attention_heads=4,
        num_key_value_heads=2,
        vocab_size=256,
        max_position_embeddings=64,
    )
    m = deferred_init(transformers.LlamaForCausalLM, cfg)
    assert is_deferred(m)
    materialize_module(m)
    ids = torch.randint(0, 256, (2, 8))
    assert m(input_ids=ids).logits.shape == (2, 8, 256)
