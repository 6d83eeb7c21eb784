# AnyPrecisionAdamW tests. Coverage model: reference
# tests/python/test_anyprecision_optimizer.py (bitwise equivalence vs
# torch.optim.AdamW in the all-fp32 / no-Kahan configuration) plus Kahan
# and low-precision state sanity checks.

import pytest
import torch

from torchdistx_amd.optimizers import AnyPrecisionAdamW


def _make_models(device):
    torch.manual_seed(2)
    model_a = torch.nn.Sequential(
        torch.nn.Linear(8, 16), torch.nn.ReLU(), torch.nn.Linear(16, 4)
    ).to(device)
    model_b = torch.nn.Sequential(
        torch.nn.Linear(8, 16), torch.nn.ReLU(), torch.nn.Linear(16, 4)
    ).to(device)
    model_b.load_state_dict(model_a.state_dict())
    return model_a, model_b


@pytest.mark.parametrize(
    "device",
    [
        "cpu",
        pytest.param("cuda", marks=pytest.mark.gpu),
    ],
)
def test_fp32_no_kahan_matches_adamw_bitwise(device) -> None:
    if device == "cuda" and not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    model_ref, model_any = _make_models(device)

    optim_ref = torch.optim.AdamW(
        model_ref.parameters(), lr=1e-2, weight_decay=0.01, foreach=False
    )
    optim_any = AnyPrecisionAdamW(
        model_any.parameters(),
        lr=1e-2,
        weight_decay=0.01,
        use_kahan_summation=False,
        momentum_dtype=torch.float32,
        variance_dtype=torch.float32,
        # Bitwise parity is an eager-path property; the fused CDNA4 kernel
        # is checked against eager with tolerances in test_gpu.py.
        use_fused=False,
    )

    for step in range(6):
        torch.manual_seed(100 + step)
        x = torch.randn(4, 8, device=device)
        for model, optim in ((model_ref, optim_ref), (model_any, optim_any)):
            optim.zero_grad()
            model(x).square().mean().backward()
            optim.step()

    for p_ref, p_any in zip(model_ref.parameters(), model_any.parameters()):
        assert torch.equal(p_ref, p_any)


def test_kahan_bf16_tracks_fp32_updates() -> None:
    # A bf16 model with Kahan compensation must track tiny repeated updates
    # that plain bf16 accumulation would lose entirely.
    steps = 1000
    p_kahan = torch.nn.Parameter(torch.ones(64, dtype=torch.bfloat16))
    p_plain = torch.nn.Parameter(torch.ones(64, dtype=torch.bfloat16))

    def run(param, use_kahan):
        opt = AnyPrecisionAdamW(
            [param],
            lr=1e-5,
            weight_decay=0.0,
            use_kahan_summation=use_kahan,
            momentum_dtype=torch.float32,
            variance_dtype=torch.float32,
            compensation_buffer_dtype=torch.bfloat16,
        )
        for _ in range(steps):
            param.grad = torch.full_like(param, 1.0)
            opt.step()

    run(p_kahan, True)
    run(p_plain, False)

    # Closed-form-ish reference in fp64: constant gradient=1 makes the Adam
    # update converge to -lr per step almost immediately.
    expected_drop = 1e-5 * steps
    kahan_drop = (1.0 - p_kahan.detach().to(torch.float64)).mean().item()
    plain_drop = (1.0 - p_plain.detach().to(torch.float64)).mean().item()

    # Kahan keeps at least ~80% of the ideal total update; plain bf16 loses
    # most of it (bf16 has ~3 decimal digits; 1e-5 steps round to nothing).
    assert kahan_drop > 0.7 * expected_drop
    assert plain_drop < 0.3 * expected_drop


def test_sparse_gradients_rejected() -> None:
    p = torch.nn.Parameter(torch.zeros(4))
    opt = AnyPrecisionAdamW([p])
    i = torch.tensor([[0], [2]]).T
    p.grad = torch.sparse_coo_tensor(i, torch.ones(2), (4,))
    with pytest.raises(RuntimeError, match="sparse"):
        opt.step()


def test_state_dtypes_follow_config() -> None:
    p = torch.nn.Parameter(torch.zeros(4))
    opt = AnyPrecisionAdamW(
        [p],
        momentum_dtype=torch.bfloat16,
        variance_dtype=torch.bfloat16,
        use_kahan_summation=True,
        compensation_buffer_dtype=torch.float16,
    )
    p.grad = torch.ones(4)
    opt.step()
    state = opt.state[p]
    assert state["exp_avg"].dtype == torch.bfloat16
    assert state["exp_avg_sq"].dtype == torch.bfloat16
    assert state["compensation"].dtype == torch.float16


def test_state_dict_roundtrip() -> None:
    torch.manual_seed(5)
    model = torch.nn.Linear(6, 6)
    opt = AnyPrecisionAdamW(model.parameters(), lr=1e-2, use_kahan_summation=True,
                            momentum_dtype=torch.float32,
                            variance_dtype=torch.float32)
    for _ in range(3):
        opt.zero_grad()
        model(torch.randn(4, 6)).square().mean().backward()
        opt.step()
    # Serialize through a real checkpoint boundary (torch.save/load), which
    # breaks the tensor aliasing a live state_dict keeps.
    import io

    buf = io.BytesIO()
    torch.save(opt.state_dict(), buf)
    buf.seek(0)
    sd = torch.load(buf)

    model2 = torch.nn.Linear(6, 6)
    model2.load_state_dict(model.state_dict())
    opt2 = AnyPrecisionAdamW(model2.parameters(), lr=1e-2, use_kahan_summation=True,
                             momentum_dtype=torch.float32,
                             variance_dtype=torch.float32)
    opt2.load_state_dict(sd)

    # One identical step after restore must produce identical params.
    torch.manual_seed(77)
    x = torch.randn(4, 6)
    for m, o in ((model, opt), (model2, opt2)):
        o.zero_grad()
        m(x).square().mean().backward()
        o.step()
    for p1, p2 in zip(model.parameters(), model2.parameters()):
        assert torch.equal(p1, p2)
