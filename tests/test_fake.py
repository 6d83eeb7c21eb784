# Fake-tensor tests. Coverage model: reference tests/python/test_fake.py
# (fake CUDA construction on CPU-only CI, guard teardown, meta_like
# metadata, error cases) plus extra coverage for ops on fake tensors,
# views, and repr.

import pytest
import torch

from torchdistx_amd.fake import fake_mode, is_fake, meta_like


def test_fake_mode_returns_cuda_tensor_if_fake_cuda_is_true() -> None:
    if torch.cuda.is_available():
        pytest.skip("Can only be tested if CUDA is not available.")

    with fake_mode(fake_cuda=True):
        a = torch.ones([10], device="cuda")

    assert a.device.type == "cuda"
    assert is_fake(a)


def test_fake_mode_raises_error_if_fake_cuda_is_false() -> None:
    if torch.cuda.is_available():
        pytest.skip("Can only be tested if CUDA is not available.")

    with pytest.raises((AssertionError, RuntimeError)):
        with fake_mode():
            torch.ones([10], device="cuda")


def test_cuda_tensor_raises_error_after_fake_mode() -> None:
    if torch.cuda.is_available():
        pytest.skip("Can only be tested if CUDA is not available.")

    with fake_mode(fake_cuda=True):
        torch.ones([10], device="cuda")

    with pytest.raises((AssertionError, RuntimeError)):
        torch.ones([10], device="cuda")


def test_fake_cpu_tensor_has_no_storage() -> None:
    with fake_mode():
        a = torch.ones([4, 4])

    assert is_fake(a)
    assert a.device.type == "cpu"
    assert a.shape == (4, 4)
    with pytest.raises(RuntimeError):
        a.untyped_storage()


def test_ops_on_fake_tensors_stay_fake() -> None:
    with fake_mode():
        a = torch.ones([4, 4])
        b = a + a
        c = b.view(16)
        d = c[:8]
    assert is_fake(b) and is_fake(c) and is_fake(d)
    assert d.shape == (8,)


def test_fake_cuda_ops_report_cuda_device() -> None:
    if torch.cuda.is_available():
        pytest.skip("Can only be tested if CUDA is not available.")
    with fake_mode(fake_cuda=True):
        a = torch.zeros([8], device="cuda")
        b = a * 2 + 1
    assert b.device.type == "cuda"
    assert is_fake(b)


def test_real_tensors_unaffected_inside_fake_mode() -> None:
    r = torch.ones([3])
    with fake_mode():
        s = r.sum()
    # `r` is real, so the computation ran for real.
    assert not is_fake(r)
    assert s.item() == pytest.approx(3.0)


def test_meta_like_returns_meta_tensor() -> None:
    with fake_mode():
        a = torch.ones([10])

    b = meta_like(a)

    assert not is_fake(b)
    assert b.device.type == "meta"
    assert b.dtype == a.dtype
    assert b.size() == a.size()
    assert b.stride() == a.stride()


def test_meta_like_raises_error_if_tensor_is_not_fake() -> None:
    a = torch.ones([10])

    with pytest.raises(ValueError):
        meta_like(a)


def test_fake_repr_mentions_fake() -> None:
    with fake_mode():
        a = torch.ones([2, 3], dtype=torch.float64)
    r = repr(a)
    assert "fake=True" in r
    assert "size=(2, 3)" in r
    # Real tensors keep the stock repr.
    assert "fake" not in repr(torch.ones([1]))


def test_fake_mode_nesting() -> None:
    with fake_mode():
        with fake_mode():
            a = torch.ones([2])
        b = torch.ones([2])
    assert is_fake(a) and is_fake(b)
    assert not is_fake(torch.ones([2]))


def test_backward_through_fake_tensors() -> None:
    # Autograd runs above the Fake key, so backward works end-to-end on
    # fake tensors and produces fake gradients.
    with fake_mode():
        x = torch.randn(4, 4, requires_grad=True)
        y = (x * 2).sum()
        y.backward()
    assert x.grad is not None
    assert is_fake(x.grad)
    assert x.grad.shape == (4, 4)
