# Fake-tensor Python API.
#
# Capability parity with the reference
# (/root/reference/src/python/torchdistx/fake.py:43-84): the fake_mode()
# context manager (with the fake_cuda escape hatch for GPU-less machines),
# is_fake, meta_like, and a Tensor.__repr__ patch so printing a storage-less
# tensor does not try to read its (nonexistent) data.

from contextlib import contextmanager
from typing import Generator

import torch

from torchdistx_amd import _C

_original_tensor_repr = torch.Tensor.__repr__


def _fake_aware_repr(tensor: torch.Tensor) -> str:
    if not _C.is_fake(tensor):
        return _original_tensor_repr(tensor)
    parts = [f"size={tuple(tensor.shape)}"]
    if tensor.dtype != torch.get_default_dtype():
        parts.append(f"dtype={tensor.dtype}")
    if tensor.device.type != "cpu":
        parts.append(f"device={tensor.device}")
    if tensor.requires_grad:
        parts.append("requires_grad=True")
    return "tensor(..., " + ", ".join(parts) + ", fake=True)"


# `Tensor.__repr__` reads storage, which fake tensors do not have; replace
# it with a fake-aware version (reference fake.py:17-40).
torch.Tensor.__repr__ = _fake_aware_repr  # type: ignore[method-assign]


@contextmanager
def fake_mode(*, fake_cuda: bool = False) -> Generator:
    """Context manager under which every newly constructed tensor is fake:
    it reports its real device and full metadata but owns no storage.

    Args:
        fake_cuda:
            Allow constructing fake "cuda" tensors even when no GPU is
            available (useful on CPU-only CI). Ignored when a GPU is
            present.
    """
    _C.enter_fake_mode(fake_cuda)
    try:
        yield
    finally:
        _C.leave_fake_mode()


def is_fake(tensor: torch.Tensor) -> bool:
    """Whether ``tensor`` is a fake (storage-less) tensor."""
    return _C.is_fake(tensor)


def meta_like(fake: torch.Tensor) -> torch.Tensor:
    """A meta tensor with the same sizes/strides/dtype as ``fake``, detached
    from any autograd history.

    Raises:
        ValueError: when ``fake`` is not a fake tensor.
    """
    return _C.meta_like(fake)
