# Drop-in compatibility alias: `torchdistx` resolves to torchdistx_amd, the
# MI355X-native implementation. Anything written against
# pytorch/torchdistx's public API — including torch.distributed.fsdp's
# built-in deferred-init support, which does `from torchdistx import
# deferred_init, fake` (torch/distributed/fsdp/_init_utils.py:55) — works
# unchanged against this package. Like the reference, the package exposes
# its API through the `fake` and `deferred_init` submodules.

__version__ = "0.3.0"

from torchdistx import deferred_init, fake  # noqa: F401
