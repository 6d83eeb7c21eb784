# torchdistx_amd — an MI355X-native framework with the capabilities of
# pytorch/torchdistx: fake tensors, deferred module initialization with
# tape replay into HBM3E through hand-written CDNA4 HIP init kernels,
# sharded materialization over RCCL/xGMI, and the SlowMo / GossipGraD /
# AnyPrecisionAdamW distributed-training utilities.
#
# Public API parity map (reference file:line):
#   fake_mode / is_fake / meta_like        — reference src/python/torchdistx/fake.py:43-84
#   deferred_init / is_deferred /
#   materialize_tensor / materialize_module — reference deferred_init.py:19-124
#   slowmo, gossip_grad, optimizers        — reference slowmo/, gossip_grad.py, optimizers/

__version__ = "0.2.0"

from torchdistx_amd.fake import fake_mode, is_fake, meta_like  # noqa: F401
from torchdistx_amd.deferred_init import (  # noqa: F401
    deferred_init,
    is_deferred,
    materialize_module,
    materialize_module_batched,
    materialize_module_parallel,
    materialize_tensor,
)

# Register the CDNA4 kernels (tdx:: ops) when the extension is built; the
# deferred-init replay engine redirects GPU init ops to them.
from torchdistx_amd import _kernels  # noqa: E402,F401
