from torchdistx_amd.parallel.fsdp2 import (  # noqa: F401
    fully_shard_deferred,
)
from torchdistx_amd.parallel.sharded_materialize import (  # noqa: F401
    assign_owners,
    materialize_experts_sharded,
    materialize_module_dim0_sharded,
    materialize_module_dtensor,
    materialize_module_distributed,
    materialize_module_tp_sharded,
    materialize_tensor_shard,
)
