# Alias of torchdistx_amd.gossip_grad (see package __init__).
from torchdistx_amd.gossip_grad import (  # noqa: F401
    INVALID_PEER,
    GossipGraDState,
    Topology,
    get_num_modules,
    gossip_grad_hook,
)
