# GossipGraD (arXiv:1803.05880) — gossip-based gradient exchange as an FSDP
# communication hook.
#
# Capability parity with the reference
# (/root/reference/src/python/torchdistx/gossip_grad.py:26-389): the
# Topology enum (CUBE hypercube via rank XOR, DISSEMINATION via +/-2^power),
# GossipGraDState with the same constructor surface and validation,
# get_num_modules, the INVALID_PEER sentinel for non-power-of-two CUBE
# configurations, and the three-stage hook: intra-node all-reduce ->
# master-rank P2P gossip -> intra-node broadcast.
#
# MI355X-native notes: all communication goes through torch.distributed c10d,
# which is RCCL on ROCm. The P2P gossip exchange is a single paired
# send/recv per backward — on the xGMI fabric of one node that maps to ONE of
# the 7 point-to-point links (each ~153 GB/s), leaving the other six links
# free, so gossiping is nearly free relative to a global all-reduce. When the
# hook runs multi-node, the same pairing maps to one NIC peer.

import math
import random
from enum import Enum, auto
from itertools import cycle

import torch
import torch.distributed as dist
from torch._C._distributed_c10d import ProcessGroup
from torch.distributed.algorithms._comm_hooks import default
from torch.distributed.fsdp import FullyShardedDataParallel as FSDP

# Sentinel for "no communication this step": in a CUBE topology with a
# non-power-of-two node count the XOR peer may fall outside the topology,
# in which case the gossip stage is skipped entirely.
INVALID_PEER = -1


class Topology(Enum):
    """Virtual topology used to pick gossip peers each period.

    CUBE: hypercube — gossip with the neighbour vertex along dimension
    ``power`` (peer index = node_rank XOR 2**power). Requires an even number
    of nodes; non-power-of-two out-of-range peers skip communication.

    DISSEMINATION: neighbour at +2**power (send) and -2**power (receive),
    walking 1, 2, 4, ... hops so information disseminates to all nodes in
    ``ceil(log2(num_nodes))`` periods.
    """

    CUBE = auto()
    DISSEMINATION = auto()


class GossipGraDState(default.DefaultState):
    """State for the GossipGraD communication hook.

    .. note:: Use with the NCCL/RCCL process-group backend and set the
        current device (``torch.cuda.set_device``) before construction,
        otherwise the paired isend/irecv can hang.

    Args:
        num_modules: number of FSDP units in the wrapped model — the hook
            fires once per unit per backward, and the topology must only
            rotate between backward passes (see :func:`get_num_modules`).
        topology: virtual topology (default: DISSEMINATION).
        local_process_group: intra-node subgroup; must be passed together
            with ``num_nodes``. Defaults to one subgroup per node.
        num_nodes: number of nodes; must be passed together with
            ``local_process_group``.
        master_process_group: group of the inter-node gossip workers.
            Defaults to the rank-0 worker of every node.
        proc_per_node: workers per node; defaults to the local subgroup size.
        random_seed: seed for the topology shuffles so every worker generates
            the same sequence (default: 2403).
    """

    def __init__(
        self,
        num_modules,
        topology=None,
        local_process_group=None,
        num_nodes=None,
        master_process_group=None,
        proc_per_node=None,
        random_seed=2403,
    ):
        if not num_modules or num_modules < 1:
            raise ValueError(
                f"num_modules must be >= 1 (got {num_modules!r}); pass "
                "get_num_modules(fsdp_model) so the topology rotation "
                "aligns with backward passes."
            )
        self.num_modules = num_modules
        self.topology = topology or Topology.DISSEMINATION

        if (local_process_group is None) != (num_nodes is None):
            raise ValueError(
                "pass local_process_group and num_nodes together, or "
                "neither (then one subgroup per node is derived)."
            )
        if local_process_group is None:
            self.local_process_group, subgroups = dist.new_subgroups()
            self.num_nodes = len(subgroups)
        else:
            if num_nodes < 1:
                raise ValueError(f"num_nodes must be >= 1, got {num_nodes}")
            self.local_process_group = local_process_group
            self.num_nodes = num_nodes

        if self.topology == Topology.CUBE and self.num_nodes % 2 != 0:
            raise ValueError(
                f"the CUBE (hypercube) topology needs an even node count; "
                f"got {self.num_nodes}. Use DISSEMINATION for odd counts."
            )

        super().__init__(self.local_process_group)

        if proc_per_node is None:
            proc_per_node = self.local_process_group.size()
        if proc_per_node < 1:
            raise ValueError(
                f"proc_per_node must be >= 1, got {proc_per_node}"
            )
        self.proc_per_node = proc_per_node

        if master_process_group is None:
            master_process_group = self._create_master_group()
        self.master_process_group = master_process_group

        self.random_seed = random_seed
        self.topologies = self._generate_topologies(self.random_seed)
        self.cur_topology = next(self.topologies)

        # Number of gossip steps to disseminate across all nodes; also the
        # topology-rotation period. Integer >= 1 even for 1 node or
        # non-power-of-two node counts.
        self.gossip_period = max(1, math.ceil(math.log(self.num_nodes, 2)))
        self.iter = 0
        self.rank = dist.get_rank()
        # Global rank of this node's master (local rank 0).
        self.master_worker = dist.get_global_rank(self.local_process_group, 0)

    def _create_master_group(self):
        """One master worker per node: global ranks 0, K, 2K, ... for K
        processes per node."""
        ranks = [i * self.proc_per_node for i in range(self.num_nodes)]
        return dist.new_group(ranks)

    def _generate_topologies(self, random_seed):
        """``num_nodes`` deterministic shuffles of the master-rank list
        [0*K, 1*K, ..., (N-1)*K], cycled forever. Master global ranks are
        stored (not node indices) so peer lookup is direct."""
        rng = random.Random(random_seed)
        masters = [i * self.proc_per_node for i in range(self.num_nodes)]
        shuffles = []
        for _ in range(self.num_nodes):
            rng.shuffle(masters)
            shuffles.append(list(masters))
        return cycle(shuffles)


def _get_send_recv_peers(state):
    """Global ranks of the send and receive peers for the current gossip
    step, from the current virtual topology and step power.

    CUBE: the same neighbour both ways (node_rank XOR 2**power); peers
    outside the topology return (INVALID_PEER, INVALID_PEER).
    DISSEMINATION: send to node_rank + 2**power, receive from
    node_rank - 2**power (mod num_nodes).
    """
    assert state.gossip_period > 0, "gossip_period must be positive"
    power = (state.iter // state.num_modules) % state.gossip_period
    node_rank = state.cur_topology.index(state.rank)

    if state.topology == Topology.CUBE:
        peer_idx = node_rank ^ (1 << power)
        if peer_idx >= len(state.cur_topology):
            return INVALID_PEER, INVALID_PEER
        peer = state.cur_topology[peer_idx]
        return peer, peer

    send_peer = state.cur_topology[(node_rank + (1 << power)) % state.num_nodes]
    recv_peer = state.cur_topology[(node_rank - (1 << power)) % state.num_nodes]
    return send_peer, recv_peer


def _gossip(state, grad, scaling_factor=0.5):
    """One gossip exchange: paired async send/recv of ``grad`` with the
    topology peers over the master process group, then average the received
    gradient into ``grad`` (scaled by ``scaling_factor`` since exactly two
    gradients are combined)."""
    send_peer, recv_peer = _get_send_recv_peers(state)
    if send_peer == INVALID_PEER or recv_peer == INVALID_PEER:
        return

    assert state.rank not in (send_peer, recv_peer), (
        f"a rank must never gossip with itself: rank={state.rank}, "
        f"send={send_peer}, recv={recv_peer}"
    )
    assert isinstance(state.master_process_group, ProcessGroup), (
        "master_process_group must be a torch.distributed ProcessGroup"
    )

    group = state.master_process_group
    recv_grad = torch.empty_like(grad)
    pair = [
        dist.P2POp(op=dist.isend, tensor=grad, peer=send_peer, group=group),
        dist.P2POp(op=dist.irecv, tensor=recv_grad, peer=recv_peer,
                   group=group),
    ]
    for req in dist.batch_isend_irecv(pair):
        req.wait()
    grad.add_(recv_grad).mul_(scaling_factor)


def get_num_modules(module: torch.nn.Module):
    """Number of FSDP modules nested in ``module`` (including itself)."""
    return len(FSDP.fsdp_modules(module))


def gossip_grad_hook(state: GossipGraDState, grad: torch.Tensor):
    """GossipGraD communication hook.

    Per FSDP-unit gradient: (1) all-reduce inside the local (intra-node)
    group; (2) the node's master rank gossips with its topology peers;
    (3) the master broadcasts the combined gradient to the local group. The
    virtual topology rotates every ``gossip_period`` backward passes (the
    per-unit ``iter`` counter is normalised by ``num_modules`` so a rotation
    never happens mid-backward).

    Register with
    ``fsdp_net.register_comm_hook(state, gossip_grad_hook)``.
    """
    if (state.iter // state.num_modules) % state.gossip_period == 0:
        state.cur_topology = next(state.topologies)

    default.allreduce_hook(state, grad)

    if not dist._rank_not_in_group(state.master_process_group):
        _gossip(state, grad)

    dist.broadcast(grad, src=state.master_worker, group=state.local_process_group)

    state.iter += 1
