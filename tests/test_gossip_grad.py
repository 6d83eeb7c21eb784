# GossipGraD tests. Coverage model: reference
# tests/python/test_comm_hooks_fsdp.py:409-651 (state validation, gossip
# numerics with pinned deterministic topologies, module counting and
# iteration bookkeeping) re-hosted on a CPU/gloo rig. "Nodes" are simulated
# by declaring small subgroups as nodes, exactly like the reference does
# with dist.new_subgroups(group_size=...).

import itertools

import pytest
import torch
import torch.distributed as dist

from tests._dist_utils import run_distributed
from torchdistx_amd.gossip_grad import (
    INVALID_PEER,
    GossipGraDState,
    Topology,
    _get_send_recv_peers,
    gossip_grad_hook,
)


class _FakeState:
    """Bare-bones stand-in for peer-math unit tests (no process group)."""

    def __init__(self, rank, num_nodes, topology, iter_=0, num_modules=1):
        self.rank = rank
        self.num_nodes = num_nodes
        self.topology = topology
        self.iter = iter_
        self.num_modules = num_modules
        self.cur_topology = list(range(num_nodes))
        import math

        self.gossip_period = max(1, math.ceil(math.log(num_nodes, 2)))


def test_cube_peers_are_symmetric_xor() -> None:
    # 8 nodes: at power p the peer of node r is r ^ 2^p, both directions.
    for power in range(3):
        for rank in range(8):
            st = _FakeState(rank, 8, Topology.CUBE, iter_=power)
            send, recv = _get_send_recv_peers(st)
            assert send == recv == rank ^ (1 << power)


def test_cube_out_of_range_peer_is_invalid() -> None:
    # 6 nodes at power 2: 3 ^ 4 = 7 >= 6 -> no communication.
    st = _FakeState(3, 6, Topology.CUBE, iter_=2)
    send, recv = _get_send_recv_peers(st)
    assert send == INVALID_PEER and recv == INVALID_PEER


def test_dissemination_peers() -> None:
    for power in range(3):
        for rank in range(8):
            st = _FakeState(rank, 8, Topology.DISSEMINATION, iter_=power)
            send, recv = _get_send_recv_peers(st)
            assert send == (rank + (1 << power)) % 8
            assert recv == (rank - (1 << power)) % 8


def test_power_rotates_with_iter_and_num_modules() -> None:
    # With 3 FSDP units per backward, the power only advances every 3 hook
    # invocations, and wraps at gossip_period.
    st = _FakeState(0, 4, Topology.DISSEMINATION, num_modules=3)
    seen = []
    for it in range(12):
        st.iter = it
        send, _ = _get_send_recv_peers(st)
        seen.append(send)
    assert seen == [1, 1, 1, 2, 2, 2, 1, 1, 1, 2, 2, 2]


def _state_validation(rank, world):
    results = {}
    try:
        GossipGraDState(num_modules=0)
    except ValueError:
        results["bad_num_modules"] = True
    try:
        GossipGraDState(num_modules=1, local_process_group=dist.group.WORLD)
    except ValueError:
        results["pg_without_nodes"] = True
    try:
        GossipGraDState(num_modules=1, num_nodes=2)
    except ValueError:
        results["nodes_without_pg"] = True
    # world=2 simulated as 1-rank "nodes": CUBE with 2 nodes is fine.
    state = GossipGraDState(
        num_modules=2,
        topology=Topology.CUBE,
        local_process_group=dist.new_group([rank]),
        num_nodes=world,
        master_process_group=dist.group.WORLD,
        proc_per_node=1,
    )
    results["gossip_period"] = state.gossip_period
    results["master_worker"] = state.master_worker
    return results


def test_state_validation_multiproc() -> None:
    results = run_distributed(_state_validation, 2)
    for rank, r in enumerate(results):
        assert r["bad_num_modules"]
        assert r["pg_without_nodes"]
        assert r["nodes_without_pg"]
        assert r["gossip_period"] == 1
        assert r["master_worker"] == rank


def _gossip_roundtrip(rank, world):
    # Every rank is its own "node"; pin the topology to the identity
    # ordering so peers are deterministic (the reference pins
    # state.topologies the same way, test_comm_hooks_fsdp.py:492-493).
    state = GossipGraDState(
        num_modules=1,
        topology=Topology.DISSEMINATION,
        local_process_group=dist.new_group([rank]),
        num_nodes=world,
        master_process_group=dist.group.WORLD,
        proc_per_node=1,
    )
    state.topologies = itertools.cycle([list(range(world))])
    state.cur_topology = list(range(world))

    grad = torch.full([4], float(rank))
    gossip_grad_hook(state, grad)

    # power 0: send to rank+1, recv from rank-1; the local all-reduce and
    # broadcast are no-ops for 1-rank nodes (DefaultState world_size uses
    # the local group). grad <- (own + recv)/2.
    expected = (rank + (rank - 1) % world) / 2
    return grad.tolist(), [expected] * 4, state.iter


@pytest.mark.parametrize("world", [2, 4])
def test_gossip_hook_numerics_multiproc(world) -> None:
    results = run_distributed(_gossip_roundtrip, world)
    for got, expected, it in results:
        assert got == pytest.approx(expected)
        assert it == 1


def _gossip_two_gpu_nodes(rank, world):
    # world=4 as two 2-rank "nodes": masters are ranks 0 and 2.
    local, _ = dist.new_subgroups(group_size=2)
    masters = dist.new_group([0, 2])
    state = GossipGraDState(
        num_modules=1,
        topology=Topology.CUBE,
        local_process_group=local,
        num_nodes=2,
        master_process_group=masters,
        proc_per_node=2,
    )
    state.topologies = itertools.cycle([[0, 2]])
    state.cur_topology = [0, 2]

    grad = torch.full([4], float(rank))
    gossip_grad_hook(state, grad)
    # Intra-node all-reduce averages within the node (DefaultState divides
    # by the local world size): node0 -> 0.5, node1 -> 2.5. CUBE gossip
    # between masters averages the two: 1.5 everywhere after broadcast.
    return grad.tolist()


def test_gossip_hierarchical_two_nodes() -> None:
    results = run_distributed(_gossip_two_gpu_nodes, 4)
    for got in results:
        assert got == pytest.approx([1.5] * 4)


def _topology_determinism(rank, world):
    state = GossipGraDState(
        num_modules=1,
        local_process_group=dist.new_group([rank]),
        num_nodes=world,
        master_process_group=dist.group.WORLD,
        proc_per_node=1,
        random_seed=777,
    )
    return [list(next(state.topologies)) for _ in range(4)]


def test_topology_sequence_identical_across_workers() -> None:
    # Every worker must generate the same rotation sequence from the seed,
    # or peers would disagree mid-training.
    results = run_distributed(_topology_determinism, 2)
    assert results[0] == results[1]
