# Distributed materialization of deferred modules over RCCL/xGMI.
#
# This is the capability BASELINE.json adds on top of the reference (which
# only materializes process-locally): turning one recorded deferred-init
# tape into initialized parameters across the 8 GPUs of an MI355X node.
#
# Three modes, chosen for the xGMI fabric (7 point-to-point links of
# ~153 GB/s per GPU vs ~6 TB/s of local HBM write bandwidth):
#
#   * "replicate" — every rank replays the full tape locally. Because the
#     tape replays the same Philox streams from the same seed, all ranks
#     produce bitwise-identical replicas with ZERO communication. On this
#     fabric local regeneration beats any broadcast: moving 7/8 of a 140 GB
#     model over xGMI costs ~40x more time than writing it locally. This is
#     the MI355X-native default for DDP-style replication.
#
#   * "shard" — tensors are assigned to ranks (greedy size-balanced,
#     deterministic); each rank materializes only the tape subgraphs of its
#     own tensors (the per-tensor call-stack replay makes the untouched
#     subgraphs free). Non-owned entries stay fake. This is the FSDP-style
#     init path: N ranks do 1/N of the HBM writes each, no communication.
#
#   * "broadcast" — shard ownership as above, but every rank ends with the
#     full module: owners materialize their tensors while non-owners
#     allocate, and each tensor is broadcast from its owner with async
#     collectives so the RCCL transfers overlap with the init kernels of
#     the tensors still being materialized. Use when ranks cannot be
#     trusted to share RNG state (e.g. mixed seeds) but must end
#     bitwise-identical.

from typing import Callable, Dict, List, Optional, Tuple

import torch
import torch.distributed as dist
from torch.nn import Module

from torchdistx_amd import _C
from torchdistx_amd.deferred_init import _restore_class


def _named_deferred_tensors(
    module: Module, buffers_only: bool
) -> List[Tuple[Module, str, torch.Tensor, bool]]:
    """(owner module, attribute key, tensor, is_param) for every deferred
    parameter/buffer, in deterministic module-traversal order."""
    out = []
    for submodule in module.modules():
        if not buffers_only:
            for key, p in submodule._parameters.items():
                if p is not None and _C.can_materialize(p):
                    out.append((submodule, key, p, True))
        for key, b in submodule._buffers.items():
            if b is not None and _C.can_materialize(b):
                out.append((submodule, key, b, False))
    return out


def _dedupe_entries(entries):
    """Tied parameters surface as several (module, key) slots holding ONE
    fake object; distributed materialization must treat them as one
    tensor (one owner, one transfer, one result object). Returns the
    unique entries, the duplicate slots, and the id->unique-entry map."""
    unique, dups, first = [], [], {}
    for e in entries:
        tid = id(e[2])
        if tid in first:
            dups.append(e)
        else:
            first[tid] = e
            unique.append(e)
    return unique, dups, first


def _relink_duplicates(dups, first) -> None:
    """Points every duplicate slot at whatever its unique counterpart's
    slot now holds (if that slot was materialized on this rank)."""
    from torchdistx_amd import _C as _core

    for submodule, key, tensor, is_param in dups:
        u_sub, u_key, _, u_is_param = first[id(tensor)]
        val = (
            u_sub._parameters[u_key]
            if u_is_param
            else u_sub._buffers[u_key]
        )
        if val is None or _core.can_materialize(val):
            continue  # unique slot not materialized on this rank (shard)
        if is_param:
            submodule._parameters[key] = val
        else:
            submodule._buffers[key] = val


def assign_owners(sizes: List[int], world_size: int) -> List[int]:
    """Greedy size-balanced deterministic assignment: largest tensors
    first, each to the currently least-loaded rank."""
    loads = [0] * world_size
    owners = [0] * len(sizes)
    order = sorted(range(len(sizes)), key=lambda i: (-sizes[i], i))
    for i in order:
        rank = min(range(world_size), key=lambda r: (loads[r], r))
        owners[i] = rank
        loads[rank] += sizes[i]
    return owners


def materialize_module_distributed(
    module: Module,
    mode: str = "replicate",
    process_group: Optional[dist.ProcessGroup] = None,
    buffers_only: bool = False,
    check_fn: Optional[Callable[[Module], bool]] = None,
) -> Dict[str, int]:
    """Materializes a deferred ``module`` across the ranks of
    ``process_group`` (default: the world). Returns {tensor index -> owner
    rank} for the shard/broadcast modes ({} for replicate).

    See the module docstring for the mode semantics.
    """
    if mode not in ("replicate", "shard", "broadcast", "allgather"):
        raise ValueError(f"unknown materialization mode: {mode!r}")

    if mode == "allgather" and dist.is_initialized():
        return _materialize_allgather(
            module, process_group, buffers_only, check_fn
        )

    if mode == "replicate" or not dist.is_initialized():
        from torchdistx_amd.deferred_init import materialize_module

        materialize_module(module, buffers_only=buffers_only, check_fn=check_fn)
        return {}

    group = process_group or dist.group.WORLD
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)

    entries = _named_deferred_tensors(module, buffers_only)
    if check_fn is not None:
        entries = [e for e in entries if check_fn(e[0])]
    entries, dups, first = _dedupe_entries(entries)
    owners = assign_owners([t.numel() for _, _, t, _ in entries], world)

    if mode == "shard":
        from torchdistx_amd.deferred_init import _batched_fill

        owned = [e for i, e in enumerate(entries) if owners[i] == rank]
        # Simple-chain owned tensors fill in one batched launch; the
        # rest replay through the tape.
        for submodule, key, tensor, is_param in _batched_fill(owned):
            mat = _C.materialize_tensor(tensor)
            if is_param:
                mat = _restore_class(tensor, mat)
                submodule._parameters[key] = mat
            else:
                submodule._buffers[key] = mat
        _relink_duplicates(dups, first)
        return {i: owners[i] for i in range(len(entries))}

    _broadcast_bucketed(entries, owners, group, rank)
    _relink_duplicates(dups, first)
    return {i: owners[i] for i in range(len(entries))}


# Bucket size for the broadcast pipeline. At ~153 GB/s per xGMI link a
# 128 MiB bucket is ~0.9 ms on the wire — deep enough to amortize
# collective launch overhead, fine-grained enough that the first transfer
# starts while almost all init kernels are still pending.
_BUCKET_BYTES = 128 << 20

# Double-buffered side streams for the bucket pipeline (created once).
_bucket_streams: List["torch.cuda.Stream"] = []


def _broadcast_bucketed(entries, owners, group, rank) -> None:
    """Owner-materialize + broadcast, pipelined: entries are packed into
    per-(owner, dtype) flat buckets; each bucket's init kernels and pack
    copies run on a side HIP stream, its RCCL broadcast is issued from
    that stream (the collective's communication stream syncs on it via
    event), and the next bucket's kernels start immediately on the other
    stream of a double-buffered pool — so transfers over xGMI overlap
    with the init kernels of everything still being materialized. Buckets
    are interleaved round-robin across owners so every rank's transfers
    engage from the start instead of serializing owner by owner."""
    # ---- plan buckets -----------------------------------------------------
    per_owner: Dict[Tuple[int, torch.dtype], List[int]] = {}
    for i, (_, _, tensor, _) in enumerate(entries):
        per_owner.setdefault((owners[i], tensor.dtype), []).append(i)

    buckets: List[Tuple[int, torch.dtype, List[int]]] = []  # (owner, dtype, idxs)
    rounds: Dict[int, int] = {}  # owner -> buckets emitted (for interleave)
    order: List[Tuple[int, int, int]] = []  # (round, owner, bucket idx)
    for (owner, dtype), idxs in sorted(
        per_owner.items(), key=lambda kv: (kv[0][0], str(kv[0][1]))
    ):
        cur: List[int] = []
        cur_bytes = 0
        for i in idxs:
            t = entries[i][2]
            nbytes = t.numel() * t.element_size()
            if cur and cur_bytes + nbytes > _BUCKET_BYTES:
                order.append((rounds.get(owner, 0), owner, len(buckets)))
                rounds[owner] = rounds.get(owner, 0) + 1
                buckets.append((owner, dtype, cur))
                cur, cur_bytes = [], 0
            cur.append(i)
            cur_bytes += nbytes
        if cur:
            order.append((rounds.get(owner, 0), owner, len(buckets)))
            rounds[owner] = rounds.get(owner, 0) + 1
            buckets.append((owner, dtype, cur))
    order.sort()

    use_streams = (
        torch.cuda.is_available()
        and entries
        and entries[0][2].is_cuda
    )
    # Stable process-lifetime streams: the caching allocator tags blocks
    # with their allocation stream, so fresh Stream objects per call
    # would orphan every cached bucket buffer and churn hipMalloc/Free
    # on each step (measured ~90x slowdown for the analogous mistake in
    # materialize_module_parallel).
    global _bucket_streams
    if use_streams and not _bucket_streams:
        _bucket_streams = [torch.cuda.Stream(), torch.cuda.Stream()]
    streams = _bucket_streams if use_streams else None
    from contextlib import nullcontext

    # ---- pipeline ---------------------------------------------------------
    pending = []  # (handle, stream) to drain at the end
    # Tied parameters (one fake object in several slots) must swap to ONE
    # materialized tensor on every rank; owners get this from
    # materialize_tensor's identity stability, receivers from this map.
    seen: Dict[int, torch.Tensor] = {}
    for k, (_, _, b) in enumerate(order):
        owner, dtype, idxs = buckets[b]
        stream = streams[k % 2] if use_streams else None
        ctx = torch.cuda.stream(stream) if use_streams else nullcontext()
        with ctx, torch.no_grad():
            numels = [entries[i][2].numel() for i in idxs]
            device = entries[idxs[0]][2].device
            flat = torch.empty(sum(numels), dtype=dtype, device=device)
            offsets = []
            off = 0
            for n in numels:
                offsets.append(off)
                off += n

            if rank == owner:
                # Init kernels + pack copies, all on this bucket's stream.
                for i, off in zip(idxs, offsets):
                    submodule, key, tensor, is_param = entries[i]
                    mat = _C.materialize_tensor(tensor)
                    flat[off : off + tensor.numel()].copy_(
                        mat.detach().view(-1)
                    )
                    _swap_entry(entries[i], mat)
            src = dist.get_global_rank(group, owner)
            # Issued from the bucket stream: RCCL's communication stream
            # waits on it (event sync inside ProcessGroupNCCL), so the
            # wire transfer starts exactly when this bucket's kernels are
            # done — while later buckets' kernels keep running.
            handle = dist.broadcast(flat, src=src, group=group, async_op=True)
            if rank != owner:
                # Unpack on the same stream, ordered after the transfer.
                if handle is not None:
                    handle.wait()
                for i, off in zip(idxs, offsets):
                    submodule, key, tensor, is_param = entries[i]
                    mat = seen.get(id(tensor))
                    if mat is None:
                        mat = torch.empty(
                            tensor.shape, dtype=tensor.dtype,
                            device=tensor.device,
                        )
                        mat.view(-1).copy_(flat[off : off + tensor.numel()])
                        seen[id(tensor)] = mat
                    _swap_entry(entries[i], mat)
                handle = None
        if handle is not None:
            pending.append(handle)

    for h in pending:
        h.wait()
    if use_streams:
        cur = torch.cuda.current_stream()
        for s in streams:
            cur.wait_stream(s)


def _swap_entry(entry, mat: torch.Tensor) -> None:
    submodule, key, tensor, is_param = entry
    if is_param:
        if mat.is_leaf and mat.requires_grad != tensor.requires_grad:
            mat.requires_grad_(tensor.requires_grad)
        mat = _restore_class(tensor, mat)
        submodule._parameters[key] = mat
    else:
        submodule._buffers[key] = mat


def materialize_experts_sharded(
    module: Module,
    expert_fqn_fragment: str = "experts",
    process_group: Optional[dist.ProcessGroup] = None,
) -> Dict[str, int]:
    """Expert-sharded materialization for MoE models (the Mixtral-style
    layout in torchdistx_amd.models): expert submodules are assigned
    round-robin to ranks and only the owner materializes them (expert
    parallelism — each rank holds its experts and nothing else), while all
    non-expert ("shared") parameters are materialized on every rank
    bitwise-identically via the partition-invariant Philox tape. No
    communication at all: ownership is decided by deterministic rank
    arithmetic, values by the pinned counters.

    Returns {expert module fqn -> owner rank}.
    """
    group = process_group or (dist.group.WORLD if dist.is_initialized() else None)
    world = dist.get_world_size(group) if group is not None else 1
    rank = dist.get_rank(group) if group is not None else 0

    # Deterministic expert enumeration by fully-qualified name.
    expert_fqns = [
        name
        for name, sub in module.named_modules()
        if f".{expert_fqn_fragment}." in f".{name}."
        and "." not in name.split(f"{expert_fqn_fragment}.")[-1]
    ]
    owners = {fqn: i % world for i, fqn in enumerate(expert_fqns)}

    all_modules = dict(module.named_modules())
    skip_ids = set()
    for fqn, owner in owners.items():
        if owner != rank:
            skip_ids.update(id(m) for m in all_modules[fqn].modules())

    from torchdistx_amd.deferred_init import materialize_module

    materialize_module(module, check_fn=lambda sub: id(sub) not in skip_ids)
    return owners


def _materialize_allgather(
    module: Module,
    process_group: Optional[dist.ProcessGroup],
    buffers_only: bool,
    check_fn: Optional[Callable[[Module], bool]],
) -> Dict[str, int]:
    """Every rank slice-materializes 1/world of each tensor's rows, then an
    async all-gather reconstructs full bitwise-identical replicas on every
    rank. Exists for completeness and comparison: on xGMI, `replicate`
    (local regeneration at HBM write speed) beats moving weights over the
    ~153 GB/s links — measure both and see. Init work per rank is 1/world;
    the collectives overlap with the remaining slice kernels."""
    group = process_group or dist.group.WORLD
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)

    entries = _named_deferred_tensors(module, buffers_only)
    if check_fn is not None:
        entries = [e for e in entries if check_fn(e[0])]
    entries, dups, first = _dedupe_entries(entries)

    handles = []
    for submodule, key, tensor, is_param in entries:
        rows = tensor.shape[0] if tensor.dim() > 0 else 1
        # Equal-size slots so all_gather_into_tensor applies; the last
        # slots may be partly past the end and are narrowed away below.
        slot = -(-rows // world)  # ceil
        start = min(rank * slot, rows)
        end = min(start + slot, rows)
        shard = materialize_tensor_shard(tensor, start, end)
        slot_shape = (slot,) + tuple(tensor.shape[1:])
        padded = torch.zeros(slot_shape, dtype=tensor.dtype,
                             device=tensor.device)
        if end > start:
            padded[: end - start] = shard.detach()
        full_padded = torch.empty((slot * world,) + tuple(tensor.shape[1:]),
                                  dtype=tensor.dtype, device=tensor.device)
        handles.append(
            dist.all_gather_into_tensor(
                full_padded, padded, group=group, async_op=True
            )
        )
        mat = full_padded.narrow(0, 0, rows).view(tensor.shape)
        if is_param:
            mat.requires_grad_(tensor.requires_grad)
            mat = _restore_class(tensor, mat)
            submodule._parameters[key] = mat
        else:
            submodule._buffers[key] = mat

    for h in handles:
        h.wait()
    _relink_duplicates(dups, first)
    return {}


def materialize_tensor_shard(
    tensor: torch.Tensor, start_row: int, end_row: int, dim: int = 0
) -> torch.Tensor:
    """Materializes indices [start_row, end_row) of a deferred tensor
    along ``dim`` without touching the rest — bitwise-equal to the
    corresponding slice of a full native materialization, at shard cost.
    dim 0 is the FSDP ``Shard(0)`` / column-parallel split; dim 1 the
    row-parallel (Megatron-style TP) split. See csrc/core/deferred_init.h
    (materializeTensorShard) for the supported tape shapes; unsupported
    tapes raise, so callers can fall back to `materialize_tensor` +
    slicing."""
    if dim < 0:
        dim += max(tensor.dim(), 1)
    try:
        shard = _C.materialize_tensor_shard(tensor, start_row, end_row, dim)
    except RuntimeError as e:
        if "slice materialization" not in str(e):
            raise
        # Not a simple init chain (e.g. a computed buffer like a RoPE
        # cache): fall back to full materialization and slice. Costs the
        # full tensor once; fine for the small computed buffers this
        # covers, and cross-rank consistency still holds (every rank
        # replays the same tape).
        full = _C.materialize_tensor(tensor)
        if full.dim() > 0:
            shard = full.narrow(dim, start_row, end_row - start_row).clone()
        else:
            shard = full
    if tensor.requires_grad:
        shard.requires_grad_(True)
    return shard


def materialize_module_dim0_sharded(
    module: Module,
    rank: Optional[int] = None,
    world_size: Optional[int] = None,
) -> Dict[str, torch.Tensor]:
    """FSDP/TP-style init: every parameter and buffer of the deferred
    ``module`` is split contiguously along dim 0 across ``world_size``
    ranks and only this rank's slice is materialized. Zero communication;
    slices are bitwise-consistent with a full materialization through the
    native kernels, so concatenating all ranks' results reconstructs the
    exact full model. N ranks can therefore initialize a model larger than
    any single device (e.g. Llama-3-405B, 812 GB bf16, across 8 GPUs).

    Returns {fully-qualified tensor name -> local shard}. The module's own
    entries are left fake (they describe the full tensors).
    """
    if rank is None:
        rank = dist.get_rank() if dist.is_initialized() else 0
    if world_size is None:
        world_size = dist.get_world_size() if dist.is_initialized() else 1

    shards: Dict[str, torch.Tensor] = {}
    for name, t in list(module.named_parameters()) + list(
        module.named_buffers()
    ):
        if not _C.can_materialize(t):
            continue
        rows = t.shape[0] if t.dim() > 0 else 1
        start = rank * rows // world_size
        end = (rank + 1) * rows // world_size
        shards[name] = materialize_tensor_shard(t, start, end)
    return shards


def materialize_module_tp_sharded(
    module: Module,
    shard_dims: Dict[str, int],
    rank: Optional[int] = None,
    world_size: Optional[int] = None,
) -> Dict[str, torch.Tensor]:
    """Tensor-parallel init: each named parameter/buffer listed in
    ``shard_dims`` (fully-qualified name -> split dim) is split
    contiguously along that dim across ``world_size`` ranks and only this
    rank's slice is materialized; names NOT listed are fully materialized
    (bitwise-replicated on every rank, e.g. row-parallel biases and
    norms). Zero communication. With the Megatron convention on
    ``nn.Linear`` weights ([out, in]): column-parallel shards dim 0,
    row-parallel shards dim 1.

    Returns {fully-qualified tensor name -> local shard or full tensor}.
    The module's own entries are left fake (they describe the full
    tensors).
    """
    if rank is None:
        rank = dist.get_rank() if dist.is_initialized() else 0
    if world_size is None:
        world_size = dist.get_world_size() if dist.is_initialized() else 1

    named = list(module.named_parameters()) + list(module.named_buffers())
    names = {name for name, _ in named}
    unknown = set(shard_dims) - names
    if unknown:
        raise ValueError(
            f"shard_dims names not found in the module: {sorted(unknown)}"
        )
    out: Dict[str, torch.Tensor] = {}
    for name, t in named:
        if not _C.can_materialize(t):
            continue
        dim = shard_dims.get(name)
        if dim is None:
            out[name] = _C.materialize_tensor(t)
            continue
        if not -max(t.dim(), 1) <= dim < max(t.dim(), 1):
            raise ValueError(
                f"shard_dims[{name!r}] = {dim} is out of range for a "
                f"{t.dim()}-d tensor"
            )
        n = t.shape[dim] if t.dim() > 0 else 1
        start = rank * n // world_size
        end = (rank + 1) * n // world_size
        out[name] = materialize_tensor_shard(t, start, end, dim)
    return out


def materialize_module_dtensor(
    module: Module,
    device_mesh,
    shard_dims: Optional[Dict[str, int]] = None,
) -> Dict[str, "torch.Tensor"]:
    """FSDP2-era init: every parameter/buffer of the deferred ``module``
    materializes directly as a ``DTensor`` over the given 1-D device
    mesh — each rank slice-materializes only its local chunk
    (torch.chunk split semantics, matching ``distribute_tensor``), so
    the full model never exists on any device. By default every tensor
    shards on dim 0 (the FSDP2 ``fully_shard`` layout); ``shard_dims``
    overrides the split dim per fully-qualified name (e.g. dim 1 for
    row-parallel TP weights), with ``None`` meaning ``Replicate()`` —
    the rank then fully materializes that tensor (bitwise-identical
    everywhere, zero communication). Returns {fqn -> DTensor}."""
    from torch.distributed.tensor import DTensor, Replicate, Shard

    if device_mesh.ndim != 1:
        raise ValueError("materialize_module_dtensor expects a 1-D mesh")
    world = device_mesh.size()
    rank = device_mesh.get_local_rank()
    shard_dims = shard_dims or {}

    out: Dict[str, torch.Tensor] = {}
    for name, t in list(module.named_parameters()) + list(
        module.named_buffers()
    ):
        if not _C.can_materialize(t):
            continue
        dim = shard_dims.get(name, 0)
        if dim is None:
            full = _C.materialize_tensor(t)
            out[name] = DTensor.from_local(
                full, device_mesh, [Replicate()], run_check=False,
                shape=t.shape, stride=t.stride(),
            )
            continue
        if dim < 0:
            dim += max(t.dim(), 1)
        n = t.shape[dim] if t.dim() > 0 else 1
        slot = -(-n // world)  # torch.chunk: ceil-size slots, short tail
        start = min(rank * slot, n)
        end = min(start + slot, n)
        local = materialize_tensor_shard(t, start, end, dim)
        out[name] = DTensor.from_local(
            local, device_mesh, [Shard(dim)], run_check=False,
            shape=t.shape, stride=t.stride(),
        )
    return out
