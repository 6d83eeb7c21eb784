# Tape introspection for deferred modules: observability the reference does
# not offer (SURVEY.md section 5 lists its observability as "None"). The
# deferred-init tape is itself a lightweight tracer; these helpers surface
# what it recorded.

from typing import Dict, Optional

import torch
from torch.nn import Module

from torchdistx_amd import _C


def record_info(tensor: torch.Tensor) -> Optional[Dict]:
    """Record metadata for a deferred tensor: the producing op's name and
    tape position, whether it already materialized, and how many recorded
    ops a materialize call would replay right now. ``None`` for real (or
    record-less) tensors."""
    return _C.record_info(tensor)


def describe_module(module: Module) -> Dict:
    """Summary of a deferred module's tape: per-tensor record info plus
    totals. Cheap enough for logging before a materialization decision."""
    tensors = {}
    total_pending = 0
    n_deferred = 0
    pending_bytes = 0
    for name, t in list(module.named_parameters()) + list(module.named_buffers()):
        info = _C.record_info(t)
        if info is None:
            continue
        info = dict(info)
        info["nbytes"] = t.numel() * t.element_size()
        tensors[name] = info
        if not info["materialized"]:
            n_deferred += 1
            pending_bytes += info["nbytes"]
        total_pending += info["pending_ops"]
    return {
        "n_recorded_tensors": len(tensors),
        "n_awaiting_materialization": n_deferred,
        "total_pending_ops": total_pending,
        "pending_bytes": pending_bytes,
        "tensors": tensors,
    }


class materialization_report:
    """Context manager timing a materialization and reporting throughput::

        with materialization_report(module) as rep:
            materialize_module(module)
        log.info("init: %(gb).1f GB in %(wall_s).3f s (%(gbps).2f GB/s)", rep)

    On entry it snapshots how many tensors/bytes are still deferred; on
    exit (after a device sync when the module lives on GPU) it fills
    ``wall_s``, ``materialized_tensors``, ``materialized_bytes``, ``gb``
    and ``gbps``. Zero overhead inside the timed region."""

    def __init__(self, module: Module):
        self._module = module

    def __enter__(self) -> Dict:
        import time

        before = describe_module(self._module)
        self._pending = {
            name: info["nbytes"]
            for name, info in before["tensors"].items()
            if not info["materialized"]
        }
        self._device_sync = any(
            t.is_cuda
            for t in list(self._module.parameters())
            + list(self._module.buffers())
        )
        self.report: Dict = {}
        self._t0 = time.perf_counter()
        return self.report

    def __exit__(self, exc_type, exc, tb) -> None:
        import time

        if exc_type is not None:
            return
        if self._device_sync:
            torch.cuda.synchronize()
        wall = time.perf_counter() - self._t0
        after = describe_module(self._module)
        # A tensor counts as materialized if it either left the record
        # set (swapped into the module, the common case) or its record
        # flipped to materialized.
        still_deferred = {
            name
            for name, info in after["tensors"].items()
            if not info["materialized"]
        }
        done = [n for n in self._pending if n not in still_deferred]
        done_bytes = sum(self._pending[n] for n in done)
        self.report.update(
            wall_s=wall,
            materialized_tensors=len(done),
            materialized_bytes=done_bytes,
            gb=done_bytes / 1e9,
            gbps=(done_bytes / 1e9 / wall) if wall > 0 else float("inf"),
        )
