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
