# Tape observability (torchdistx_amd.utils.tape).

from torch.nn import Linear

from torchdistx_amd import deferred_init, materialize_module
from torchdistx_amd.models import TINY, build_model
from torchdistx_amd.utils import describe_module, record_info


def test_record_info_lifecycle() -> None:
    m = deferred_init(Linear, 4, 4)
    info = record_info(m.weight)
    assert info is not None
    assert not info["materialized"]
    assert info["pending_ops"] >= 2  # empty + uniform_ at minimum
    assert info["op_name"]

    materialize_module(m)
    assert record_info(m.weight) is None  # real tensors carry no record


def test_describe_module_counts() -> None:
    m = deferred_init(build_model, TINY)
    d = describe_module(m)
    n_tensors = sum(1 for _ in m.parameters()) + sum(1 for _ in m.buffers())
    assert d["n_recorded_tensors"] == n_tensors
    assert d["n_awaiting_materialization"] == n_tensors
    assert d["total_pending_ops"] >= n_tensors

    materialize_module(m)
    d2 = describe_module(m)
    assert d2["n_awaiting_materialization"] == 0
    assert d2["n_recorded_tensors"] == 0  # records dropped with the fakes


def test_materialization_report() -> None:
    from torchdistx_amd.utils.tape import materialization_report

    m = deferred_init(Linear, 64, 64)
    with materialization_report(m) as rep:
        materialize_module(m)
    assert rep["materialized_tensors"] == 2
    # 64*64*4 + 64*4 bytes
    assert rep["materialized_bytes"] == 64 * 64 * 4 + 64 * 4
    assert rep["wall_s"] > 0 and rep["gbps"] > 0

    # A second report on an already-real module measures nothing.
    with materialization_report(m) as rep2:
        pass
    assert rep2["materialized_tensors"] == 0
