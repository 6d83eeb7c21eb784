// Python bindings for the fake-tensor and deferred-init cores.
//
// Capability parity with the reference binding layer
// (/root/reference/src/python/torchdistx/_C/{module,fake,deferred_init}.cc):
// the same eight entry points (enter/leave_deferred_init,
// enter/leave_fake_mode, is_fake, can_materialize, materialize_tensor,
// meta_like), GIL release around materialization, ValueError translation,
// and CUDA lazy-init suppression for the fake-cuda mode. Python-object
// identity across repeated materializations comes from THPVariable_Wrap's
// pyobj-slot reuse (one TensorImpl <-> one Python object); Parameter-class
// preservation is handled in the Python API layer.

#include <torch/extension.h>

#include <torch/csrc/utils/device_lazy_init.h>

#include "deferred_init.h"
#include "fake.h"
#include "native_redirect.h"

namespace {

// Tracks whether we disabled CUDA lazy-init so nested fake modes restore it
// exactly once.
thread_local size_t fake_cuda_lazy_init_suppressions = 0;

void enterFakeModeBinding(bool fake_cuda) {
  if (fake_cuda) {
    // Keep tensor construction from eagerly initializing the (possibly
    // absent) HIP runtime while fake "cuda" tensors are being built.
    torch::utils::set_requires_device_init(at::kCUDA, false);
    ++fake_cuda_lazy_init_suppressions;
  }
  tdx::enterFakeMode(fake_cuda);
}

void leaveFakeModeBinding() {
  tdx::leaveFakeMode();
  if (!tdx::isFakeModeActive() && fake_cuda_lazy_init_suppressions > 0) {
    fake_cuda_lazy_init_suppressions = 0;
    torch::utils::set_requires_device_init(at::kCUDA, true);
  }
}

}  // namespace

PYBIND11_MODULE(_C, m) {
  m.doc() = "torchdistx_amd native core: fake tensors + deferred init";

  pybind11::register_exception_translator([](std::exception_ptr e) {
    try {
      if (e) {
        std::rethrow_exception(e);
      }
    } catch (const c10::ValueError& err) {
      PyErr_SetString(PyExc_ValueError, err.what_without_backtrace());
    }
    // Other exception types fall through to the previously registered
    // translators (pybind runs translators in reverse registration order).
  });

  m.def("enter_fake_mode", &enterFakeModeBinding, pybind11::arg("fake_cuda"),
        "Enter fake-tensor mode on this thread (nestable). With fake_cuda, "
        "fake \"cuda\" tensors can be built without a GPU runtime.");
  m.def("leave_fake_mode", &leaveFakeModeBinding,
        "Leave the innermost fake mode.");

  m.def("is_fake", [](const at::Tensor& t) { return tdx::isFake(t); },
        "Whether the tensor is a storage-less fake tensor.");

  m.def("meta_like", [](const at::Tensor& t) { return tdx::metaLike(t); },
        "Detached meta tensor with the fake tensor's metadata; raises "
        "ValueError for non-fake inputs.");

  m.def("enter_deferred_init", &tdx::enterDeferredInit,
        "Enter deferred-init recording on this thread (nestable).");
  m.def("leave_deferred_init", &tdx::leaveDeferredInit,
        "Leave the innermost deferred-init context.");

  m.def("can_materialize",
        [](const at::Tensor& t) { return tdx::canMaterialize(t); },
        "Whether the tensor is fake AND carries a replayable tape record.");

  m.def("set_native_init", &tdx::setNativeInitEnabled,
        pybind11::arg("enabled"));
  m.def("native_init_enabled", &tdx::nativeInitEnabled);
  m.def("set_native_init_cpu", &tdx::setNativeInitCpuEnabled,
        pybind11::arg("enabled"));
  m.def("native_init_cpu_enabled", &tdx::nativeInitCpuEnabled);

  m.def("record_info", [](const at::Tensor& t) -> pybind11::object {
    auto info = tdx::recordInfo(t);
    if (!info.has_value()) {
      return pybind11::none();
    }
    pybind11::dict d;
    d["op_nr"] = info->op_nr;
    d["output_index"] = info->output_index;
    d["materialized"] = info->materialized;
    d["op_name"] = info->op_name;
    d["pending_ops"] = info->pending_ops;
    return d;
  });

  m.def("materialize_tensor", [](const at::Tensor& t) {
    at::Tensor out;
    {
      pybind11::gil_scoped_release release;
      out = tdx::materializeTensor(t);
    }
    return out;
  }, "Replays the tensor's tape segment (GIL released); identity for real "
     "tensors, stable object across repeated calls and aliases.");

  m.def("tensor_init_plan", [](const at::Tensor& t) -> pybind11::object {
    auto plan = tdx::tensorInitPlan(t);
    if (!plan.has_value()) {
      return pybind11::none();
    }
    static const char* kKinds[] = {"factory", "uniform", "normal",
                                   "bernoulli", "fill", "zero"};
    pybind11::dict d;
    d["kind"] = kKinds[static_cast<int>(plan->kind)];
    d["p0"] = plan->p0;
    d["p1"] = plan->p1;
    d["seed"] = static_cast<int64_t>(plan->seed);
    d["offset"] = static_cast<int64_t>(plan->offset);
    d["sizes"] = plan->sizes;
    d["dtype"] = pybind11::cast(plan->dtype);
    d["device"] = pybind11::cast(plan->device);
    d["requires_grad"] = plan->requires_grad;
    return d;
  }, "Reduced replay plan of a simple init chain (the final whole-tensor "
     "value step), or None when the tape is not that simple. Feeds the "
     "batched replay planner (materialize_module_batched).");

  m.def("materialize_tensor_shard",
        [](const at::Tensor& t, int64_t start_row, int64_t end_row,
           int64_t dim) {
          at::Tensor out;
          {
            pybind11::gil_scoped_release release;
            out = tdx::materializeTensorShard(t, start_row, end_row, dim);
          }
          return out;
        },
        pybind11::arg("tensor"), pybind11::arg("start_row"),
        pybind11::arg("end_row"), pybind11::arg("dim") = 0,
        "Materializes indices [start_row, end_row) of the deferred tensor "
        "along `dim` alone, bitwise-equal to that slice of a full native "
        "materialization (see docs/distributed_materialization.md).");
}
