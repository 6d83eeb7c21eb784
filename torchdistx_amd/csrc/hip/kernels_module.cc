// Python module for the CDNA4 kernel extension (torchdistx_amd._K).
// Importing this module also registers the tdx:: init ops with the
// dispatcher (TORCH_LIBRARY in init_kernels.hip), which is what the
// deferred-init replay redirect looks for.

#include <torch/extension.h>

namespace tdx {

void anyprecision_adamw_step(at::Tensor& param,
                             const at::Tensor& grad,
                             at::Tensor& exp_avg,
                             at::Tensor& exp_avg_sq,
                             std::optional<at::Tensor> compensation,
                             double lr,
                             double beta1,
                             double beta2,
                             double eps,
                             double weight_decay,
                             double step_size,
                             double bias_correction2_sqrt);

void batched_init_launch(std::vector<at::Tensor> tensors,
                         std::vector<int64_t> dists,
                         std::vector<double> p0s,
                         std::vector<double> p1s,
                         std::vector<int64_t> seeds,
                         std::vector<int64_t> offsets);

void anyprecision_adamw_batched_step(
    std::vector<at::Tensor> params,
    std::vector<at::Tensor> grads,
    std::vector<at::Tensor> exp_avgs,
    std::vector<at::Tensor> exp_avg_sqs,
    std::vector<std::optional<at::Tensor>> compensations,
    double lr,
    double beta1,
    double beta2,
    double eps,
    double weight_decay,
    std::vector<double> step_sizes,
    std::vector<double> bias_correction2_sqrts);

}  // namespace tdx

PYBIND11_MODULE(_K, m) {
  m.doc() = "torchdistx_amd CDNA4 kernels (gfx950)";

  m.def("has_init_kernels", [] { return true; });
  m.def("has_anyprecision_adamw", [] { return true; });
  m.def("has_anyprecision_adamw_batched", [] { return true; });
  m.def("has_batched_init", [] { return true; });

  m.def("batched_init_", &tdx::batched_init_launch,
        pybind11::arg("tensors"), pybind11::arg("dists"),
        pybind11::arg("p0s"), pybind11::arg("p1s"), pybind11::arg("seeds"),
        pybind11::arg("offsets"),
        "One launch filling every listed tensor with its reduced init "
        "plan (dist: 0 uniform, 1 normal, 2 bernoulli, 3 fill, 4 zero); "
        "bitwise-identical to the per-tensor tdx kernels.");

  m.def("anyprecision_adamw_batched_",
        &tdx::anyprecision_adamw_batched_step, pybind11::arg("params"),
        pybind11::arg("grads"), pybind11::arg("exp_avgs"),
        pybind11::arg("exp_avg_sqs"), pybind11::arg("compensations"),
        pybind11::arg("lr"), pybind11::arg("beta1"), pybind11::arg("beta2"),
        pybind11::arg("eps"), pybind11::arg("weight_decay"),
        pybind11::arg("step_sizes"),
        pybind11::arg("bias_correction2_sqrts"),
        "One-launch AdamW update for a whole list of parameters (kahan "
        "optional per tensor via the compensations list).");

  m.def(
      "anyprecision_adamw_",
      [](at::Tensor param, at::Tensor grad, at::Tensor exp_avg,
         at::Tensor exp_avg_sq, std::optional<at::Tensor> compensation,
         double lr, double beta1, double beta2, double eps,
         double weight_decay, double step_size,
         double bias_correction2_sqrt) {
        tdx::anyprecision_adamw_step(
            param, grad, exp_avg, exp_avg_sq, std::move(compensation), lr,
            beta1, beta2, eps, weight_decay, step_size,
            bias_correction2_sqrt);
      },
      pybind11::arg("param"), pybind11::arg("grad"), pybind11::arg("exp_avg"),
      pybind11::arg("exp_avg_sq"), pybind11::arg("compensation"),
      pybind11::arg("lr"), pybind11::arg("beta1"), pybind11::arg("beta2"),
      pybind11::arg("eps"), pybind11::arg("weight_decay"),
      pybind11::arg("step_size"), pybind11::arg("bias_correction2_sqrt"));
}
