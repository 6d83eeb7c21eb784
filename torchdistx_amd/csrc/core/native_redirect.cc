#include "native_redirect.h"

#include <atomic>
#include <cstdlib>
#include <cstring>

namespace tdx {
namespace {

std::atomic<bool> native_init_enabled{true};
std::atomic<bool> native_init_cpu_enabled{false};

struct Redirect {
  const char* aten_name;
  const char* aten_overload;
  const char* tdx_name;
  bool needs_float_dtype;  // RNG/fill kernels cover f32/bf16/f16 only
};

constexpr Redirect kRedirects[] = {
    {"aten::uniform_", "", "uniform_", true},
    {"aten::normal_", "", "normal_", true},
    {"aten::bernoulli_", "float", "bernoulli_", true},
    {"aten::fill_", "Scalar", "fill_", true},
    {"aten::zero_", "", "zero_", false},
    {"aten::copy_", "", "copy_", true},
};

bool requireNative() {
  static const bool required = std::getenv("TDX_REQUIRE_NATIVE_INIT") != nullptr;
  return required;
}

}  // namespace

void setNativeInitEnabled(bool enabled) noexcept {
  native_init_enabled.store(enabled, std::memory_order_relaxed);
}

bool nativeInitEnabled() noexcept {
  return native_init_enabled.load(std::memory_order_relaxed);
}

void setNativeInitCpuEnabled(bool enabled) noexcept {
  native_init_cpu_enabled.store(enabled, std::memory_order_relaxed);
}

bool nativeInitCpuEnabled() noexcept {
  return native_init_cpu_enabled.load(std::memory_order_relaxed);
}

bool tryNativeInitRedirect(
    const c10::OperatorHandle& op,
    torch::jit::Stack& stack,
    const std::optional<std::pair<uint64_t, uint64_t>>& philox) {
  if (!nativeInitEnabled()) {
    return false;
  }
  const auto& name = op.schema().operator_name();
  const Redirect* redirect = nullptr;
  for (const Redirect& r : kRedirects) {
    if (name.name == r.aten_name && name.overload_name == r.aten_overload) {
      redirect = &r;
      break;
    }
  }
  if (redirect == nullptr) {
    return false;
  }

  // These schemas all take `self` as the first (and mutated) argument; the
  // replay stack holds exactly the op's arguments at this point.
  if (stack.empty() || !stack.front().isTensor()) {
    return false;
  }
  const at::Tensor& self = stack.front().toTensor();
  if (!self.defined() || !self.is_contiguous()) {
    return false;
  }
  if (!self.is_cuda() && !(self.is_cpu() && nativeInitCpuEnabled())) {
    return false;
  }
  if (redirect->needs_float_dtype) {
    auto st = self.scalar_type();
    if (st != at::kFloat && st != at::kBFloat16 && st != at::kHalf) {
      return false;
    }
  }
  if (std::strcmp(redirect->tdx_name, "copy_") == 0) {
    // tdx::copy_ covers the same-device contiguous same-shape case only.
    if (stack.size() < 2 || !stack[1].isTensor()) {
      return false;
    }
    const at::Tensor& src = stack[1].toTensor();
    auto sst = src.scalar_type();
    if (!src.defined() || src.device() != self.device() ||
        !src.is_contiguous() || !src.sizes().equals(self.sizes()) ||
        (sst != at::kFloat && sst != at::kBFloat16 && sst != at::kHalf)) {
      return false;
    }
  }

  auto target = c10::Dispatcher::singleton().findOp(
      {std::string("tdx::") + redirect->tdx_name, ""});
  if (!target.has_value()) {
    TORCH_CHECK(!requireNative(),
                "TDX_REQUIRE_NATIVE_INIT is set but the torchdistx_amd._K "
                "kernel extension is not loaded; refusing to fall back to "
                "stock ATen kernels for `",
                name, "`.");
    return false;
  }
  const bool is_rng = std::strcmp(redirect->tdx_name, "uniform_") == 0 ||
                      std::strcmp(redirect->tdx_name, "normal_") == 0 ||
                      std::strcmp(redirect->tdx_name, "bernoulli_") == 0;
  if (is_rng) {
    // An explicitly passed generator must be honored; only the default
    // generator's stream is replaced by the pinned counter-based one.
    // The generator slot differs per op (uniform_/normal_: arg 3,
    // bernoulli_.float: arg 2), so scan for one.
    for (const c10::IValue& v : stack) {
      if (v.isGenerator()) {
        return false;
      }
    }
    // tdx RNG schemas carry two trailing optional args: the Philox seed
    // and counter offset pinned at record time (partition-invariant init).
    if (philox.has_value()) {
      stack.emplace_back(static_cast<int64_t>(philox->first));
      stack.emplace_back(static_cast<int64_t>(philox->second));
    } else {
      stack.emplace_back();
      stack.emplace_back();
    }
  }
  target->callBoxed(&stack);
  return true;
}

}  // namespace tdx
