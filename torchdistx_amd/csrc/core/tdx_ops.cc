// tdx:: op schemas and their CPU reference implementations.
//
// The schemas live here (in the always-loaded core extension) so the CUDA
// implementations in the _K kernel extension can register against them,
// and so the CPU reference paths below are available everywhere — they
// implement the same counter-based value layout as the CDNA4 kernels
// (one Philox4x32-10 call per 16-byte element group, keyed by the group
// index), which is what makes slice materialization testable on GPU-less
// CI. Bitwise determinism is per device type: CPU results match CPU
// results; GPU results match GPU results. (The uniform transform is pure
// integer+fma math and matches across devices in practice; the normal
// transform uses each device's fastest transcendentals.)

#include <cmath>
#include <optional>
#include <type_traits>

#include <ATen/ATen.h>
#include <torch/library.h>

#include "philox.h"

namespace tdx {
namespace {

template <typename T>
struct GroupTraits;
template <>
struct GroupTraits<float> {
  static constexpr int kElems = 4;
};
template <>
struct GroupTraits<at::BFloat16> {
  static constexpr int kElems = 8;
};
template <>
struct GroupTraits<at::Half> {
  static constexpr int kElems = 8;
};

// RNG kinds shared with the CDNA4 kernels.
enum class RngKind { kUniform, kNormal, kBernoulli };

// Fills vals[kElems] for element group `g` — the same mapping the CDNA4
// kernels use (csrc/hip/init_kernels.hip rng_kernel). Normals draw from
// Philox4x32-7, uniforms and bernoulli from Philox4x32-10 (see philox.h).
template <typename T, RngKind kKind>
void groupValues(uint64_t g, float a, float b, uint64_t seed,
                 uint64_t offset, float* vals) {
  constexpr bool kNormal = kKind == RngKind::kNormal;
  philox::U4 bits = kNormal ? philox::philox7(seed, g, offset)
                            : philox::philox10(seed, g, offset);
  if constexpr (GroupTraits<T>::kElems == 4) {
    float u[4] = {philox::u32_to_uniform(bits.x),
                  philox::u32_to_uniform(bits.y),
                  philox::u32_to_uniform(bits.z),
                  philox::u32_to_uniform(bits.w)};
    if constexpr (kKind == RngKind::kUniform) {
      for (int j = 0; j < 4; ++j) {
        vals[j] = std::fmaf(u[j], b, a);
      }
    } else if constexpr (kKind == RngKind::kBernoulli) {
      for (int j = 0; j < 4; ++j) {
        vals[j] = u[j] < a ? 1.0f : 0.0f;
      }
    } else {
      for (int p = 0; p < 2; ++p) {
        float u1 = std::max(u[p * 2], 1.1754944e-38f);
        float r = std::sqrt(-2.0f * std::log(u1));
        float ang = 6.2831853071795865f * u[p * 2 + 1];
        vals[p * 2 + 0] = std::fmaf(r * std::cos(ang), b, a);
        vals[p * 2 + 1] = std::fmaf(r * std::sin(ang), b, a);
      }
    }
  } else {
    uint32_t words[4] = {bits.x, bits.y, bits.z, bits.w};
    for (int j = 0; j < 4; ++j) {
      float lo = philox::u16_to_uniform(words[j]);
      float hi = philox::u16_to_uniform(words[j] >> 16);
      if constexpr (kKind == RngKind::kUniform) {
        vals[j * 2 + 0] = std::fmaf(lo, b, a);
        vals[j * 2 + 1] = std::fmaf(hi, b, a);
      } else if constexpr (kKind == RngKind::kBernoulli) {
        vals[j * 2 + 0] = lo < a ? 1.0f : 0.0f;
        vals[j * 2 + 1] = hi < a ? 1.0f : 0.0f;
      } else {
        float u1 = std::max(lo, 1.1754944e-38f);
        float r = std::sqrt(-2.0f * std::log(u1));
        float ang = 6.2831853071795865f * hi;
        vals[j * 2 + 0] = std::fmaf(r * std::cos(ang), b, a);
        vals[j * 2 + 1] = std::fmaf(r * std::sin(ang), b, a);
      }
    }
  }
}

// Writes elements [start, end) of the virtual full tensor into
// out[0 .. end-start), reproducing the group-indexed counter layout.
template <typename T, RngKind kKind>
void cpuPhiloxRange(T* out, int64_t start, int64_t end, float a, float b,
                    uint64_t seed, uint64_t offset) {
  constexpr int kElems = GroupTraits<T>::kElems;
  const int64_t g_first = start / kElems;
  const int64_t g_last = (end + kElems - 1) / kElems;
  float vals[kElems];
  for (int64_t g = g_first; g < g_last; ++g) {
    groupValues<T, kKind>(static_cast<uint64_t>(g), a, b, seed, offset,
                          vals);
    const int64_t base = g * kElems;
    const int64_t lo = std::max(base, start);
    const int64_t hi = std::min(base + kElems, end);
    for (int64_t e = lo; e < hi; ++e) {
      out[e - start] = static_cast<T>(vals[e - base]);
    }
  }
}

template <RngKind kKind>
void cpuPhiloxDispatch(at::Tensor& self, int64_t start, int64_t end,
                       float a, float b, uint64_t seed, uint64_t offset) {
  TORCH_CHECK(self.is_contiguous(),
              "tdx CPU init requires contiguous tensors");
  TORCH_CHECK(self.numel() == end - start,
              "shard numel must equal end - start");
  switch (self.scalar_type()) {
    case at::kFloat:
      cpuPhiloxRange<float, kKind>(self.data_ptr<float>(), start, end, a,
                                     b, seed, offset);
      break;
    case at::kBFloat16:
      cpuPhiloxRange<at::BFloat16, kKind>(self.data_ptr<at::BFloat16>(),
                                            start, end, a, b, seed, offset);
      break;
    case at::kHalf:
      cpuPhiloxRange<at::Half, kKind>(self.data_ptr<at::Half>(), start,
                                        end, a, b, seed, offset);
      break;
    default:
      TORCH_CHECK(false, "tdx init supports float32/bf16/fp16, got ",
                  self.scalar_type());
  }
}

// ---- CPU impls of the full-tensor ops -------------------------------------

at::Tensor& cpu_uniform_(at::Tensor& self, double from, double to,
                         std::optional<at::Generator> generator,
                         std::optional<int64_t> seed,
                         std::optional<int64_t> offset) {
  if (seed.has_value() && offset.has_value()) {
    cpuPhiloxDispatch<RngKind::kUniform>(self, 0, self.numel(),
                             static_cast<float>(from),
                             static_cast<float>(to - from),
                             static_cast<uint64_t>(*seed),
                             static_cast<uint64_t>(*offset));
    return self;
  }
  return self.uniform_(from, to, std::move(generator));
}

at::Tensor& cpu_normal_(at::Tensor& self, double mean, double std,
                        std::optional<at::Generator> generator,
                        std::optional<int64_t> seed,
                        std::optional<int64_t> offset) {
  TORCH_CHECK(std >= 0.0, "normal_ expects std >= 0.0, but found std=", std);
  if (seed.has_value() && offset.has_value()) {
    cpuPhiloxDispatch<RngKind::kNormal>(self, 0, self.numel(), static_cast<float>(mean),
                            static_cast<float>(std),
                            static_cast<uint64_t>(*seed),
                            static_cast<uint64_t>(*offset));
    return self;
  }
  return self.normal_(mean, std, std::move(generator));
}

at::Tensor& cpu_bernoulli_(at::Tensor& self, double p,
                           std::optional<at::Generator> generator,
                           std::optional<int64_t> seed,
                           std::optional<int64_t> offset) {
  TORCH_CHECK(0.0 <= p && p <= 1.0,
              "bernoulli_ expects 0 <= p <= 1, but found p=", p);
  if (seed.has_value() && offset.has_value()) {
    cpuPhiloxDispatch<RngKind::kBernoulli>(
        self, 0, self.numel(), static_cast<float>(p), 0.0f,
        static_cast<uint64_t>(*seed), static_cast<uint64_t>(*offset));
    return self;
  }
  return self.bernoulli_(p, std::move(generator));
}

at::Tensor& cpu_fill_(at::Tensor& self, const at::Scalar& value) {
  return self.fill_(value);
}

at::Tensor& cpu_zero_(at::Tensor& self) {
  return self.zero_();
}

at::Tensor& cpu_copy_(at::Tensor& self, const at::Tensor& src,
                      bool non_blocking) {
  return self.copy_(src, non_blocking);
}

// ---- shard ops (CPU impls; CUDA impls in csrc/hip/init_kernels.hip) -------

at::Tensor& cpu_uniform_shard_(at::Tensor& shard, int64_t start, int64_t end,
                               double from, double to, int64_t seed,
                               int64_t offset) {
  cpuPhiloxDispatch<RngKind::kUniform>(shard, start, end, static_cast<float>(from),
                           static_cast<float>(to - from),
                           static_cast<uint64_t>(seed),
                           static_cast<uint64_t>(offset));
  return shard;
}

at::Tensor& cpu_normal_shard_(at::Tensor& shard, int64_t start, int64_t end,
                              double mean, double std, int64_t seed,
                              int64_t offset) {
  cpuPhiloxDispatch<RngKind::kNormal>(shard, start, end, static_cast<float>(mean),
                          static_cast<float>(std),
                          static_cast<uint64_t>(seed),
                          static_cast<uint64_t>(offset));
  return shard;
}

at::Tensor& cpu_bernoulli_shard_(at::Tensor& shard, int64_t start,
                                 int64_t end, double p, int64_t seed,
                                 int64_t offset) {
  cpuPhiloxDispatch<RngKind::kBernoulli>(
      shard, start, end, static_cast<float>(p), 0.0f,
      static_cast<uint64_t>(seed), static_cast<uint64_t>(offset));
  return shard;
}

// Windowed shard (any-dim slices): the shard is n_blocks contiguous
// global ranges of block_len elements, block r starting at global
// element g_off + r * g_stride — `full.narrow(dim, s, len)` flattened.
// Same counter layout as the flat range, so each block is bitwise the
// matching stretch of the full tensor (CUDA twin:
// csrc/hip/init_kernels.hip rng_shard_window_kernel).
template <RngKind kKind>
void cpuPhiloxWindow(at::Tensor& shard, int64_t n_blocks, int64_t block_len,
                     int64_t g_stride, int64_t g_off, float a, float b,
                     uint64_t seed, uint64_t offset) {
  TORCH_CHECK(shard.is_contiguous(),
              "tdx CPU init requires contiguous tensors");
  TORCH_CHECK(n_blocks >= 0 && block_len >= 0 && g_stride >= block_len &&
                  g_off >= 0,
              "invalid shard window");
  TORCH_CHECK(shard.numel() == n_blocks * block_len,
              "shard numel must equal n_blocks * block_len");
  auto run = [&](auto* out) {
    using T = std::remove_pointer_t<decltype(out)>;
    for (int64_t r = 0; r < n_blocks; ++r) {
      const int64_t lo = g_off + r * g_stride;
      cpuPhiloxRange<T, kKind>(out + r * block_len, lo, lo + block_len, a,
                               b, seed, offset);
    }
  };
  switch (shard.scalar_type()) {
    case at::kFloat:
      run(shard.data_ptr<float>());
      break;
    case at::kBFloat16:
      run(shard.data_ptr<at::BFloat16>());
      break;
    case at::kHalf:
      run(shard.data_ptr<at::Half>());
      break;
    default:
      TORCH_CHECK(false, "tdx CPU init supports float32/bf16/fp16, got ",
                  shard.scalar_type());
  }
}

at::Tensor& cpu_uniform_shard_win_(at::Tensor& shard, int64_t n_blocks,
                                   int64_t block_len, int64_t g_stride,
                                   int64_t g_off, double from, double to,
                                   int64_t seed, int64_t offset) {
  cpuPhiloxWindow<RngKind::kUniform>(
      shard, n_blocks, block_len, g_stride, g_off, static_cast<float>(from),
      static_cast<float>(to - from), static_cast<uint64_t>(seed),
      static_cast<uint64_t>(offset));
  return shard;
}

at::Tensor& cpu_normal_shard_win_(at::Tensor& shard, int64_t n_blocks,
                                  int64_t block_len, int64_t g_stride,
                                  int64_t g_off, double mean, double std,
                                  int64_t seed, int64_t offset) {
  cpuPhiloxWindow<RngKind::kNormal>(
      shard, n_blocks, block_len, g_stride, g_off, static_cast<float>(mean),
      static_cast<float>(std), static_cast<uint64_t>(seed),
      static_cast<uint64_t>(offset));
  return shard;
}

at::Tensor& cpu_bernoulli_shard_win_(at::Tensor& shard, int64_t n_blocks,
                                     int64_t block_len, int64_t g_stride,
                                     int64_t g_off, double p, int64_t seed,
                                     int64_t offset) {
  cpuPhiloxWindow<RngKind::kBernoulli>(
      shard, n_blocks, block_len, g_stride, g_off, static_cast<float>(p),
      0.0f, static_cast<uint64_t>(seed), static_cast<uint64_t>(offset));
  return shard;
}

TORCH_LIBRARY(tdx, m) {
  m.def(
      "uniform_(Tensor(a!) self, float from=0., float to=1., *, "
      "Generator? generator=None, int? seed=None, int? offset=None) "
      "-> Tensor(a!)");
  m.def(
      "normal_(Tensor(a!) self, float mean=0., float std=1., *, "
      "Generator? generator=None, int? seed=None, int? offset=None) "
      "-> Tensor(a!)");
  m.def(
      "bernoulli_(Tensor(a!) self, float p=0.5, *, "
      "Generator? generator=None, int? seed=None, int? offset=None) "
      "-> Tensor(a!)");
  m.def("fill_(Tensor(a!) self, Scalar value) -> Tensor(a!)");
  m.def("zero_(Tensor(a!) self) -> Tensor(a!)");
  m.def(
      "copy_(Tensor(a!) self, Tensor src, bool non_blocking=False) "
      "-> Tensor(a!)");
  m.def(
      "uniform_shard_(Tensor(a!) shard, int start, int end, float from=0., "
      "float to=1., *, int seed, int offset) -> Tensor(a!)");
  m.def(
      "normal_shard_(Tensor(a!) shard, int start, int end, float mean=0., "
      "float std=1., *, int seed, int offset) -> Tensor(a!)");
  m.def(
      "bernoulli_shard_(Tensor(a!) shard, int start, int end, float p=0.5, "
      "*, int seed, int offset) -> Tensor(a!)");
  m.def(
      "uniform_shard_win_(Tensor(a!) shard, int n_blocks, int block_len, "
      "int g_stride, int g_off, float from=0., float to=1., *, int seed, "
      "int offset) -> Tensor(a!)");
  m.def(
      "normal_shard_win_(Tensor(a!) shard, int n_blocks, int block_len, "
      "int g_stride, int g_off, float mean=0., float std=1., *, int seed, "
      "int offset) -> Tensor(a!)");
  m.def(
      "bernoulli_shard_win_(Tensor(a!) shard, int n_blocks, int block_len, "
      "int g_stride, int g_off, float p=0.5, *, int seed, int offset) "
      "-> Tensor(a!)");
}

TORCH_LIBRARY_IMPL(tdx, CPU, m) {
  m.impl("uniform_", cpu_uniform_);
  m.impl("normal_", cpu_normal_);
  m.impl("fill_", cpu_fill_);
  m.impl("zero_", cpu_zero_);
  m.impl("copy_", cpu_copy_);
  m.impl("uniform_shard_", cpu_uniform_shard_);
  m.impl("normal_shard_", cpu_normal_shard_);
  m.impl("bernoulli_", cpu_bernoulli_);
  m.impl("bernoulli_shard_", cpu_bernoulli_shard_);
  m.impl("uniform_shard_win_", cpu_uniform_shard_win_);
  m.impl("normal_shard_win_", cpu_normal_shard_win_);
  m.impl("bernoulli_shard_win_", cpu_bernoulli_shard_win_);
}

}  // namespace
}  // namespace tdx
