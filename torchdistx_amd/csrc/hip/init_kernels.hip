// CDNA4 (gfx950) parameter-init kernels for tape replay into HBM3E.
//
// These are the hand-written HIP kernels behind the ops the deferred-init
// tape records for standard module construction:
// uniform_ / normal_ (counter-based Philox4x32: 10 rounds for uniforms,
// 7 for the VALU-bound normals — see philoxN below), fill_ and zero_.
// They register as the `tdx::` op namespace; the deferred-init replay
// engine redirects the recorded aten:: init ops to them when the target
// tensor lives on the GPU (csrc/core/native_redirect.cc).
//
// Design (per the CDNA4 kernel playbook for memory-bound elementwise ops):
//  * 256-thread blocks (4 wave64s), grid-stride loops capped at 16384
//    blocks (measured sweet spot for this ALU-heavy streaming shape;
//    see kMaxBlocks below);
//  * every store is 16 bytes per lane (float4 / 8 x bf16) — 1 KiB per wave
//    per instruction, the HBM coalescing sweet spot; scalar tail only for
//    the last partial group;
//  * counter-based Philox keyed on (seed, element-group) so any slice of a
//    tensor can be generated independently and reproducibly — this is what
//    makes sharded materialization embarrassingly parallel across ranks;
//  * no LDS: the kernels are pure streaming writes, so tiles would add
//    nothing (LDS is for reuse, of which there is none).

#include <hip/hip_runtime.h>

#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>
#include <ATen/hip/HIPGeneratorImpl.h>
#include <torch/library.h>

#include <limits>

#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

namespace tdx {
namespace {

constexpr int kBlock = 256;
// Measured on gfx950 (scripts/rng_tune.hip): this ALU-heavy streaming
// kernel keeps improving past the usual ~2048-block guideline — at 16384
// blocks the one-philox-per-store uniform kernel reaches ~5.5 TB/s on a
// 4 GiB bf16 fill (pure-store ceiling: 6.4 TB/s via hipMemset; the
// Philox-7 + log2-Box-Muller normals are VALU-bound at ~3.5 TB/s once
// clocks are warm — profiles/dvfs_ramp_note.md).
constexpr int kMaxBlocks = 16384;

// ---------------------------------------------------------------------------
// Philox4x32-10 (standard constants), producing 4 x uint32 per invocation.
// Kept in sync with the CPU reference in csrc/core/philox.h (identical
// constants and integer pipeline).
// ---------------------------------------------------------------------------

__device__ __forceinline__ uint2 mulhilo32(uint32_t a, uint32_t b) {
  uint2 r;
  r.x = a * b;
  r.y = __umulhi(a, b);
  return r;
}

// Round-count-templated Philox4x32, kept in sync with the CPU reference
// in csrc/core/philox.h. Uniforms use the standard 10 rounds (the kernel
// is store-bound, extra rounds are free); normals use 7 — Philox4x32-7
// passes the full BigCrush battery (Salmon et al., SC'11, Table 2) and
// the transcendental-heavy normal kernel is VALU-bound, where the saved
// rounds are a measured ~13% fill-rate win (profiles/rng_tune_r7.log).
template <int kRounds>
__device__ __forceinline__ uint4 philoxN(uint64_t seed,
                                         uint64_t subsequence,
                                         uint64_t offset) {
  constexpr uint32_t kW0 = 0x9E3779B9u;
  constexpr uint32_t kW1 = 0xBB67AE85u;
  constexpr uint32_t kM0 = 0xD2511F53u;
  constexpr uint32_t kM1 = 0xCD9E8D57u;

  uint32_t k0 = static_cast<uint32_t>(seed);
  uint32_t k1 = static_cast<uint32_t>(seed >> 32);
  uint4 c = make_uint4(static_cast<uint32_t>(offset),
                       static_cast<uint32_t>(offset >> 32),
                       static_cast<uint32_t>(subsequence),
                       static_cast<uint32_t>(subsequence >> 32));
#pragma unroll
  for (int round = 0; round < kRounds; ++round) {
    uint2 r0 = mulhilo32(kM0, c.x);
    uint2 r1 = mulhilo32(kM1, c.z);
    c = make_uint4(r1.y ^ c.y ^ k0, r1.x, r0.y ^ c.w ^ k1, r0.x);
    k0 += kW0;
    k1 += kW1;
  }
  return c;
}

// uint32 -> [0, 1) float with a 24-bit mantissa (matches the resolution of
// PyTorch's uniform transformation).
__device__ __forceinline__ float u32_to_uniform(uint32_t x) {
  return static_cast<float>(x >> 8) * (1.0f / 16777216.0f);
}

// uint16 -> [0, 1) float. For 16-bit output dtypes (8-bit bf16 / 11-bit
// fp16 mantissas) 16 random bits per sample are ample, and they halve the
// Philox work per byte: one philox10 call yields a full 16-byte store.
__device__ __forceinline__ float u16_to_uniform(uint32_t x) {
  return static_cast<float>(x & 0xffffu) * (1.0f / 65536.0f);
}

// Two uniforms -> two standard normals (Box-Muller). -2*ln(u) is computed
// as -2*ln2*log2(u): __log2f maps straight to v_log_f32, measurably
// faster than __logf's wrapper (2.69 -> 3.15 TB/s on the bf16 normal
// kernel, scripts/rng_tune.hip).
__device__ __forceinline__ float2 box_muller(float u1, float u2) {
  // Guard u1 away from 0 so log stays finite.
  u1 = fmaxf(u1, 1.1754944e-38f);
  float r = sqrtf(-1.3862943611198906f * __log2f(u1));
  float s, c;
  __sincosf(6.2831853071795865f * u2, &s, &c);
  return make_float2(r * c, r * s);
}

template <typename T>
__device__ __forceinline__ T from_float(float v) {
  // Pin the f32 rounding step: without this barrier the compiler fuses
  // fma+convert into v_fma_mixlo_f16 (ONE rounding, straight to fp16)
  // in some loop shapes but not others, so the same element's bits
  // depended on which code path (vectorized main loop vs elementwise
  // boundary) stored it — a 1-ulp fp16 divergence that broke the
  // slice/full bitwise invariant (caught by the 40k-seed GPU slice
  // fuzz). bf16/f32 paths never fused; the barrier costs one cvt that
  // was already in the unfused schedule.
  asm volatile("" : "+v"(v));
  if constexpr (std::is_same_v<T, __hip_bfloat16>) {
    return __float2bfloat16(v);
  } else if constexpr (std::is_same_v<T, __half>) {
    return __float2half(v);
  } else {
    return v;
  }
}

// Elements per thread-group-iteration: 16-bit types pack 8 elements into
// one 16-byte store; 32-bit types pack 4 — one philox10 per store either
// way (16-bit dtypes draw 16 random bits per sample).
template <typename T>
struct VecTraits;
template <>
struct VecTraits<float> {
  static constexpr int kElems = 4;
  using Vec = float4;
};
template <>
struct VecTraits<__hip_bfloat16> {
  static constexpr int kElems = 8;
  struct alignas(16) Vec {
    __hip_bfloat16 v[8];
  };
};
template <>
struct VecTraits<__half> {
  static constexpr int kElems = 8;
  struct alignas(16) Vec {
    __half v[8];
  };
};

enum class Dist { kUniform, kNormal, kBernoulli };

// Computes the VecTraits<T>::kElems values of element group `g` — ONE
// philox10 with counter = the 16-byte group index turned into 4 fp32
// samples (24-bit uniforms) or 8 bf16/fp16 samples (16-bit uniforms,
// matched to the output mantissa). Shared by the full-tensor and the
// shard kernels so a shard is bitwise a slice of the full tensor.
template <typename T, Dist kDist>
__device__ __forceinline__ void rngGroupValues(uint64_t g,
                                               float a,
                                               float b,
                                               uint64_t seed,
                                               uint64_t offset,
                                               float* vals) {
  uint4 bits = kDist == Dist::kNormal ? philoxN<7>(seed, g, offset)
                                      : philoxN<10>(seed, g, offset);
  if constexpr (VecTraits<T>::kElems == 4) {
    float u[4] = {u32_to_uniform(bits.x), u32_to_uniform(bits.y),
                  u32_to_uniform(bits.z), u32_to_uniform(bits.w)};
    if constexpr (kDist == Dist::kUniform) {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        vals[j] = fmaf(u[j], b, a);
      }
    } else if constexpr (kDist == Dist::kBernoulli) {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        vals[j] = u[j] < a ? 1.0f : 0.0f;
      }
    } else {
      float2 n01 = box_muller(u[0], u[1]);
      float2 n23 = box_muller(u[2], u[3]);
      vals[0] = fmaf(n01.x, b, a);
      vals[1] = fmaf(n01.y, b, a);
      vals[2] = fmaf(n23.x, b, a);
      vals[3] = fmaf(n23.y, b, a);
    }
  } else {
    uint32_t words[4] = {bits.x, bits.y, bits.z, bits.w};
    if constexpr (kDist == Dist::kUniform) {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        vals[j * 2 + 0] = fmaf(u16_to_uniform(words[j]), b, a);
        vals[j * 2 + 1] = fmaf(u16_to_uniform(words[j] >> 16), b, a);
      }
    } else if constexpr (kDist == Dist::kBernoulli) {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        vals[j * 2 + 0] = u16_to_uniform(words[j]) < a ? 1.0f : 0.0f;
        vals[j * 2 + 1] = u16_to_uniform(words[j] >> 16) < a ? 1.0f : 0.0f;
      }
    } else {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float2 nj = box_muller(u16_to_uniform(words[j]),
                               u16_to_uniform(words[j] >> 16));
        vals[j * 2 + 0] = fmaf(nj.x, b, a);
        vals[j * 2 + 1] = fmaf(nj.y, b, a);
      }
    }
  }
}

// Indexed by GROUP, not element: full 16-byte groups take the branchless
// main loop (no per-iteration tail compare — worth ~10% on the
// VALU-bound normal kernel), the single partial tail group is handled by
// one thread. Group counts stay 32-bit up to 2^31 groups (= 32 GiB of
// bf16), so real tensors never pay 64-bit loop arithmetic.
template <typename T, Dist kDist, typename IdxT>
__global__ void rng_kernel(T* __restrict__ out,
                           IdxT n_full_groups,
                           uint32_t tail_elems,
                           float a,
                           float b,
                           uint64_t seed,
                           uint64_t offset) {
  constexpr int kElems = VecTraits<T>::kElems;
  using Vec = typename VecTraits<T>::Vec;

  const IdxT stride = static_cast<IdxT>(gridDim.x) * blockDim.x;
  for (IdxT g = blockIdx.x * static_cast<IdxT>(blockDim.x) + threadIdx.x;
       g < n_full_groups; g += stride) {
    float vals[kElems];
    rngGroupValues<T, kDist>(static_cast<uint64_t>(g), a, b, seed, offset,
                             vals);
    Vec v;
    T* vp = reinterpret_cast<T*>(&v);
#pragma unroll
    for (int j = 0; j < kElems; ++j) {
      vp[j] = from_float<T>(vals[j]);
    }
    *reinterpret_cast<Vec*>(out + static_cast<uint64_t>(g) * kElems) = v;
  }
  if (tail_elems != 0 && blockIdx.x == 0 && threadIdx.x == 0) {
    const uint64_t g = n_full_groups;
    float vals[kElems];
    rngGroupValues<T, kDist>(g, a, b, seed, offset, vals);
    for (uint32_t j = 0; j < tail_elems; ++j) {
      out[g * kElems + j] = from_float<T>(vals[j]);
    }
  }
}

template <typename T>
__global__ void fill_kernel(T* __restrict__ out, int64_t n, float value_f) {
  constexpr int kElems = VecTraits<T>::kElems;
  using Vec = typename VecTraits<T>::Vec;

  const int64_t n_groups = (n + kElems - 1) / kElems;
  const int64_t stride = static_cast<int64_t>(gridDim.x) * blockDim.x;

  const T value = from_float<T>(value_f);
  Vec v;
  T* vp = reinterpret_cast<T*>(&v);
#pragma unroll
  for (int j = 0; j < kElems; ++j) {
    vp[j] = value;
  }

  for (int64_t g = blockIdx.x * static_cast<int64_t>(blockDim.x) + threadIdx.x;
       g < n_groups; g += stride) {
    const int64_t base = g * kElems;
    if (base + kElems <= n) {
      *reinterpret_cast<Vec*>(out + base) = v;
    } else {
      for (int64_t j = 0; base + j < n; ++j) {
        out[base + j] = value;
      }
    }
  }
}

int numBlocks(int64_t n_groups) {
  int64_t blocks = (n_groups + kBlock - 1) / kBlock;
  return static_cast<int>(std::min<int64_t>(blocks, kMaxBlocks));
}

// Reserves a Philox offset window on the generator; 4 counters per launch
// is enough because the per-element-group subsequence already separates
// streams within a launch.
uint64_t acquireSeedOffset(const std::optional<at::Generator>& generator,
                           uint64_t* seed) {
  auto gen = at::get_generator_or_default<at::CUDAGeneratorImpl>(
      generator, at::cuda::detail::getDefaultCUDAGenerator());
  std::lock_guard<std::mutex> lock(gen->mutex_);
  auto state = gen->philox_engine_inputs(4);
  *seed = state.first;
  return state.second;
}

template <Dist kDist>
void launchRng(at::Tensor& self,
               double p0,
               double p1,
               const std::optional<at::Generator>& generator,
               std::optional<int64_t> pinned_seed,
               std::optional<int64_t> pinned_offset) {
  TORCH_CHECK(self.is_contiguous(),
              "tdx init kernels require contiguous tensors");
  const int64_t n = self.numel();
  if (n == 0) {
    return;
  }
  uint64_t seed = 0;
  uint64_t offset = 0;
  if (pinned_seed.has_value() && pinned_offset.has_value()) {
    // Replay with record-time-pinned Philox state: the result is the same
    // no matter which rank replays this op, or in what order.
    seed = static_cast<uint64_t>(*pinned_seed);
    offset = static_cast<uint64_t>(*pinned_offset);
  } else {
    offset = acquireSeedOffset(generator, &seed);
  }
  auto stream = at::cuda::getCurrentCUDAStream();

  // p0/p1 are (from, to) for uniform, (mean, std) for normal, and
  // (p, unused) for bernoulli; uniform/normal apply a + b * sample,
  // bernoulli thresholds the raw uniform against a.
  float a = static_cast<float>(p0);
  float b = kDist == Dist::kUniform ? static_cast<float>(p1 - p0)
                                    : static_cast<float>(p1);

  auto launch = [&](auto type_tag) {
    using T = decltype(type_tag);
    constexpr int kElems = VecTraits<T>::kElems;
    const int64_t n_full = n / kElems;
    const uint32_t tail = static_cast<uint32_t>(n % kElems);
    const int64_t n_groups = n_full + (tail != 0 ? 1 : 0);
    if (n_full <= std::numeric_limits<uint32_t>::max() / 2) {
      hipLaunchKernelGGL((rng_kernel<T, kDist, uint32_t>),
                         dim3(numBlocks(n_groups)), dim3(kBlock), 0,
                         stream.stream(),
                         reinterpret_cast<T*>(self.data_ptr()),
                         static_cast<uint32_t>(n_full), tail, a, b, seed,
                         offset);
    } else {
      hipLaunchKernelGGL((rng_kernel<T, kDist, uint64_t>),
                         dim3(numBlocks(n_groups)), dim3(kBlock), 0,
                         stream.stream(),
                         reinterpret_cast<T*>(self.data_ptr()),
                         static_cast<uint64_t>(n_full), tail, a, b, seed,
                         offset);
    }
    C10_HIP_KERNEL_LAUNCH_CHECK();
  };
  switch (self.scalar_type()) {
    case at::kFloat:
      launch(float{});
      break;
    case at::kBFloat16:
      launch(__hip_bfloat16{});
      break;
    case at::kHalf:
      launch(__half{});
      break;
    default:
      TORCH_CHECK(false, "tdx RNG init kernels support float32/bf16/fp16, ",
                  "got ", self.scalar_type());
  }
}

at::Tensor& tdx_uniform_(at::Tensor& self,
                         double from,
                         double to,
                         std::optional<at::Generator> generator,
                         std::optional<int64_t> seed,
                         std::optional<int64_t> offset) {
  // double (and other unsupported dtypes) never reach here: the redirect
  // layer filters, and direct callers get a clear dispatch error.
  launchRng<Dist::kUniform>(self, from, to, generator, seed, offset);
  return self;
}

at::Tensor& tdx_normal_(at::Tensor& self,
                        double mean,
                        double std,
                        std::optional<at::Generator> generator,
                        std::optional<int64_t> seed,
                        std::optional<int64_t> offset) {
  TORCH_CHECK(std >= 0.0, "normal_ expects std >= 0.0, but found std=", std);
  launchRng<Dist::kNormal>(self, mean, std, generator, seed, offset);
  return self;
}

// Shard kernel: writes elements [start, end) of the virtual full tensor
// into out[0 .. end-start). Interior groups use full 16-byte stores;
// boundary groups store elementwise.
template <typename T, Dist kDist>
__global__ void rng_shard_kernel(T* __restrict__ out,
                                 int64_t start,
                                 int64_t end,
                                 float a,
                                 float b,
                                 uint64_t seed,
                                 uint64_t offset) {
  constexpr int kElems = VecTraits<T>::kElems;
  using Vec = typename VecTraits<T>::Vec;

  const int64_t g_first = start / kElems;
  const int64_t g_last = (end + kElems - 1) / kElems;
  const int64_t stride = static_cast<int64_t>(gridDim.x) * blockDim.x;
  // The vector path stores Vec packs at out + (base - start); base is a
  // multiple of kElems, so the destination is 16-byte aligned only when
  // `start` is too. Unaligned shards (odd row sizes) store elementwise.
  const bool vec_aligned = (start % kElems) == 0;

  if (vec_aligned) {
    // Branchless main loop over the full interior groups (aligned start
    // means there is no partial head group); the single partial tail
    // group is handled by one thread below.
    const int64_t g_int_hi = end / kElems;
    for (int64_t g = g_first +
             blockIdx.x * static_cast<int64_t>(blockDim.x) + threadIdx.x;
         g < g_int_hi; g += stride) {
      float vals[kElems];
      rngGroupValues<T, kDist>(static_cast<uint64_t>(g), a, b, seed,
                               offset, vals);
      Vec v;
      T* vp = reinterpret_cast<T*>(&v);
#pragma unroll
      for (int j = 0; j < kElems; ++j) {
        vp[j] = from_float<T>(vals[j]);
      }
      *reinterpret_cast<Vec*>(out + (g * kElems - start)) = v;
    }
    if (blockIdx.x == 0 && threadIdx.x == 0 && end % kElems != 0 &&
        g_int_hi >= g_first) {
      float vals[kElems];
      rngGroupValues<T, kDist>(static_cast<uint64_t>(g_int_hi), a, b, seed,
                               offset, vals);
      const int64_t base = g_int_hi * kElems;
      for (int64_t e = base > start ? base : start; e < end; ++e) {
        out[e - start] = from_float<T>(vals[e - base]);
      }
    }
    return;
  }

  for (int64_t g = g_first +
           blockIdx.x * static_cast<int64_t>(blockDim.x) + threadIdx.x;
       g < g_last; g += stride) {
    float vals[kElems];
    rngGroupValues<T, kDist>(static_cast<uint64_t>(g), a, b, seed, offset,
                             vals);
    const int64_t base = g * kElems;
    const int64_t lo = base > start ? base : start;
    const int64_t hi = base + kElems < end ? base + kElems : end;
    for (int64_t e = lo; e < hi; ++e) {
      out[e - start] = from_float<T>(vals[e - base]);
    }
  }
}

template <Dist kDist>
void launchRngShard(at::Tensor& shard,
                    int64_t start,
                    int64_t end,
                    double p0,
                    double p1,
                    int64_t seed,
                    int64_t offset) {
  TORCH_CHECK(shard.is_contiguous(),
              "tdx shard kernels require contiguous tensors");
  TORCH_CHECK(shard.numel() == end - start,
              "shard numel must equal end - start");
  if (shard.numel() == 0) {
    return;
  }
  float a = static_cast<float>(p0);
  float b = kDist == Dist::kUniform ? static_cast<float>(p1 - p0)
                                    : static_cast<float>(p1);
  auto stream = at::cuda::getCurrentCUDAStream();
  auto launch = [&](auto type_tag) {
    using T = decltype(type_tag);
    const int64_t n_groups =
        (end + VecTraits<T>::kElems - 1) / VecTraits<T>::kElems -
        start / VecTraits<T>::kElems;
    hipLaunchKernelGGL((rng_shard_kernel<T, kDist>),
                       dim3(numBlocks(n_groups)), dim3(kBlock), 0,
                       stream.stream(),
                       reinterpret_cast<T*>(shard.data_ptr()), start, end, a,
                       b, static_cast<uint64_t>(seed),
                       static_cast<uint64_t>(offset));
    C10_HIP_KERNEL_LAUNCH_CHECK();
  };
  switch (shard.scalar_type()) {
    case at::kFloat:
      launch(float{});
      break;
    case at::kBFloat16:
      launch(__hip_bfloat16{});
      break;
    case at::kHalf:
      launch(__half{});
      break;
    default:
      TORCH_CHECK(false, "tdx shard kernels support float32/bf16/fp16, got ",
                  shard.scalar_type());
  }
}

at::Tensor& tdx_uniform_shard_(at::Tensor& shard, int64_t start, int64_t end,
                               double from, double to, int64_t seed,
                               int64_t offset) {
  launchRngShard<Dist::kUniform>(shard, start, end, from, to, seed, offset);
  return shard;
}

at::Tensor& tdx_normal_shard_(at::Tensor& shard, int64_t start, int64_t end,
                              double mean, double std, int64_t seed,
                              int64_t offset) {
  TORCH_CHECK(std >= 0.0, "normal_ expects std >= 0.0, but found std=", std);
  launchRngShard<Dist::kNormal>(shard, start, end, mean, std, seed, offset);
  return shard;
}

at::Tensor& tdx_bernoulli_(at::Tensor& self,
                           double p,
                           std::optional<at::Generator> generator,
                           std::optional<int64_t> seed,
                           std::optional<int64_t> offset) {
  TORCH_CHECK(0.0 <= p && p <= 1.0,
              "bernoulli_ expects 0 <= p <= 1, but found p=", p);
  launchRng<Dist::kBernoulli>(self, p, 0.0, generator, seed, offset);
  return self;
}

at::Tensor& tdx_bernoulli_shard_(at::Tensor& shard, int64_t start,
                                 int64_t end, double p, int64_t seed,
                                 int64_t offset) {
  TORCH_CHECK(0.0 <= p && p <= 1.0,
              "bernoulli_ expects 0 <= p <= 1, but found p=", p);
  launchRngShard<Dist::kBernoulli>(shard, start, end, p, 0.0, seed, offset);
  return shard;
}

// Windowed shard kernel: materializes `full.narrow(dim, s, len)` of a
// contiguous virtual full tensor for any dim — the slice is n_blocks
// contiguous global ranges of block_len elements, block r starting at
// global element g_off + r * g_stride (dim 0 is the n_blocks == 1
// special case, served by rng_shard_kernel above). Shard flat index i
// maps to global element
//   g(i) = g_off + (i / block_len) * g_stride + (i % block_len),
// and the value of global element g is lane g % kElems of Philox group
// g / kElems — identical to the full-tensor kernel, so any-dim slices
// are bitwise sub-tensors of the full materialization. When g_off,
// block_len and g_stride are all group-multiples (the common case:
// trailing dims of transformer weights are multiples of 8) every group
// of the window is whole and the destination is 16-byte aligned, so the
// fast path is group-indexed with vector stores like the flat kernel;
// odd geometries store elementwise.
// IdxT is the SHARD-side loop index type: 32-bit up to 2^31 shard
// groups/elements (the flat-kernel design rule — real shards never pay
// 64-bit divides, the one VALU-expensive op here); only the GLOBAL
// group id widens to 64-bit, with a multiply, not a divide.
template <typename T, Dist kDist, typename IdxT>
__global__ void rng_shard_window_kernel(T* __restrict__ out,
                                        IdxT n_blocks,
                                        IdxT block_len,
                                        int64_t g_stride,
                                        int64_t g_off,
                                        float a,
                                        float b,
                                        uint64_t seed,
                                        uint64_t offset) {
  constexpr int kElems = VecTraits<T>::kElems;
  using Vec = typename VecTraits<T>::Vec;
  const IdxT stride = static_cast<IdxT>(gridDim.x) * blockDim.x;
  const bool vec_aligned = (g_off % kElems) == 0 &&
                           (block_len % kElems) == 0 &&
                           (g_stride % kElems) == 0;
  if (vec_aligned) {
    const IdxT gpb = block_len / kElems;  // groups per window block
    const IdxT n_groups = n_blocks * gpb;
    const uint64_t g0 = static_cast<uint64_t>(g_off) / kElems;
    const uint64_t gs = static_cast<uint64_t>(g_stride) / kElems;
    for (IdxT j = blockIdx.x * static_cast<IdxT>(blockDim.x) + threadIdx.x;
         j < n_groups; j += stride) {
      const IdxT blk = j / gpb;
      const IdxT within = j - blk * gpb;
      const uint64_t g = g0 + static_cast<uint64_t>(blk) * gs + within;
      float vals[kElems];
      rngGroupValues<T, kDist>(g, a, b, seed, offset, vals);
      Vec v;
      T* vp = reinterpret_cast<T*>(&v);
#pragma unroll
      for (int e = 0; e < kElems; ++e) {
        vp[e] = from_float<T>(vals[e]);
      }
      *reinterpret_cast<Vec*>(out + static_cast<uint64_t>(j) * kElems) = v;
    }
    return;
  }
  const IdxT n = n_blocks * block_len;
  for (IdxT i = blockIdx.x * static_cast<IdxT>(blockDim.x) + threadIdx.x;
       i < n; i += stride) {
    const IdxT blk = i / block_len;
    const IdxT rem = i - blk * block_len;
    const uint64_t ge = static_cast<uint64_t>(g_off) +
                        static_cast<uint64_t>(blk) * g_stride + rem;
    const uint64_t g = ge / kElems;
    float vals[kElems];
    rngGroupValues<T, kDist>(g, a, b, seed, offset, vals);
    out[i] = from_float<T>(vals[ge - g * kElems]);
  }
}

template <Dist kDist>
void launchRngShardWindow(at::Tensor& shard, int64_t n_blocks,
                          int64_t block_len, int64_t g_stride, int64_t g_off,
                          double p0, double p1, int64_t seed,
                          int64_t offset) {
  TORCH_CHECK(shard.is_contiguous(),
              "tdx shard kernels require contiguous tensors");
  TORCH_CHECK(n_blocks >= 0 && block_len >= 0 && g_stride >= block_len &&
                  g_off >= 0,
              "invalid shard window");
  TORCH_CHECK(shard.numel() == n_blocks * block_len,
              "shard numel must equal n_blocks * block_len");
  if (shard.numel() == 0) {
    return;
  }
  float a = static_cast<float>(p0);
  float b = kDist == Dist::kUniform ? static_cast<float>(p1 - p0)
                                    : static_cast<float>(p1);
  auto stream = at::cuda::getCurrentCUDAStream();
  auto launch = [&](auto type_tag) {
    using T = decltype(type_tag);
    const int64_t n_work =
        (shard.numel() + VecTraits<T>::kElems - 1) / VecTraits<T>::kElems;
    auto launch_idx = [&](auto idx_tag) {
      using IdxT = decltype(idx_tag);
      hipLaunchKernelGGL((rng_shard_window_kernel<T, kDist, IdxT>),
                         dim3(numBlocks(n_work)), dim3(kBlock), 0,
                         stream.stream(),
                         reinterpret_cast<T*>(shard.data_ptr()),
                         static_cast<IdxT>(n_blocks),
                         static_cast<IdxT>(block_len), g_stride, g_off, a,
                         b, static_cast<uint64_t>(seed),
                         static_cast<uint64_t>(offset));
      C10_HIP_KERNEL_LAUNCH_CHECK();
    };
    if (shard.numel() <= std::numeric_limits<int32_t>::max()) {
      launch_idx(uint32_t{});
    } else {
      launch_idx(int64_t{});
    }
  };
  switch (shard.scalar_type()) {
    case at::kFloat:
      launch(float{});
      break;
    case at::kBFloat16:
      launch(__hip_bfloat16{});
      break;
    case at::kHalf:
      launch(__half{});
      break;
    default:
      TORCH_CHECK(false, "tdx shard kernels support float32/bf16/fp16, got ",
                  shard.scalar_type());
  }
}

at::Tensor& tdx_uniform_shard_win_(at::Tensor& shard, int64_t n_blocks,
                                   int64_t block_len, int64_t g_stride,
                                   int64_t g_off, double from, double to,
                                   int64_t seed, int64_t offset) {
  launchRngShardWindow<Dist::kUniform>(shard, n_blocks, block_len, g_stride,
                                       g_off, from, to, seed, offset);
  return shard;
}

at::Tensor& tdx_normal_shard_win_(at::Tensor& shard, int64_t n_blocks,
                                  int64_t block_len, int64_t g_stride,
                                  int64_t g_off, double mean, double std,
                                  int64_t seed, int64_t offset) {
  TORCH_CHECK(std >= 0.0, "normal_ expects std >= 0.0, but found std=", std);
  launchRngShardWindow<Dist::kNormal>(shard, n_blocks, block_len, g_stride,
                                      g_off, mean, std, seed, offset);
  return shard;
}

at::Tensor& tdx_bernoulli_shard_win_(at::Tensor& shard, int64_t n_blocks,
                                     int64_t block_len, int64_t g_stride,
                                     int64_t g_off, double p, int64_t seed,
                                     int64_t offset) {
  TORCH_CHECK(0.0 <= p && p <= 1.0,
              "bernoulli_ expects 0 <= p <= 1, but found p=", p);
  launchRngShardWindow<Dist::kBernoulli>(shard, n_blocks, block_len, g_stride,
                                         g_off, p, 0.0, seed, offset);
  return shard;
}

at::Tensor& tdx_fill_(at::Tensor& self, const at::Scalar& value) {
  TORCH_CHECK(self.is_contiguous(),
              "tdx init kernels require contiguous tensors");
  const int64_t n = self.numel();
  if (n == 0) {
    return self;
  }
  auto stream = at::cuda::getCurrentCUDAStream();
  auto launch = [&](auto type_tag) {
    using T = decltype(type_tag);
    const int64_t n_groups =
        (n + VecTraits<T>::kElems - 1) / VecTraits<T>::kElems;
    hipLaunchKernelGGL(fill_kernel<T>, dim3(numBlocks(n_groups)),
                       dim3(kBlock), 0, stream.stream(),
                       reinterpret_cast<T*>(self.data_ptr()), n,
                       value.to<float>());
    C10_HIP_KERNEL_LAUNCH_CHECK();
  };
  switch (self.scalar_type()) {
    case at::kFloat:
      launch(float{});
      break;
    case at::kBFloat16:
      launch(__hip_bfloat16{});
      break;
    case at::kHalf:
      launch(__half{});
      break;
    default:
      TORCH_CHECK(false, "tdx fill kernel supports float32/bf16/fp16, got ",
                  self.scalar_type());
  }
  return self;
}

// Vectorized cast-copy: dst[i] = cast(src[i]) for contiguous same-shape
// GPU tensors across {f32, bf16, f16}. 8 elements per thread per
// iteration; the narrow side uses 16-byte packs, the f32 side 2x16-byte.
template <typename TDst, typename TSrc, typename IdxT>
__global__ void cast_copy_kernel(TDst* __restrict__ dst,
                                 const TSrc* __restrict__ src,
                                 IdxT n) {
  struct alignas(16) PackDst {
    TDst v[8];
  };
  struct alignas(16) PackSrc {
    TSrc v[8];
  };
  const IdxT n_groups = (n + 7) / 8;
  const IdxT stride = static_cast<IdxT>(gridDim.x) * blockDim.x;
  for (IdxT g = blockIdx.x * static_cast<IdxT>(blockDim.x) + threadIdx.x;
       g < n_groups; g += stride) {
    const IdxT base = g * 8;
    if (base + 8 <= n) {
      PackSrc in = *reinterpret_cast<const PackSrc*>(src + base);
      PackDst out;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f;
        if constexpr (std::is_same_v<TSrc, float>) {
          f = in.v[j];
        } else if constexpr (std::is_same_v<TSrc, __hip_bfloat16>) {
          f = __bfloat162float(in.v[j]);
        } else {
          f = __half2float(in.v[j]);
        }
        out.v[j] = from_float<TDst>(f);
      }
      *reinterpret_cast<PackDst*>(dst + base) = out;
    } else {
      for (IdxT j = 0; base + j < n; ++j) {
        float f;
        if constexpr (std::is_same_v<TSrc, float>) {
          f = src[base + j];
        } else if constexpr (std::is_same_v<TSrc, __hip_bfloat16>) {
          f = __bfloat162float(src[base + j]);
        } else {
          f = __half2float(src[base + j]);
        }
        dst[base + j] = from_float<TDst>(f);
      }
    }
  }
}

template <typename F>
bool dispatchCopyType(c10::ScalarType st, F&& f) {
  switch (st) {
    case at::kFloat:
      f(float{});
      return true;
    case at::kBFloat16:
      f(__hip_bfloat16{});
      return true;
    case at::kHalf:
      f(__half{});
      return true;
    default:
      return false;
  }
}

at::Tensor& tdx_copy_(at::Tensor& self,
                      const at::Tensor& src,
                      bool non_blocking) {
  (void)non_blocking;  // same-device async copy; stream-ordered anyway
  TORCH_CHECK(self.is_contiguous() && src.is_contiguous() &&
                  self.sizes().equals(src.sizes()) && self.is_cuda() &&
                  src.is_cuda(),
              "tdx::copy_ requires contiguous same-shape GPU tensors");
  const int64_t n = self.numel();
  if (n == 0) {
    return self;
  }
  // Same dtype: a straight device memcpy is the fastest path.
  if (self.scalar_type() == src.scalar_type()) {
    auto stream = at::cuda::getCurrentCUDAStream();
    C10_HIP_CHECK(hipMemcpyAsync(self.data_ptr(), src.data_ptr(),
                                 n * self.element_size(),
                                 hipMemcpyDeviceToDevice, stream.stream()));
    return self;
  }
  auto stream = at::cuda::getCurrentCUDAStream();
  bool ok = dispatchCopyType(self.scalar_type(), [&](auto dt) {
    using TDst = decltype(dt);
    dispatchCopyType(src.scalar_type(), [&](auto st2) {
      using TSrc = decltype(st2);
      const int64_t n_groups = (n + 7) / 8;
      if (n <= std::numeric_limits<uint32_t>::max() / 2) {
        hipLaunchKernelGGL((cast_copy_kernel<TDst, TSrc, uint32_t>),
                           dim3(numBlocks(n_groups)), dim3(kBlock), 0,
                           stream.stream(),
                           reinterpret_cast<TDst*>(self.data_ptr()),
                           reinterpret_cast<const TSrc*>(src.data_ptr()),
                           static_cast<uint32_t>(n));
      } else {
        hipLaunchKernelGGL((cast_copy_kernel<TDst, TSrc, uint64_t>),
                           dim3(numBlocks(n_groups)), dim3(kBlock), 0,
                           stream.stream(),
                           reinterpret_cast<TDst*>(self.data_ptr()),
                           reinterpret_cast<const TSrc*>(src.data_ptr()),
                           static_cast<uint64_t>(n));
      }
      C10_HIP_KERNEL_LAUNCH_CHECK();
    });
  });
  TORCH_CHECK(ok, "tdx::copy_ supports float32/bf16/fp16");
  return self;
}

at::Tensor& tdx_zero_(at::Tensor& self) {
  if (!self.is_contiguous() || self.numel() == 0) {
    return self.zero_();
  }
  // A pure byte clear: hand it to the copy engine via memset.
  auto stream = at::cuda::getCurrentCUDAStream();
  C10_HIP_CHECK(hipMemsetAsync(self.data_ptr(), 0,
                               self.numel() * self.element_size(),
                               stream.stream()));
  return self;
}

// ---------------------------------------------------------------------------
// Batched init: ONE launch fills a whole list of tensors (the reduced
// value steps of simple init chains, see tensorInitPlan). A 70B replica
// is ~560 per-tensor launches; at ~8 us of launch+ramp each, batching
// them recovers ~4.5 ms of a ~53 ms step. Virtual blocks of kInitEPB
// elements map to tensors via a prefix table (same dispatch shape as
// the batched AdamW kernel); each tensor keeps its own Philox
// (seed, offset) and group indexing from 0, so bits are identical to
// the per-tensor kernels.
// ---------------------------------------------------------------------------

constexpr int kInitEPB = 16384;  // elements per virtual block

struct InitTensorMeta {
  void* ptr;
  int64_t n;
  int32_t dist;   // 0 uniform, 1 normal, 2 bernoulli, 3 fill, 4 zero
  int32_t dtype;  // 0 f32, 1 bf16, 2 f16
  float a, b;     // (from, range) / (mean, std) / (p, -) / (value, -)
  uint64_t seed, offset;
};

template <typename T, Dist kDist>
__device__ __forceinline__ void fillRangeVec(T* __restrict__ out,
                                             int64_t begin, int64_t end,
                                             int64_t n, float a, float b,
                                             uint64_t seed,
                                             uint64_t offset) {
  constexpr int kElems = VecTraits<T>::kElems;
  using Vec = typename VecTraits<T>::Vec;
  // begin is kInitEPB-aligned (hence kElems-aligned); only the last
  // vblock can carry a partial tail group — handled by one thread after
  // the branchless full-group loop.
  const int64_t full_end = begin + ((end - begin) / kElems) * kElems;
  for (int64_t base = begin + threadIdx.x * kElems; base < full_end;
       base += blockDim.x * kElems) {
    float vals[kElems];
    rngGroupValues<T, kDist>(static_cast<uint64_t>(base / kElems), a, b,
                             seed, offset, vals);
    Vec v;
    T* vp = reinterpret_cast<T*>(&v);
#pragma unroll
    for (int j = 0; j < kElems; ++j) {
      vp[j] = from_float<T>(vals[j]);
    }
    *reinterpret_cast<Vec*>(out + base) = v;
  }
  if (threadIdx.x == 0 && full_end < end) {
    float vals[kElems];
    rngGroupValues<T, kDist>(static_cast<uint64_t>(full_end / kElems), a,
                             b, seed, offset, vals);
    for (int64_t e = full_end; e < end; ++e) {
      out[e] = from_float<T>(vals[e - full_end]);
    }
  }
}

template <typename T>
__device__ __forceinline__ void fillRangeConst(T* __restrict__ out,
                                               int64_t begin, int64_t end,
                                               int64_t n, float value) {
  constexpr int kElems = VecTraits<T>::kElems;
  using Vec = typename VecTraits<T>::Vec;
  const T v = from_float<T>(value);
  Vec pack;
  T* vp = reinterpret_cast<T*>(&pack);
#pragma unroll
  for (int j = 0; j < kElems; ++j) {
    vp[j] = v;
  }
  for (int64_t base = begin + threadIdx.x * kElems; base < end;
       base += blockDim.x * kElems) {
    if (base + kElems <= n) {
      *reinterpret_cast<Vec*>(out + base) = pack;
    } else {
      for (int64_t j = 0; base + j < n; ++j) {
        out[base + j] = v;
      }
    }
  }
}

template <typename T>
__device__ __forceinline__ void initDispatchDist(const InitTensorMeta& mt,
                                                 int64_t begin,
                                                 int64_t end) {
  T* out = static_cast<T*>(mt.ptr);
  switch (mt.dist) {
    case 0:
      fillRangeVec<T, Dist::kUniform>(out, begin, end, mt.n, mt.a, mt.b,
                                      mt.seed, mt.offset);
      break;
    case 1:
      fillRangeVec<T, Dist::kNormal>(out, begin, end, mt.n, mt.a, mt.b,
                                     mt.seed, mt.offset);
      break;
    case 2:
      fillRangeVec<T, Dist::kBernoulli>(out, begin, end, mt.n, mt.a, mt.b,
                                        mt.seed, mt.offset);
      break;
    case 3:
      fillRangeConst<T>(out, begin, end, mt.n, mt.a);
      break;
    default:
      fillRangeConst<T>(out, begin, end, mt.n, 0.0f);
      break;
  }
}

// blob layout: int64 prefix[n_tensors] then InitTensorMeta[n_tensors].
__global__ void batched_init_kernel(const uint8_t* __restrict__ blob,
                                    int32_t n_tensors,
                                    int64_t total_vblocks) {
  const int64_t* prefix = reinterpret_cast<const int64_t*>(blob);
  const InitTensorMeta* metas = reinterpret_cast<const InitTensorMeta*>(
      blob + sizeof(int64_t) * n_tensors);

  for (int64_t vb = blockIdx.x; vb < total_vblocks; vb += gridDim.x) {
    int lo = 0, hi = n_tensors - 1;
    while (lo < hi) {
      const int mid = (lo + hi + 1) >> 1;
      if (prefix[mid] <= vb) {
        lo = mid;
      } else {
        hi = mid - 1;
      }
    }
    const InitTensorMeta mt = metas[lo];
    const int64_t begin = (vb - prefix[lo]) * kInitEPB;
    const int64_t end = begin + kInitEPB < mt.n ? begin + kInitEPB : mt.n;
    switch (mt.dtype) {
      case 0:
        initDispatchDist<float>(mt, begin, end);
        break;
      case 1:
        initDispatchDist<__hip_bfloat16>(mt, begin, end);
        break;
      default:
        initDispatchDist<__half>(mt, begin, end);
        break;
    }
  }
}

}  // namespace

void batched_init_launch(std::vector<at::Tensor> tensors,
                         std::vector<int64_t> dists,
                         std::vector<double> p0s,
                         std::vector<double> p1s,
                         std::vector<int64_t> seeds,
                         std::vector<int64_t> offsets) {
  const size_t n_tensors = tensors.size();
  TORCH_CHECK(n_tensors > 0 && dists.size() == n_tensors &&
                  p0s.size() == n_tensors && p1s.size() == n_tensors &&
                  seeds.size() == n_tensors && offsets.size() == n_tensors,
              "batched init: list length mismatch");

  const size_t prefix_bytes = sizeof(int64_t) * n_tensors;
  const size_t blob_bytes =
      prefix_bytes + sizeof(InitTensorMeta) * n_tensors;
  at::Tensor host_blob = at::empty(
      {static_cast<int64_t>(blob_bytes)},
      at::TensorOptions().dtype(at::kByte).pinned_memory(true));
  auto* prefix = reinterpret_cast<int64_t*>(host_blob.data_ptr());
  auto* metas = reinterpret_cast<InitTensorMeta*>(
      static_cast<uint8_t*>(host_blob.data_ptr()) + prefix_bytes);

  int64_t total_vblocks = 0;
  for (size_t t = 0; t < n_tensors; ++t) {
    const at::Tensor& x = tensors[t];
    TORCH_CHECK(x.is_cuda() && x.is_contiguous(),
                "batched init: contiguous GPU tensors only");
    InitTensorMeta& mt = metas[t];
    mt.ptr = x.data_ptr();
    mt.n = x.numel();
    mt.dist = static_cast<int32_t>(dists[t]);
    TORCH_CHECK(0 <= mt.dist && mt.dist <= 4, "batched init: bad dist");
    switch (x.scalar_type()) {
      case at::kFloat:
        mt.dtype = 0;
        break;
      case at::kBFloat16:
        mt.dtype = 1;
        break;
      case at::kHalf:
        mt.dtype = 2;
        break;
      default:
        TORCH_CHECK(false, "batched init supports f32/bf16/f16, got ",
                    x.scalar_type());
    }
    const double p0 = p0s[t];
    const double p1 = p1s[t];
    // Same (a, b) mapping as launchRng.
    mt.a = static_cast<float>(p0);
    mt.b = mt.dist == 0 ? static_cast<float>(p1 - p0)
                        : static_cast<float>(p1);
    mt.seed = static_cast<uint64_t>(seeds[t]);
    mt.offset = static_cast<uint64_t>(offsets[t]);
    prefix[t] = total_vblocks;
    total_vblocks += (mt.n + kInitEPB - 1) / kInitEPB;
  }

  at::Tensor dev_blob =
      host_blob.to(tensors[0].device(), /*non_blocking=*/true);
  const int grid =
      static_cast<int>(std::min<int64_t>(total_vblocks, 16384));
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(batched_init_kernel, dim3(grid), dim3(kBlock), 0,
                     stream.stream(),
                     static_cast<const uint8_t*>(dev_blob.const_data_ptr()),
                     static_cast<int32_t>(n_tensors), total_vblocks);
  C10_HIP_KERNEL_LAUNCH_CHECK();
}

namespace {

// Schemas are defined by the core extension (csrc/core/tdx_ops.cc); this
// extension contributes the CUDA implementations.
TORCH_LIBRARY_IMPL(tdx, CUDA, m) {
  m.impl("uniform_", tdx_uniform_);
  m.impl("normal_", tdx_normal_);
  m.impl("fill_", tdx_fill_);
  m.impl("zero_", tdx_zero_);
  m.impl("copy_", tdx_copy_);
  m.impl("uniform_shard_", tdx_uniform_shard_);
  m.impl("normal_shard_", tdx_normal_shard_);
  m.impl("bernoulli_", tdx_bernoulli_);
  m.impl("bernoulli_shard_", tdx_bernoulli_shard_);
  m.impl("uniform_shard_win_", tdx_uniform_shard_win_);
  m.impl("normal_shard_win_", tdx_normal_shard_win_);
  m.impl("bernoulli_shard_win_", tdx_bernoulli_shard_win_);
}

}  // namespace
}  // namespace tdx
