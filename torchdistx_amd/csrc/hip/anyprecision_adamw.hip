// Fused AnyPrecisionAdamW step for CDNA4.
//
// The eager optimizer (optimizers/anyprecision_optimizer.py) launches ~10
// elementwise kernels per parameter, each streaming the full tensor through
// HBM. This kernel performs the whole update — decoupled weight decay,
// lerp momentum, variance update, bias-corrected denom, optional
// Kahan-compensated weight update — in ONE pass: 4-5 reads + 3-4 writes per
// element instead of ~20, which is the entire win for a memory-bound op.
//
// All arithmetic is fp32; tensor elements are converted at load/store per
// their runtime dtype (f32 / bf16 / f16 in any combination). The per-access
// dtype switch is wave-uniform and fully predicted, so on this
// memory-bound kernel it costs nothing measurable; a vectorized
// specialization for the flagship (bf16 param / f32 momentum / bf16
// variance) layout is the next optimization step.

#include <hip/hip_runtime.h>

#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

namespace tdx {
namespace {

enum class DT : int { kF32 = 0, kBF16 = 1, kF16 = 2 };

__device__ __forceinline__ float loadf(const void* p, DT dt, int64_t i) {
  switch (dt) {
    case DT::kF32:
      return static_cast<const float*>(p)[i];
    case DT::kBF16:
      return __bfloat162float(static_cast<const __hip_bfloat16*>(p)[i]);
    default:
      return __half2float(static_cast<const __half*>(p)[i]);
  }
}

// Pins the f32 rounding step before narrowing conversions, so every
// code path (generic, vectorized, batched) rounds identically — the
// compiler otherwise fuses fma+convert (v_fma_mix*) in some shapes only
// (see init_kernels.hip from_float).
__device__ __forceinline__ float roundBarrier(float v) {
  asm volatile("" : "+v"(v));
  return v;
}

__device__ __forceinline__ void storef(void* p, DT dt, int64_t i, float v) {
  v = roundBarrier(v);
  switch (dt) {
    case DT::kF32:
      static_cast<float*>(p)[i] = v;
      break;
    case DT::kBF16:
      static_cast<__hip_bfloat16*>(p)[i] = __float2bfloat16(v);
      break;
    default:
      static_cast<__half*>(p)[i] = __float2half(v);
      break;
  }
}

// Vector packs: 8 elements per thread per iteration, 16-byte accesses for
// 16-bit types, 2x16-byte for fp32.
template <typename T>
struct alignas(16) Pack8Aligned {
  T v[8];
};

template <typename T>
__device__ __forceinline__ void load8(const T* p, int64_t base, float* out) {
  Pack8Aligned<T> pk = *reinterpret_cast<const Pack8Aligned<T>*>(p + base);
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    if constexpr (std::is_same_v<T, float>) {
      out[j] = pk.v[j];
    } else if constexpr (std::is_same_v<T, __hip_bfloat16>) {
      out[j] = __bfloat162float(pk.v[j]);
    } else {
      out[j] = __half2float(pk.v[j]);
    }
  }
}

template <typename T>
__device__ __forceinline__ void store8(T* p, int64_t base, const float* in) {
  Pack8Aligned<T> pk;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const float vj = roundBarrier(in[j]);
    if constexpr (std::is_same_v<T, float>) {
      pk.v[j] = vj;
    } else if constexpr (std::is_same_v<T, __hip_bfloat16>) {
      pk.v[j] = __float2bfloat16(vj);
    } else {
      pk.v[j] = __float2half(vj);
    }
  }
  *reinterpret_cast<Pack8Aligned<T>*>(p + base) = pk;
}

struct AdamWScalars {
  int64_t n;
  float lr;
  float beta1;
  float beta2;
  float eps;
  float weight_decay;
  float step_size;
  float bias_correction2_sqrt;
};

// Vectorized fused step: 8 elements per thread per grid-stride iteration,
// full 16-byte (bf16/fp16) or 2x16-byte (fp32) accesses per tensor. The
// update math is identical to the generic kernel below.
template <typename Tp, typename Tg, typename Tm, typename Tv, typename Tc,
          bool kKahan>
__global__ void adamw_vec_kernel(Tp* __restrict__ param,
                                 const Tg* __restrict__ grad,
                                 Tm* __restrict__ exp_avg,
                                 Tv* __restrict__ exp_avg_sq,
                                 Tc* __restrict__ compensation,
                                 AdamWScalars s) {
  const int64_t n_groups = s.n / 8;  // full groups; tail by generic kernel
  const int64_t stride = static_cast<int64_t>(gridDim.x) * blockDim.x;
  for (int64_t g = blockIdx.x * static_cast<int64_t>(blockDim.x) + threadIdx.x;
       g < n_groups; g += stride) {
    const int64_t base = g * 8;
    float p[8], gr[8], m[8], v[8];
    load8(param, base, p);
    load8(grad, base, gr);
    load8(exp_avg, base, m);
    load8(exp_avg_sq, base, v);
    float c[8];
    if constexpr (kKahan) {
      load8(compensation, base, c);
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      if (s.weight_decay != 0.0f) {
        p[j] *= 1.0f - s.lr * s.weight_decay;
      }
      m[j] += (1.0f - s.beta1) * (gr[j] - m[j]);
      v[j] = v[j] * s.beta2 + gr[j] * gr[j] * (1.0f - s.beta2);
      const float denom = sqrtf(v[j]) / s.bias_correction2_sqrt + s.eps;
      const float update = -s.step_size * (m[j] / denom);
      if constexpr (kKahan) {
        c[j] += update;
        const float prev = p[j];
        float p_new = p[j] + c[j];
        // Round-trip through the storage dtype to keep the Kahan
        // remainder exact w.r.t. what is actually stored.
        Tp stored;
        const float p_new_r = roundBarrier(p_new);
        if constexpr (std::is_same_v<Tp, float>) {
          stored = p_new_r;
        } else if constexpr (std::is_same_v<Tp, __hip_bfloat16>) {
          stored = __float2bfloat16(p_new_r);
        } else {
          stored = __float2half(p_new_r);
        }
        float p_rounded;
        if constexpr (std::is_same_v<Tp, float>) {
          p_rounded = stored;
        } else if constexpr (std::is_same_v<Tp, __hip_bfloat16>) {
          p_rounded = __bfloat162float(stored);
        } else {
          p_rounded = __half2float(stored);
        }
        c[j] += prev - p_rounded;
        p[j] = p_new;
      } else {
        p[j] += update;
      }
    }
    store8(param, base, p);
    store8(exp_avg, base, m);
    store8(exp_avg_sq, base, v);
    if constexpr (kKahan) {
      store8(compensation, base, c);
    }
  }
}

struct AdamWArgs {
  void* param;
  const void* grad;
  void* exp_avg;
  void* exp_avg_sq;
  void* compensation;  // nullptr -> plain update
  int64_t n;
  int64_t i0;  // first element to process (tail launches after vec path)
  DT dt_param;
  DT dt_grad;
  DT dt_m;
  DT dt_v;
  DT dt_c;
  float lr;
  float beta1;
  float beta2;
  float eps;
  float weight_decay;
  float step_size;
  float bias_correction2_sqrt;
};

__global__ void anyprecision_adamw_kernel(AdamWArgs args) {
  const int64_t stride = static_cast<int64_t>(gridDim.x) * blockDim.x;
  const bool kahan = args.compensation != nullptr;

  for (int64_t i = args.i0 +
           blockIdx.x * static_cast<int64_t>(blockDim.x) + threadIdx.x;
       i < args.n; i += stride) {
    float p = loadf(args.param, args.dt_param, i);
    const float g = loadf(args.grad, args.dt_grad, i);
    float m = loadf(args.exp_avg, args.dt_m, i);
    float v = loadf(args.exp_avg_sq, args.dt_v, i);

    if (args.weight_decay != 0.0f) {
      p *= 1.0f - args.lr * args.weight_decay;
    }
    m += (1.0f - args.beta1) * (g - m);
    v = v * args.beta2 + g * g * (1.0f - args.beta2);

    const float denom =
        sqrtf(v) / args.bias_correction2_sqrt + args.eps;
    const float update = -args.step_size * (m / denom);

    if (kahan) {
      // Kahan: fold the accumulated update into the (low-precision) param
      // and keep the rounded-away remainder in the compensation buffer.
      float c = loadf(args.compensation, args.dt_c, i) + update;
      const float prev = p;
      storef(args.param, args.dt_param, i, p + c);
      const float p_stored = loadf(args.param, args.dt_param, i);
      c += prev - p_stored;
      storef(args.compensation, args.dt_c, i, c);
    } else {
      p += update;
      storef(args.param, args.dt_param, i, p);
    }
    storef(args.exp_avg, args.dt_m, i, m);
    storef(args.exp_avg_sq, args.dt_v, i, v);
  }
}

// {f32, bf16} type dispatch for the vectorized path (fp16 falls back to
// the generic kernel to bound template instantiation count).
template <typename F>
bool dispatch2(DT dt, F&& f) {
  if (dt == DT::kF32) {
    f(float{});
    return true;
  }
  if (dt == DT::kBF16) {
    f(__hip_bfloat16{});
    return true;
  }
  return false;
}

bool aligned16(const void* p) {
  return (reinterpret_cast<uintptr_t>(p) & 0xf) == 0;
}

// Launches the vectorized kernel for the full 8-element groups when every
// tensor is 16-byte aligned and no tensor is fp16. Returns the number of
// elements handled (0 when the fast path does not apply).
int64_t tryLaunchVec(const AdamWArgs& args, int grid, int block,
                     hipStream_t stream) {
  const int64_t n_full = args.n & ~int64_t{7};
  if (n_full == 0 || !aligned16(args.param) || !aligned16(args.grad) ||
      !aligned16(args.exp_avg) || !aligned16(args.exp_avg_sq) ||
      (args.compensation != nullptr && !aligned16(args.compensation))) {
    return 0;
  }
  AdamWScalars s{n_full,         args.lr,       args.beta1,
                 args.beta2,     args.eps,      args.weight_decay,
                 args.step_size, args.bias_correction2_sqrt};
  bool launched = false;
  dispatch2(args.dt_param, [&](auto tp) {
    using Tp = decltype(tp);
    dispatch2(args.dt_grad, [&](auto tg) {
      using Tg = decltype(tg);
      dispatch2(args.dt_m, [&](auto tm) {
        using Tm = decltype(tm);
        dispatch2(args.dt_v, [&](auto tv) {
          using Tv = decltype(tv);
          if (args.compensation == nullptr) {
            hipLaunchKernelGGL(
                (adamw_vec_kernel<Tp, Tg, Tm, Tv, float, false>), dim3(grid),
                dim3(block), 0, stream,
                static_cast<Tp*>(args.param),
                static_cast<const Tg*>(args.grad),
                static_cast<Tm*>(args.exp_avg),
                static_cast<Tv*>(args.exp_avg_sq), nullptr, s);
            launched = true;
          } else {
            dispatch2(args.dt_c, [&](auto tc) {
              using Tc = decltype(tc);
              hipLaunchKernelGGL(
                  (adamw_vec_kernel<Tp, Tg, Tm, Tv, Tc, true>), dim3(grid),
                  dim3(block), 0, stream,
                  static_cast<Tp*>(args.param),
                  static_cast<const Tg*>(args.grad),
                  static_cast<Tm*>(args.exp_avg),
                  static_cast<Tv*>(args.exp_avg_sq),
                  static_cast<Tc*>(args.compensation), s);
              launched = true;
            });
          }
        });
      });
    });
  });
  return launched ? n_full : 0;
}

// ---------------------------------------------------------------------------
// Batched (multi-tensor) step: one launch updates every parameter of an
// optimizer group. A transformer optimizer step is hundreds of tensors,
// most of them small — per-tensor launches cost ~5-10 us each, which for
// a 70B model's ~723 parameters is milliseconds of pure launch overhead.
// Virtual blocks of kBatchEPB elements are assigned to tensors via a
// prefix-sum table; each real block grid-strides over virtual blocks and
// binary-searches its tensor.
// ---------------------------------------------------------------------------

constexpr int kBatchEPB = 2048;  // elements per virtual block (256 thr x 8)

struct BatchTensorMeta {
  void* param;
  const void* grad;
  void* exp_avg;
  void* exp_avg_sq;
  void* compensation;  // nullptr -> plain update
  int64_t n;
  int32_t dt_param, dt_grad, dt_m, dt_v, dt_c;
  float step_size;
  float bias_correction2_sqrt;
};

struct BatchHeader {
  int32_t n_tensors;
  int64_t total_vblocks;
  float lr, beta1, beta2, eps, weight_decay;
};

// blob layout: int64 prefix[n_tensors] then BatchTensorMeta[n_tensors].
__global__ void adamw_batched_kernel(const uint8_t* __restrict__ blob,
                                     BatchHeader h) {
  const int64_t* prefix = reinterpret_cast<const int64_t*>(blob);
  const BatchTensorMeta* metas = reinterpret_cast<const BatchTensorMeta*>(
      blob + sizeof(int64_t) * h.n_tensors);

  for (int64_t vb = blockIdx.x; vb < h.total_vblocks; vb += gridDim.x) {
    // Largest t with prefix[t] <= vb.
    int lo = 0, hi = h.n_tensors - 1;
    while (lo < hi) {
      const int mid = (lo + hi + 1) >> 1;
      if (prefix[mid] <= vb) {
        lo = mid;
      } else {
        hi = mid - 1;
      }
    }
    const BatchTensorMeta mt = metas[lo];
    const int64_t base = (vb - prefix[lo]) * kBatchEPB;
    const bool kahan = mt.compensation != nullptr;
#pragma unroll
    for (int j = 0; j < kBatchEPB / 256; ++j) {
      const int64_t i = base + j * 256 + threadIdx.x;
      if (i >= mt.n) {
        break;
      }
      float p = loadf(mt.param, static_cast<DT>(mt.dt_param), i);
      const float g = loadf(mt.grad, static_cast<DT>(mt.dt_grad), i);
      float m = loadf(mt.exp_avg, static_cast<DT>(mt.dt_m), i);
      float v = loadf(mt.exp_avg_sq, static_cast<DT>(mt.dt_v), i);

      if (h.weight_decay != 0.0f) {
        p *= 1.0f - h.lr * h.weight_decay;
      }
      m += (1.0f - h.beta1) * (g - m);
      v = v * h.beta2 + g * g * (1.0f - h.beta2);
      const float denom = sqrtf(v) / mt.bias_correction2_sqrt + h.eps;
      const float update = -mt.step_size * (m / denom);

      if (kahan) {
        float c = loadf(mt.compensation, static_cast<DT>(mt.dt_c), i) +
                  update;
        const float prev = p;
        storef(mt.param, static_cast<DT>(mt.dt_param), i, p + c);
        const float p_stored =
            loadf(mt.param, static_cast<DT>(mt.dt_param), i);
        c += prev - p_stored;
        storef(mt.compensation, static_cast<DT>(mt.dt_c), i, c);
      } else {
        p += update;
        storef(mt.param, static_cast<DT>(mt.dt_param), i, p);
      }
      storef(mt.exp_avg, static_cast<DT>(mt.dt_m), i, m);
      storef(mt.exp_avg_sq, static_cast<DT>(mt.dt_v), i, v);
    }
  }
}

DT toDT(const at::Tensor& t) {
  switch (t.scalar_type()) {
    case at::kFloat:
      return DT::kF32;
    case at::kBFloat16:
      return DT::kBF16;
    case at::kHalf:
      return DT::kF16;
    default:
      TORCH_CHECK(false,
                  "fused AnyPrecisionAdamW supports float32/bf16/fp16, got ",
                  t.scalar_type());
  }
}

}  // namespace

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
                             double bias_correction2_sqrt) {
  TORCH_CHECK(param.is_cuda() && grad.is_cuda(),
              "fused AnyPrecisionAdamW requires GPU tensors");
  const at::Tensor* checked[] = {&param, &grad, &exp_avg, &exp_avg_sq};
  for (const at::Tensor* t : checked) {
    // Wrapper subclasses (DTensor, ...) reach C++ as storage-less impls;
    // dereferencing their null device pointer would be a GPU memory
    // fault, so refuse them here rather than in the kernel.
    TORCH_CHECK(t->unsafeGetTensorImpl()->has_storage() &&
                    t->const_data_ptr() != nullptr,
                "fused AnyPrecisionAdamW requires plain dense tensors "
                "(got a storage-less tensor, e.g. a DTensor wrapper)");
    TORCH_CHECK(t->is_contiguous(),
                "fused AnyPrecisionAdamW requires contiguous tensors");
    TORCH_CHECK(t->numel() == param.numel(),
                "fused AnyPrecisionAdamW: size mismatch");
  }

  AdamWArgs args{};
  args.param = param.data_ptr();
  args.grad = grad.data_ptr();
  args.exp_avg = exp_avg.data_ptr();
  args.exp_avg_sq = exp_avg_sq.data_ptr();
  args.compensation =
      compensation.has_value() ? compensation->data_ptr() : nullptr;
  args.n = param.numel();
  args.dt_param = toDT(param);
  args.dt_grad = toDT(grad);
  args.dt_m = toDT(exp_avg);
  args.dt_v = toDT(exp_avg_sq);
  args.dt_c = compensation.has_value() ? toDT(*compensation) : DT::kF32;
  args.lr = static_cast<float>(lr);
  args.beta1 = static_cast<float>(beta1);
  args.beta2 = static_cast<float>(beta2);
  args.eps = static_cast<float>(eps);
  args.weight_decay = static_cast<float>(weight_decay);
  args.step_size = static_cast<float>(step_size);
  args.bias_correction2_sqrt = static_cast<float>(bias_correction2_sqrt);

  const int block = 256;
  const int grid = static_cast<int>(
      std::min<int64_t>((args.n + block - 1) / block, 8192));
  auto stream = at::cuda::getCurrentCUDAStream();
  args.i0 = tryLaunchVec(args, grid, block, stream.stream());
  if (args.i0 > 0) {
    C10_HIP_KERNEL_LAUNCH_CHECK();
  }
  if (args.i0 < args.n) {
    hipLaunchKernelGGL(anyprecision_adamw_kernel, dim3(grid), dim3(block), 0,
                       stream.stream(), args);
    C10_HIP_KERNEL_LAUNCH_CHECK();
  }
}

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
    std::vector<double> bias_correction2_sqrts) {
  const size_t n_tensors = params.size();
  TORCH_CHECK(n_tensors > 0, "batched AdamW: empty tensor list");
  TORCH_CHECK(grads.size() == n_tensors && exp_avgs.size() == n_tensors &&
                  exp_avg_sqs.size() == n_tensors &&
                  compensations.size() == n_tensors &&
                  step_sizes.size() == n_tensors &&
                  bias_correction2_sqrts.size() == n_tensors,
              "batched AdamW: list length mismatch");

  const size_t prefix_bytes = sizeof(int64_t) * n_tensors;
  const size_t blob_bytes =
      prefix_bytes + sizeof(BatchTensorMeta) * n_tensors;
  at::Tensor host_blob = at::empty(
      {static_cast<int64_t>(blob_bytes)},
      at::TensorOptions().dtype(at::kByte).pinned_memory(true));
  auto* prefix = reinterpret_cast<int64_t*>(host_blob.data_ptr());
  auto* metas = reinterpret_cast<BatchTensorMeta*>(
      static_cast<uint8_t*>(host_blob.data_ptr()) + prefix_bytes);

  int64_t total_vblocks = 0;
  for (size_t t = 0; t < n_tensors; ++t) {
    const at::Tensor& p = params[t];
    const at::Tensor* checked[] = {&p, &grads[t], &exp_avgs[t],
                                   &exp_avg_sqs[t]};
    for (const at::Tensor* x : checked) {
      TORCH_CHECK(x->is_cuda() && x->is_contiguous() &&
                      x->unsafeGetTensorImpl()->has_storage() &&
                      x->const_data_ptr() != nullptr &&
                      x->numel() == p.numel(),
                  "batched AdamW: tensors must be plain dense contiguous "
                  "GPU tensors of equal numel");
    }
    BatchTensorMeta& mt = metas[t];
    mt.param = p.data_ptr();
    mt.grad = grads[t].data_ptr();
    mt.exp_avg = exp_avgs[t].data_ptr();
    mt.exp_avg_sq = exp_avg_sqs[t].data_ptr();
    mt.compensation =
        compensations[t].has_value() ? compensations[t]->data_ptr() : nullptr;
    mt.n = p.numel();
    mt.dt_param = static_cast<int32_t>(toDT(p));
    mt.dt_grad = static_cast<int32_t>(toDT(grads[t]));
    mt.dt_m = static_cast<int32_t>(toDT(exp_avgs[t]));
    mt.dt_v = static_cast<int32_t>(toDT(exp_avg_sqs[t]));
    mt.dt_c = compensations[t].has_value()
                  ? static_cast<int32_t>(toDT(*compensations[t]))
                  : static_cast<int32_t>(DT::kF32);
    mt.step_size = static_cast<float>(step_sizes[t]);
    mt.bias_correction2_sqrt =
        static_cast<float>(bias_correction2_sqrts[t]);
    prefix[t] = total_vblocks;
    total_vblocks += (mt.n + kBatchEPB - 1) / kBatchEPB;
  }

  at::Tensor dev_blob =
      host_blob.to(params[0].device(), /*non_blocking=*/true);

  BatchHeader h;
  h.n_tensors = static_cast<int32_t>(n_tensors);
  h.total_vblocks = total_vblocks;
  h.lr = static_cast<float>(lr);
  h.beta1 = static_cast<float>(beta1);
  h.beta2 = static_cast<float>(beta2);
  h.eps = static_cast<float>(eps);
  h.weight_decay = static_cast<float>(weight_decay);

  const int grid =
      static_cast<int>(std::min<int64_t>(total_vblocks, 16384));
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(adamw_batched_kernel, dim3(grid), dim3(256), 0,
                     stream.stream(),
                     static_cast<const uint8_t*>(dev_blob.const_data_ptr()),
                     h);
  C10_HIP_KERNEL_LAUNCH_CHECK();
}

}  // namespace tdx
