// Standalone tuning probe for the Philox uniform kernel: sweeps
// elements-per-thread and grid size on a 4 GiB bf16 buffer.
// Build/run on the GPU box:
//   hipcc --offload-arch=gfx950 -O3 -std=c++17 scripts/rng_tune.hip -o /tmp/rng_tune && /tmp/rng_tune
#include <hip/hip_bf16.h>
#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>

#define CHECK(x)                                            \
  do {                                                      \
    hipError_t e = (x);                                     \
    if (e != hipSuccess) {                                  \
      printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__); \
      return 1;                                             \
    }                                                       \
  } while (0)

__device__ __forceinline__ uint2 mulhilo32(uint32_t a, uint32_t b) {
  uint2 r;
  r.x = a * b;
  r.y = __umulhi(a, b);
  return r;
}

__device__ __forceinline__ uint4 philox10(uint64_t seed, uint64_t subseq,
                                          uint64_t offset) {
  constexpr uint32_t kW0 = 0x9E3779B9u, kW1 = 0xBB67AE85u;
  constexpr uint32_t kM0 = 0xD2511F53u, kM1 = 0xCD9E8D57u;
  uint32_t k0 = (uint32_t)seed, k1 = (uint32_t)(seed >> 32);
  uint4 c = make_uint4((uint32_t)offset, (uint32_t)(offset >> 32),
                       (uint32_t)subseq, (uint32_t)(subseq >> 32));
#pragma unroll
  for (int r = 0; r < 10; ++r) {
    uint2 r0 = mulhilo32(kM0, c.x);
    uint2 r1 = mulhilo32(kM1, c.z);
    c = make_uint4(r1.y ^ c.y ^ k0, r1.x, r0.y ^ c.w ^ k1, r0.x);
    k0 += kW0;
    k1 += kW1;
  }
  return c;
}

// Round-count-parametrized Philox (7 rounds passes BigCrush per Salmon
// et al. 2011, Table 2; 10 is PyTorch's margin choice). Measures how much
// of the normal-kernel plateau is Philox VALU work.
template <int R>
__device__ __forceinline__ uint4 philoxR(uint64_t seed, uint64_t subseq,
                                         uint64_t offset) {
  constexpr uint32_t kW0 = 0x9E3779B9u, kW1 = 0xBB67AE85u;
  constexpr uint32_t kM0 = 0xD2511F53u, kM1 = 0xCD9E8D57u;
  uint32_t k0 = (uint32_t)seed, k1 = (uint32_t)(seed >> 32);
  uint4 c = make_uint4((uint32_t)offset, (uint32_t)(offset >> 32),
                       (uint32_t)subseq, (uint32_t)(subseq >> 32));
#pragma unroll
  for (int r = 0; r < R; ++r) {
    uint2 r0 = mulhilo32(kM0, c.x);
    uint2 r1 = mulhilo32(kM1, c.z);
    c = make_uint4(r1.y ^ c.y ^ k0, r1.x, r0.y ^ c.w ^ k1, r0.x);
    k0 += kW0;
    k1 += kW1;
  }
  return c;
}

__device__ __forceinline__ float u2f(uint32_t x) {
  return (float)(x >> 8) * (1.0f / 16777216.0f);
}

struct alignas(16) V8 {
  __hip_bfloat16 v[8];
};

// EPT = elements per thread per grid-stride iteration (multiple of 8).
template <int EPT>
__global__ void rng_bf16(__hip_bfloat16* __restrict__ out, uint32_t n8,
                         float a, float b, uint64_t seed, uint64_t offset) {
  // n8 = number of 8-element groups; 32-bit indexing (n < 2^31 groups).
  constexpr int G = EPT / 8;  // 16B stores per iteration
  const uint32_t stride = gridDim.x * blockDim.x;
  for (uint32_t g0 = blockIdx.x * blockDim.x + threadIdx.x; g0 * G < n8;
       g0 += stride) {
#pragma unroll
    for (int s = 0; s < G; ++s) {
      uint32_t g = g0 * G + s;
      if (g >= n8) break;
      float vals[8];
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        uint4 bits = philox10(seed, (uint64_t)g * 2 + c, offset);
        vals[c * 4 + 0] = fmaf(u2f(bits.x), b, a);
        vals[c * 4 + 1] = fmaf(u2f(bits.y), b, a);
        vals[c * 4 + 2] = fmaf(u2f(bits.z), b, a);
        vals[c * 4 + 3] = fmaf(u2f(bits.w), b, a);
      }
      V8 v;
#pragma unroll
      for (int j = 0; j < 8; ++j) v.v[j] = __float2bfloat16(vals[j]);
      *reinterpret_cast<V8*>(out + (uint64_t)g * 8) = v;
    }
  }
}

// Contiguous-per-thread variant: thread handles G consecutive groups.
template <int G>
__global__ void rng_bf16_wide(__hip_bfloat16* __restrict__ out, uint32_t n8,
                              float a, float b, uint64_t seed,
                              uint64_t offset) {
  const uint32_t stride = gridDim.x * blockDim.x;
  for (uint32_t t = blockIdx.x * blockDim.x + threadIdx.x; t * G < n8;
       t += stride) {
    uint4 bits[2 * G];
#pragma unroll
    for (int c = 0; c < 2 * G; ++c) {
      uint32_t g = t * G + c / 2;
      bits[c] = philox10(seed, (uint64_t)g * 2 + (c & 1), offset);
    }
#pragma unroll
    for (int s = 0; s < G; ++s) {
      uint32_t g = t * G + s;
      if (g >= n8) break;
      V8 v;
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        uint4 bb = bits[s * 2 + c];
        v.v[c * 4 + 0] = __float2bfloat16(fmaf(u2f(bb.x), b, a));
        v.v[c * 4 + 1] = __float2bfloat16(fmaf(u2f(bb.y), b, a));
        v.v[c * 4 + 2] = __float2bfloat16(fmaf(u2f(bb.z), b, a));
        v.v[c * 4 + 3] = __float2bfloat16(fmaf(u2f(bb.w), b, a));
      }
      *reinterpret_cast<V8*>(out + (uint64_t)g * 8) = v;
    }
  }
}

__device__ __forceinline__ float2 bm(float u1, float u2) {
  u1 = fmaxf(u1, 1.1754944e-38f);
  float r = sqrtf(-2.0f * __logf(u1));
  float sn, cs;
  __sincosf(6.2831853071795865f * u2, &sn, &cs);
  return make_float2(r * cs, r * sn);
}

// normal bf16: ONE philox -> 8 samples via 16-bit uniforms (kernel v2).
__global__ void normal_bf16_u16(__hip_bfloat16* __restrict__ out, uint32_t n8,
                                float a, float b, uint64_t seed,
                                uint64_t offset) {
  const uint32_t stride = gridDim.x * blockDim.x;
  for (uint32_t g = blockIdx.x * blockDim.x + threadIdx.x; g < n8;
       g += stride) {
    uint4 bits = philox10(seed, g, offset);
    uint32_t w[4] = {bits.x, bits.y, bits.z, bits.w};
    V8 v;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float2 nj = bm((float)(w[j] & 0xffffu) * (1.0f / 65536.0f),
                     (float)(w[j] >> 16) * (1.0f / 65536.0f));
      v.v[j * 2 + 0] = __float2bfloat16(fmaf(nj.x, b, a));
      v.v[j * 2 + 1] = __float2bfloat16(fmaf(nj.y, b, a));
    }
    *reinterpret_cast<V8*>(out + (uint64_t)g * 8) = v;
  }
}

// Philox-round sweep of the shipped v2-log2 shape.
template <int R>
__global__ void normal_bf16_rounds(__hip_bfloat16* __restrict__ out,
                                   uint32_t n8, float a, float b,
                                   uint64_t seed, uint64_t offset) {
  const uint32_t stride = gridDim.x * blockDim.x;
  for (uint32_t g = blockIdx.x * blockDim.x + threadIdx.x; g < n8;
       g += stride) {
    uint4 bits = philoxR<R>(seed, g, offset);
    uint32_t w[4] = {bits.x, bits.y, bits.z, bits.w};
    V8 v;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float2 nj = bm((float)(w[j] & 0xffffu) * (1.0f / 65536.0f),
                     (float)(w[j] >> 16) * (1.0f / 65536.0f));
      v.v[j * 2 + 0] = __float2bfloat16(fmaf(nj.x, b, a));
      v.v[j * 2 + 1] = __float2bfloat16(fmaf(nj.y, b, a));
    }
    *reinterpret_cast<V8*>(out + (uint64_t)g * 8) = v;
  }
}

template <int R>
__global__ void uniform_bf16_rounds(__hip_bfloat16* __restrict__ out,
                                    uint32_t n8, float a, float b,
                                    uint64_t seed, uint64_t offset) {
  const uint32_t stride = gridDim.x * blockDim.x;
  for (uint32_t g = blockIdx.x * blockDim.x + threadIdx.x; g < n8;
       g += stride) {
    uint4 bits = philoxR<R>(seed, g, offset);
    uint32_t w[4] = {bits.x, bits.y, bits.z, bits.w};
    V8 v;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      v.v[j * 2 + 0] = __float2bfloat16(
          fmaf((float)(w[j] & 0xffffu) * (1.0f / 65536.0f), b, a));
      v.v[j * 2 + 1] = __float2bfloat16(
          fmaf((float)(w[j] >> 16) * (1.0f / 65536.0f), b, a));
    }
    *reinterpret_cast<V8*>(out + (uint64_t)g * 8) = v;
  }
}

// normal bf16 with TWO philox and 24-bit uniforms (kernel v1 style).
__global__ void normal_bf16_u32(__hip_bfloat16* __restrict__ out, uint32_t n8,
                                float a, float b, uint64_t seed,
                                uint64_t offset) {
  const uint32_t stride = gridDim.x * blockDim.x;
  for (uint32_t g = blockIdx.x * blockDim.x + threadIdx.x; g < n8;
       g += stride) {
    V8 v;
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      uint4 bits = philox10(seed, (uint64_t)g * 2 + c, offset);
      float2 n01 = bm(u2f(bits.x), u2f(bits.y));
      float2 n23 = bm(u2f(bits.z), u2f(bits.w));
      v.v[c * 4 + 0] = __float2bfloat16(fmaf(n01.x, b, a));
      v.v[c * 4 + 1] = __float2bfloat16(fmaf(n01.y, b, a));
      v.v[c * 4 + 2] = __float2bfloat16(fmaf(n23.x, b, a));
      v.v[c * 4 + 3] = __float2bfloat16(fmaf(n23.y, b, a));
    }
    *reinterpret_cast<V8*>(out + (uint64_t)g * 8) = v;
  }
}

// uniform bf16 v2: ONE philox -> 8 samples via 16-bit uniforms.
__global__ void uniform_bf16_u16(__hip_bfloat16* __restrict__ out,
                                 uint32_t n8, float a, float b, uint64_t seed,
                                 uint64_t offset) {
  const uint32_t stride = gridDim.x * blockDim.x;
  for (uint32_t g = blockIdx.x * blockDim.x + threadIdx.x; g < n8;
       g += stride) {
    uint4 bits = philox10(seed, g, offset);
    uint32_t w[4] = {bits.x, bits.y, bits.z, bits.w};
    V8 v;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      v.v[j * 2 + 0] =
          __float2bfloat16(fmaf((float)(w[j] & 0xffffu) * (1.0f / 65536.0f), b, a));
      v.v[j * 2 + 1] =
          __float2bfloat16(fmaf((float)(w[j] >> 16) * (1.0f / 65536.0f), b, a));
    }
    *reinterpret_cast<V8*>(out + (uint64_t)g * 8) = v;
  }
}

// ILP variant: two independent groups per iteration so the transcendental
// chains of two Box-Muller batches interleave.
__global__ void normal_bf16_u16x2(__hip_bfloat16* __restrict__ out,
                                  uint32_t n8, float a, float b,
                                  uint64_t seed, uint64_t offset) {
  const uint32_t stride = gridDim.x * blockDim.x;
  const uint32_t half = (n8 + 1) / 2;
  for (uint32_t t = blockIdx.x * blockDim.x + threadIdx.x; t < half;
       t += stride) {
    uint32_t g0 = t;
    uint32_t g1 = t + half;
    uint4 b0 = philox10(seed, g0, offset);
    uint4 b1 = philox10(seed, g1, offset);
    uint32_t w0[4] = {b0.x, b0.y, b0.z, b0.w};
    uint32_t w1[4] = {b1.x, b1.y, b1.z, b1.w};
    V8 v0, v1;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float2 n0 = bm((float)(w0[j] & 0xffffu) * (1.0f / 65536.0f),
                     (float)(w0[j] >> 16) * (1.0f / 65536.0f));
      float2 n1 = bm((float)(w1[j] & 0xffffu) * (1.0f / 65536.0f),
                     (float)(w1[j] >> 16) * (1.0f / 65536.0f));
      v0.v[j * 2 + 0] = __float2bfloat16(fmaf(n0.x, b, a));
      v0.v[j * 2 + 1] = __float2bfloat16(fmaf(n0.y, b, a));
      v1.v[j * 2 + 0] = __float2bfloat16(fmaf(n1.x, b, a));
      v1.v[j * 2 + 1] = __float2bfloat16(fmaf(n1.y, b, a));
    }
    *reinterpret_cast<V8*>(out + (uint64_t)g0 * 8) = v0;
    if (g1 < n8) {
      *reinterpret_cast<V8*>(out + (uint64_t)g1 * 8) = v1;
    }
  }
}

// Cheap-log variant: ln(u) for u = m * 2^-16 with integer m>=1 can be
// computed from the float bit pattern: ln(u) = (log2(m) - 16) * ln2, and
// log2(m) = exponent + log2(mantissa); v_log_f32 IS that — so instead try
// replacing sincos with a single sin via angle-sum identity? Keep simple:
// variant replacing __logf with __log2f * ln2 (same instr count, sanity).
__global__ void normal_bf16_log2(__hip_bfloat16* __restrict__ out,
                                 uint32_t n8, float a, float b,
                                 uint64_t seed, uint64_t offset) {
  const uint32_t stride = gridDim.x * blockDim.x;
  for (uint32_t g = blockIdx.x * blockDim.x + threadIdx.x; g < n8;
       g += stride) {
    uint4 bits = philox10(seed, g, offset);
    uint32_t w[4] = {bits.x, bits.y, bits.z, bits.w};
    V8 v;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float u1 = fmaxf((float)(w[j] & 0xffffu) * (1.0f / 65536.0f),
                       1.1754944e-38f);
      float r = sqrtf(-1.3862943611f * __log2f(u1));
      float sn, cs;
      __sincosf(6.2831853071795865f * ((float)(w[j] >> 16) * (1.0f / 65536.0f)),
                &sn, &cs);
      v.v[j * 2 + 0] = __float2bfloat16(fmaf(r * cs, b, a));
      v.v[j * 2 + 1] = __float2bfloat16(fmaf(r * sn, b, a));
    }
    *reinterpret_cast<V8*>(out + (uint64_t)g * 8) = v;
  }
}

// Split-phase x2: both philox results first, then all transcendental
// chains back-to-back so the scheduler can interleave them.
__global__ void normal_bf16_split2(__hip_bfloat16* __restrict__ out,
                                   uint32_t n8, float a, float b,
                                   uint64_t seed, uint64_t offset) {
  const uint32_t stride = gridDim.x * blockDim.x;
  const uint32_t half = (n8 + 1) / 2;
  for (uint32_t t = blockIdx.x * blockDim.x + threadIdx.x; t < half;
       t += stride) {
    uint32_t g0 = t, g1 = t + half;
    uint4 b0 = philox10(seed, g0, offset);
    uint4 b1 = philox10(seed, g1, offset);
    uint32_t w[8] = {b0.x, b0.y, b0.z, b0.w, b1.x, b1.y, b1.z, b1.w};
    float r[8], sn[8], cs[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float u1 = fmaxf((float)(w[j] & 0xffffu) * (1.0f / 65536.0f),
                       1.1754944e-38f);
      r[j] = sqrtf(-1.3862943611f * __log2f(u1));
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      __sincosf(6.2831853071795865f * ((float)(w[j] >> 16) * (1.0f / 65536.0f)),
                &sn[j], &cs[j]);
    }
    V8 v0, v1;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      v0.v[j * 2 + 0] = __float2bfloat16(fmaf(r[j] * cs[j], b, a));
      v0.v[j * 2 + 1] = __float2bfloat16(fmaf(r[j] * sn[j], b, a));
      v1.v[j * 2 + 0] = __float2bfloat16(fmaf(r[4 + j] * cs[4 + j], b, a));
      v1.v[j * 2 + 1] = __float2bfloat16(fmaf(r[4 + j] * sn[4 + j], b, a));
    }
    *reinterpret_cast<V8*>(out + (uint64_t)g0 * 8) = v0;
    if (g1 < n8) {
      *reinterpret_cast<V8*>(out + (uint64_t)g1 * 8) = v1;
    }
  }
}

// Combined candidate: rounds=7 Philox + log2 Box-Muller (the two wins
// measured separately in round 1, never together).
template <int R>
__global__ void normal_bf16_rlog2(__hip_bfloat16* __restrict__ out,
                                  uint32_t n8, float a, float b,
                                  uint64_t seed, uint64_t offset) {
  const uint32_t stride = gridDim.x * blockDim.x;
  for (uint32_t g = blockIdx.x * blockDim.x + threadIdx.x; g < n8;
       g += stride) {
    uint4 bits = philoxR<R>(seed, g, offset);
    uint32_t w[4] = {bits.x, bits.y, bits.z, bits.w};
    V8 v;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float u1 = fmaxf((float)(w[j] & 0xffffu) * (1.0f / 65536.0f),
                       1.1754944e-38f);
      float r = sqrtf(-1.3862943611f * __log2f(u1));
      float sn, cs;
      __sincosf(
          6.2831853071795865f * ((float)(w[j] >> 16) * (1.0f / 65536.0f)),
          &sn, &cs);
      v.v[j * 2 + 0] = __float2bfloat16(fmaf(r * cs, b, a));
      v.v[j * 2 + 1] = __float2bfloat16(fmaf(r * sn, b, a));
    }
    *reinterpret_cast<V8*>(out + (uint64_t)g * 8) = v;
  }
}

// r7 + log2 + split-phase x2 (transcendental chains of two groups
// back-to-back for ILP).
template <int R>
__global__ void normal_bf16_rsplit2(__hip_bfloat16* __restrict__ out,
                                    uint32_t n8, float a, float b,
                                    uint64_t seed, uint64_t offset) {
  const uint32_t stride = gridDim.x * blockDim.x;
  const uint32_t half = (n8 + 1) / 2;
  for (uint32_t t = blockIdx.x * blockDim.x + threadIdx.x; t < half;
       t += stride) {
    uint32_t g0 = t, g1 = t + half;
    uint4 b0 = philoxR<R>(seed, g0, offset);
    uint4 b1 = philoxR<R>(seed, g1, offset);
    uint32_t w[8] = {b0.x, b0.y, b0.z, b0.w, b1.x, b1.y, b1.z, b1.w};
    float r[8], sn[8], cs[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float u1 = fmaxf((float)(w[j] & 0xffffu) * (1.0f / 65536.0f),
                       1.1754944e-38f);
      r[j] = sqrtf(-1.3862943611f * __log2f(u1));
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      __sincosf(
          6.2831853071795865f * ((float)(w[j] >> 16) * (1.0f / 65536.0f)),
          &sn[j], &cs[j]);
    }
    V8 v0, v1;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      v0.v[j * 2 + 0] = __float2bfloat16(fmaf(r[j] * cs[j], b, a));
      v0.v[j * 2 + 1] = __float2bfloat16(fmaf(r[j] * sn[j], b, a));
      v1.v[j * 2 + 0] = __float2bfloat16(fmaf(r[4 + j] * cs[4 + j], b, a));
      v1.v[j * 2 + 1] = __float2bfloat16(fmaf(r[4 + j] * sn[4 + j], b, a));
    }
    *reinterpret_cast<V8*>(out + (uint64_t)g0 * 8) = v0;
    if (g1 < n8) {
      *reinterpret_cast<V8*>(out + (uint64_t)g1 * 8) = v1;
    }
  }
}

template <typename K>
double benchB(K kernel, __hip_bfloat16* buf, uint32_t n8, int blocks,
              int iters, int threads) {
  hipEvent_t e0, e1;
  hipEventCreate(&e0);
  hipEventCreate(&e1);
  kernel<<<blocks, threads>>>(buf, n8, -1.f, 2.f, 42, 4);  // warm
  hipDeviceSynchronize();
  hipEventRecord(e0);
  for (int i = 0; i < iters; ++i) {
    kernel<<<blocks, threads>>>(buf, n8, -1.f, 2.f, 42, 4);
  }
  hipEventRecord(e1);
  hipEventSynchronize(e1);
  float ms;
  hipEventElapsedTime(&ms, e0, e1);
  double bytes = (double)n8 * 16.0 * iters;
  return bytes / (ms / 1e3) / 1e12;  // TB/s
}

template <typename K>
double bench(K kernel, __hip_bfloat16* buf, uint32_t n8, int blocks,
             int iters) {
  hipEvent_t e0, e1;
  hipEventCreate(&e0);
  hipEventCreate(&e1);
  kernel<<<blocks, 256>>>(buf, n8, -1.f, 2.f, 42, 4);  // warm
  hipDeviceSynchronize();
  hipEventRecord(e0);
  for (int i = 0; i < iters; ++i) {
    kernel<<<blocks, 256>>>(buf, n8, -1.f, 2.f, 42, 4);
  }
  hipEventRecord(e1);
  hipEventSynchronize(e1);
  float ms;
  hipEventElapsedTime(&ms, e0, e1);
  double bytes = (double)n8 * 16.0 * iters;
  return bytes / (ms / 1e3) / 1e12;  // TB/s
}

int main() {
  const uint64_t bytes = 4ull << 30;
  const uint32_t n8 = (uint32_t)(bytes / 16);
  __hip_bfloat16* buf;
  CHECK(hipMalloc(&buf, bytes));

  for (int blocks : {4096, 8192, 16384}) {
    printf("blocks=%d\n", blocks);
    printf("  uniform 2xphilox (v1): %.2f TB/s\n", bench(rng_bf16<8>, buf, n8, blocks, 5));
    printf("  uniform 1xphilox (v2): %.2f TB/s\n", bench(uniform_bf16_u16, buf, n8, blocks, 5));
    printf("  normal  2xphilox (v1): %.2f TB/s\n", bench(normal_bf16_u32, buf, n8, blocks, 5));
    printf("  normal  1xphilox (v2): %.2f TB/s\n", bench(normal_bf16_u16, buf, n8, blocks, 5));
    printf("  normal  v2 ILPx2     : %.2f TB/s\n", bench(normal_bf16_u16x2, buf, n8, blocks, 5));
    printf("  normal  v2 log2      : %.2f TB/s\n", bench(normal_bf16_log2, buf, n8, blocks, 5));
    printf("  normal  split2 t256  : %.2f TB/s\n", benchB(normal_bf16_split2, buf, n8, blocks, 5, 256));
    printf("  normal  split2 t512  : %.2f TB/s\n", benchB(normal_bf16_split2, buf, n8, blocks/2, 5, 512));
    printf("  normal  log2 t512    : %.2f TB/s\n", benchB(normal_bf16_log2, buf, n8, blocks/2, 5, 512));
    printf("  normal  log2 t128    : %.2f TB/s\n", benchB(normal_bf16_log2, buf, n8, blocks*2, 5, 128));
    printf("  normal  rounds=10    : %.2f TB/s\n", bench(normal_bf16_rounds<10>, buf, n8, blocks, 5));
    printf("  normal  rounds=7     : %.2f TB/s\n", bench(normal_bf16_rounds<7>, buf, n8, blocks, 5));
    printf("  uniform rounds=10    : %.2f TB/s\n", bench(uniform_bf16_rounds<10>, buf, n8, blocks, 5));
    printf("  uniform rounds=7     : %.2f TB/s\n", bench(uniform_bf16_rounds<7>, buf, n8, blocks, 5));
    printf("  normal  r7+log2      : %.2f TB/s\n", bench(normal_bf16_rlog2<7>, buf, n8, blocks, 5));
    printf("  normal  r7+log2 t512 : %.2f TB/s\n", benchB(normal_bf16_rlog2<7>, buf, n8, blocks/2, 5, 512));
    printf("  normal  r7+split2    : %.2f TB/s\n", bench(normal_bf16_rsplit2<7>, buf, n8, blocks, 5));
    printf("  normal  r7+split2 512: %.2f TB/s\n", benchB(normal_bf16_rsplit2<7>, buf, n8, blocks/2, 5, 512));
  }
  // memset reference ceiling
  hipEvent_t e0, e1;
  hipEventCreate(&e0);
  hipEventCreate(&e1);
  hipEventRecord(e0);
  for (int i = 0; i < 5; ++i) hipMemsetAsync(buf, 0, bytes);
  hipEventRecord(e1);
  hipEventSynchronize(e1);
  float ms;
  hipEventElapsedTime(&ms, e0, e1);
  printf("hipMemset      : %.2f TB/s\n", (double)bytes * 5 / (ms / 1e3) / 1e12);
  hipFree(buf);
  return 0;
}
