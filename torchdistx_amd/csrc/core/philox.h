// Philox4x32-10 (standard constants) shared between the CDNA4 kernels and
// the CPU reference implementation of the tdx init ops. Header-only and
// host/device-compilable: the integer pipeline and the uniform conversions
// are bit-identical wherever they run; transcendental transforms
// (Box-Muller) are intentionally NOT shared — each device uses its fastest
// implementation, and bitwise determinism is promised per device type, not
// across device types.

#pragma once

#include <cstdint>

#if defined(__HIPCC__)
#define TDX_HD __host__ __device__ __forceinline__
#else
#define TDX_HD inline
#endif

namespace tdx {
namespace philox {

struct U4 {
  uint32_t x, y, z, w;
};

TDX_HD uint32_t mulhi32(uint32_t a, uint32_t b) {
#if defined(__HIP_DEVICE_COMPILE__)
  return __umulhi(a, b);
#else
  return static_cast<uint32_t>(
      (static_cast<uint64_t>(a) * static_cast<uint64_t>(b)) >> 32);
#endif
}

// Round-count-templated Philox4x32. The uniform path uses the standard
// 10 rounds (it is store-bound, extra rounds are free); the normal path
// uses 7 rounds — Philox4x32-7 passes the full BigCrush battery (Salmon
// et al., "Parallel random numbers: as easy as 1, 2, 3", SC'11, Table 2;
// 10 is the authors' safety margin) and the normal kernel is VALU-bound,
// where the 3 saved rounds are a measured ~13% fill-rate win on gfx950
// (profiles/rng_tune_r7.log).
template <int kRounds>
TDX_HD U4 philox(uint64_t seed, uint64_t subsequence, uint64_t offset) {
  constexpr uint32_t kW0 = 0x9E3779B9u;
  constexpr uint32_t kW1 = 0xBB67AE85u;
  constexpr uint32_t kM0 = 0xD2511F53u;
  constexpr uint32_t kM1 = 0xCD9E8D57u;

  uint32_t k0 = static_cast<uint32_t>(seed);
  uint32_t k1 = static_cast<uint32_t>(seed >> 32);
  U4 c{static_cast<uint32_t>(offset), static_cast<uint32_t>(offset >> 32),
       static_cast<uint32_t>(subsequence),
       static_cast<uint32_t>(subsequence >> 32)};
#if defined(__HIP_DEVICE_COMPILE__)
#pragma unroll
#endif
  for (int round = 0; round < kRounds; ++round) {
    uint32_t lo0 = kM0 * c.x;
    uint32_t hi0 = mulhi32(kM0, c.x);
    uint32_t lo1 = kM1 * c.z;
    uint32_t hi1 = mulhi32(kM1, c.z);
    c = U4{hi1 ^ c.y ^ k0, lo1, hi0 ^ c.w ^ k1, lo0};
    k0 += kW0;
    k1 += kW1;
  }
  return c;
}

TDX_HD U4 philox10(uint64_t seed, uint64_t subsequence, uint64_t offset) {
  return philox<10>(seed, subsequence, offset);
}

TDX_HD U4 philox7(uint64_t seed, uint64_t subsequence, uint64_t offset) {
  return philox<7>(seed, subsequence, offset);
}

// uint32 -> [0, 1) float, 24-bit resolution (fp32 outputs).
TDX_HD float u32_to_uniform(uint32_t x) {
  return static_cast<float>(x >> 8) * (1.0f / 16777216.0f);
}

// uint16 -> [0, 1) float, 16-bit resolution (bf16/fp16 outputs).
TDX_HD float u16_to_uniform(uint32_t x) {
  return static_cast<float>(x & 0xffffu) * (1.0f / 65536.0f);
}

}  // namespace philox
}  // namespace tdx
