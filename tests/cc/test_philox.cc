// Standalone C++ unit tests for the shared Philox header (the integer
// pipeline both the CPU reference impls and the CDNA4 kernels compile).
// No torch dependency: built and run by tests/test_cc_units.py and by
// scripts/ci_local.sh with a plain g++.
//
// (The reference repo planned C++ unit tests but never added them —
// reference CMakeLists.txt:104-106 "#TODO: Add catch2 tests.")

#include <cmath>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <set>

#include "../../torchdistx_amd/csrc/core/philox.h"

static int failures = 0;
#define CHECK(cond)                                                     \
  do {                                                                  \
    if (!(cond)) {                                                      \
      std::printf("FAIL %s:%d: %s\n", __FILE__, __LINE__, #cond);       \
      ++failures;                                                       \
    }                                                                   \
  } while (0)

using tdx::philox::philox10;
using tdx::philox::philox7;
using tdx::philox::u16_to_uniform;
using tdx::philox::u32_to_uniform;

int main() {
  // Determinism + sensitivity: same inputs same bits; any input change
  // flips the output.
  auto a = philox10(42, 7, 3);
  auto b = philox10(42, 7, 3);
  CHECK(a.x == b.x && a.y == b.y && a.z == b.z && a.w == b.w);
  auto c = philox10(42, 8, 3);
  CHECK(a.x != c.x || a.y != c.y || a.z != c.z || a.w != c.w);
  auto d = philox10(43, 7, 3);
  CHECK(a.x != d.x || a.y != d.y);
  auto e = philox10(42, 7, 4);
  CHECK(a.x != e.x || a.y != e.y);

  // 7 and 10 rounds are different generators (the normal path uses 7).
  auto r7 = philox7(42, 7, 3);
  CHECK(r7.x != a.x || r7.y != a.y);

  // No short-cycle / trivial collisions over a counter sweep.
  std::set<uint64_t> seen;
  for (uint64_t g = 0; g < 4096; ++g) {
    auto v = philox10(123456789, g, 0);
    seen.insert((static_cast<uint64_t>(v.x) << 32) | v.y);
  }
  CHECK(seen.size() == 4096);

  // Uniform conversions: ranges and resolutions.
  CHECK(u32_to_uniform(0) == 0.0f);
  CHECK(u32_to_uniform(0xffffffffu) < 1.0f);
  CHECK(u32_to_uniform(0xffffffffu) > 0.9999f);
  CHECK(u16_to_uniform(0) == 0.0f);
  CHECK(u16_to_uniform(0xffffu) < 1.0f);
  // 24-bit resolution: consecutive high-byte values map to distinct floats.
  CHECK(u32_to_uniform(0x100u) != u32_to_uniform(0x200u));

  // Statistics over 1M draws from the 7-round generator (the normal
  // path): mean ~0.5, variance ~1/12.
  double sum = 0.0, sq = 0.0;
  const int n = 1 << 20;
  for (int g = 0; g < n / 4; ++g) {
    auto v = philox7(99, static_cast<uint64_t>(g), 4);
    for (uint32_t w : {v.x, v.y, v.z, v.w}) {
      double u = u32_to_uniform(w);
      sum += u;
      sq += u * u;
    }
  }
  const double mean = sum / n;
  const double var = sq / n - mean * mean;
  CHECK(std::fabs(mean - 0.5) < 0.002);
  CHECK(std::fabs(var - 1.0 / 12.0) < 0.002);

  if (failures == 0) {
    std::printf("philox unit tests: all passed\n");
    return 0;
  }
  std::printf("philox unit tests: %d failures\n", failures);
  return 1;
}
