// Replay-time redirect of recorded aten:: init ops to the hand-written
// CDNA4 kernels registered under the tdx:: namespace by the _K extension
// (csrc/hip/init_kernels.hip).
//
// The tape records the ops module constructors actually emit
// (aten::uniform_ / normal_ / fill_ / zero_); when the replay target lives
// on the GPU and the kernel extension is loaded, the replay engine calls
// the tdx:: implementation instead of the stock ATen kernel. This keeps
// torch's global op table untouched (no behavioral change for code outside
// this framework) while every materialization path runs the native
// kernels.

#pragma once

#include <cstdint>
#include <optional>
#include <utility>

#include <ATen/core/dispatch/Dispatcher.h>
#include <ATen/core/stack.h>

namespace tdx {

// Attempts to run `op`'s recorded call through a tdx:: kernel. Returns
// true when the redirect ran (outputs are on the stack); false when the
// caller should fall through to the stock dispatch. Throws when the
// TDX_REQUIRE_NATIVE_INIT environment variable is set, the op is
// redirectable, the tensor is on the GPU, and the kernel extension is not
// loaded (fail-loud mode for GPU CI).
bool tryNativeInitRedirect(
    const c10::OperatorHandle& op,
    torch::jit::Stack& stack,
    const std::optional<std::pair<uint64_t, uint64_t>>& philox = std::nullopt);

void setNativeInitEnabled(bool enabled) noexcept;
bool nativeInitEnabled() noexcept;

// Opt-in: also redirect CPU-target init ops to the tdx CPU reference
// implementations (pinned-Philox layout). Off by default so plain CPU
// replay stays bitwise-equal to eager construction; turn on when CPU
// materialization must be slice-consistent with materialize_tensor_shard.
void setNativeInitCpuEnabled(bool enabled) noexcept;
bool nativeInitCpuEnabled() noexcept;

}  // namespace tdx
