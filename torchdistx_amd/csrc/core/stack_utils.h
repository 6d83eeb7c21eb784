// Boxed-stack tensor visitation helpers for the fake / deferred-init
// dispatch handlers.
//
// Capability parity with the reference stack utilities
// (/root/reference/src/cc/torchdistx/stack_utils.{h,cc}): iterate and
// convert every tensor in a torch::jit::Stack argument region, including
// tensors nested inside TensorLists / optional lists, in a stable visit
// order shared by recording and replay.

#pragma once

#include <functional>

#include <ATen/core/ivalue.h>
#include <ATen/core/stack.h>

namespace tdx {

// Visits every tensor (defined or not) reachable from the IValues in
// [begin, end) of `stack`, in deterministic order: stack slot order, and
// within a (possibly nested) list, element order. The visitor may mutate the
// tensor in place via the returned reference semantics of the setter form.
using TensorVisitor = std::function<void(const at::Tensor&)>;

// Read-only visit.
void visitTensors(const torch::jit::Stack& stack,
                  size_t begin,
                  size_t end,
                  const TensorVisitor& visitor);

// Mutating visit: `fn` receives each tensor and returns its replacement
// (return the input unchanged to keep it). Lists are updated in place.
using TensorMapFn = std::function<at::Tensor(const at::Tensor&)>;

void mapTensors(torch::jit::Stack& stack,
                size_t begin,
                size_t end,
                const TensorMapFn& fn);

// Deep-copies the IValues in [begin, end): container IValues (lists,
// tuples, dicts) are copied structurally so later caller-side mutation of
// the originals cannot corrupt a recorded call frame; tensors and other
// leaf values are copied by reference. Throws if a value of a mutable,
// un-recordable type (e.g. a Storage, a Future) is encountered —
// mirroring the reference's immutable-stack-type validation
// (reference deferred_init.cc:227-254).
std::vector<c10::IValue> copyStackRegion(const torch::jit::Stack& stack,
                                         size_t begin,
                                         size_t end);

}  // namespace tdx
