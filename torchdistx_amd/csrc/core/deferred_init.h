// Deferred module initialization: record every op performed on fake tensors
// on an in-memory tape, then replay the tape to materialize real tensors.
//
// Capability parity with the reference deferred-init core
// (/root/reference/src/cc/torchdistx/deferred_init.{h,cc}): a boxed fallback
// on the pre-autograd `DeferredInit` dispatch key, an Op/OpNode tape with
// chronological (op_nr) ordering, storage-alias-aware in-place replay,
// view keep-alives, external-tensor version-counter checks, a
// VariableHooks proxy recording `.data` reads/writes, and identity-stable
// repeated materialization.

#pragma once

#include <ATen/ATen.h>

namespace tdx {

// TLS-scoped deferred-init mode; nestable.
void enterDeferredInit();
void leaveDeferredInit();
bool isDeferredInitActive() noexcept;

// RAII: temporarily disable deferred-init recording on this thread.
class NoDeferredInit {
 public:
  NoDeferredInit();
  ~NoDeferredInit();
  NoDeferredInit(const NoDeferredInit&) = delete;
  NoDeferredInit& operator=(const NoDeferredInit&) = delete;

 private:
  bool prev_;
};

// True when `tensor` is a fake tensor carrying a deferred-init record (i.e.
// materializeTensor can reconstruct it).
bool canMaterialize(const at::Tensor& tensor) noexcept;

// Replays the relevant tape segment and returns the real tensor. Identity
// rules: a non-fake input is returned as-is; repeated calls (and aliased
// fakes) yield the same underlying TensorImpl. Raises c10::ValueError for a
// fake tensor with no record.
at::Tensor materializeTensor(const at::Tensor& tensor);

// Introspection of a deferred tensor's tape record (observability beyond
// the reference, which exposes nothing). All fields are cheap to compute
// except pending_ops, which runs the same call-stack analysis
// materialization would.
struct RecordInfo {
  uint64_t op_nr = 0;          // tape position of the producing op
  size_t output_index = 0;     // which output of that op
  bool materialized = false;   // producer already replayed?
  std::string op_name;         // producer op (empty once replayed)
  size_t pending_ops = 0;      // ops a materialize call would replay now
};

// std::nullopt when `tensor` carries no deferred-init record.
std::optional<RecordInfo> recordInfo(const at::Tensor& tensor);

// Reduced replay plan of a simple init chain (see tensorInitPlan).
struct InitPlan {
  enum class Kind { kFactory, kUniform, kNormal, kBernoulli, kFill, kZero };
  Kind kind = Kind::kFactory;
  double p0 = 0.0;
  double p1 = 0.0;
  uint64_t seed = 0;
  uint64_t offset = 0;
  std::vector<int64_t> sizes;
  c10::ScalarType dtype = c10::ScalarType::Float;
  c10::Device device{c10::DeviceType::CPU};
  bool requires_grad = false;
};

std::optional<InitPlan> tensorInitPlan(const at::Tensor& tensor);

// Slice materialization: materializes indices [start_row, end_row) of
// the deferred tensor along `dim` WITHOUT materializing the rest,
// bitwise-equal to the corresponding slice of a full materialization
// (per device type). dim 0 serves FSDP `Shard(0)` and column-parallel
// weights; dim 1 serves row-parallel (Megatron-style TP) weights.
// Requires the tensor's tape to be a "simple init chain" — a factory
// (empty/zeros/ones/full) followed by whole-tensor in-place init ops
// (uniform_/normal_/fill_/zero_) and aliasing pass-throughs
// (detach/variable_data) — which is exactly what module constructors
// record; throws a descriptive error otherwise so callers can fall back
// to full materialization. This is the FSDP/TP init primitive: N ranks
// materialize disjoint slices of a model larger than any single device,
// with zero communication.
at::Tensor materializeTensorShard(const at::Tensor& tensor,
                                  int64_t start_row,
                                  int64_t end_row,
                                  int64_t dim = 0);

}  // namespace tdx
