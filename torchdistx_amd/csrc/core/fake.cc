// See fake.h. Reference behavior map:
//   FakeTensorImpl / key-set computation   — reference fake.cc:69-150
//   shallow copy / detach semantics        — reference fake.cc:196-237
//   boxed Fake fallback                    — reference fake.cc:257-548
//   fake mode TLS + fake CUDA device guard — reference fake.cc:554-623

#include "fake.h"

#include <mutex>

#include <ATen/core/dispatch/Dispatcher.h>
#include <c10/core/DeviceGuard.h>
#include <c10/core/impl/DeviceGuardImplInterface.h>
#include <c10/core/impl/FakeGuardImpl.h>
#include <c10/core/impl/LocalDispatchKeySet.h>
#include <torch/library.h>

#include "stack_utils.h"

namespace tdx {
namespace {

constexpr c10::DispatchKey kFakeKey = c10::DispatchKey::Fake;

// An index-less "cuda" device resolves to the current device at allocation
// time in eager mode; fake tensors must report the same concrete index
// (FSDP and friends compare devices exactly).
c10::Device canonicalFakeDevice(c10::Device d) {
  if (d.is_cuda() && !d.has_index()) {
    const auto* guard =
        c10::impl::device_guard_impl_registry[static_cast<size_t>(
                                                  c10::DeviceType::CUDA)]
            .load();
    c10::DeviceIndex index = 0;
    if (guard != nullptr && guard->deviceCount() > 0) {
      index = guard->getDevice().index();
    }
    d = c10::Device{c10::DeviceType::CUDA, index};
  }
  return d;
}

c10::DispatchKeySet computeFakeKeySet(const at::Tensor& meta,
                                      c10::Device fake_device) {
  c10::DispatchKey backend =
      c10::computeDispatchKey(meta.scalar_type(), meta.layout(), fake_device);
  c10::BackendComponent component = c10::toBackendComponent(backend);
  auto ks = c10::DispatchKeySet{backend};
  ks = ks | c10::getAutogradRelatedKeySetFromBackend(component);
  ks = ks | c10::getAutocastRelatedKeySetFromBackend(component);
  return ks.add(kFakeKey);
}

}  // namespace

FakeTensorImpl::FakeTensorImpl(at::Tensor meta, c10::Device fake_device)
    : TensorImpl(computeFakeKeySet(meta, canonicalFakeDevice(fake_device)),
                 meta.dtype(),
                 canonicalFakeDevice(fake_device)),
      meta_{std::move(meta)},
      fake_device_{canonicalFakeDevice(fake_device)} {
  TORCH_INTERNAL_ASSERT(meta_.is_meta(),
                        "FakeTensorImpl requires a meta shadow tensor.");
  set_storage_access_should_throw();
  refreshFromMeta();
}

void FakeTensorImpl::refreshFromMeta() {
  set_sizes_and_strides(meta_.sizes(), meta_.strides());
  set_storage_offset(meta_.storage_offset());
  data_type_ = meta_.dtype();
}

template <typename VariableVersion>
c10::intrusive_ptr<c10::TensorImpl> FakeTensorImpl::shallowCopyAndDetachImpl(
    VariableVersion&& version_counter,
    bool allow_tensor_metadata_change) const {
  // Clone the shadow meta impl as well, so the detached fake tensor has an
  // independent meta tensor for metadata mutation (reference fake.cc:196-237).
  at::Tensor meta_copy{meta_.unsafeGetTensorImpl()->shallow_copy_and_detach(
      /*version_counter=*/0, /*allow_tensor_metadata_change=*/true)};
  auto impl = c10::make_intrusive<FakeTensorImpl>(std::move(meta_copy),
                                                  fake_device_);
  impl->set_version_counter(std::forward<VariableVersion>(version_counter));
  impl->set_allow_tensor_metadata_change(allow_tensor_metadata_change);
  return impl;
}

c10::intrusive_ptr<c10::TensorImpl> FakeTensorImpl::shallow_copy_and_detach(
    const c10::VariableVersion& version_counter,
    bool allow_tensor_metadata_change) const {
  return shallowCopyAndDetachImpl(version_counter,
                                  allow_tensor_metadata_change);
}

c10::intrusive_ptr<c10::TensorImpl> FakeTensorImpl::shallow_copy_and_detach(
    c10::VariableVersion&& version_counter,
    bool allow_tensor_metadata_change) const {
  return shallowCopyAndDetachImpl(std::move(version_counter),
                                  allow_tensor_metadata_change);
}

void FakeTensorImpl::shallow_copy_from(
    const c10::intrusive_ptr<TensorImpl>& impl) {
  auto* src = dynamic_cast<FakeTensorImpl*>(impl.get());
  TORCH_CHECK(src != nullptr,
              "`set_data` on a fake tensor requires the new value to be a "
              "fake tensor as well.");
  meta_ = src->meta_;
  fake_device_ = src->fake_device_;
  dispatch_data_ = src->dispatch_data_;
  key_set_ = src->key_set();
  device_opt_ = src->device();
  refreshFromMeta();
}

FakeTensorImpl* asFake(const at::Tensor& tensor) noexcept {
  if (!tensor.defined() ||
      !tensor.unsafeGetTensorImpl()->key_set().has(kFakeKey)) {
    return nullptr;
  }
  return dynamic_cast<FakeTensorImpl*>(tensor.unsafeGetTensorImpl());
}

bool isFake(const at::Tensor& tensor) noexcept {
  return asFake(tensor) != nullptr;
}

at::Tensor makeFake(at::Tensor meta, c10::Device fake_device) {
  return at::detail::make_tensor<FakeTensorImpl>(std::move(meta),
                                                 fake_device);
}

at::Tensor metaLike(const at::Tensor& fake) {
  auto* impl = asFake(fake);
  TORCH_CHECK_VALUE(impl != nullptr,
                    "`fake` was expected to be a fake tensor.");
  return at::Tensor{impl->meta_tensor().unsafeGetTensorImpl()
                        ->shallow_copy_and_detach(
                            /*version_counter=*/0,
                            /*allow_tensor_metadata_change=*/true)};
}

// ---------------------------------------------------------------------------
// The boxed Fake fallback.
// ---------------------------------------------------------------------------

namespace {

bool isDeviceType(const c10::TypePtr& type) {
  if (type->kind() == c10::TypeKind::DeviceObjType) {
    return true;
  }
  if (type->kind() == c10::TypeKind::OptionalType) {
    return type->castRaw<c10::OptionalType>()->getElementType()->kind() ==
           c10::TypeKind::DeviceObjType;
  }
  return false;
}

// Per-call state of one fake dispatch.
struct FakeCall {
  const c10::OperatorHandle& op;
  torch::jit::Stack* stack;
  size_t args_begin;
  size_t num_args;

  bool has_tensor_arg = false;
  bool has_fake_arg = false;
  std::optional<c10::Device> tensor_device;    // first fake, else first real
  std::optional<size_t> device_arg_slot;       // schema slot of `device`
  std::optional<c10::Device> explicit_device;  // non-None value of that slot
  // meta impl -> fake tensor, for un-mapping in-place / pass-through returns.
  std::unordered_map<c10::TensorImpl*, at::Tensor> meta_to_fake;
};

// A `device` schema argument names the output device only for ops that
// actually consume one at dispatch time: a BackendSelect kernel, or a
// TensorOptions argument pack (the consecutive dtype/layout/device/
// pin_memory run codegen emits for factories). An argument that merely
// happens to be named `device` is not a factory device (reference
// fake.cc:370-414; documented rules 1-2 in the reference docs
// fake_tensor_and_deferred_init.rst:120-137).
bool opConsumesDeviceArg(const c10::OperatorHandle& op) {
  if (op.hasKernelForDispatchKey(c10::DispatchKey::BackendSelect)) {
    return true;
  }
  const auto& args = op.schema().arguments();
  for (size_t i = 0; i + 3 < args.size(); ++i) {
    if (args[i].name() == "dtype" && args[i + 1].name() == "layout" &&
        args[i + 2].name() == "device" &&
        args[i + 3].name() == "pin_memory") {
      return true;
    }
  }
  return false;
}

void assessOp(FakeCall& call) {
  const auto& schema = call.op.schema();

  visitTensors(
      *call.stack, call.args_begin, call.args_begin + call.num_args,
      [&](const at::Tensor& t) {
        if (!t.defined()) {
          return;
        }
        call.has_tensor_arg = true;
        if (isFake(t)) {
          call.has_fake_arg = true;
        }
        // 0-dim CPU tensors are exempt from the common-device rule:
        // they are ubiquitous as scalar operands of GPU expressions
        // (reference fake.cc:314-316, 346-368).
        if (t.dim() == 0 && t.device().is_cpu()) {
          return;
        }
        c10::Device d = t.device();
        if (!call.tensor_device.has_value()) {
          call.tensor_device = d;
        } else {
          TORCH_CHECK(*call.tensor_device == d, "`", schema.operator_name(),
                      "` was called on fake tensors with arguments spread "
                      "over two devices (", *call.tensor_device, " and ", d,
                      "); fake dispatch needs one common device (0-dim CPU "
                      "scalars excepted).");
        }
      });

  if (!opConsumesDeviceArg(call.op)) {
    return;
  }
  const auto& args = schema.arguments();
  for (size_t i = 0; i < args.size(); ++i) {
    if (args[i].name() == "device" && isDeviceType(args[i].type())) {
      call.device_arg_slot = i;
      const c10::IValue& v = (*call.stack)[call.args_begin + i];
      if (v.isDevice()) {
        call.explicit_device = v.toDevice();
      }
      break;
    }
  }
}

// Output-device heuristic (reference fake.cc:419-432; documented rules in
// the reference docs fake_tensor_and_deferred_init.rst:120-137):
//   1. an explicit `device` argument wins;
//   2. else the common device of the (fake-first) tensor arguments;
//   3. else CPU.
c10::Device outputDevice(const FakeCall& call) {
  if (call.explicit_device.has_value()) {
    return *call.explicit_device;
  }
  if (call.tensor_device.has_value()) {
    return *call.tensor_device;
  }
  return c10::Device{c10::DeviceType::CPU};
}

void fakeHandler(const c10::OperatorHandle& op,
                 c10::DispatchKeySet ks,
                 torch::jit::Stack* stack) {
  c10::impl::ExcludeDispatchKeyGuard no_reentry{kFakeKey};

  const auto& schema = op.schema();
  size_t num_args = schema.arguments().size();
  size_t num_rets = schema.returns().size();

  FakeCall call{op, stack, stack->size() - num_args, num_args};
  assessOp(call);

  bool should_fake = call.has_fake_arg || call.device_arg_slot.has_value() ||
                     !call.has_tensor_arg;
  if (!should_fake) {
    // Real tensors while fake mode is merely enabled: business as usual.
    op.redispatchBoxed(
        ks & c10::DispatchKeySet{c10::DispatchKeySet::FULL_AFTER, kFakeKey},
        stack);
    return;
  }

  c10::Device out_device = outputDevice(call);

  // The meta backend performs the actual (shape/stride/dtype) computation.
  TORCH_CHECK(op.hasComputedKernelForDispatchKey(c10::DispatchKey::Meta),
              "`", schema.operator_name(),
              "` has no meta-backend kernel, so it cannot run on fake "
              "tensors.");
  if (!out_device.is_meta()) {
    c10::DispatchKey backend_key = c10::computeDispatchKey(
        std::nullopt, std::nullopt, out_device);
    TORCH_CHECK(op.hasComputedKernelForDispatchKey(backend_key),
                "`", schema.operator_name(), "` has no kernel for '",
                out_device, "', so its fake result could never be "
                "materialized on that device.");
  }

  // Swap fake arguments for their meta shadows, remembering the mapping so
  // in-place and pass-through returns can be unmapped afterwards.
  mapTensors(*stack, call.args_begin, call.args_begin + num_args,
             [&](const at::Tensor& t) -> at::Tensor {
               if (auto* fake = asFake(t)) {
                 const at::Tensor& shadow = fake->meta_tensor();
                 call.meta_to_fake.emplace(shadow.unsafeGetTensorImpl(), t);
                 return shadow;
               }
               return t;
             });
  if (call.device_arg_slot.has_value()) {
    (*stack)[call.args_begin + *call.device_arg_slot] =
        c10::IValue{c10::Device{c10::DeviceType::Meta}};
  }

  op.redispatchBoxed(c10::DispatchKeySet{c10::DispatchKey::Meta}, stack);

  // Wrap returned meta tensors as fake tensors on the inferred device. An
  // in-place return (a meta impl we mapped on the way in) reuses its
  // original fake tensor with refreshed metadata.
  mapTensors(*stack, stack->size() - num_rets, stack->size(),
             [&](const at::Tensor& t) -> at::Tensor {
               if (!t.defined() || !t.is_meta()) {
                 return t;
               }
               auto it = call.meta_to_fake.find(t.unsafeGetTensorImpl());
               if (it != call.meta_to_fake.end()) {
                 asFake(it->second)->refreshFromMeta();
                 return it->second;
               }
               return makeFake(t, out_device);
             });
}

TORCH_LIBRARY_IMPL(_, Fake, m) {
  m.fallback(torch::CppFunction::makeFromBoxedFunction<&fakeHandler>());
}

// ---------------------------------------------------------------------------
// Fake mode TLS + fake CUDA device guard.
// ---------------------------------------------------------------------------

thread_local size_t fake_mode_level = 0;

// When the cuda/HIP device-guard slot is empty (CPU-only torch builds) a
// no-op guard is installed for the duration of fake mode so "cuda" Device
// objects and key sets can be constructed without a GPU (the reference's
// "we basically lie to PyTorch" trick, fake.cc:554-586). On PyTorch-ROCm the
// HIP guard is registered even without a visible GPU, so this is usually a
// no-op here and only matters for portability.
std::mutex guard_mutex;
size_t fake_cuda_entries = 0;
bool installed_fake_cuda_guard = false;

std::atomic<const c10::impl::DeviceGuardImplInterface*>& cudaGuardSlot() {
  return c10::impl::device_guard_impl_registry[static_cast<size_t>(
      c10::DeviceType::CUDA)];
}

void pushFakeCudaGuard() {
  std::lock_guard<std::mutex> lock{guard_mutex};
  if (fake_cuda_entries++ == 0 && cudaGuardSlot().load() == nullptr) {
    static c10::impl::FakeGuardImpl<c10::DeviceType::CUDA> noop_guard;
    cudaGuardSlot().store(&noop_guard);
    installed_fake_cuda_guard = true;
  }
}

void popFakeCudaGuard() {
  std::lock_guard<std::mutex> lock{guard_mutex};
  if (fake_cuda_entries > 0 && --fake_cuda_entries == 0 &&
      installed_fake_cuda_guard) {
    cudaGuardSlot().store(nullptr);
    installed_fake_cuda_guard = false;
  }
}

thread_local bool fake_cuda_active = false;

}  // namespace

void enterFakeMode(bool fake_cuda) {
  if (fake_mode_level++ == 0) {
    c10::impl::tls_set_dispatch_key_included(kFakeKey, true);
    if (fake_cuda) {
      pushFakeCudaGuard();
      fake_cuda_active = true;
    }
  }
}

void leaveFakeMode() {
  TORCH_CHECK(fake_mode_level > 0, "Not in fake mode.");
  if (--fake_mode_level == 0) {
    c10::impl::tls_set_dispatch_key_included(kFakeKey, false);
    if (fake_cuda_active) {
      popFakeCudaGuard();
      fake_cuda_active = false;
    }
  }
}

bool isFakeModeActive() noexcept {
  return fake_mode_level > 0;
}

}  // namespace tdx
