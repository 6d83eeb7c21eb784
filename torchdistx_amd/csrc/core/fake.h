// Fake tensors: storage-less tensors that report a real device.
//
// Capability parity with the reference fake-tensor core
// (/root/reference/src/cc/torchdistx/fake.{h,cc}): a TensorImpl subclass
// holding a shadow *meta* tensor used for all shape/dtype computation, a
// boxed fallback on the (post-autograd) `Fake` dispatch key that diverts
// every op to PyTorch's meta backend and re-wraps the results, a TLS fake
// mode, a per-dispatch-key side table used by the deferred-init tape, and a
// "fake device" escape hatch so fake cuda/hip tensors can be built on
// GPU-less CI machines.
//
// MI355X-native redesign notes: this build targets PyTorch-ROCm, where
// `cuda` IS the HIP backend. torch 2.10 still reserves the out-of-core
// `DispatchKey::Fake` / `DispatchKey::DeferredInit` slots
// (c10/core/DispatchKey.h:230,387), so no private-key tricks are needed.

#pragma once

#include <memory>
#include <unordered_map>

#include <ATen/ATen.h>
#include <c10/core/TensorImpl.h>

namespace tdx {

// A tensor impl with no storage that reports `fake_device` while delegating
// every dispatched computation to its shadow meta tensor.
class FakeTensorImpl : public c10::TensorImpl {
 public:
  FakeTensorImpl(at::Tensor meta, c10::Device fake_device);

  const at::Tensor& meta_tensor() const noexcept {
    return meta_;
  }

  c10::Device fake_device() const noexcept {
    return fake_device_;
  }

  // Re-syncs sizes / strides / dtype / storage-offset from the shadow meta
  // tensor after an in-place meta op may have changed them.
  void refreshFromMeta();

  // Per-dispatch-key side data; the deferred-init tape stores its
  // TensorRecord here (reference fake.cc:118-121, deferred_init.cc:699-711).
  void setData(c10::DispatchKey key, std::shared_ptr<void> data) {
    dispatch_data_[key] = std::move(data);
  }
  std::shared_ptr<void> getData(c10::DispatchKey key) const {
    auto it = dispatch_data_.find(key);
    return it == dispatch_data_.end() ? nullptr : it->second;
  }
  bool hasData(c10::DispatchKey key) const {
    return dispatch_data_.count(key) != 0;
  }

  c10::intrusive_ptr<TensorImpl> shallow_copy_and_detach(
      const c10::VariableVersion& version_counter,
      bool allow_tensor_metadata_change) const override;
  c10::intrusive_ptr<TensorImpl> shallow_copy_and_detach(
      c10::VariableVersion&& version_counter,
      bool allow_tensor_metadata_change) const override;
  void shallow_copy_from(const c10::intrusive_ptr<TensorImpl>& impl) override;

 private:
  template <typename VariableVersion>
  c10::intrusive_ptr<TensorImpl> shallowCopyAndDetachImpl(
      VariableVersion&& version_counter,
      bool allow_tensor_metadata_change) const;

  at::Tensor meta_;
  c10::Device fake_device_;
  std::unordered_map<c10::DispatchKey, std::shared_ptr<void>> dispatch_data_;
};

// nullptr when `tensor` is not fake.
FakeTensorImpl* asFake(const at::Tensor& tensor) noexcept;

bool isFake(const at::Tensor& tensor) noexcept;

// Wraps `meta` (a meta tensor) as a new fake tensor on `fake_device`.
at::Tensor makeFake(at::Tensor meta, c10::Device fake_device);

// Detached clone of the shadow meta tensor (no autograd history).
at::Tensor metaLike(const at::Tensor& fake);

// TLS fake mode: while entered, newly constructed tensors are fake.
// `fake_cuda` permits constructing fake "cuda" tensors on machines where the
// cuda/HIP runtime has no devices (GPU-less CI).
void enterFakeMode(bool fake_cuda);
void leaveFakeMode();
bool isFakeModeActive() noexcept;

}  // namespace tdx
