#include "stack_utils.h"

#include <c10/util/Exception.h>

namespace tdx {
namespace {

void visitIValue(const c10::IValue& v, const TensorVisitor& visitor) {
  if (v.isTensor()) {
    visitor(v.toTensor());
  } else if (v.isTensorList()) {
    for (const at::Tensor& t : v.toTensorList()) {
      visitor(t);
    }
  } else if (v.isOptionalTensorList()) {
    for (const std::optional<at::Tensor>& t : v.toOptionalTensorList()) {
      if (t.has_value()) {
        visitor(*t);
      }
    }
  } else if (v.isList()) {
    for (const c10::IValue& e : v.toListRef()) {
      visitIValue(e, visitor);
    }
  } else if (v.isTuple()) {
    for (const c10::IValue& e : v.toTupleRef().elements()) {
      visitIValue(e, visitor);
    }
  } else if (v.isGenericDict()) {
    for (const auto& kv : v.toGenericDict()) {
      visitIValue(kv.key(), visitor);
      visitIValue(kv.value(), visitor);
    }
  }
}

c10::IValue mapIValue(const c10::IValue& v, const TensorMapFn& fn) {
  if (v.isTensor()) {
    return fn(v.toTensor());
  }
  if (v.isTensorList()) {
    auto lst = v.toTensorList();
    for (size_t i = 0; i < lst.size(); ++i) {
      lst.set(i, fn(lst.get(i)));
    }
    return v;
  }
  if (v.isOptionalTensorList()) {
    auto lst = v.toOptionalTensorList();
    for (size_t i = 0; i < lst.size(); ++i) {
      std::optional<at::Tensor> e = lst.get(i);
      if (e.has_value()) {
        lst.set(i, std::optional<at::Tensor>(fn(*e)));
      }
    }
    return v;
  }
  if (v.isList()) {
    auto lst = v.toList();
    for (size_t i = 0; i < lst.size(); ++i) {
      lst.set(i, mapIValue(lst.get(i), fn));
    }
    return v;
  }
  if (v.isTuple()) {
    auto elems = v.toTupleRef().elements().vec();
    bool changed = false;
    for (auto& e : elems) {
      c10::IValue mapped = mapIValue(e, fn);
      changed = changed || !mapped.isSameIdentity(e);
      e = std::move(mapped);
    }
    if (changed) {
      return c10::ivalue::Tuple::create(std::move(elems));
    }
    return v;
  }
  if (v.isGenericDict()) {
    auto dict = v.toGenericDict();
    for (const auto& kv : dict) {
      c10::IValue mapped = mapIValue(kv.value(), fn);
      if (!mapped.isSameIdentity(kv.value())) {
        dict.insert_or_assign(kv.key(), std::move(mapped));
      }
    }
    return v;
  }
  return v;
}

// Structural copy of one IValue. Containers are rebuilt; leaves (tensors,
// scalars, strings, devices, generators, ...) are shared.
c10::IValue copyIValue(const c10::IValue& v) {
  if (v.isList()) {
    auto src = v.toList();
    c10::impl::GenericList dst{src.elementType()};
    dst.reserve(src.size());
    for (const c10::IValue& e : src) {
      dst.push_back(copyIValue(e));
    }
    return dst;
  }
  if (v.isTuple()) {
    std::vector<c10::IValue> elems;
    for (const c10::IValue& e : v.toTupleRef().elements()) {
      elems.push_back(copyIValue(e));
    }
    return c10::ivalue::Tuple::create(std::move(elems));
  }
  if (v.isGenericDict()) {
    auto src = v.toGenericDict();
    c10::impl::GenericDict dst{src.keyType(), src.valueType()};
    for (const auto& kv : src) {
      dst.insert(copyIValue(kv.key()), copyIValue(kv.value()));
    }
    return dst;
  }
  // Leaf values a deferred-init tape can hold verbatim are shared; anything
  // with observable mutability that cannot be snapshotted is rejected with a
  // clear error so a recording failure is loud, not silent.
  TORCH_CHECK(
      !v.isStorage() && !v.isFuture() && !v.isRRef() && !v.isPyObject() &&
          !v.isCapsule() && !v.isQuantizer() && !v.isStream(),
      "Cannot record an operation with an argument of type `",
      v.tagKind(),
      "` in a deferred-init context.");
  return v;
}

}  // namespace

void visitTensors(const torch::jit::Stack& stack,
                  size_t begin,
                  size_t end,
                  const TensorVisitor& visitor) {
  for (size_t i = begin; i < end; ++i) {
    visitIValue(stack[i], visitor);
  }
}

void mapTensors(torch::jit::Stack& stack,
                size_t begin,
                size_t end,
                const TensorMapFn& fn) {
  for (size_t i = begin; i < end; ++i) {
    stack[i] = mapIValue(stack[i], fn);
  }
}

std::vector<c10::IValue> copyStackRegion(const torch::jit::Stack& stack,
                                         size_t begin,
                                         size_t end) {
  std::vector<c10::IValue> out;
  out.reserve(end - begin);
  for (size_t i = begin; i < end; ++i) {
    out.push_back(copyIValue(stack[i]));
  }
  return out;
}

}  // namespace tdx
