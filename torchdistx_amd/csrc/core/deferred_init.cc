// See deferred_init.h. Reference behavior map (file:line into
// /root/reference/src/cc/torchdistx/deferred_init.cc):
//   recording fallback + stack copy      — :732-862, :88-100
//   Op / OpNode / TensorRecord tape      — :157-462
//   view keep-alives                     — :126-154, :431-462
//   materialization (call-stack build)   — :506-667
//   VariableHooks proxy                  — :889-948, :1050-1128
//
// The tape is re-designed rather than translated: one mutable TensorRecord
// per fake impl (restamped in place), per-storage AliasGroups owning the
// family's writer nodes (an acyclic replacement for the reference's view
// keep-alives), record-time history snapshots for cross-family reads, and
// a replay collection that takes a family's full writer set, closes over
// dependencies + each writer's chronological prefix, and pulls in (to
// fixpoint) readers whose inputs a collected later writer would clobber.
// Validated differentially against eager execution by tests/test_tape_fuzz
// (thousands of random aliasing programs).

#include "deferred_init.h"

#include <algorithm>
#include <atomic>
#include <condition_variable>
#include <cstring>
#include <functional>
#include <memory>
#include <mutex>
#include <optional>
#include <unordered_set>
#include <vector>

#include <ATen/CPUGeneratorImpl.h>
#include <ATen/Context.h>
#include <ATen/ThreadLocalState.h>
#include <c10/core/GradMode.h>
#include <ATen/core/dispatch/Dispatcher.h>
#include <ATen/core/VariableHooksInterface.h>
#include <c10/core/DefaultDtype.h>
#include <c10/core/impl/LocalDispatchKeySet.h>
#include <torch/library.h>

#include "fake.h"
#include "native_redirect.h"
#include "stack_utils.h"

namespace tdx {
namespace {

constexpr c10::DispatchKey kDeferredKey = c10::DispatchKey::DeferredInit;
constexpr c10::DispatchKey kFakeKey = c10::DispatchKey::Fake;

struct OpNode;

// Producer edge: which output of which node materializes a given fake.
struct OpOutputDescriptor {
  std::shared_ptr<OpNode> node;
  size_t index = 0;  // tensor-visit-order index into the node's outputs
};

// All tensors sharing one (meta) storage share one AliasGroup, which owns
// every node that wrote to that storage. Any surviving member of the alias
// family therefore keeps the whole family's tape segments alive — a view
// can outlive its base (and vice versa) without losing in-place writes —
// and, unlike record<->record keep-alive edges, the ownership graph is
// acyclic: groups own nodes, nodes own dependency nodes, records own their
// group. When the last fake of the family dies, the group and its nodes
// (including any materialized outputs they cache) are released.
struct AliasGroup {
  std::vector<std::shared_ptr<OpNode>> writers;
};

// Per-fake-tensor record, stored in the impl's dispatch-key side table.
// Mutable: an in-place op restamps `desc` in place, so aliased tensors
// observe the rewrite through their shared group.
struct TensorRecord {
  OpOutputDescriptor desc;
  std::shared_ptr<AliasGroup> group;
};

// One recorded call frame.
struct RecordedOp {
  std::string name;
  // Contract: consumes the arguments from the stack and leaves exactly the
  // op's return values on it (OperatorHandle::callBoxed semantics).
  std::function<void(torch::jit::Stack&)> run;
  std::vector<c10::IValue> args;  // deep-copied; fake slots -> undefined
  at::ThreadLocalState tls;       // captured with DeferredInit excluded
  // Pinned Philox (seed, counter offset) for RNG ops; also consumed by the
  // slice-materialization fast path (materializeTensorShard).
  std::optional<std::pair<uint64_t, uint64_t>> philox;
};

// Pinned Philox state for recorded RNG ops: derived from the default
// generator's seed and the op's tape position at RECORD time, so replaying
// ANY subset of the tape on ANY rank produces identical bits
// (partition-invariant initialization — the property that makes sharded /
// broadcast materialization bitwise-equal to local materialization when
// the native CDNA4 kernels run the replay).
constexpr uint64_t kPhiloxStridePerOp = 4;

// RNG session, one per outermost deferred_init on each thread (sessions
// are per-thread by construction: deferred_level is thread_local).
//
// A session is partitioned into SEGMENTS by the mutations the user makes
// to the default CPU generator. Recording never consumes the generator,
// so at record time of each RNG op the full generator state is hashed:
// an unchanged hash means "same segment, next slot", a changed hash
// means the user re-seeded (or otherwise perturbed) the generator mid
// construction and a new segment begins there. Each segment derives:
//
//   * a 64-bit Philox nonce = the state hash. Pins are therefore a pure
//     function of the generator state in effect when the op was
//     recorded: torch.manual_seed(S) — before deferred_init OR inside
//     the module builder — fixes the native init bits, exactly like
//     eager init is a function of the generator state at construction,
//     and independently of any ambient (e.g. per-rank) seeding that the
//     in-builder seed overrides. Within one tape, replaying any subset
//     on any rank gives identical bits (partition-invariant sharded
//     materialization).
//   * a state snapshot. CPU stock replay draws from a cursor initialized
//     to the segment snapshot (not from the ambient generator), so a
//     full in-tape-order CPU materialization is bitwise-equal to eager
//     construction, and replay never perturbs the ambient stream.
//
// Session EXIT advances the live CPU generator by one 64-bit draw, the
// way eager construction consumes it: a second session without
// re-seeding hashes a different state and draws an independent stream
// (two unseeded models never collide), while a session entered right
// after torch.manual_seed(S) still sees the pristine state S — so its
// pins and cursors match an eager construction from S bitwise.
struct RngSession {
  bool has_segment = false;
  uint64_t segment_hash = 0;   // raw state hash of the current segment
  uint64_t nonce = 0;          // collision-salted Philox seed in use
  uint64_t next_slot = 1;      // Philox slot within the current segment
  std::vector<uint64_t> raw_hashes;  // for registry release at session end
  // Default CPU generator state; seeded from segment snapshots and
  // advanced only by this session's CPU stock replay, in tape order.
  // Guarded by cursor_mutex: node execution runs without the tape lock,
  // and the swap-draw-swap of the process-global default generator must
  // be atomic per op (stock-CPU RNG replay therefore serializes per
  // session — it consumes one generator stream by definition; the
  // pinned-Philox native paths parallelize freely).
  at::Tensor replay_cursor;
  std::mutex cursor_mutex;
};

thread_local std::shared_ptr<RngSession> current_rng_session;

// Live-segment registry: two sessions OPEN AT THE SAME TIME (concurrent
// threads) that hash the same generator state would otherwise pin the
// same stream — recording consumes nothing, so the shared state cannot
// tell them apart. Concurrent collisions get a salted nonce (and a
// salted replay snapshot); sequential sessions release their entries at
// exit, so re-seeding to the same state later still reproduces exactly.
std::mutex segment_registry_mutex;
std::unordered_map<uint64_t, uint64_t> live_segment_uses;

uint64_t mixSalt(uint64_t h, uint64_t salt) {
  for (int b = 0; b < 8; ++b) {
    h = (h ^ ((salt >> (8 * b)) & 0xff)) * 1099511628211ull;
  }
  return h;
}

bool isRngOpName(const std::string& name) {
  return name == "aten::uniform_" || name == "aten::normal_" ||
         name == "aten::bernoulli_";
}

// Word-at-a-time mix over a generator-state blob (splitmix-style):
// change detection + nonce derivation, not cryptography. Runs once per
// recorded RNG op over the ~5 KB Mersenne state, so byte-at-a-time FNV
// (~5 us per op) would be a measurable recording cost.
uint64_t hashStateBytes(const uint8_t* bytes, int64_t n) {
  uint64_t h = 0x9E3779B97F4A7C15ull ^ static_cast<uint64_t>(n);
  int64_t i = 0;
  for (; i + 8 <= n; i += 8) {
    uint64_t w;
    std::memcpy(&w, bytes + i, 8);
    h ^= w;
    h *= 0xBF58476D1CE4E5B9ull;
    h ^= h >> 29;
  }
  uint64_t tail = 0;
  for (int j = 0; i < n; ++i, ++j) {
    tail |= static_cast<uint64_t>(bytes[i]) << (8 * j);
  }
  h ^= tail;
  h *= 0x94D049BB133111EBull;
  h ^= h >> 32;
  return h;
}

void beginRngSession() {
  current_rng_session = std::make_shared<RngSession>();
}

void endRngSession() {
  // Registry release and the exit-advance must be ONE atomic step with
  // respect to other threads opening segments: otherwise a concurrent
  // thread can snapshot the pre-advance generator state, lose the race
  // to this release, and register the same state hash with salt 0 —
  // pinning the stream this session already used.
  std::lock_guard<std::mutex> reg_lock(segment_registry_mutex);
  if (current_rng_session != nullptr) {
    for (uint64_t h : current_rng_session->raw_hashes) {
      auto it = live_segment_uses.find(h);
      if (it != live_segment_uses.end() && --(it->second) == 0) {
        live_segment_uses.erase(it);
      }
    }
  }
  current_rng_session = nullptr;
  // Consume one draw so the next unseeded session sees a new state
  // (recorded ops keep the session alive through their own references).
  auto gen = at::globalContext().defaultGenerator(c10::DeviceType::CPU);
  std::lock_guard<std::mutex> lock(gen.mutex());
  gen.get<at::CPUGeneratorImpl>()->random64();
}

// Pins (seed, counter-offset) for every recorded RNG op. The pins drive
// the tdx Philox kernels on GPU replay and the slice-materialization fast
// path on any device; plain CPU replay ignores them (it draws from the
// segment's replay cursor for eager bit-parity). When this op opens a new
// segment, `segment_start` receives the segment's generator-state
// snapshot (undefined otherwise).
std::optional<std::pair<uint64_t, uint64_t>> pinPhiloxForOp(
    const std::string& name, at::Tensor* segment_start) {
  if (!isRngOpName(name) || current_rng_session == nullptr) {
    return std::nullopt;
  }
  RngSession& session = *current_rng_session;
  // The registry lock brackets the state snapshot AND the segment
  // registration, and session exit advances the generator under the
  // same lock — so "state hash is live in the registry" and "state has
  // been advanced past" are mutually exclusive, with no window where a
  // second thread can re-pin a just-closed session's stream (gen.mutex
  // nests inside it, same order as endRngSession).
  std::lock_guard<std::mutex> reg_lock(segment_registry_mutex);
  auto gen = at::globalContext().defaultGenerator(c10::DeviceType::CPU);
  at::Tensor state;
  {
    std::lock_guard<std::mutex> lock(gen.mutex());
    state = gen.get_state();  // returns a fresh copy
  }
  const uint64_t h =
      hashStateBytes(state.const_data_ptr<uint8_t>(), state.numel());
  if (!session.has_segment || h != session.segment_hash) {
    session.has_segment = true;
    session.segment_hash = h;
    session.next_slot = 1;
    const uint64_t salt = live_segment_uses[h]++;
    session.raw_hashes.push_back(h);
    session.nonce = salt == 0 ? h : mixSalt(h, salt);
    if (salt != 0) {
      // Concurrent collision: give the CPU stock path a distinct stream
      // too, by snapshotting a generator seeded with the salted nonce.
      auto salted = at::make_generator<at::CPUGeneratorImpl>(session.nonce);
      state = salted.get_state();
    }
    *segment_start = std::move(state);
  }
  return std::make_pair(session.nonce,
                        session.next_slot++ * kPhiloxStridePerOp);
}

struct InputSlot {
  std::optional<OpOutputDescriptor> desc;  // set when the arg was fake
  std::optional<uint32_t> external_version;  // set for real tensor args
  // For fake inputs read from a *different* storage family than this op
  // writes: a record-time snapshot of that family's writers so far. Replay
  // of this op must first replay these (they produced the value the op
  // observed), and holding them here keeps cross-family history alive even
  // after the other family's fakes are gone. Acyclic by construction:
  // every held node has a smaller op_nr.
  std::vector<std::shared_ptr<OpNode>> history;
  // Inference tensors have no version counter, so their mutations cannot
  // be tracked; replay rejects them with a clear error.
  bool inference = false;
};

struct AliasGroup;

struct OpNode {
  uint64_t op_nr = 0;
  std::optional<RecordedOp> op;          // dropped after replay
  std::vector<InputSlot> input_slots;    // tensor visit order over args
  std::vector<c10::Storage> output_storages;  // meta storages, visit order
  std::vector<std::weak_ptr<OpNode>> dependents;
  // The storage families this node writes (weak: groups own nodes, never
  // the reverse). Replay collection closes over each collected writer's
  // chronological prefix in these families, so a writer forced into a
  // replay set (e.g. by the clobber rule) can never run before an earlier
  // writer of the same storage.
  std::vector<std::weak_ptr<AliasGroup>> write_groups;
  std::vector<at::Tensor> outputs;  // real tensors after replay
  bool materialized = false;  // Done: outputs valid, frame freed
  // Another thread is executing this node's op with tape_mutex released;
  // threads whose call stacks share the node wait on tape_cv instead of
  // double-replaying. All transitions happen under tape_mutex.
  bool running = false;
};

std::atomic<uint64_t> next_op_nr{1};

// Serializes tape STRUCTURE access: recording (recordOp), call-stack
// construction, node state transitions and recordInfo. Materialization
// runs with the GIL released, so without this a second thread could
// record into or replay from the same node graph mid-walk. The lock is
// NOT held while a node's op actually executes — replay releases it
// around the kernel/compute work and marks the node `running`, so
// disjoint subgraphs materialize concurrently on multiple threads (and
// multiple HIP streams) while threads whose closures share a node wait
// on tape_cv for its single execution. Recursive because a terminal op
// (aten::item) can materialize while a recording is on the stack.
std::recursive_mutex tape_mutex;
std::condition_variable_any tape_cv;

std::shared_ptr<TensorRecord> getRecord(FakeTensorImpl* fake) {
  return std::static_pointer_cast<TensorRecord>(fake->getData(kDeferredKey));
}

void setRecord(FakeTensorImpl* fake, std::shared_ptr<TensorRecord> rec) {
  fake->setData(kDeferredKey, std::move(rec));
}

uint32_t tensorVersion(const at::Tensor& t) {
  return t.unsafeGetTensorImpl()->version_counter().current_version();
}

bool storagesAlias(const c10::Storage& a, const c10::Storage& b) {
  return a && b && a.is_alias_of(b);
}

// ---------------------------------------------------------------------------
// Recording.
// ---------------------------------------------------------------------------

thread_local size_t deferred_level = 0;

bool shouldRecordHere() {
  return deferred_level > 0 &&
         !c10::impl::tls_is_dispatch_key_excluded(kDeferredKey);
}

// Creates an OpNode out of a recorded call frame. `args` is the deep-copied
// pre-execution argument region; `stack`/`rets_begin` point at the
// post-execution return values, whose fake tensors get stamped with this
// node's records.
void recordOp(std::string name,
              std::function<void(torch::jit::Stack&)> run,
              std::vector<c10::IValue> args,
              at::ThreadLocalState tls,
              torch::jit::Stack& stack,
              size_t rets_begin,
              std::optional<std::pair<uint64_t, uint64_t>> philox =
                  std::nullopt) {
  std::lock_guard<std::recursive_mutex> lock{tape_mutex};
  auto node = std::make_shared<OpNode>();
  node->op_nr = next_op_nr.fetch_add(1, std::memory_order_relaxed);

  // Input pass: capture producer descriptors for fake args (replacing them
  // with undefined placeholders in the saved frame) and version counters
  // for external (real) tensor args.
  std::vector<std::shared_ptr<TensorRecord>> input_records;
  std::vector<c10::Storage> input_storages;
  std::vector<size_t> fake_slot_idx;  // input_slots index per fake input
  {
    torch::jit::Stack args_stack{args.begin(), args.end()};
    mapTensors(args_stack, 0, args_stack.size(),
               [&](const at::Tensor& t) -> at::Tensor {
                 if (auto* fake = asFake(t)) {
                   auto rec = getRecord(fake);
                   TORCH_CHECK(
                       rec != nullptr,
                       "A fake tensor constructed outside of a deferred-init "
                       "context (or stripped of its record) was used in `",
                       name, "` and cannot be recorded.");
                   fake_slot_idx.push_back(node->input_slots.size());
                   node->input_slots.push_back(InputSlot{rec->desc, {}, {}});
                   rec->desc.node->dependents.emplace_back(node);
                   input_records.push_back(std::move(rec));
                   input_storages.push_back(fake->meta_tensor().storage());
                   return at::Tensor{};
                 }
                 if (t.defined() && t.is_inference()) {
                   InputSlot slot;
                   slot.inference = true;
                   node->input_slots.push_back(std::move(slot));
                 } else if (t.defined()) {
                   node->input_slots.push_back(
                       InputSlot{{}, tensorVersion(t), {}});
                 } else {
                   node->input_slots.push_back(InputSlot{});
                 }
                 return t;
               });
    args.assign(args_stack.begin(), args_stack.end());
  }

  node->op = RecordedOp{std::move(name), std::move(run), std::move(args),
                        std::move(tls), philox};

  // Output pass: stamp (or restamp) each fake output's record.
  visitTensors(stack, rets_begin, stack.size(), [&](const at::Tensor& t) {
    size_t out_idx = node->output_storages.size();
    auto* fake = asFake(t);
    if (fake == nullptr) {
      node->output_storages.emplace_back();
      return;
    }
    const c10::Storage& out_storage = fake->meta_tensor().storage();
    node->output_storages.push_back(out_storage);

    auto rec = getRecord(fake);
    if (rec == nullptr) {
      rec = std::make_shared<TensorRecord>();
      setRecord(fake, rec);
    }
    // Restamp IN PLACE: aliased records observe the rewrite through the
    // shared group (fills the role of the reference's view_records_,
    // deferred_init.cc:126-154).
    rec->desc = OpOutputDescriptor{node, out_idx};

    // Alias-group wiring: adopt the group of any aliased fake input (a
    // view adopts its base's family), else keep/create our own, and
    // register this node as a writer of the family's storage.
    if (rec->group == nullptr) {
      for (size_t i = 0; i < input_records.size(); ++i) {
        if (storagesAlias(out_storage, input_storages[i]) &&
            input_records[i]->group != nullptr) {
          rec->group = input_records[i]->group;
          break;
        }
      }
      if (rec->group == nullptr) {
        rec->group = std::make_shared<AliasGroup>();
      }
    }
    if (rec->group->writers.empty() ||
        rec->group->writers.back() != node) {
      rec->group->writers.push_back(node);
      node->write_groups.emplace_back(rec->group);
    }
  });

  // History snapshots: for every fake input whose storage this op does NOT
  // write (a pure read from another family), capture that family's writer
  // list as of now.
  for (size_t i = 0; i < input_records.size(); ++i) {
    bool writes_it = false;
    for (const c10::Storage& out : node->output_storages) {
      if (storagesAlias(out, input_storages[i])) {
        writes_it = true;
        break;
      }
    }
    if (!writes_it && input_records[i]->group != nullptr) {
      node->input_slots[fake_slot_idx[i]].history =
          input_records[i]->group->writers;
    }
  }
}

bool isTerminalOp(const c10::OperatorHandle& op) {
  const auto& name = op.schema().operator_name().name;
  return name == "aten::item" || name == "aten::_local_scalar_dense";
}

// Whole-tensor in-place init ops `(Tensor(a!) self, ...scalars...) ->
// Tensor(a!)`: on a fake self they change no metadata and produce no new
// tensor, so recording can skip the fake-layer walk AND the meta
// redispatch entirely. This matters enormously for throughput: PyTorch
// has no C++ meta kernels for uniform_/normal_ — the Meta key lands in
// the Python-dispatcher _refs decomposition at ~0.5 ms per op, which
// made these two ops ~85% of the host-side cost of recording a 70B tape.
bool isInPlaceInitOp(const c10::OperatorName& name) {
  if (name.name == "aten::uniform_" || name.name == "aten::normal_" ||
      name.name == "aten::zero_") {
    return name.overload_name.empty();
  }
  if (name.name == "aten::fill_") {
    return name.overload_name == "Scalar";
  }
  if (name.name == "aten::bernoulli_") {
    return name.overload_name == "float";
  }
  return false;
}

void deferredInitHandler(const c10::OperatorHandle& op,
                         c10::DispatchKeySet ks,
                         torch::jit::Stack* stack) {
  c10::impl::ExcludeDispatchKeyGuard no_reentry{kDeferredKey};

  const auto& schema = op.schema();
  size_t num_args = schema.arguments().size();
  size_t args_begin = stack->size() - num_args;

  bool has_fake_arg = false;
  visitTensors(*stack, args_begin, args_begin + num_args,
               [&](const at::Tensor& t) {
                 if (auto* fake = asFake(t)) {
                   has_fake_arg = true;
                   TORCH_CHECK(
                       getRecord(fake) != nullptr,
                       "A fake tensor without a deferred-init record was "
                       "passed to `", schema.operator_name(),
                       "`. Only tensors constructed inside `deferred_init()` "
                       "can be used here.");
                 }
               });

  const auto below =
      ks & c10::DispatchKeySet{c10::DispatchKeySet::FULL_AFTER, kDeferredKey};

  // Terminal ops need real data (e.g. aten::item): materialize the fake
  // arguments eagerly and run for real (reference deferred_init.cc:813-815).
  if (has_fake_arg && isTerminalOp(op)) {
    mapTensors(*stack, args_begin, args_begin + num_args,
               [&](const at::Tensor& t) -> at::Tensor {
                 return isFake(t) ? materializeTensor(t) : t;
               });
    op.redispatchBoxed(below, stack);
    return;
  }

  auto saved = copyStackRegion(*stack, args_begin, args_begin + num_args);
  at::ThreadLocalState tls;  // DeferredInit is excluded in this snapshot

  // Fast path for whole-tensor in-place init on a fake self: the op
  // neither changes metadata nor creates tensors, so skip the fake-layer
  // redispatch (and with it the expensive Python-dispatcher meta
  // decompositions of uniform_/normal_). Validation that eager would do
  // on entry is performed here so errors still surface at record time.
  bool fast_inplace = false;
  if (has_fake_arg && isInPlaceInitOp(schema.operator_name()) &&
      (*stack)[args_begin].isTensor()) {
    const at::Tensor& self = (*stack)[args_begin].toTensor();
    // Autograd-visible mutations (in-place on a requires_grad tensor
    // under grad mode) must keep going through the autograd layer so
    // eager's semantics — the leaf-mutation error, version bumps —
    // surface at record time exactly as they would eagerly.
    if (isFake(self) &&
        (!self.requires_grad() || !c10::GradMode::is_enabled())) {
      const auto& opname = schema.operator_name().name;
      if (opname == "aten::uniform_" || opname == "aten::normal_") {
        TORCH_CHECK(c10::isFloatingType(self.scalar_type()), "`", opname,
                    "` expects a floating-point tensor, got ",
                    self.scalar_type());
      }
      if (opname == "aten::normal_" && num_args >= 3 &&
          (*stack)[args_begin + 2].isDouble()) {
        const double std = (*stack)[args_begin + 2].toDouble();
        TORCH_CHECK(std >= 0.0,
                    "normal_ expects std >= 0.0, but found std=", std);
      }
      if (opname == "aten::bernoulli_" && num_args >= 2 &&
          (*stack)[args_begin + 1].isDouble()) {
        const double p = (*stack)[args_begin + 1].toDouble();
        TORCH_CHECK(0.0 <= p && p <= 1.0,
                    "bernoulli_ expects 0 <= p <= 1, but found p=", p);
      }
      fast_inplace = true;
    }
  }

  if (fast_inplace) {
    c10::IValue self_iv = (*stack)[args_begin];
    stack->resize(args_begin);
    stack->push_back(std::move(self_iv));
  } else {
    // Factory calls (no tensor inputs) with dtype=None resolve the dtype
    // at execution time from the process-global default. Pin the default
    // that is in effect NOW into the recorded frame, so replay is
    // faithful even after torch.set_default_dtype() changes (for
    // non-factories None means "follow the input tensor" and must stay
    // None).
    bool has_tensor_arg = false;
    visitTensors(*stack, args_begin, args_begin + num_args,
                 [&](const at::Tensor& t) {
                   has_tensor_arg = has_tensor_arg || t.defined();
                 });
    if (!has_tensor_arg) {
      const auto& schema_args = schema.arguments();
      for (size_t i = 0; i < schema_args.size(); ++i) {
        if (schema_args[i].name() == "dtype" && saved[i].isNone()) {
          // NB: schema .type() erases ScalarType to int; the semantic
          // type lives in .real_type().
          const auto* type = schema_args[i].real_type().get();
          if (type->kind() == c10::TypeKind::OptionalType &&
              type->castRaw<c10::OptionalType>()->getElementType()->kind() ==
                  c10::TypeKind::ScalarTypeType) {
            saved[i] = c10::IValue{c10::get_default_dtype_as_scalartype()};
          }
        }
      }
    }

    // Execute through the fake layer: shape/dtype work happens on the
    // meta backend and fake tensors come back.
    op.redispatchBoxed(below.add(kFakeKey), stack);
  }

  size_t rets_begin = stack->size() - schema.returns().size();
  bool has_fake_ret = false;
  visitTensors(*stack, rets_begin, stack->size(), [&](const at::Tensor& t) {
    has_fake_ret = has_fake_ret || isFake(t);
  });

  if (has_fake_arg || has_fake_ret) {
    at::Tensor segment_start;
    auto philox = pinPhiloxForOp(schema.operator_name().name, &segment_start);
    // RNG ops additionally carry their session (and, for the first op of
    // a segment, the segment's generator-state snapshot) for the CPU
    // replay cursor.
    std::shared_ptr<RngSession> session =
        philox.has_value() ? current_rng_session : nullptr;
    recordOp(
        schema.operator_name().name,
        [handle = op, philox, session,
         segment_start](torch::jit::Stack& s) {
          // Replay hot path: recorded init ops whose target lives on the
          // GPU run through the hand-written CDNA4 kernels (tdx::) when
          // the _K extension is loaded. The pinned Philox state makes the
          // result independent of which rank replays which subset.
          if (tryNativeInitRedirect(handle, s, philox)) {
            return;
          }
          if (session != nullptr && !s.empty() && s.front().isTensor()) {
            // CPU stock replay of a session RNG op: draw from the
            // segment's replay cursor instead of the ambient generator,
            // so in-tape-order materialization reproduces eager bits and
            // the ambient stream is left untouched. Ops recorded with an
            // explicit generator keep consuming that generator.
            const at::Tensor& self = s.front().toTensor();
            bool explicit_gen = false;
            for (const c10::IValue& v : s) {
              explicit_gen = explicit_gen || v.isGenerator();
            }
            if (self.defined() && self.is_cpu() && !explicit_gen) {
              std::lock_guard<std::mutex> cursor_lock{session->cursor_mutex};
              if (segment_start.defined()) {
                session->replay_cursor = segment_start;
              }
              if (session->replay_cursor.defined()) {
                auto gen = at::globalContext().defaultGenerator(
                    c10::DeviceType::CPU);
                auto swap_state = [&gen](const at::Tensor& st) {
                  std::lock_guard<std::mutex> lock(gen.mutex());
                  at::Tensor prev = gen.get_state();
                  gen.set_state(st);
                  return prev;
                };
                at::Tensor ambient = swap_state(session->replay_cursor);
                handle.callBoxed(s);
                session->replay_cursor = swap_state(ambient);
                return;
              }
              // Out-of-tape-order replay reached a mid-segment op before
              // its segment head: CPU bits for such partial replays are
              // unspecified; draw from the ambient generator.
            }
          }
          handle.callBoxed(s);
        },
        std::move(saved), std::move(tls), *stack, rets_begin, philox);
  }
}

TORCH_LIBRARY_IMPL(_, DeferredInit, m) {
  m.fallback(torch::CppFunction::makeFromBoxedFunction<&deferredInitHandler>());
}

// ---------------------------------------------------------------------------
// VariableHooks proxy: `.data` reads and writes bypass the dispatcher, so a
// recording proxy is swapped in for the duration of deferred-init
// (reference deferred_init.cc:889-948, 1050-1128; design note in the
// reference docs fake_tensor_and_deferred_init.rst:153-188).
// ---------------------------------------------------------------------------

class ProxyVariableHooks final : public at::impl::VariableHooksInterface {
 public:
  explicit ProxyVariableHooks(at::impl::VariableHooksInterface* inner)
      : inner_{inner} {}

  at::TensorBase variable_data(const at::TensorBase& self) const override;
  void set_data(const at::TensorBase& self,
                const at::TensorBase& new_data) const override;
  at::TensorBase data(const at::TensorBase& self) const override {
    // Route through variable_data so `.data` reads are recorded too.
    return variable_data(self);
  }

  at::TensorBase tensor_data(const at::TensorBase& t) const override {
    return inner_->tensor_data(t);
  }
  const std::shared_ptr<torch::autograd::Node>& grad_fn(
      const at::TensorBase& t) const override {
    return inner_->grad_fn(t);
  }
  unsigned _register_hook(
      const at::TensorBase& t,
      std::function<at::TensorBase(const at::TensorBase&)> hook)
      const override {
    return inner_->_register_hook(t, std::move(hook));
  }
  void remove_hook(const at::TensorBase& t, unsigned pos) const override {
    inner_->remove_hook(t, pos);
  }
  bool is_view(const at::TensorBase& t) const override {
    return inner_->is_view(t);
  }
  const at::TensorBase& base(const at::TensorBase& t) const override {
    return inner_->base(t);
  }
  const std::string& name(const at::TensorBase& t) const override {
    return inner_->name(t);
  }
  bool is_leaf(const at::TensorBase& t) const override {
    return inner_->is_leaf(t);
  }
  int64_t output_nr(const at::TensorBase& t) const override {
    return inner_->output_nr(t);
  }
  int64_t _version(const at::TensorBase& t) const override {
    return inner_->_version(t);
  }
  void retain_grad(const at::TensorBase& t) const override {
    inner_->retain_grad(t);
  }
  bool retains_grad(const at::TensorBase& t) const override {
    return inner_->retains_grad(t);
  }
  void _backward(const at::Tensor& t,
                 at::TensorList inputs,
                 const std::optional<at::Tensor>& gradient,
                 std::optional<bool> keep_graph,
                 bool create_graph) const override {
    inner_->_backward(t, inputs, gradient, keep_graph, create_graph);
  }
  void requires_grad_(const at::TensorBase& t, bool value) const override {
    inner_->requires_grad_(t, value);
  }
  void basic_autograd_not_implemented_fallback(
      const c10::OperatorHandle& op,
      c10::DispatchKeySet dispatch_keys,
      torch::jit::Stack* stack) const override {
    inner_->basic_autograd_not_implemented_fallback(op, dispatch_keys, stack);
  }
  std::optional<c10::ScalarType> grad_dtype(
      const at::TensorBase& t) const override {
    return inner_->grad_dtype(t);
  }
  void set_grad_dtype(
      const at::TensorBase& t,
      const std::optional<c10::ScalarType>& dtype) const override {
    inner_->set_grad_dtype(t, dtype);
  }

  at::impl::VariableHooksInterface* inner() const noexcept {
    return inner_;
  }

  void setInner(at::impl::VariableHooksInterface* inner) noexcept {
    inner_ = inner;
  }

 private:
  at::impl::VariableHooksInterface* inner_;
};

std::mutex hooks_mutex;
size_t hooks_refcount = 0;
at::impl::VariableHooksInterface* saved_hooks = nullptr;
// Never freed: a thread that fetched the global hooks pointer just
// before the last session left (e.g. replaying a recorded
// tdx::variable_data op, which calls through the hooks) may still be
// mid-call when the stock hooks are restored — deleting the proxy there
// is a use-after-free. The proxy forwards to the real hooks and records
// nothing outside an active session, so keeping the one object alive
// for the process lifetime is both safe and semantically inert.
ProxyVariableHooks* proxy_hooks = nullptr;

void installProxyHooks() {
  std::lock_guard<std::mutex> lock{hooks_mutex};
  if (hooks_refcount++ == 0) {
    at::impl::VariableHooksInterface* current = at::impl::GetVariableHooks();
    TORCH_INTERNAL_ASSERT(current != proxy_hooks);
    saved_hooks = current;
    if (proxy_hooks == nullptr) {
      proxy_hooks = new ProxyVariableHooks{saved_hooks};
    } else {
      proxy_hooks->setInner(saved_hooks);
    }
    at::impl::SetVariableHooks(proxy_hooks);
  }
}

void removeProxyHooks() {
  std::lock_guard<std::mutex> lock{hooks_mutex};
  TORCH_INTERNAL_ASSERT(hooks_refcount > 0);
  if (--hooks_refcount == 0) {
    at::impl::SetVariableHooks(saved_hooks);
  }
}

at::TensorBase ProxyVariableHooks::variable_data(
    const at::TensorBase& self) const {
  at::TensorBase out_base = inner_->variable_data(self);
  at::Tensor self_t{self};
  if (!shouldRecordHere() || !isFake(self_t)) {
    return out_base;
  }
  auto* self_fake = asFake(self_t);
  if (getRecord(self_fake) == nullptr) {
    return out_base;
  }
  at::Tensor out{out_base};
  torch::jit::Stack rets{c10::IValue{out}};
  std::vector<c10::IValue> args{c10::IValue{self_t}};
  recordOp(
      "tdx::variable_data",
      [](torch::jit::Stack& s) {
        at::Tensor in = s.back().toTensor();
        s.pop_back();
        s.emplace_back(in.variable_data());
      },
      std::move(args), at::ThreadLocalState{}, rets, 0);
  return out;
}

void ProxyVariableHooks::set_data(const at::TensorBase& self,
                                  const at::TensorBase& new_data) const {
  at::Tensor self_t{self};
  at::Tensor new_t{new_data};
  bool record = shouldRecordHere() && isFake(self_t) &&
                getRecord(asFake(self_t)) != nullptr;
  std::vector<c10::IValue> args;
  if (record) {
    // Capture the pre-mutation producer of `self` before the impl swap.
    args = {c10::IValue{self_t}, c10::IValue{new_t}};
  }
  inner_->set_data(self, new_data);
  if (!record) {
    return;
  }
  torch::jit::Stack rets{c10::IValue{self_t}};
  recordOp(
      "tdx::set_data",
      [](torch::jit::Stack& s) {
        at::Tensor nd = s.back().toTensor();
        s.pop_back();
        at::Tensor slf = s.back().toTensor();
        s.pop_back();
        slf.set_data(nd);
        s.emplace_back(slf);
      },
      std::move(args), at::ThreadLocalState{}, rets, 0);
}

// ---------------------------------------------------------------------------
// Materialization.
// ---------------------------------------------------------------------------

// Does `node` produce (as one of its outputs) a tensor aliasing `storage`?
bool outputsAlias(const OpNode& node, const c10::Storage& storage) {
  for (const c10::Storage& s : node.output_storages) {
    if (storagesAlias(s, storage)) {
      return true;
    }
  }
  return false;
}

// The ordered replay set for the tensor behind `rec` (fills the role of
// the reference's buildCallStack + getLastInPlaceOpNode,
// deferred_init.cc:506-622, but is driven by the alias group instead of a
// dependents walk — a dependents walk cannot reach in-place writes made to
// a *base* after a view was taken, since those hang off the base's
// producer, not the view node):
//   1. every writer of the target's storage family (from the AliasGroup),
//      with their dependency closures and cross-family history snapshots;
//   2. to fixpoint, dependents of collected nodes whose inputs a collected
//      later writer would clobber — they must replay now, in chronological
//      position, or they would later observe post-overwrite values.
struct CallStack {
  std::vector<std::shared_ptr<OpNode>> nodes;
  std::unordered_set<OpNode*> members;

  void addWithDeps(const std::shared_ptr<OpNode>& n) {
    if (!members.insert(n.get()).second) {
      return;
    }
    nodes.push_back(n);
    for (const InputSlot& slot : n->input_slots) {
      if (slot.desc.has_value()) {
        addWithDeps(slot.desc->node);
      }
      for (const auto& h : slot.history) {
        addWithDeps(h);
      }
    }
    // Chronological-prefix closure: every earlier writer of every family
    // this node writes must replay before it.
    for (const auto& wg : n->write_groups) {
      if (auto group = wg.lock()) {
        for (const auto& w : group->writers) {
          if (w->op_nr < n->op_nr) {
            addWithDeps(w);
          }
        }
      }
    }
  }
};

std::vector<std::shared_ptr<OpNode>> buildCallStack(const TensorRecord& rec) {
  CallStack cs;
  uint64_t horizon = 0;
  if (rec.group != nullptr) {
    for (const auto& w : rec.group->writers) {
      horizon = std::max(horizon, w->op_nr);
      cs.addWithDeps(w);
    }
  } else {
    cs.addWithDeps(rec.desc.node);
    horizon = rec.desc.node->op_nr;
  }

  bool changed = true;
  while (changed) {
    changed = false;
    // Snapshot: addWithDeps mutates cs.nodes.
    std::vector<std::shared_ptr<OpNode>> snapshot = cs.nodes;
    for (const auto& n : snapshot) {
      for (const auto& wd : n->dependents) {
        auto d = wd.lock();
        if (!d || cs.members.count(d.get()) != 0 || d->materialized ||
            d->op_nr > horizon) {
          continue;
        }
        // Clobber rule: `d` reads an output of a collected node whose
        // storage a collected later op overwrites.
        bool include = false;
        for (const InputSlot& slot : d->input_slots) {
          if (!slot.desc.has_value() ||
              cs.members.count(slot.desc->node.get()) == 0) {
            continue;
          }
          const c10::Storage& s =
              slot.desc->node->output_storages[slot.desc->index];
          if (!s) {
            continue;
          }
          for (const auto& w : cs.nodes) {
            if (w->op_nr > d->op_nr && w.get() != slot.desc->node.get() &&
                outputsAlias(*w, s)) {
              include = true;
              break;
            }
          }
          if (include) {
            break;
          }
        }
        if (include) {
          cs.addWithDeps(d);
          changed = true;
        }
      }
    }
  }

  std::sort(cs.nodes.begin(), cs.nodes.end(),
            [](const auto& a, const auto& b) { return a->op_nr < b->op_nr; });
  return cs.nodes;
}

// Rebuilds a replay stack for `node` from its saved frame, substituting
// materialized dependency outputs for fake placeholders and validating
// external tensors (reference materializeArguments,
// deferred_init.cc:640-667).
torch::jit::Stack materializeArguments(OpNode& node) {
  torch::jit::Stack stack{node.op->args.begin(), node.op->args.end()};
  size_t visit = 0;
  mapTensors(stack, 0, stack.size(), [&](const at::Tensor& t) -> at::Tensor {
    const InputSlot& slot = node.input_slots.at(visit++);
    if (slot.desc.has_value()) {
      const OpOutputDescriptor& dep = *slot.desc;
      TORCH_INTERNAL_ASSERT(dep.node->materialized,
                            "deferred-init replay ordering bug: dependency "
                            "not yet materialized");
      return dep.node->outputs.at(dep.index);
    }
    TORCH_CHECK(!slot.inference,
                "An inference tensor was used while recording `",
                node.op->name,
                "`; in-place updates to inference tensors cannot be "
                "tracked, so the tape cannot be replayed faithfully.");
    if (t.defined() && slot.external_version.has_value()) {
      TORCH_CHECK(tensorVersion(t) == *slot.external_version,
                  "The external tensor used by the recorded op `",
                  node.op->name,
                  "` was modified in place after recording; the tape can no "
                  "longer be replayed faithfully.");
    }
    return t;
  });
  return stack;
}

// Ensures `node` is Done before returning. Called with `lock` held (at
// recursion depth 1); releases the tape lock around the op execution so
// other threads can build call stacks, record, and run disjoint nodes
// concurrently. If another thread is already executing the node, waits
// for it instead of double-replaying.
void replayNode(std::unique_lock<std::recursive_mutex>& lock,
                const std::shared_ptr<OpNode>& node) {
  while (node->running) {
    tape_cv.wait(lock);
  }
  if (node->materialized) {
    return;
  }
  TORCH_CHECK(node->op.has_value(),
              "This deferred-init tape segment has already been freed and "
              "cannot be replayed again.");
  // Dependencies are Done (the caller replays in chronological order and
  // waits on in-flight nodes), so the argument stack can be built now.
  torch::jit::Stack stack = materializeArguments(*node);
  node->running = true;
  lock.unlock();
  try {
    at::ThreadLocalStateGuard tls_guard{node->op->tls};
    // Replay must run for real regardless of any ambient fake/deferred
    // mode captured in the snapshot or active on this thread.
    c10::impl::ExcludeDispatchKeyGuard no_deferred{kDeferredKey};
    c10::impl::ExcludeDispatchKeyGuard no_fake{kFakeKey};
    node->op->run(stack);
  } catch (...) {
    lock.lock();
    node->running = false;
    tape_cv.notify_all();
    throw;
  }
  lock.lock();
  node->running = false;
  node->outputs.clear();
  visitTensors(stack, 0, stack.size(),
               [&](const at::Tensor& t) { node->outputs.push_back(t); });
  node->materialized = true;
  // Free the tape incrementally: drop the call frame and the dependency
  // edges (reference detachDependencies, deferred_init.cc:524).
  node->op.reset();
  node->input_slots.clear();
  node->input_slots.shrink_to_fit();
  tape_cv.notify_all();
}

}  // namespace

void enterDeferredInit() {
  if (deferred_level++ == 0) {
    beginRngSession();
    c10::impl::tls_set_dispatch_key_included(kDeferredKey, true);
    installProxyHooks();
  }
}

void leaveDeferredInit() {
  TORCH_CHECK(deferred_level > 0, "Not in a deferred-init context.");
  if (--deferred_level == 0) {
    c10::impl::tls_set_dispatch_key_included(kDeferredKey, false);
    removeProxyHooks();
    endRngSession();
  }
}

bool isDeferredInitActive() noexcept {
  return deferred_level > 0;
}

NoDeferredInit::NoDeferredInit()
    : prev_{c10::impl::tls_is_dispatch_key_excluded(kDeferredKey)} {
  c10::impl::tls_set_dispatch_key_excluded(kDeferredKey, true);
}

NoDeferredInit::~NoDeferredInit() {
  c10::impl::tls_set_dispatch_key_excluded(kDeferredKey, prev_);
}

bool canMaterialize(const at::Tensor& tensor) noexcept {
  auto* fake = asFake(tensor);
  return fake != nullptr && getRecord(fake) != nullptr;
}

std::optional<RecordInfo> recordInfo(const at::Tensor& tensor) {
  std::lock_guard<std::recursive_mutex> lock{tape_mutex};
  auto* fake = asFake(tensor);
  if (fake == nullptr) {
    return std::nullopt;
  }
  auto rec = getRecord(fake);
  if (rec == nullptr) {
    return std::nullopt;
  }
  RecordInfo info;
  const auto& node = rec->desc.node;
  info.op_nr = node->op_nr;
  info.output_index = rec->desc.index;
  info.materialized = node->materialized;
  if (node->op.has_value()) {
    info.op_name = node->op->name;
  }
  if (!node->materialized) {
    for (const auto& n : buildCallStack(*rec)) {
      if (!n->materialized) {
        ++info.pending_ops;
      }
    }
  }
  return info;
}

namespace {

bool chainNameIs(const std::string& name, const char* prefix) {
  return name.rfind(prefix, 0) == 0;
}

// One value-affecting step of a simple init chain.
struct ChainStep {
  enum class Kind {
    kFactory,
    kUniform,
    kNormal,
    kBernoulli,
    kFill,
    kZero,
    kPass,
    // Elementwise op with only scalar parameters (erfinv_/mul_/add_/
    // clamp_/...): slicing commutes with it, so a shard just applies the
    // same op. This is what makes trunc_normal_-style init chains
    // (uniform_ -> erfinv_ -> mul_ -> add_ -> clamp_) slice-
    // materializable.
    kPointwise,
  } kind;
  double p0 = 0.0;  // from / mean / fill value
  double p1 = 0.0;  // to / std
  std::optional<std::pair<uint64_t, uint64_t>> philox;
  std::vector<c10::IValue> pointwise_args;  // non-self args for kPointwise
  // Replay closure of the recorded op (boxed call), for kPointwise.
  std::function<void(torch::jit::Stack&)> pointwise_run;
};

// In-place elementwise ops whose only parameters are scalars; slicing
// commutes with all of them.
bool isPointwiseScalarOpName(const std::string& name) {
  static const char* kNames[] = {
      "aten::erfinv_",    "aten::mul_",       "aten::add_",
      "aten::sub_",       "aten::div_",       "aten::clamp_",
      "aten::clamp_min_", "aten::clamp_max_", "aten::neg_",
      "aten::abs_",       "aten::sqrt_",      "aten::reciprocal_",
  };
  for (const char* n : kNames) {
    if (name == n) {
      return true;
    }
  }
  return false;
}

bool isScalarLikeArg(const c10::IValue& v) {
  if (v.isNone() || v.isScalar() || v.isDouble() || v.isInt() ||
      v.isBool()) {
    return true;
  }
  // Python scalars often reach the dispatcher as wrapped-number 0-dim
  // tensors (`t.mul_(2.0)` -> mul_.Tensor). A real 0-dim tensor operand
  // commutes with slicing just like a Scalar; an UNDEFINED tensor here
  // is a nulled-out fake placeholder (a tensor dependency the shard path
  // cannot substitute) and must be rejected.
  if (v.isTensor()) {
    const at::Tensor& t = v.toTensor();
    return t.defined() && t.dim() == 0 && !isFake(t);
  }
  return false;
}

ChainStep classifyChainNode(const OpNode& node) {
  TORCH_CHECK(node.op.has_value(),
              "slice materialization: the tape segment for `",
              (node.materialized ? "an already-materialized op"
                                 : "a freed op"),
              "` is gone; materialize the tensor fully instead.");
  const RecordedOp& op = *node.op;
  ChainStep step;
  auto scalarArg = [&](size_t i) -> double {
    const c10::IValue& v = op.args.at(i);
    if (v.isDouble()) {
      return v.toDouble();
    }
    if (v.isInt()) {
      return static_cast<double>(v.toInt());
    }
    if (v.isScalar()) {
      return v.toScalar().toDouble();
    }
    TORCH_CHECK(false, "slice materialization: unexpected argument type `",
                v.tagKind(), "` in `", op.name, "`");
  };
  if (chainNameIs(op.name, "aten::empty")) {
    step.kind = ChainStep::Kind::kFactory;
  } else if (chainNameIs(op.name, "aten::zeros")) {
    step.kind = ChainStep::Kind::kZero;
  } else if (chainNameIs(op.name, "aten::ones")) {
    step.kind = ChainStep::Kind::kFill;
    step.p0 = 1.0;
  } else if (chainNameIs(op.name, "aten::full")) {
    step.kind = ChainStep::Kind::kFill;
    step.p0 = scalarArg(1);
  } else if (op.name == "aten::uniform_" || op.name == "aten::normal_" ||
             op.name == "aten::bernoulli_") {
    for (size_t i = 1; i < op.args.size(); ++i) {
      TORCH_CHECK(!op.args[i].isGenerator(),
                  "slice materialization: `", op.name,
                  "` was recorded with an explicit generator, which the "
                  "counter-based shard path cannot honor; materialize the "
                  "tensor fully instead.");
    }
    if (op.name == "aten::bernoulli_") {
      step.kind = ChainStep::Kind::kBernoulli;
      step.p0 = scalarArg(1);
    } else {
      step.kind = op.name == "aten::uniform_" ? ChainStep::Kind::kUniform
                                              : ChainStep::Kind::kNormal;
      step.p0 = scalarArg(1);
      step.p1 = scalarArg(2);
    }
    step.philox = op.philox;
  } else if (op.name == "aten::fill_") {
    step.kind = ChainStep::Kind::kFill;
    step.p0 = scalarArg(1);
  } else if (op.name == "aten::zero_") {
    step.kind = ChainStep::Kind::kZero;
  } else if (op.name == "aten::detach" || op.name == "tdx::variable_data" ||
             op.name == "aten::alias") {
    step.kind = ChainStep::Kind::kPass;
  } else if (isPointwiseScalarOpName(op.name) &&
             std::all_of(op.args.begin() + 1, op.args.end(),
                         isScalarLikeArg)) {
    step.kind = ChainStep::Kind::kPointwise;
    step.pointwise_args.assign(op.args.begin() + 1, op.args.end());
    step.pointwise_run = op.run;
  } else {
    TORCH_CHECK(false, "slice materialization: `", op.name,
                "` is not a whole-tensor init op; materialize the tensor "
                "fully instead.");
  }
  return step;
}

// A dim-d slice of a contiguous tensor, flattened: n_blocks contiguous
// global element ranges of block_len elements, block r starting at
// g_off + r * g_stride. dim 0 is the n_blocks == 1 case and takes the
// flat shard ops (whose boundary handling is tuned for it); dim > 0
// takes the windowed ops.
struct ShardWindow {
  int64_t n_blocks;
  int64_t block_len;
  int64_t g_stride;
  int64_t g_off;
};

void applyShardStep(const ChainStep& step, at::Tensor& shard,
                    const ShardWindow& w) {
  switch (step.kind) {
    case ChainStep::Kind::kFactory:
    case ChainStep::Kind::kPass:
      return;
    case ChainStep::Kind::kPointwise: {
      torch::jit::Stack s;
      s.emplace_back(shard);
      for (const c10::IValue& a : step.pointwise_args) {
        s.push_back(a);
      }
      c10::impl::ExcludeDispatchKeyGuard no_deferred{kDeferredKey};
      c10::impl::ExcludeDispatchKeyGuard no_fake{kFakeKey};
      step.pointwise_run(s);
      shard = s.back().toTensor();
      return;
    }
    case ChainStep::Kind::kZero:
      shard.zero_();
      return;
    case ChainStep::Kind::kFill:
      shard.fill_(step.p0);
      return;
    case ChainStep::Kind::kUniform:
    case ChainStep::Kind::kNormal:
    case ChainStep::Kind::kBernoulli: {
      TORCH_CHECK(step.philox.has_value(),
                  "slice materialization: this RNG op carries no pinned "
                  "Philox state (was it recorded by an older build?)");
      static const auto uniform_shard =
          c10::Dispatcher::singleton()
              .findSchemaOrThrow("tdx::uniform_shard_", "")
              .typed<at::Tensor&(at::Tensor&, int64_t, int64_t, double,
                                 double, int64_t, int64_t)>();
      static const auto normal_shard =
          c10::Dispatcher::singleton()
              .findSchemaOrThrow("tdx::normal_shard_", "")
              .typed<at::Tensor&(at::Tensor&, int64_t, int64_t, double,
                                 double, int64_t, int64_t)>();
      static const auto bernoulli_shard =
          c10::Dispatcher::singleton()
              .findSchemaOrThrow("tdx::bernoulli_shard_", "")
              .typed<at::Tensor&(at::Tensor&, int64_t, int64_t, double,
                                 int64_t, int64_t)>();
      static const auto uniform_shard_win =
          c10::Dispatcher::singleton()
              .findSchemaOrThrow("tdx::uniform_shard_win_", "")
              .typed<at::Tensor&(at::Tensor&, int64_t, int64_t, int64_t,
                                 int64_t, double, double, int64_t,
                                 int64_t)>();
      static const auto normal_shard_win =
          c10::Dispatcher::singleton()
              .findSchemaOrThrow("tdx::normal_shard_win_", "")
              .typed<at::Tensor&(at::Tensor&, int64_t, int64_t, int64_t,
                                 int64_t, double, double, int64_t,
                                 int64_t)>();
      static const auto bernoulli_shard_win =
          c10::Dispatcher::singleton()
              .findSchemaOrThrow("tdx::bernoulli_shard_win_", "")
              .typed<at::Tensor&(at::Tensor&, int64_t, int64_t, int64_t,
                                 int64_t, double, int64_t, int64_t)>();
      const auto& [seed, offset] = *step.philox;
      const auto s64 = static_cast<int64_t>(seed);
      const auto o64 = static_cast<int64_t>(offset);
      if (w.n_blocks == 1) {
        const int64_t start = w.g_off;
        const int64_t end = w.g_off + w.block_len;
        if (step.kind == ChainStep::Kind::kUniform) {
          uniform_shard.call(shard, start, end, step.p0, step.p1, s64, o64);
        } else if (step.kind == ChainStep::Kind::kNormal) {
          normal_shard.call(shard, start, end, step.p0, step.p1, s64, o64);
        } else {
          bernoulli_shard.call(shard, start, end, step.p0, s64, o64);
        }
      } else if (step.kind == ChainStep::Kind::kUniform) {
        uniform_shard_win.call(shard, w.n_blocks, w.block_len, w.g_stride,
                               w.g_off, step.p0, step.p1, s64, o64);
      } else if (step.kind == ChainStep::Kind::kNormal) {
        normal_shard_win.call(shard, w.n_blocks, w.block_len, w.g_stride,
                              w.g_off, step.p0, step.p1, s64, o64);
      } else {
        bernoulli_shard_win.call(shard, w.n_blocks, w.block_len, w.g_stride,
                                 w.g_off, step.p0, s64, o64);
      }
      return;
    }
  }
}

}  // namespace

// Reduces a deferred tensor's tape chain to its final whole-tensor value
// step, when the chain is that simple. Every simple-chain value step
// (uniform/normal/bernoulli/fill/zero) overwrites the whole tensor, so
// only the LAST one determines the bits: batched replay can then fill
// many tensors with ONE kernel launch. Returns nullopt for anything
// else (pointwise tails, views, cross-tensor deps, explicit
// generators); callers fall back to ordinary materialization.
std::optional<InitPlan> tensorInitPlan(const at::Tensor& tensor) {
  auto* fake = asFake(tensor);
  if (fake == nullptr) {
    return std::nullopt;
  }
  auto rec = getRecord(fake);
  if (rec == nullptr) {
    return std::nullopt;
  }
  const at::Tensor& meta = fake->meta_tensor();
  if (!meta.is_contiguous()) {
    return std::nullopt;
  }
  std::lock_guard<std::recursive_mutex> lock{tape_mutex};
  InitPlan plan;
  try {
    auto nodes = buildCallStack(*rec);
    bool have_value = false;
    for (const auto& n : nodes) {
      ChainStep step = classifyChainNode(*n);
      switch (step.kind) {
        case ChainStep::Kind::kFactory:
        case ChainStep::Kind::kPass:
          break;
        case ChainStep::Kind::kPointwise:
          return std::nullopt;  // needs sequential application
        case ChainStep::Kind::kZero:
          plan.kind = InitPlan::Kind::kZero;
          have_value = true;
          break;
        case ChainStep::Kind::kFill:
          plan.kind = InitPlan::Kind::kFill;
          plan.p0 = step.p0;
          have_value = true;
          break;
        case ChainStep::Kind::kUniform:
        case ChainStep::Kind::kNormal:
        case ChainStep::Kind::kBernoulli:
          if (!step.philox.has_value()) {
            return std::nullopt;
          }
          plan.kind = step.kind == ChainStep::Kind::kUniform
                          ? InitPlan::Kind::kUniform
                          : (step.kind == ChainStep::Kind::kNormal
                                 ? InitPlan::Kind::kNormal
                                 : InitPlan::Kind::kBernoulli);
          plan.p0 = step.p0;
          plan.p1 = step.p1;
          plan.seed = step.philox->first;
          plan.offset = step.philox->second;
          have_value = true;
          break;
      }
    }
    if (!have_value) {
      plan.kind = InitPlan::Kind::kFactory;  // allocate-only (empty)
    }
  } catch (const c10::Error&) {
    return std::nullopt;  // not a simple chain
  }
  plan.sizes.assign(meta.sizes().begin(), meta.sizes().end());
  plan.dtype = meta.scalar_type();
  plan.device = fake->fake_device();
  plan.requires_grad = tensor.requires_grad();
  return plan;
}

at::Tensor materializeTensorShard(const at::Tensor& tensor,
                                  int64_t start_row,
                                  int64_t end_row,
                                  int64_t dim) {
  auto* fake = asFake(tensor);
  TORCH_CHECK_VALUE(fake != nullptr && getRecord(fake) != nullptr,
                    "`tensor` is not a deferred tensor.");
  const at::Tensor& meta = fake->meta_tensor();
  TORCH_CHECK(meta.is_contiguous(),
              "slice materialization requires a contiguous tensor");
  TORCH_CHECK(0 <= dim && (dim < meta.dim() || (dim == 0 && meta.dim() == 0)),
              "invalid slice dim ", dim, " for a ", meta.dim(),
              "-d tensor");
  const int64_t dim_len = meta.dim() == 0 ? 1 : meta.size(dim);
  TORCH_CHECK(0 <= start_row && start_row <= end_row && end_row <= dim_len,
              "invalid slice range [", start_row, ", ", end_row,
              ") for size ", dim_len, " along dim ", dim);
  // full.narrow(dim, start, end - start) flattened = a strided window
  // over the full element space: prod(sizes[:dim]) blocks of
  // (end - start) * prod(sizes[dim+1:]) contiguous elements each.
  int64_t inner = 1;  // prod(sizes[dim+1:])
  for (int64_t d = dim + 1; d < meta.dim(); ++d) {
    inner *= meta.size(d);
  }
  int64_t n_blocks = 1;  // prod(sizes[:dim])
  for (int64_t d = 0; d < dim; ++d) {
    n_blocks *= meta.size(d);
  }
  const ShardWindow w{n_blocks, (end_row - start_row) * inner,
                      dim_len * inner, start_row * inner};

  std::lock_guard<std::recursive_mutex> lock{tape_mutex};
  auto rec = getRecord(fake);
  auto nodes = buildCallStack(*rec);
  std::vector<ChainStep> steps;
  steps.reserve(nodes.size());
  for (const auto& n : nodes) {
    steps.push_back(classifyChainNode(*n));
  }
  TORCH_CHECK(!steps.empty() &&
                  (steps.front().kind == ChainStep::Kind::kFactory ||
                   steps.front().kind == ChainStep::Kind::kZero ||
                   steps.front().kind == ChainStep::Kind::kFill),
              "slice materialization: the tape does not start with a "
              "factory op");

  std::vector<int64_t> shard_sizes(meta.sizes().begin(), meta.sizes().end());
  if (!shard_sizes.empty()) {
    shard_sizes[dim] = end_row - start_row;
  }
  at::Tensor shard = at::empty(
      shard_sizes, at::TensorOptions()
                       .dtype(meta.scalar_type())
                       .device(fake->fake_device()));
  for (const ChainStep& step : steps) {
    applyShardStep(step, shard, w);
  }
  return shard;
}

at::Tensor materializeTensor(const at::Tensor& tensor) {
  auto* fake = asFake(tensor);
  if (fake == nullptr) {
    return tensor;  // identity for real tensors
  }
  auto rec = getRecord(fake);
  TORCH_CHECK_VALUE(rec != nullptr,
                    "`tensor` is fake but carries no deferred-init record, "
                    "so it cannot be materialized.");

  std::unique_lock<std::recursive_mutex> lock{tape_mutex};
  // Capture the producer descriptor while holding the lock: a concurrent
  // recording thread may restamp `rec->desc` while this thread's replay
  // runs unlocked, and the pre-restamp value is the one this call stack
  // materializes (a legal serialization of the two racing calls).
  OpOutputDescriptor desc = rec->desc;
  for (const auto& n : buildCallStack(*rec)) {
    replayNode(lock, n);
  }

  at::Tensor out = desc.node->outputs.at(desc.index);
  // requires_grad_() is untraceable on purpose; re-apply the fake tensor's
  // final autograd flag explicitly (reference deferred_init.cc:713-729).
  if (out.is_leaf() && out.requires_grad() != tensor.requires_grad()) {
    out.set_requires_grad(tensor.requires_grad());
  }
  return out;
}

}  // namespace tdx
