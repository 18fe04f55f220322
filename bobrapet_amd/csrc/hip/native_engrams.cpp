// GIL-free built-in engram bodies for the bobraccel NativeLane.
//
// The DAG core (_core.so) dispatches built-in engram steps here through a
// C++ function table (csrc/core/native_lane.h) instead of the Python
// launcher: the whole body — input parsing, cached tables, HIP kernel
// launch, payload registration — runs on the core's loop thread with NO
// GIL; completion is event-polled (hipEventQuery), so sub-millisecond
// stories stop being Python-bound.
//
// Implemented kinds:
//   1 = embed            (the parallel8 branch body: gather+pool+L2norm)
//   2 = allgather-join   (world==1 local join: concat + register)
// Output tensors live in a lane-held registry keyed "native/t-<n>"; the
// Python TensorStore resolves those keys through native_tensor_get (so
// mixed native/Python stories and tests keep working), and the core's
// gc_run frees them via free_key.
#include <ATen/ATen.h>
#include <c10/hip/HIPCachingAllocator.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <deque>
#include <mutex>
#include <string>
#include <unordered_map>
#include <vector>
#include <chrono>

#include "../core/jvalue.h"
#include "../core/native_lane.h"

extern "C" void launch_embed_pool(void*, void*, const void*, const void*, int,
                                  int, int, int, hipStream_t);

namespace {

using bobraccel::JArray;
using bobraccel::JObject;
using bobraccel::JValue;
using bobraccel::NativeLane;

double now_s() {
  return std::chrono::duration<double>(
             std::chrono::steady_clock::now().time_since_epoch())
      .count();
}

int64_t jint(const JValue& v, const char* key, int64_t dflt) {
  if (!v.is_object()) return dflt;
  const JObject& o = v.as_object();
  auto it = o.find(key);
  if (it == o.end()) return dflt;
  if (it->second.is_int()) return it->second.as_int();
  if (it->second.is_double()) return (int64_t)it->second.as_double();
  return dflt;
}

struct Entry {
  at::Tensor t;
  hipEvent_t ready = nullptr;  // recorded on the producing stream
};

struct Ticket {
  hipEvent_t ev = nullptr;  // non-owning (registry owns)
  JValue out;
  double t0 = 0;
};

struct LaneState {
  std::mutex mu;
  std::unordered_map<std::string, Entry> registry;
  std::deque<std::string> order;  // FIFO eviction
  std::unordered_map<long, Ticket> tickets;
  long next_key = 1;
  long next_ticket = 1;
  // cached synthetic tables/ids (mirror engrams/embed.py caches)
  std::unordered_map<std::string, at::Tensor> cache;
  std::vector<std::vector<c10::hip::HIPStream>> streams;  // per device
  int rr = 0;
  long launches = 0;
  long fallbacks = 0;
  long failures = 0;
  std::string last_error;
};

LaneState& S() {
  static LaneState s;
  return s;
}

constexpr size_t kRegistryCap = 16384;

c10::hip::HIPStream pick_stream(int device) {
  LaneState& s = S();
  if ((int)s.streams.size() <= device) s.streams.resize(device + 1);
  auto& pool = s.streams[device];
  if (pool.empty()) {
    for (int i = 0; i < 4; ++i)
      pool.push_back(c10::hip::getStreamFromPool(false, device));
  }
  return pool[(s.rr++) & 3];
}

std::string register_tensor(at::Tensor t, hipEvent_t ev) {  // mu held
  LaneState& s = S();
  std::string key = "native/t-" + std::to_string(s.next_key++);
  if (s.order.size() >= kRegistryCap) {
    const std::string& old = s.order.front();
    auto it = s.registry.find(old);
    if (it != s.registry.end()) {
      if (it->second.ready) hipEventDestroy(it->second.ready);
      s.registry.erase(it);
    }
    s.order.pop_front();
  }
  s.registry[key] = Entry{std::move(t), ev};
  s.order.push_back(key);
  return key;
}

JValue tensor_ref(const std::string& key, const at::Tensor& t) {
  JObject meta;
  meta["key"] = key;
  meta["kind"] = std::string("tensor");
  meta["dtype"] = std::string(t.scalar_type() == at::kBFloat16 ? "bfloat16"
                              : t.scalar_type() == at::kFloat  ? "float32"
                                                               : "other");
  JArray shape;
  for (auto d : t.sizes()) shape.push_back((int64_t)d);
  meta["shape"] = std::move(shape);
  meta["device"] = std::string("cuda:") + std::to_string(t.get_device());
  meta["size"] = (int64_t)(t.numel() * t.element_size());
  JObject ref;
  ref["$storageRef"] = std::move(meta);
  return JValue(std::move(ref));
}

at::Tensor cached_table(int64_t vocab, int64_t dim, int64_t seed, int device) {
  LaneState& s = S();  // mu held
  std::string key = "tab/" + std::to_string(vocab) + "/" + std::to_string(dim) +
                    "/" + std::to_string(seed) + "/" + std::to_string(device);
  auto it = s.cache.find(key);
  if (it != s.cache.end()) return it->second;
  auto gen = at::detail::createCPUGenerator((uint64_t)seed);
  at::Tensor cpu = at::empty({vocab, dim}, at::dtype(at::kFloat));
  cpu.normal_(0.0, 0.05, gen);
  at::Tensor t = cpu.to(at::device(at::kCUDA).dtype(at::kBFloat16),
                        /*non_blocking=*/false);
  if (t.get_device() != device)
    t = t.to(at::Device(at::kCUDA, device));
  s.cache[key] = t;
  return t;
}

at::Tensor cached_ids(int64_t vocab, int64_t batch, int64_t seq, int64_t sd,
                      int device) {
  LaneState& s = S();  // mu held
  std::string key = "ids/" + std::to_string(vocab) + "/" + std::to_string(batch) +
                    "/" + std::to_string(seq) + "/" + std::to_string(sd) + "/" +
                    std::to_string(device);
  auto it = s.cache.find(key);
  if (it != s.cache.end()) return it->second;
  auto gen = at::detail::createCPUGenerator((uint64_t)sd);
  at::Tensor cpu = at::randint(0, vocab, {batch, seq}, gen,
                               at::dtype(at::kLong));
  at::Tensor t = cpu.to(at::device(at::Device(at::kCUDA, device)).dtype(at::kInt));
  s.cache[key] = t;
  return t;
}

// ---------------------------------------------------------------------------

long embed_launch(const JValue& cfg, const JValue& inp, int device) {
  LaneState& s = S();
  std::lock_guard<std::mutex> g(s.mu);
  const int64_t dim = jint(inp, "dim", jint(cfg, "dim", 4096));
  const int64_t vocab = jint(inp, "vocab", jint(cfg, "vocab", 32000));
  const int64_t seed = jint(cfg, "seed", 7);
  if (inp.is_object() && inp.as_object().count("ids"))
    return 0;  // explicit ids: Python path handles it
  const int64_t batch = jint(inp, "batch", jint(cfg, "batch", 32));
  const int64_t seq = jint(inp, "seqLen", jint(cfg, "seqLen", 128));
  const int64_t sd = jint(inp, "seed", 0);
  if (dim % 8 != 0 || dim > 8192) return 0;

  at::Tensor table = cached_table(vocab, dim, seed, device);
  at::Tensor ids = cached_ids(vocab, batch, seq, sd, device);

  auto stream = pick_stream(device);
  c10::hip::setCurrentHIPStream(stream);  // allocations tag this stream
  at::Tensor out = at::empty({batch, dim},
                             at::device(at::Device(at::kCUDA, device))
                                 .dtype(at::kBFloat16));
  const int nchunk = (int)((seq + 7) / 8);
  at::Tensor pooled = at::zeros({batch * nchunk, dim},
                                at::device(at::Device(at::kCUDA, device))
                                    .dtype(at::kFloat));
  launch_embed_pool(out.data_ptr(), pooled.data_ptr(), table.data_ptr(),
                    ids.data_ptr(), (int)batch, (int)seq, (int)dim, (int)vocab,
                    stream.stream());
  hipEvent_t ev;
  hipEventCreateWithFlags(&ev, hipEventDisableTiming);
  hipEventRecord(ev, stream.stream());

  std::string key = register_tensor(out, ev);
  JObject o;
  o["batch"] = (int64_t)batch;
  o["dim"] = (int64_t)dim;
  o["embeddings"] = tensor_ref(key, out);
  long tk = s.next_ticket++;
  s.tickets[tk] = Ticket{ev, JValue(std::move(o)), now_s()};
  return tk;
}

long join_launch(const JValue& cfg, const JValue& inp, int device) {
  (void)cfg;
  LaneState& s = S();
  std::lock_guard<std::mutex> g(s.mu);
  if (!inp.is_object()) return 0;
  const JObject& o = inp.as_object();
  std::vector<std::string> keys;
  auto collect_ref = [&](const JValue& ref) -> bool {
    if (!ref.is_object()) return false;
    auto it = ref.as_object().find("$storageRef");
    if (it == ref.as_object().end() || !it->second.is_object()) return false;
    auto kit = it->second.as_object().find("key");
    if (kit == it->second.as_object().end() || !kit->second.is_string())
      return false;
    const std::string& k = kit->second.as_string();
    if (k.rfind("native/", 0) != 0) return false;  // mixed story: fallback
    keys.push_back(k);
    return true;
  };
  auto bit = o.find("branches");
  if (bit != o.end() && bit->second.is_object()) {
    for (const auto& [name, bout] : bit->second.as_object()) {  // sorted
      if (!bout.is_object()) continue;
      auto e = bout.as_object().find("embeddings");
      if (e == bout.as_object().end())
        e = bout.as_object().find("logits");
      if (e == bout.as_object().end()) continue;
      if (!collect_ref(e->second)) return 0;
    }
  } else {
    auto rit = o.find("refs");
    if (rit == o.end() || !rit->second.is_array()) return 0;
    for (const auto& r : rit->second.as_array())
      if (!collect_ref(r)) return 0;
  }
  if (keys.empty()) return 0;

  std::vector<at::Tensor> parts;
  std::vector<hipEvent_t> waits;
  for (const std::string& k : keys) {
    auto it = s.registry.find(k);
    if (it == s.registry.end()) {
      // a lane-minted ref no longer held: failing loudly beats handing
      // lane tensors to the Python join (mixed-runtime hazard)
      s.failures++;
      s.last_error = "native join: ref " + k + " not in the lane registry";
      return -1;
    }
    parts.push_back(it->second.t.reshape({-1, it->second.t.size(-1)}));
    if (it->second.ready) waits.push_back(it->second.ready);
  }
  const int dev = parts[0].get_device();
  auto stream = pick_stream(dev);
  for (hipEvent_t ev : waits) hipStreamWaitEvent(stream.stream(), ev, 0);
  c10::hip::setCurrentHIPStream(stream);
  at::Tensor local = at::cat(parts, 0);
  // allocator stream-use records via the HIP API (Tensor::record_stream
  // rejects the raw HIP stream under the masquerading layer)
  for (auto& p : parts)
    c10::hip::HIPCachingAllocator::recordStream(p.storage().data_ptr(), stream);
  // world == 1 (the Python glue only routes the join natively then):
  // the all-gather is the identity — `joined` IS the local concat
  hipEvent_t ev;
  hipEventCreateWithFlags(&ev, hipEventDisableTiming);
  hipEventRecord(ev, stream.stream());
  std::string key = register_tensor(local, ev);

  JObject out;
  out["rows"] = (int64_t)local.size(0);
  out["dim"] = (int64_t)local.size(-1);
  out["worldRows"] = (int64_t)local.size(0);
  out["world"] = (int64_t)1;
  out["joined"] = tensor_ref(key, local);
  long tk = s.next_ticket++;
  s.tickets[tk] = Ticket{ev, JValue(std::move(out)), now_s()};
  return tk;
}

long fail_ticket(const std::string& msg) {
  LaneState& s = S();  // mu held by caller? no — take it
  std::lock_guard<std::mutex> g(s.mu);
  s.failures++;
  s.last_error = msg;
  long tk = s.next_ticket++;
  Ticket t;
  t.ev = nullptr;
  t.out = JValue(msg);  // poll() reports it as the error
  s.tickets[tk] = std::move(t);
  return tk;  // a positive ticket that polls straight to failure
}

long lane_launch(void*, int kind, const JValue* cfg, const JValue* input,
                 int device) {
  long r = 0;
  try {
    if (kind == 1)
      r = embed_launch(*cfg, *input, device);
    else if (kind == 2)
      r = join_launch(*cfg, *input, device);
  } catch (const std::exception& e) {
    // internal error mid-flight: fail the step loudly (falling back to
    // the Python body after lane side effects mixes runtimes)
    return fail_ticket(std::string("native lane: ") + e.what());
  }
  LaneState& s = S();
  std::string err;
  {
    std::lock_guard<std::mutex> g(s.mu);
    if (r > 0) s.launches++;
    if (r == 0) s.fallbacks++;
    if (r >= 0) return r;
    err = s.last_error;
  }
  return fail_ticket(err);  // takes the mutex itself
}

int lane_poll(void*, long ticket, JValue* out, std::string* err) {
  LaneState& s = S();
  std::lock_guard<std::mutex> g(s.mu);
  auto it = s.tickets.find(ticket);
  if (it == s.tickets.end()) {
    *err = "unknown native ticket";
    return -1;
  }
  if (it->second.ev == nullptr) {  // pre-failed ticket
    *err = it->second.out.is_string() ? it->second.out.as_string()
                                      : "native lane failure";
    s.tickets.erase(it);
    return -1;
  }
  hipError_t rc = hipEventQuery(it->second.ev);
  if (rc == hipErrorNotReady) return 0;
  if (rc != hipSuccess) {
    *err = std::string("hip error: ") + hipGetErrorString(rc);
    s.tickets.erase(it);
    return -1;
  }
  JValue o = std::move(it->second.out);
  if (o.is_object())
    o.as_object()["latencyMs"] = (now_s() - it->second.t0) * 1000.0;
  *out = std::move(o);
  s.tickets.erase(it);
  return 1;
}

void lane_free_key(void*, const char* key) {
  LaneState& s = S();
  std::lock_guard<std::mutex> g(s.mu);
  auto it = s.registry.find(key);
  if (it == s.registry.end()) return;
  if (it->second.ready) hipEventDestroy(it->second.ready);
  s.registry.erase(it);
}

NativeLane g_lane{1, nullptr, &lane_launch, &lane_poll, &lane_free_key};

}  // namespace

// ---- hooks for the torch binding layer (same .so) -------------------------

const void* bobra_native_lane_ptr() { return &g_lane; }

bool bobra_native_tensor_get(const std::string& key, at::Tensor* out) {
  LaneState& s = S();
  at::Tensor t;
  hipEvent_t ev = nullptr;
  {
    std::lock_guard<std::mutex> g(s.mu);
    auto it = s.registry.find(key);
    if (it == s.registry.end()) return false;
    t = it->second.t;
    ev = it->second.ready;
  }
  // make the payload safe to read on the caller's stream; sync OUTSIDE
  // the lane mutex so a slow event never stalls the core's launch path
  if (ev) (void)hipEventSynchronize(ev);
  *out = t;
  return true;
}

size_t bobra_native_registry_size() {
  LaneState& s = S();
  std::lock_guard<std::mutex> g(s.mu);
  return s.registry.size();
}

void bobra_native_lane_stats(long* launches, long* fallbacks, long* failures,
                             std::string* last_error) {
  LaneState& s = S();
  std::lock_guard<std::mutex> g(s.mu);
  *launches = s.launches;
  *fallbacks = s.fallbacks;
  *failures = s.failures;
  *last_error = s.last_error;
}
