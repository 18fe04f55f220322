// ThreadSanitizer driver for the bobraccel native engine (SURVEY §5.2:
// "TSan-clean C++ scheduler").  Pure C++ — the pytest suite runs the same
// engine under Python, but an LD_PRELOAD TSan over an uninstrumented
// CPython cannot model the GIL/clockwait handoffs (GCC-11's libtsan has
// no pthread_cond_clockwait interceptor), so the sanitizer run lives
// here with the WHOLE binary instrumented.
//
// Exercises, concurrently: submits from 4 threads, engram completions
// from 2 worker threads (through the queued launcher), the native-lane
// ticket path (stub lane: tickets complete after 2 polls), gate
// decisions, cancels, run_status/run_count snapshots, wait() and
// gc_run().  Build + run:  bash tests/tsan/run_tsan.sh
#include <atomic>
#include <condition_variable>
#include <cstdio>
#include <deque>
#include <mutex>
#include <thread>
#include <unordered_map>
#include <vector>

#include "engine.h"
#include "native_lane.h"

using namespace bobraccel;

// ---- queued launcher: the engine invokes it on the loop thread (mu_
// held); completions must come from other threads, as in production
struct LaunchReq {
  uint64_t run;
  int step;
  int branch;
  uint32_t attempt;
};
static std::mutex q_mu;
static std::condition_variable q_cv;
static std::deque<LaunchReq> q;
static std::atomic<bool> q_done{false};

// ---- stub native lane: tickets complete after two polls
struct FakeLane {
  std::mutex mu;
  std::unordered_map<long, int> polls;
  long next = 1;
};
static long fl_launch(void* self, int kind, const JValue* cfg,
                      const JValue* input, int device) {
  auto* fl = static_cast<FakeLane*>(self);
  std::lock_guard<std::mutex> g(fl->mu);
  long t = fl->next++;
  fl->polls[t] = 0;
  (void)kind;
  (void)cfg;
  (void)input;
  (void)device;
  return t;
}
static int fl_poll(void* self, long ticket, JValue* out, std::string* err) {
  auto* fl = static_cast<FakeLane*>(self);
  std::lock_guard<std::mutex> g(fl->mu);
  auto it = fl->polls.find(ticket);
  if (it == fl->polls.end()) {
    *err = "unknown ticket";
    return -1;
  }
  if (++it->second < 2) return 0;
  fl->polls.erase(it);
  JObject o;
  o["native"] = true;
  *out = JValue(std::move(o));
  return 1;
}
static void fl_free(void* self, const char* key) {
  (void)self;
  (void)key;
}

static PlanStep engram_step(std::string name, std::vector<int> deps,
                            int native_kind = 0) {
  PlanStep s;
  s.name = std::move(name);
  s.kind = StepKind::Engram;
  s.engram = "echo";
  s.native_kind = native_kind;
  s.deps = std::move(deps);
  s.dep_allow_failure.assign(s.deps.size(), false);
  return s;
}

static Plan finish(Plan p) {
  p.dependents.assign(p.steps.size(), {});
  for (size_t i = 0; i < p.steps.size(); ++i)
    for (int d : p.steps[i].deps) p.dependents[d].push_back((int)i);
  return p;
}

int main() {
  NativeEngine eng;
  FakeLane fl;
  NativeLane lane;
  lane.self = &fl;
  lane.launch = &fl_launch;
  lane.poll = &fl_poll;
  lane.free_key = &fl_free;
  eng.set_native_lane(&lane);
  eng.set_devices({0});
  eng.set_launcher([](uint64_t run, int step, int branch, uint32_t attempt,
                      const std::string& engram, const std::string& sname,
                      const JValue& with) {
    (void)engram;
    (void)sname;
    (void)with;
    std::lock_guard<std::mutex> g(q_mu);
    q.push_back({run, step, branch, attempt});
    q_cv.notify_all();
  });

  // plan A: linear engram chain with a sleep and a parallel fan-out
  Plan a;
  a.name = "chain";
  a.steps.push_back(engram_step("s0", {}));
  {
    PlanStep sl;
    sl.name = "nap";
    sl.kind = StepKind::Sleep;
    sl.sleep_duration = 0.001;
    sl.deps = {0};
    sl.dep_allow_failure = {false};
    a.steps.push_back(std::move(sl));
  }
  {
    PlanStep par;
    par.name = "fan";
    par.kind = StepKind::Parallel;
    par.deps = {1};
    par.dep_allow_failure = {false};
    for (int b = 0; b < 4; ++b) par.branches.push_back(engram_step("b" + std::to_string(b), {}));
    a.steps.push_back(std::move(par));
  }
  a.steps.push_back(engram_step("tail", {2}));
  int plan_a = eng.register_plan(finish(std::move(a)));

  // plan B: native-lane step + gate
  Plan b;
  b.name = "lane-gate";
  b.steps.push_back(engram_step("embed", {}, /*native_kind=*/1));
  {
    PlanStep g2;
    g2.name = "approve";
    g2.kind = StepKind::Gate;
    g2.deps = {0};
    g2.dep_allow_failure = {false};
    b.steps.push_back(std::move(g2));
  }
  b.steps.push_back(engram_step("after", {1}));
  int plan_b = eng.register_plan(finish(std::move(b)));

  // plan C: executeStory of plan A
  Plan c;
  c.name = "outer";
  {
    PlanStep xs;
    xs.name = "sub";
    xs.kind = StepKind::ExecuteStory;
    xs.target_plan = plan_a;
    c.steps.push_back(std::move(xs));
  }
  int plan_c = eng.register_plan(finish(std::move(c)));

  eng.start();

  // engram workers: complete queued launches (some after a tiny delay)
  std::vector<std::thread> workers;
  for (int w = 0; w < 2; ++w) {
    workers.emplace_back([&eng, w] {
      for (;;) {
        LaunchReq r;
        {
          std::unique_lock<std::mutex> g(q_mu);
          q_cv.wait(g, [] { return q_done.load() || !q.empty(); });
          if (q.empty()) {
            if (q_done.load()) return;
            continue;
          }
          r = q.front();
          q.pop_front();
        }
        JObject out;
        out["v"] = (int64_t)(r.step + w);
        eng.complete_engram(r.run, r.step, r.branch, r.attempt, 0,
                            JValue(std::move(out)), "");
      }
    });
  }

  const int kRuns = 150;
  std::vector<uint64_t> ids[4];
  std::vector<std::thread> subs;
  for (int t = 0; t < 4; ++t) {
    subs.emplace_back([&, t] {
      for (int i = 0; i < kRuns; ++i) {
        int plan = (i % 3 == 0) ? plan_c : (i % 3 == 1) ? plan_a : plan_b;
        JObject in;
        in["i"] = (int64_t)i;
        ids[t].push_back(eng.submit(plan, JValue(std::move(in))));
      }
    });
  }
  // gate decider + canceller + status pollers race the submitters
  std::atomic<bool> stop_aux{false};
  std::thread gates([&] {
    while (!stop_aux.load()) {
      // blanket-approve: decide_gate on non-gate steps / unknown runs is a
      // no-op, and the id space covers executeStory children too
      for (uint64_t id = 1; id < 2500; ++id)
        eng.decide_gate(id, 1, true);
      std::this_thread::sleep_for(std::chrono::milliseconds(2));
    }
  });
  std::thread statuses([&] {
    while (!stop_aux.load()) {
      for (uint64_t id = 1; id < 60; ++id) (void)eng.run_status(id);
      (void)eng.run_count();
      std::this_thread::sleep_for(std::chrono::milliseconds(1));
    }
  });
  std::thread cancels([&] {
    for (uint64_t id = 3; id < 40; id += 7) {
      eng.cancel(id);
      std::this_thread::sleep_for(std::chrono::milliseconds(1));
    }
  });

  for (auto& s : subs) s.join();
  // wait for every run; keep approving gates meanwhile (the gates thread
  // covers ids that appeared after its last sweep)
  bool all_ok = true;
  for (int t = 0; t < 4; ++t)
    for (uint64_t id : ids[t])
      if (!eng.wait(id, 10.0)) all_ok = false;
  stop_aux.store(true);
  gates.join();
  statuses.join();
  cancels.join();

  // verify terminality + gc a slice concurrently with status reads
  size_t before = eng.run_count();
  std::thread gc1([&] {
    for (int t = 0; t < 2; ++t)
      for (uint64_t id : ids[t]) eng.gc_run(id);
  });
  std::thread gc2([&] {
    for (int t = 2; t < 4; ++t)
      for (uint64_t id : ids[t]) eng.gc_run(id);
  });
  gc1.join();
  gc2.join();

  q_done.store(true);
  q_cv.notify_all();
  for (auto& w : workers) w.join();
  eng.stop();

  std::printf("tsan driver: %d runs, all_waited=%d, runs before gc=%zu after=%zu\n",
              4 * kRuns, (int)all_ok, before, eng.run_count());
  return all_ok ? 0 : 1;
}
