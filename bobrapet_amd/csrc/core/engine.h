// bobraccel: the native DAG run engine (the MI355X replacement for the
// reference's controller-runtime reconcile machinery — SURVEY.md §2.6).
//
// One loop thread owns all run state; events (submissions, engram
// completions, timers, gate decisions) are queued; a tick is an in-memory
// graph pass with the same state-machine semantics as the Python engine
// (engine/dag.py): needs + if/requires readiness, primitives, retry with
// exit classes, fail-fast, terminal-phase-wins merges.  Engram steps are
// dispatched through a launcher callback (the binding layer routes them to
// the GPU worker slots); completion events flow back via complete_engram.
//
// Scope: the batch fast path (main steps + output template).  Stories with
// compensations/finally/streaming run on the Python engine; the Python
// front end picks the path per story.
#pragma once

#include <atomic>
#include <chrono>
#include <condition_variable>
#include <cstdint>
#include <deque>
#include <functional>
#include <mutex>
#include <queue>
#include <random>
#include <thread>
#include <unordered_map>

#include "expr.h"
#include "native_lane.h"
#include "jvalue.h"

namespace bobraccel {

enum class Phase : uint8_t {
  Pending = 0,
  Running,
  Succeeded,
  Failed,
  Finished,
  Canceled,
  Paused,
  Timeout,
  Skipped,
  Blocked,
  Compensated,
};

inline bool is_terminal(Phase p) {
  switch (p) {
    case Phase::Succeeded:
    case Phase::Failed:
    case Phase::Finished:
    case Phase::Canceled:
    case Phase::Timeout:
    case Phase::Skipped:
    case Phase::Compensated:
      return true;
    default:
      return false;
  }
}

inline const char* phase_name(Phase p) {
  switch (p) {
    case Phase::Pending: return "Pending";
    case Phase::Running: return "Running";
    case Phase::Succeeded: return "Succeeded";
    case Phase::Failed: return "Failed";
    case Phase::Finished: return "Finished";
    case Phase::Canceled: return "Canceled";
    case Phase::Paused: return "Paused";
    case Phase::Timeout: return "Timeout";
    case Phase::Skipped: return "Skipped";
    case Phase::Blocked: return "Blocked";
    case Phase::Compensated: return "Compensated";
  }
  return "?";
}

enum class StepKind : uint8_t {
  Engram = 0,
  Condition,
  Sleep,
  Stop,
  Wait,
  Gate,
  Parallel,
  ExecuteStory,
};

struct RetryCfg {
  int max_retries = 0;
  double delay = 1.0;
  double max_delay = 60.0;
  int jitter_pct = 10;
  int backoff = 0;  // 0 exponential, 1 linear, 2 constant
};

struct PlanStep {
  std::string name;
  StepKind kind = StepKind::Engram;
  std::vector<int> deps;
  std::vector<bool> dep_allow_failure;  // parallel to deps
  ExprPtr if_expr;
  std::vector<ExprPtr> requires_;
  TNodePtr with_tpl;
  bool allow_failure = false;
  RetryCfg retry;
  double timeout = 0;  // seconds, 0 = none
  // primitive parameters
  double sleep_duration = 0;
  ExprPtr until;
  double wait_timeout = 0;
  double poll_interval = 0.02;
  bool on_timeout_skip = false;  // false → Timeout/fail, true → Skipped
  Phase stop_phase = Phase::Succeeded;
  std::vector<PlanStep> branches;  // Parallel
  std::string engram;              // launcher routing key
  int native_kind = 0;             // >0: dispatch via the NativeLane
  JValue native_cfg;               // engram config for the lane
  ExprPtr post_exec;               // postExecution condition over {output}
  std::string post_exec_msg;
  int target_plan = -1;            // ExecuteStory
};

struct Plan {
  std::string name;
  std::vector<PlanStep> steps;  // main ++ compensations ++ finally
  std::vector<std::vector<int>> dependents;
  TNodePtr output_tpl;
  bool fail_fast = true;
  double story_timeout = 0;
  int concurrency = 0;  // max concurrently Running/Paused steps (0 = unlimited)
  // 3-phase layout (reference: dag.go:482-511 main→compensation→finally)
  int n_main = -1;  // -1 → every step is main
  int n_comp = 0;
  int n_fin = 0;
  int main_end() const { return n_main < 0 ? (int)steps.size() : n_main; }
  int comp_end() const { return main_end() + n_comp; }
};

struct StepState {
  Phase phase = Phase::Pending;
  JValue output;
  std::string error;
  int retries = 0;
  uint32_t attempt = 0;  // bumped per launch; stale completions ignored
  double started = 0, finished = 0;
};

struct Run {
  uint64_t id = 0;
  int plan_id = -1;
  JValue inputs;
  Phase phase = Phase::Pending;
  std::vector<StepState> states;
  std::vector<std::vector<StepState>> branch_states;
  std::unordered_map<int, int> gates;  // step → 0 pending / 1 approved / 2 rejected
  bool cancel_requested = false;
  int exec_phase = 0;  // 0 main, 1 compensation, 2 finally
  bool stop_seen = false;
  Phase stop_phase = Phase::Succeeded;
  int failure_step = -1;
  JValue output;
  std::string error;
  double started = 0, finished = 0;
  uint64_t parent_run = 0;
  int parent_step = -1;
  int parent_branch = -1;
};

// launcher(run_id, step_index, branch_index(-1), attempt, engram_name,
//          step_name, resolved_input) — called FROM THE LOOP THREAD; must
//          not block.
using EngramLauncher =
    std::function<void(uint64_t, int, int, uint32_t, const std::string&,
                       const std::string&, const JValue&)>;

class NativeEngine {
 public:
  NativeEngine() = default;
  ~NativeEngine() { stop(); }

  int register_plan(Plan plan) {
    std::lock_guard<std::mutex> g(mu_);
    plans_.push_back(std::move(plan));
    return (int)plans_.size() - 1;
  }

  void set_launcher(EngramLauncher fn) { launcher_ = std::move(fn); }

  // GIL-free built-in engram lane (native_lane.h); devices for placement
  void set_native_lane(const NativeLane* lane) {
    std::lock_guard<std::mutex> g(mu_);
    lane_ = lane;
  }
  void set_devices(std::vector<int> devs) {
    std::lock_guard<std::mutex> g(mu_);
    devices_ = std::move(devs);
  }

  void start() {
    bool expected = false;
    if (!running_.compare_exchange_strong(expected, true)) return;
    loop_ = std::thread([this] { this->loop(); });
  }

  void stop() {
    bool expected = true;
    if (!running_.compare_exchange_strong(expected, false)) return;
    {
      std::lock_guard<std::mutex> g(mu_);
      cv_.notify_all();
    }
    if (loop_.joinable()) loop_.join();
  }

  uint64_t submit(int plan_id, JValue inputs, uint64_t parent_run = 0,
                  int parent_step = -1, int parent_branch = -1) {
    std::lock_guard<std::mutex> g(mu_);
    uint64_t id = next_run_++;
    Run run;
    run.id = id;
    run.plan_id = plan_id;
    run.inputs = std::move(inputs);
    run.states.resize(plans_[plan_id].steps.size());
    run.branch_states.resize(plans_[plan_id].steps.size());
    run.parent_run = parent_run;
    run.parent_step = parent_step;
    run.parent_branch = parent_branch;
    runs_.emplace(id, std::move(run));
    events_.push_back({EvKind::Tick, id, 0, 0, 0, JValue(), ""});
    cv_.notify_all();
    return id;
  }

  void complete_engram(uint64_t run_id, int step, int branch, uint32_t attempt,
                       int exit_code, JValue output, std::string error) {
    std::lock_guard<std::mutex> g(mu_);
    events_.push_back({EvKind::EngramDone, run_id, step, branch, attempt,
                       std::move(output), std::move(error), exit_code});
    cv_.notify_all();
  }

  void decide_gate(uint64_t run_id, int step, bool approved) {
    std::lock_guard<std::mutex> g(mu_);
    auto it = runs_.find(run_id);
    if (it == runs_.end()) return;
    it->second.gates[step] = approved ? 1 : 2;
    events_.push_back({EvKind::Tick, run_id, 0, 0, 0, JValue(), ""});
    cv_.notify_all();
  }

  void cancel(uint64_t run_id) {
    std::lock_guard<std::mutex> g(mu_);
    auto it = runs_.find(run_id);
    if (it == runs_.end()) return;
    it->second.cancel_requested = true;
    events_.push_back({EvKind::Tick, run_id, 0, 0, 0, JValue(), ""});
    cv_.notify_all();
  }

  bool wait(uint64_t run_id, double timeout_s) {
    std::unique_lock<std::mutex> g(mu_);
    auto pred = [&] {
      auto it = runs_.find(run_id);
      return it == runs_.end() || is_terminal(it->second.phase);
    };
    if (timeout_s <= 0) {
      done_cv_.wait(g, pred);
      return true;
    }
#ifdef BOBRA_TSAN_COMPAT
    // see timed_wait(): clockwait is invisible to GCC-11 libtsan and
    // corrupts the mutex model for every later report
    double deadline = now() + timeout_s;
    while (!pred()) {
      double dt = deadline - now();
      if (dt <= 0) return pred();
      struct timespec ts;
      clock_gettime(CLOCK_REALTIME, &ts);
      long ns = ts.tv_nsec + (long)(dt * 1e9);
      ts.tv_sec += ns / 1000000000L;
      ts.tv_nsec = ns % 1000000000L;
      pthread_cond_timedwait(done_cv_.native_handle(),
                             g.mutex()->native_handle(), &ts);
    }
    return true;
#else
    return done_cv_.wait_for(
        g, std::chrono::duration<double>(timeout_s), pred);
#endif
  }

  // snapshot for the host language
  JValue run_status(uint64_t run_id) {
    std::lock_guard<std::mutex> g(mu_);
    auto it = runs_.find(run_id);
    if (it == runs_.end()) return JValue();
    const Run& run = it->second;
    const Plan& plan = plans_[run.plan_id];
    JObject steps;
    for (size_t i = 0; i < run.states.size(); ++i) {
      const StepState& st = run.states[i];
      // never-launched compensation/finally steps have no state surface
      // (reference parity: unreconciled steps don't appear in stepStates)
      if ((int)i >= plan.main_end() && st.phase == Phase::Pending &&
          is_terminal(run.phase))
        continue;
      JObject s;
      s["phase"] = phase_name(st.phase);
      s["output"] = st.output;
      if (!st.error.empty()) s["error"] = st.error;
      s["retries"] = (int64_t)st.retries;
      if (st.started) s["startedAt"] = st.started;
      if (st.finished) s["finishedAt"] = st.finished;
      steps[plan.steps[i].name] = std::move(s);
    }
    JObject out;
    out["phase"] = phase_name(run.phase);
    out["steps"] = std::move(steps);
    out["output"] = run.output;
    if (!run.error.empty()) out["error"] = run.error;
    if (run.started) out["startedAt"] = run.started;
    if (run.finished) out["finishedAt"] = run.finished;
    return out;
  }

  size_t run_count() {
    std::lock_guard<std::mutex> g(mu_);
    return runs_.size();
  }

  void gc_run(uint64_t run_id) {
    // erase a TERMINAL run and its terminal descendants (executeStory
    // children submitted internally carry parent_run links) — the retention
    // role of the Python engine's TTL cleanup for the fast path
    std::lock_guard<std::mutex> g(mu_);
    auto it = runs_.find(run_id);
    if (it == runs_.end() || !is_terminal(it->second.phase)) return;
    std::vector<uint64_t> doomed{run_id};
    bool grew = true;
    while (grew) {
      grew = false;
      for (const auto& [id, r] : runs_) {
        if (r.parent_run == 0 || !is_terminal(r.phase)) continue;
        bool parent_doomed = false, self_doomed = false;
        for (uint64_t d : doomed) {
          parent_doomed |= d == r.parent_run;
          self_doomed |= d == id;
        }
        if (parent_doomed && !self_doomed) {
          doomed.push_back(id);
          grew = true;
        }
      }
    }
    if (lane_ && lane_->free_key) {
      for (uint64_t d : doomed) {
        auto rit = runs_.find(d);
        if (rit == runs_.end()) continue;
        free_native_keys(rit->second.output);
        for (auto& st : rit->second.states) free_native_keys(st.output);
        for (auto& bs : rit->second.branch_states)
          for (auto& st : bs) free_native_keys(st.output);
      }
    }
    for (uint64_t d : doomed) runs_.erase(d);
  }

  // release lane-held payloads referenced by a gc'd run's outputs
  void free_native_keys(const JValue& v) {  // mu_ held
    if (v.is_object()) {
      const JObject& o = v.as_object();
      auto it = o.find("$storageRef");
      if (it != o.end() && it->second.is_object()) {
        const JObject& r = it->second.as_object();
        auto k = r.find("key");
        if (k != r.end() && k->second.is_string()) {
          const std::string& key = k->second.as_string();
          if (key.rfind("native/", 0) == 0)
            lane_->free_key(lane_->self, key.c_str());
        }
      }
      for (const auto& [kk, vv] : o) free_native_keys(vv);
    } else if (v.is_array()) {
      for (const auto& e : v.as_array()) free_native_keys(e);
    }
  }

 private:
  enum class EvKind { Tick, EngramDone, Timer };
  struct Event {
    EvKind kind;
    uint64_t run_id;
    int step;
    int branch;
    uint32_t attempt;
    JValue output;
    std::string error;
    int exit_code = 0;
    int timer_tag = 0;  // 0 generic, 1 sleep, 2 wait-poll, 3 step-timeout,
                        // 4 retry, 5 story-timeout, 6 gate-deadline
  };
  struct Timer {
    double at;
    uint64_t seq;
    Event ev;
    bool operator<(const Timer& o) const {
      return at > o.at || (at == o.at && seq > o.seq);  // min-heap
    }
  };

  static double now() {
    return std::chrono::duration<double>(
               std::chrono::steady_clock::now().time_since_epoch())
        .count();
  }

  void arm(double at, Event ev) {  // mu_ held
    timers_.push(Timer{at, timer_seq_++, std::move(ev)});
  }

  void loop();
  void tick(Run& run);

  // condition_variable::wait_for uses pthread_cond_clockwait (steady
  // clock) on this glibc, which GCC-11's libtsan does NOT intercept —
  // a TSan build then believes the mutex was never released and every
  // later access reports as a race.  Under BOBRA_TSAN_COMPAT wait on
  // the REALTIME clock through the intercepted pthread_cond_timedwait;
  // the production build keeps the steady-clock wait (REALTIME is
  // jump-sensitive, acceptable only for the sanitizer run).
  void timed_wait(std::unique_lock<std::mutex>& g, double seconds) {
#ifdef BOBRA_TSAN_COMPAT
    struct timespec ts;
    clock_gettime(CLOCK_REALTIME, &ts);
    long ns = ts.tv_nsec + (long)(seconds * 1e9);
    ts.tv_sec += ns / 1000000000L;
    ts.tv_nsec = ns % 1000000000L;
    pthread_cond_timedwait(cv_.native_handle(), g.mutex()->native_handle(),
                           &ts);
#else
    cv_.wait_for(g, std::chrono::duration<double>(seconds));
#endif
  }
  void sync_primitives(Run& run, const Plan& plan);
  bool phase_pass(Run& run, const Plan& plan);
  bool steps_settled(Run& run, const Plan& plan, int begin, int end,
                     bool failure, bool fail_fast);
  bool active_range(Run& run, const Plan& plan, int* begin, int* end);
  void launch_step(Run& run, const Plan& plan, int idx);
  void launch_branch(Run& run, const Plan& plan, int idx, int bidx);
  void maybe_finalize(Run& run, const Plan& plan);
  void handle_engram_done(const Event& ev);
  void handle_timer(const Event& ev);
  JObject build_scope(const Run& run, const Plan& plan);
  int readiness(Run& run, const Plan& plan, int idx, bool failure, JObject* scope);
  bool has_failure(Run& run, const Plan& plan);
  void run_terminal(Run& run);

  std::mutex mu_;
  std::condition_variable cv_;
  std::condition_variable done_cv_;
  std::deque<Event> events_;
  std::priority_queue<Timer> timers_;
  uint64_t timer_seq_ = 0;
  std::vector<Plan> plans_;
  std::unordered_map<uint64_t, Run> runs_;
  uint64_t next_run_ = 1;
  EngramLauncher launcher_;
  const NativeLane* lane_ = nullptr;
  std::vector<int> devices_;
  struct NTicket {
    long ticket;
    uint64_t run;
    int step;
    int branch;
    uint32_t attempt;
  };
  std::vector<NTicket> ntickets_;
  std::thread loop_;
  std::atomic<bool> running_{false};
  std::mt19937 rng_{12345};
};

}  // namespace bobraccel
