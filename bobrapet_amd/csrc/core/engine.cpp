// bobraccel engine implementation: the native DAG state machine.
// Semantics mirror bobrapet_amd/engine/dag.py (which mirrors the reference
// DAG reconciler — internal/controller/runs/dag.go); the Python engine's
// test suite also runs against this core through the binding layer.
#include "engine.h"

namespace bobraccel {

namespace {
constexpr int kReady = 0, kWaitDeps = 1, kSkip = 2;
}

void NativeEngine::loop() {
  std::unique_lock<std::mutex> g(mu_);
  while (running_.load()) {
    // fire due timers
    double t = now();
    while (!timers_.empty() && timers_.top().at <= t) {
      Event ev = timers_.top().ev;
      timers_.pop();
      events_.push_back(std::move(ev));
    }
    // poll native-lane tickets (hipEventQuery behind the fn pointer):
    // completed engram work becomes EngramDone events without the GIL
    if (!ntickets_.empty() && lane_) {
      for (size_t i = 0; i < ntickets_.size();) {
        JValue out;
        std::string err;
        int rc = lane_->poll(lane_->self, ntickets_[i].ticket, &out, &err);
        if (rc == 0) {
          ++i;
          continue;
        }
        const NTicket tk = ntickets_[i];
        ntickets_[i] = ntickets_.back();
        ntickets_.pop_back();
        events_.push_back({EvKind::EngramDone, tk.run, tk.step, tk.branch,
                           tk.attempt, std::move(out), std::move(err),
                           rc == 1 ? 0 : 2});
      }
    }
    if (events_.empty()) {
      if (!ntickets_.empty()) {
        // GPU work in flight: nap briefly, then re-poll
        timed_wait(g, 20e-6);
      } else if (timers_.empty()) {
        cv_.wait(g);
      } else {
        double dt = timers_.top().at - now();
        if (dt > 0) timed_wait(g, dt);
      }
      continue;
    }
    Event ev = std::move(events_.front());
    events_.pop_front();
    switch (ev.kind) {
      case EvKind::Tick: {
        auto it = runs_.find(ev.run_id);
        if (it != runs_.end()) tick(it->second);
        break;
      }
      case EvKind::EngramDone:
        handle_engram_done(ev);
        break;
      case EvKind::Timer:
        handle_timer(ev);
        break;
    }
  }
}

// mu_ held throughout (launcher callbacks are queued and invoked after the
// state updates of this tick, still on the loop thread, without mu_).
void NativeEngine::tick(Run& run) {
  if (is_terminal(run.phase)) return;
  const Plan& plan = plans_[run.plan_id];
  if (run.phase == Phase::Pending) {
    run.phase = Phase::Running;
    run.started = now();
    if (plan.story_timeout > 0) {
      Event ev{EvKind::Timer, run.id, 0, 0, 0, JValue(), ""};
      ev.timer_tag = 5;
      arm(run.started + plan.story_timeout, std::move(ev));
    }
  }

  if (run.cancel_requested) {
    double t = now();
    for (auto& st : run.states) {
      if (st.phase == Phase::Pending || st.phase == Phase::Blocked) {
        st.phase = Phase::Skipped;
        st.finished = t;
      } else if (!is_terminal(st.phase)) {
        st.phase = Phase::Canceled;  // engram completions arriving later are
        st.finished = t;             // ignored (terminal wins)
      }
    }
    run.phase = Phase::Canceled;
    run.finished = t;
    run_terminal(run);
    return;
  }

  sync_primitives(run, plan);
  for (size_t i = 0; i < plan.steps.size() + 1; ++i) {
    if (!phase_pass(run, plan)) break;
    sync_primitives(run, plan);
  }
  maybe_finalize(run, plan);
}

void NativeEngine::sync_primitives(Run& run, const Plan& plan) {
  double t = now();
  for (size_t i = 0; i < plan.steps.size(); ++i) {
    const PlanStep& step = plan.steps[i];
    StepState& st = run.states[i];
    if (is_terminal(st.phase)) continue;
    if (step.kind == StepKind::Parallel && st.phase == Phase::Running) {
      auto& kids = run.branch_states[i];
      bool all_done = !kids.empty();
      for (auto& k : kids)
        if (!is_terminal(k.phase)) all_done = false;
      if (!all_done) continue;
      JObject branches;
      std::string failed;
      for (size_t b = 0; b < kids.size(); ++b) {
        branches[step.branches[b].name] = kids[b].output;
        bool hard = kids[b].phase == Phase::Failed ||
                    kids[b].phase == Phase::Timeout ||
                    kids[b].phase == Phase::Canceled;
        if (hard && !step.branches[b].allow_failure) {
          if (!failed.empty()) failed += ",";
          failed += step.branches[b].name;
        }
      }
      JObject out;
      out["branches"] = std::move(branches);
      st.output = std::move(out);
      st.finished = t;
      if (!failed.empty()) {
        st.phase = Phase::Failed;
        st.error = "parallel branches failed: " + failed;
      } else {
        st.phase = Phase::Succeeded;
      }
    } else if (step.kind == StepKind::Gate && st.phase == Phase::Paused) {
      auto git = run.gates.find((int)i);
      int decision = git == run.gates.end() ? 0 : git->second;
      if (decision == 1) {
        JObject out;
        out["approved"] = true;
        st.output = std::move(out);
        st.phase = Phase::Succeeded;
        st.finished = t;
      } else if (decision == 2) {
        st.phase = Phase::Failed;
        st.error = "gate rejected";
        st.finished = t;
      }
    } else if (step.kind == StepKind::Wait && st.phase == Phase::Running) {
      // Event-driven wakeup: an until-condition only references run state
      // (steps.*, inputs), which changes exclusively on ticks — so re-check
      // here and the wait resolves on the same tick its dependency
      // completes.  The armed poll timer (tag 2) stays as a fallback and
      // no-ops once the phase is terminal.
      JObject scope = build_scope(run, plan);
      bool done = false;
      try {
        done = step.until && eval_expr(*step.until, scope).truthy();
      } catch (const ExprError&) {
      }
      if (done) {
        st.phase = Phase::Succeeded;
        st.finished = t;
      }
    } else if (step.kind == StepKind::ExecuteStory &&
               st.phase == Phase::Running && st.output.is_object()) {
      const JValue child_id = st.output.get("childRun");
      if (!child_id.is_int()) continue;
      auto cit = runs_.find((uint64_t)child_id.as_int());
      if (cit == runs_.end() || !is_terminal(cit->second.phase)) continue;
      const Run& child = cit->second;
      JObject out;
      out["childRun"] = child_id;
      out["output"] = child.output;
      out["phase"] = phase_name(child.phase);
      st.output = std::move(out);
      st.finished = t;
      if (child.phase == Phase::Succeeded) {
        st.phase = Phase::Succeeded;
      } else {
        st.phase = Phase::Failed;
        st.error = std::string("sub-story finished ") + phase_name(child.phase);
      }
    }
  }
}

bool NativeEngine::has_failure(Run& run, const Plan& plan) {
  if (run.failure_step >= 0) return true;
  for (int i = 0; i < plan.main_end(); ++i) {  // failure = a MAIN step failed
    Phase p = run.states[i].phase;
    bool hard = p == Phase::Failed || p == Phase::Timeout || p == Phase::Canceled;
    if (hard && !plan.steps[i].allow_failure) {
      run.failure_step = i;
      return true;
    }
  }
  return false;
}

bool NativeEngine::steps_settled(Run& run, const Plan& plan, int begin, int end,
                                 bool failure, bool fail_fast) {
  // mirrors dag.py _steps_settled: terminal everywhere, or under fail-fast
  // with a failure present every non-terminal step is unstarted (skip it)
  double t = now();
  bool settled = true;
  for (int i = begin; i < end; ++i) {
    StepState& st = run.states[i];
    if (is_terminal(st.phase)) continue;
    if (failure && fail_fast) {
      if (st.phase == Phase::Running || st.phase == Phase::Paused) return false;
      st.phase = Phase::Skipped;
      st.error = "skipped by fail-fast";
      st.finished = t;
      continue;
    }
    settled = false;
  }
  return settled;
}

bool NativeEngine::active_range(Run& run, const Plan& plan, int* begin, int* end) {
  // main → compensation (on failure/cancel) → finally (always)
  bool failure = has_failure(run, plan) || run.cancel_requested;
  if (!steps_settled(run, plan, 0, plan.main_end(), failure, plan.fail_fast)) {
    *begin = 0;
    *end = plan.main_end();
    run.exec_phase = 0;
    return true;
  }
  failure = has_failure(run, plan) || run.cancel_requested;
  if (failure && plan.n_comp > 0 &&
      !steps_settled(run, plan, plan.main_end(), plan.comp_end(), false, false)) {
    *begin = plan.main_end();
    *end = plan.comp_end();
    run.exec_phase = 1;
    return true;
  }
  if (plan.n_fin > 0 &&
      !steps_settled(run, plan, plan.comp_end(), (int)plan.steps.size(), false, false)) {
    *begin = plan.comp_end();
    *end = (int)plan.steps.size();
    run.exec_phase = 2;
    return true;
  }
  return false;
}

JObject NativeEngine::build_scope(const Run& run, const Plan& plan) {
  JObject steps;
  for (size_t i = 0; i < run.states.size(); ++i) {
    const StepState& st = run.states[i];
    JObject s;
    s["phase"] = phase_name(st.phase);
    s["output"] = st.output;
    s["retries"] = (int64_t)st.retries;
    if (!st.error.empty()) s["error"] = st.error;
    steps[plan.steps[i].name] = std::move(s);
  }
  JObject scope;
  scope["inputs"] = run.inputs;
  scope["steps"] = std::move(steps);
  JObject r;
  r["id"] = (int64_t)run.id;
  r["phase"] = phase_name(run.phase);
  scope["run"] = std::move(r);
  return scope;
}

int NativeEngine::readiness(Run& run, const Plan& plan, int idx, bool failure,
                            JObject* scope) {
  const PlanStep& step = plan.steps[idx];
  if (failure && plan.fail_fast) return kSkip;
  for (size_t d = 0; d < step.deps.size(); ++d) {
    const StepState& ds = run.states[step.deps[d]];
    if (!is_terminal(ds.phase)) return kWaitDeps;
    if (ds.phase == Phase::Skipped) return kSkip;
    bool hard = ds.phase == Phase::Failed || ds.phase == Phase::Timeout ||
                ds.phase == Phase::Canceled;
    if (hard && !step.dep_allow_failure[d]) return kSkip;
  }
  if (step.if_expr || !step.requires_.empty()) {
    if (scope->empty()) *scope = build_scope(run, plan);
    try {
      if (step.if_expr && !eval_expr(*step.if_expr, *scope).truthy())
        return kSkip;
      for (const auto& req : step.requires_)
        if (eval_expr(*req, *scope).is_null()) return kSkip;
    } catch (const ExprError&) {
      return kSkip;
    }
  }
  return kReady;
}

bool NativeEngine::phase_pass(Run& run, const Plan& plan) {
  int begin = 0, end = 0;
  if (!active_range(run, plan, &begin, &end)) return false;
  const bool main_phase = begin == 0;
  // fail-fast/stop semantics apply only to the MAIN phase; compensation
  // and finally steps run precisely BECAUSE of a failure or stop
  bool failure = main_phase && has_failure(run, plan);
  bool progressed = false;
  JObject scope;  // built lazily, invalidated after each launch
  int active = 0;
  if (plan.concurrency > 0)
    for (const auto& st : run.states)
      if (st.phase == Phase::Running || st.phase == Phase::Paused) ++active;
  double t = now();
  for (size_t i = (size_t)begin; i < (size_t)end; ++i) {
    StepState& st = run.states[i];
    if (st.phase != Phase::Pending) continue;
    if (run.stop_seen && main_phase) {
      st.phase = Phase::Skipped;
      st.error = "skipped by stop";
      st.finished = t;
      progressed = true;
      continue;
    }
    int verdict = readiness(run, plan, (int)i, failure, &scope);
    if (verdict == kSkip) {
      st.phase = Phase::Skipped;
      st.finished = t;
      progressed = true;
      continue;
    }
    if (verdict != kReady) continue;
    if (plan.concurrency > 0 && active >= plan.concurrency) continue;
    launch_step(run, plan, (int)i);
    ++active;
    progressed = true;
    failure = main_phase && has_failure(run, plan);
    scope.clear();  // immediate completions may have changed step outputs
  }
  return progressed;
}

void NativeEngine::launch_step(Run& run, const Plan& plan, int idx) {
  const PlanStep& step = plan.steps[idx];
  StepState& st = run.states[idx];
  st.started = now();
  JObject scope = build_scope(run, plan);
  JValue with;
  if (step.with_tpl) {
    try {
      with = eval_template(*step.with_tpl, scope);
    } catch (const ExprError& e) {
      st.phase = Phase::Failed;
      st.error = std::string("template: ") + e.what();
      st.finished = now();
      return;
    }
  }
  switch (step.kind) {
    case StepKind::Condition: {
      bool result = true;
      if (step.if_expr) {
        // condition expression is stored in if_expr for condition steps
        try {
          result = eval_expr(*step.if_expr, scope).truthy();
        } catch (const ExprError&) {
          result = false;
        }
      }
      JObject out;
      out["result"] = result;
      st.output = std::move(out);
      st.phase = Phase::Succeeded;
      st.finished = now();
      break;
    }
    case StepKind::Sleep: {
      st.phase = Phase::Running;
      double dur = step.sleep_duration;
      if (with.is_object() && with.get("duration").is_number())
        dur = with.get("duration").as_double();
      Event ev{EvKind::Timer, run.id, idx, -1, st.attempt, JValue(), ""};
      ev.timer_tag = 1;
      arm(now() + dur, std::move(ev));
      break;
    }
    case StepKind::Stop: {
      run.stop_seen = true;
      run.stop_phase = step.stop_phase;
      JObject out;
      out["phase"] = phase_name(step.stop_phase);
      st.output = std::move(out);
      st.phase = Phase::Succeeded;
      st.finished = now();
      break;
    }
    case StepKind::Wait: {
      st.phase = Phase::Running;
      bool done = false;
      try {
        done = step.until && eval_expr(*step.until, scope).truthy();
      } catch (const ExprError&) {
      }
      if (done) {
        st.phase = Phase::Succeeded;
        st.finished = now();
        break;
      }
      if (step.wait_timeout > 0) {
        Event dl{EvKind::Timer, run.id, idx, -1, st.attempt, JValue(), ""};
        dl.timer_tag = 6;
        arm(st.started + step.wait_timeout, std::move(dl));
      }
      Event ev{EvKind::Timer, run.id, idx, -1, st.attempt, JValue(), ""};
      ev.timer_tag = 2;
      arm(now() + step.poll_interval, std::move(ev));
      break;
    }
    case StepKind::Gate: {
      st.phase = Phase::Paused;
      run.gates.emplace(idx, 0);
      if (step.wait_timeout > 0) {
        Event ev{EvKind::Timer, run.id, idx, -1, st.attempt, JValue(), ""};
        ev.timer_tag = 6;
        arm(now() + step.wait_timeout, std::move(ev));
      }
      break;
    }
    case StepKind::Parallel: {
      st.phase = Phase::Running;
      auto& kids = run.branch_states[idx];
      kids.resize(step.branches.size());
      for (size_t b = 0; b < step.branches.size(); ++b)
        launch_branch(run, plan, idx, (int)b);
      break;
    }
    case StepKind::ExecuteStory: {
      if (step.target_plan < 0) {
        st.phase = Phase::Failed;
        st.error = "executeStory: unknown target plan";
        st.finished = now();
        break;
      }
      st.phase = Phase::Running;
      // inline submit (mu_ already held by the loop)
      uint64_t id = next_run_++;
      Run child;
      child.id = id;
      child.plan_id = step.target_plan;
      child.inputs = with.is_object() ? with.get("with") : JValue();
      if (child.inputs.is_null() && with.is_object()) child.inputs = with;
      child.states.resize(plans_[step.target_plan].steps.size());
      child.branch_states.resize(plans_[step.target_plan].steps.size());
      child.parent_run = run.id;
      child.parent_step = idx;
      runs_.emplace(id, std::move(child));
      JObject out;
      out["childRun"] = (int64_t)id;
      st.output = std::move(out);
      events_.push_back({EvKind::Tick, id, 0, 0, 0, JValue(), ""});
      break;
    }
    case StepKind::Engram: {
      st.phase = Phase::Running;
      st.attempt += 1;
      if (step.timeout > 0) {
        Event ev{EvKind::Timer, run.id, idx, -1, st.attempt, JValue(), ""};
        ev.timer_tag = 3;
        arm(now() + step.timeout, std::move(ev));
      }
      if (step.native_kind > 0 && lane_) {
        const int dev =
            devices_.empty()
                ? 0
                : devices_[(run.id * 131 + (uint64_t)idx * 31) % devices_.size()];
        long tk = lane_->launch(lane_->self, step.native_kind,
                                &step.native_cfg, &with, dev);
        if (tk > 0) {
          ntickets_.push_back({tk, run.id, idx, -1, st.attempt});
          break;
        }
      }
      if (launcher_)
        launcher_(run.id, idx, -1, st.attempt, step.engram, step.name, with);
      else {
        st.phase = Phase::Failed;
        st.error = "no engram launcher registered";
        st.finished = now();
      }
      break;
    }
  }
}

void NativeEngine::launch_branch(Run& run, const Plan& plan, int idx, int b) {
  const PlanStep& branch = plan.steps[idx].branches[b];
  StepState& st = run.branch_states[idx][b];
  st.started = now();
  JObject scope = build_scope(run, plan);
  switch (branch.kind) {
    case StepKind::Condition: {
      bool result = true;
      if (branch.if_expr) {
        try {
          result = eval_expr(*branch.if_expr, scope).truthy();
        } catch (const ExprError&) {
          result = false;
        }
      }
      JObject out;
      out["result"] = result;
      st.output = std::move(out);
      st.phase = Phase::Succeeded;
      st.finished = now();
      break;
    }
    case StepKind::Sleep: {
      st.phase = Phase::Running;
      Event ev{EvKind::Timer, run.id, idx, b, st.attempt, JValue(), ""};
      ev.timer_tag = 1;
      arm(now() + branch.sleep_duration, std::move(ev));
      break;
    }
    case StepKind::Engram: {
      st.phase = Phase::Running;
      st.attempt += 1;
      JValue with;
      if (branch.with_tpl) {
        try {
          with = eval_template(*branch.with_tpl, scope);
        } catch (const ExprError& e) {
          st.phase = Phase::Failed;
          st.error = std::string("template: ") + e.what();
          st.finished = now();
          return;
        }
      }
      if (branch.native_kind > 0 && lane_) {
        const int dev =
            devices_.empty()
                ? 0
                : devices_[(run.id * 131 + (uint64_t)idx * 31 + (uint64_t)b * 7 + 1) %
                           devices_.size()];
        long tk = lane_->launch(lane_->self, branch.native_kind,
                                &branch.native_cfg, &with, dev);
        if (tk > 0) {
          ntickets_.push_back({tk, run.id, idx, b, st.attempt});
          break;
        }
      }
      if (launcher_)
        launcher_(run.id, idx, b, st.attempt, branch.engram,
                  plan.steps[idx].name + "/" + branch.name, with);
      else {
        st.phase = Phase::Failed;
        st.error = "no engram launcher registered";
        st.finished = now();
      }
      break;
    }
    default: {
      st.phase = Phase::Failed;
      st.error = "unsupported branch kind";
      st.finished = now();
    }
  }
}

void NativeEngine::handle_engram_done(const Event& ev) {
  auto it = runs_.find(ev.run_id);
  if (it == runs_.end()) return;
  Run& run = it->second;
  const Plan& plan = plans_[run.plan_id];
  if (ev.step < 0 || ev.step >= (int)run.states.size()) return;
  StepState& st = ev.branch < 0 ? run.states[ev.step]
                                : run.branch_states[ev.step][ev.branch];
  if (is_terminal(st.phase) || st.attempt != ev.attempt) return;  // stale
  double t = now();
  // exit classes: 0 success, 1 retry, 2 terminal, 3 rateLimited, other unknown
  int code = ev.exit_code;
  const PlanStep& pstep = ev.branch < 0 ? plan.steps[ev.step]
                                        : plan.steps[ev.step].branches[ev.branch];
  if (code == 0 && pstep.post_exec) {
    // postExecution check (reference: steprun_controller.go:2050-2124):
    // condition over the scope + this output; falsy → terminal failure
    JObject scope = build_scope(run, plan);
    scope["output"] = ev.output;
    bool ok = false;
    try {
      ok = eval_expr(*pstep.post_exec, scope).truthy();
    } catch (const ExprError&) {
      ok = false;
    }
    if (!ok) {
      st.phase = Phase::Failed;
      st.output = ev.output;
      st.error = pstep.post_exec_msg.empty() ? "postExecution condition failed"
                                             : pstep.post_exec_msg;
      st.finished = t;
      tick(run);
      return;
    }
  }
  if (code == 0) {
    st.phase = Phase::Succeeded;
    st.output = ev.output;
    st.finished = t;
  } else {
    const RetryCfg& rc =
        ev.branch < 0 ? plan.steps[ev.step].retry
                      : plan.steps[ev.step].branches[ev.branch].retry;
    bool retryable = code == 1 || code == 3 || (code != 2);
    bool burns = code == 1 || code == 3;  // unknown doesn't burn budget
    bool can = retryable &&
               (!burns || st.retries < rc.max_retries) &&
               (burns || st.retries < 50) && code != 2;
    if (can && !run.cancel_requested) {
      if (burns) st.retries += 1;
      double base = rc.delay;
      double d = rc.backoff == 0 ? base * (double)(1ull << std::min(st.retries > 0 ? st.retries - 1 : 0, 30))
               : rc.backoff == 1 ? base * st.retries
                                 : base;
      if (code == 3 && d < 5.0) d = 5.0;
      if (rc.max_delay > 0 && d > rc.max_delay) d = rc.max_delay;
      if (rc.jitter_pct > 0) {
        std::uniform_real_distribution<double> dist(
            1.0 - rc.jitter_pct / 100.0, 1.0 + rc.jitter_pct / 100.0);
        d *= dist(rng_);
      }
      Event rev{EvKind::Timer, ev.run_id, ev.step, ev.branch, st.attempt,
                JValue(), ""};
      rev.timer_tag = 4;
      arm(t + d, std::move(rev));
    } else {
      st.phase = Phase::Failed;
      st.error = ev.error.empty() ? ("exit code " + std::to_string(code))
                                  : ev.error;
      st.finished = t;
    }
  }
  tick(run);
}

void NativeEngine::handle_timer(const Event& ev) {
  auto it = runs_.find(ev.run_id);
  if (it == runs_.end()) return;
  Run& run = it->second;
  const Plan& plan = plans_[run.plan_id];
  double t = now();
  if (ev.timer_tag == 5) {  // story timeout
    if (is_terminal(run.phase)) return;
    for (auto& st : run.states)
      if (!is_terminal(st.phase)) {
        st.phase = Phase::Timeout;
        st.finished = t;
      }
    run.phase = Phase::Timeout;
    run.error = "story timeout";
    run.finished = t;
    run_terminal(run);
    return;
  }
  if (ev.step < 0 || ev.step >= (int)run.states.size()) return;
  StepState& st = ev.branch < 0 ? run.states[ev.step]
                                : run.branch_states[ev.step].size() > (size_t)ev.branch
                                      ? run.branch_states[ev.step][ev.branch]
                                      : run.states[ev.step];
  const PlanStep& step = plan.steps[ev.step];
  switch (ev.timer_tag) {
    case 1:  // sleep done
      if (st.phase == Phase::Running && st.attempt == ev.attempt) {
        JObject out;
        out["slept"] = true;
        st.output = std::move(out);
        st.phase = Phase::Succeeded;
        st.finished = t;
      }
      break;
    case 2: {  // wait poll
      if (st.phase != Phase::Running) return;
      JObject scope = build_scope(run, plan);
      bool done = false;
      try {
        done = step.until && eval_expr(*step.until, scope).truthy();
      } catch (const ExprError&) {
      }
      if (done) {
        st.phase = Phase::Succeeded;
        st.finished = t;
      } else {
        Event ev2{EvKind::Timer, ev.run_id, ev.step, ev.branch, ev.attempt,
                  JValue(), ""};
        ev2.timer_tag = 2;
        arm(t + step.poll_interval, std::move(ev2));
        return;  // no tick needed
      }
      break;
    }
    case 3:  // step timeout
      if (!is_terminal(st.phase) && st.attempt == ev.attempt) {
        st.phase = Phase::Timeout;
        st.error = "step timeout";
        st.finished = t;
      }
      break;
    case 4: {  // retry due
      if (is_terminal(st.phase) || run.cancel_requested) break;
      if (ev.branch < 0)
        launch_step(run, plan, ev.step);
      else
        launch_branch(run, plan, ev.step, ev.branch);
      break;
    }
    case 6:  // wait/gate deadline
      if (st.phase == Phase::Running || st.phase == Phase::Paused) {
        st.phase = step.on_timeout_skip ? Phase::Skipped : Phase::Timeout;
        st.error = step.on_timeout_skip ? "" : "timed out";
        st.finished = t;
      }
      break;
  }
  tick(run);
}

void NativeEngine::maybe_finalize(Run& run, const Plan& plan) {
  int b = 0, e = 0;
  if (active_range(run, plan, &b, &e)) return;  // a phase still has work
  // unreached compensation steps (no failure) stay Pending with no start
  // time — the Python engine / reference leave them without state at all,
  // and run_status omits them for the same surface
  bool failure = has_failure(run, plan);
  bool comp_ok = plan.n_comp > 0;
  for (int i = plan.main_end(); i < plan.comp_end(); ++i) {
    Phase p = run.states[i].phase;
    if (p != Phase::Succeeded && p != Phase::Skipped) comp_ok = false;
  }
  (void)0;
  bool fin_failed = false;
  for (int i = plan.comp_end(); i < (int)plan.steps.size(); ++i) {
    Phase p = run.states[i].phase;
    bool hard = p == Phase::Failed || p == Phase::Timeout || p == Phase::Canceled;
    if (hard && !plan.steps[i].allow_failure) fin_failed = true;
  }
  if (run.cancel_requested)
    run.phase = Phase::Canceled;
  else if (run.stop_seen)
    run.phase = run.stop_phase;
  else if (failure && plan.n_comp > 0 && comp_ok)
    run.phase = Phase::Compensated;
  else if (failure || fin_failed)
    run.phase = Phase::Failed;
  else
    run.phase = Phase::Succeeded;
  if (failure && run.failure_step >= 0)
    run.error = plan.steps[run.failure_step].name + ": " +
                run.states[run.failure_step].error;
  if (run.phase == Phase::Succeeded && plan.output_tpl) {
    JObject scope = build_scope(run, plan);
    try {
      run.output = eval_template(*plan.output_tpl, scope);
    } catch (const ExprError& e) {
      run.phase = Phase::Failed;
      run.error = std::string("output template: ") + e.what();
    }
  }
  run.finished = now();
  run_terminal(run);
}

void NativeEngine::run_terminal(Run& run) {
  if (run.parent_run != 0) {
    events_.push_back(
        {EvKind::Tick, run.parent_run, 0, 0, 0, JValue(), ""});
  }
  done_cv_.notify_all();
}

}  // namespace bobraccel
