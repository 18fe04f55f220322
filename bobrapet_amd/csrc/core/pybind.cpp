// Python bindings for bobraccel (the native DAG core).
//
// Deadlock discipline: every engine entry point releases the GIL before
// taking the engine mutex (call_guard<gil_scoped_release>), and the
// launcher callback (invoked from the loop thread) re-acquires the GIL via
// pybind's functional wrapper — so GIL holders never wait on the engine
// mutex and the loop thread only takes the GIL while the mutex is held by
// itself alone.
#include <pybind11/functional.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "engine.h"

namespace py = pybind11;
using namespace bobraccel;

// ---------------------------------------------------------------------------
// JValue <-> Python
// ---------------------------------------------------------------------------

static void py_incref(void* p) {
  py::gil_scoped_acquire g;
  Py_INCREF((PyObject*)p);
}
static void py_decref(void* p) {
  py::gil_scoped_acquire g;
  Py_DECREF((PyObject*)p);
}

static JValue to_jvalue(const py::handle& obj) {
  if (obj.is_none()) return JValue();
  if (py::isinstance<py::bool_>(obj)) return JValue(obj.cast<bool>());
  if (py::isinstance<py::int_>(obj)) return JValue(obj.cast<int64_t>());
  if (py::isinstance<py::float_>(obj)) return JValue(obj.cast<double>());
  if (py::isinstance<py::str>(obj)) return JValue(obj.cast<std::string>());
  if (py::isinstance<py::list>(obj) || py::isinstance<py::tuple>(obj)) {
    JArray arr;
    for (const auto& item : obj) arr.push_back(to_jvalue(item));
    return JValue(std::move(arr));
  }
  if (py::isinstance<py::dict>(obj)) {
    JObject o;
    for (const auto& item : obj.cast<py::dict>())
      o[py::str(item.first).cast<std::string>()] = to_jvalue(item.second);
    return JValue(std::move(o));
  }
  // anything else (tensors, storage refs as objects) rides through opaque
  return JValue(Opaque((void*)obj.ptr(), py_incref, py_decref));
}

static py::object to_python(const JValue& v) {
  if (v.is_null()) return py::none();
  if (v.is_bool()) return py::bool_(v.as_bool());
  if (v.is_int()) return py::int_(v.as_int());
  if (v.is_double()) return py::float_(v.as_double());
  if (v.is_string()) return py::str(v.as_string());
  if (v.is_array()) {
    py::list out;
    for (const auto& item : v.as_array()) out.append(to_python(item));
    return out;
  }
  if (v.is_object()) {
    py::dict out;
    for (const auto& [k, val] : v.as_object()) out[py::str(k)] = to_python(val);
    return out;
  }
  if (v.is_opaque())
    return py::reinterpret_borrow<py::object>((PyObject*)v.as_opaque().ptr);
  return py::none();
}

// ---------------------------------------------------------------------------
// AST (templating.parser tuples) -> ExprNode
// ---------------------------------------------------------------------------

static ExprPtr to_expr(const py::handle& node);

static std::vector<ExprPtr> to_expr_list(const py::handle& seq) {
  std::vector<ExprPtr> out;
  for (const auto& item : seq) out.push_back(to_expr(item));
  return out;
}

static ExprPtr to_expr(const py::handle& node) {
  auto t = node.cast<py::tuple>();
  std::string tag = t[0].cast<std::string>();
  auto n = std::make_shared<ExprNode>();
  if (tag == "const") {
    n->op = Op::Const;
    n->constant = to_jvalue(t[1]);
  } else if (tag == "var") {
    n->op = Op::Var;
    n->str = t[1].cast<std::string>();
  } else if (tag == "get") {
    n->op = Op::Get;
    n->children.push_back(to_expr(t[1]));
    n->str = py::str(t[2]).cast<std::string>();
  } else if (tag == "index") {
    n->op = Op::Index;
    n->children.push_back(to_expr(t[1]));
    n->children.push_back(to_expr(t[2]));
  } else if (tag == "and" || tag == "or") {
    n->op = tag == "and" ? Op::And : Op::Or;
    n->children.push_back(to_expr(t[1]));
    n->children.push_back(to_expr(t[2]));
  } else if (tag == "not") {
    n->op = Op::Not;
    n->children.push_back(to_expr(t[1]));
  } else if (tag == "cmp" || tag == "bin") {
    n->op = tag == "cmp" ? Op::Cmp : Op::Bin;
    n->str = t[1].cast<std::string>();
    n->children.push_back(to_expr(t[2]));
    n->children.push_back(to_expr(t[3]));
  } else if (tag == "neg") {
    n->op = Op::Neg;
    n->children.push_back(to_expr(t[1]));
  } else if (tag == "cond") {
    n->op = Op::Cond;
    n->children.push_back(to_expr(t[1]));
    n->children.push_back(to_expr(t[2]));
    n->children.push_back(to_expr(t[3]));
  } else if (tag == "list") {
    n->op = Op::List;
    n->children = to_expr_list(t[1]);
  } else if (tag == "map") {
    n->op = Op::Map;
    for (const auto& pair : t[1]) {
      auto kv = pair.cast<py::tuple>();
      // key node is ('const', str)
      n->strs.push_back(kv[0].cast<py::tuple>()[1].cast<std::string>());
      n->children.push_back(to_expr(kv[1]));
    }
  } else if (tag == "call") {
    n->op = Op::Call;
    n->str = t[1].cast<std::string>();
    n->children = to_expr_list(t[2]);
  } else if (tag == "method") {
    n->op = Op::Method;
    n->str = t[2].cast<std::string>();
    n->children.push_back(to_expr(t[1]));
    for (const auto& a : t[3]) n->children.push_back(to_expr(a));
  } else {
    throw std::runtime_error("unknown AST tag: " + tag);
  }
  return n;
}

// Template trees from the Python compiler: nested
//   ("lit", value) | ("expr", ast) | ("parts", [(lit, ast|None), ...])
// | ("arr", [tnode...]) | ("obj", [(key, tnode)...])
static TNodePtr to_tnode(const py::handle& node) {
  auto t = node.cast<py::tuple>();
  std::string tag = t[0].cast<std::string>();
  auto n = std::make_shared<TNode>();
  if (tag == "lit") {
    n->kind = TNode::Literal;
    n->literal = to_jvalue(t[1]);
  } else if (tag == "expr") {
    n->kind = TNode::Single;
    n->expr = to_expr(t[1]);
  } else if (tag == "parts") {
    n->kind = TNode::Parts;
    for (const auto& pair : t[1]) {
      auto kv = pair.cast<py::tuple>();
      ExprPtr e = kv[1].is_none() ? nullptr : to_expr(kv[1]);
      n->parts.emplace_back(kv[0].cast<std::string>(), e);
    }
  } else if (tag == "arr") {
    n->kind = TNode::Arr;
    for (const auto& item : t[1]) n->items.push_back(to_tnode(item));
  } else if (tag == "obj") {
    n->kind = TNode::Obj;
    for (const auto& pair : t[1]) {
      auto kv = pair.cast<py::tuple>();
      n->fields.emplace_back(kv[0].cast<std::string>(), to_tnode(kv[1]));
    }
  } else {
    throw std::runtime_error("unknown template tag: " + tag);
  }
  return n;
}

// ---------------------------------------------------------------------------
// Plan construction from Python dicts (built by runtime/native.py)
// ---------------------------------------------------------------------------

static PlanStep to_step(const py::dict& d) {
  PlanStep s;
  s.name = d["name"].cast<std::string>();
  s.kind = (StepKind)d["kind"].cast<int>();
  if (d.contains("deps")) s.deps = d["deps"].cast<std::vector<int>>();
  if (d.contains("depAllowFailure"))
    s.dep_allow_failure = d["depAllowFailure"].cast<std::vector<bool>>();
  s.dep_allow_failure.resize(s.deps.size(), false);
  if (d.contains("if")) s.if_expr = to_expr(d["if"]);
  if (d.contains("requires"))
    for (const auto& r : d["requires"]) s.requires_.push_back(to_expr(r));
  if (d.contains("with")) s.with_tpl = to_tnode(d["with"]);
  if (d.contains("allowFailure")) s.allow_failure = d["allowFailure"].cast<bool>();
  if (d.contains("retry")) {
    auto r = d["retry"].cast<py::dict>();
    if (r.contains("maxRetries")) s.retry.max_retries = r["maxRetries"].cast<int>();
    if (r.contains("delay")) s.retry.delay = r["delay"].cast<double>();
    if (r.contains("maxDelay")) s.retry.max_delay = r["maxDelay"].cast<double>();
    if (r.contains("jitterPct")) s.retry.jitter_pct = r["jitterPct"].cast<int>();
    if (r.contains("backoff")) s.retry.backoff = r["backoff"].cast<int>();
  }
  if (d.contains("timeout")) s.timeout = d["timeout"].cast<double>();
  if (d.contains("sleepDuration"))
    s.sleep_duration = d["sleepDuration"].cast<double>();
  if (d.contains("until")) s.until = to_expr(d["until"]);
  if (d.contains("waitTimeout")) s.wait_timeout = d["waitTimeout"].cast<double>();
  if (d.contains("pollInterval"))
    s.poll_interval = d["pollInterval"].cast<double>();
  if (d.contains("onTimeoutSkip"))
    s.on_timeout_skip = d["onTimeoutSkip"].cast<bool>();
  if (d.contains("stopPhase")) {
    std::string p = d["stopPhase"].cast<std::string>();
    s.stop_phase = p == "Failed" ? Phase::Failed
                 : p == "Finished" ? Phase::Finished
                                   : Phase::Succeeded;
  }
  if (d.contains("engram")) s.engram = d["engram"].cast<std::string>();
  if (d.contains("nativeKind")) s.native_kind = d["nativeKind"].cast<int>();
  if (d.contains("nativeCfg")) s.native_cfg = to_jvalue(d["nativeCfg"]);
  if (d.contains("targetPlan")) s.target_plan = d["targetPlan"].cast<int>();
  if (d.contains("postExec")) s.post_exec = to_expr(d["postExec"]);
  if (d.contains("postExecMsg"))
    s.post_exec_msg = d["postExecMsg"].cast<std::string>();
  if (d.contains("branches"))
    for (const auto& b : d["branches"])
      s.branches.push_back(to_step(b.cast<py::dict>()));
  return s;
}

static Plan to_plan(const py::dict& d) {
  Plan p;
  p.name = d["name"].cast<std::string>();
  for (const auto& s : d["steps"]) p.steps.push_back(to_step(s.cast<py::dict>()));
  p.dependents.resize(p.steps.size());
  for (size_t i = 0; i < p.steps.size(); ++i)
    for (int dep : p.steps[i].deps) p.dependents[dep].push_back((int)i);
  if (d.contains("output")) p.output_tpl = to_tnode(d["output"]);
  if (d.contains("failFast")) p.fail_fast = d["failFast"].cast<bool>();
  if (d.contains("storyTimeout"))
    p.story_timeout = d["storyTimeout"].cast<double>();
  if (d.contains("concurrency")) p.concurrency = d["concurrency"].cast<int>();
  if (d.contains("nMain")) p.n_main = d["nMain"].cast<int>();
  if (d.contains("nComp")) p.n_comp = d["nComp"].cast<int>();
  if (d.contains("nFin")) p.n_fin = d["nFin"].cast<int>();
  return p;
}

PYBIND11_MODULE(_core, m) {
  m.doc() = "bobraccel — native DAG run engine for bobrapet_amd";

  py::class_<NativeEngine>(m, "NativeEngine")
      .def(py::init<>())
      .def("register_plan",
           [](NativeEngine& e, const py::dict& plan) {
             return e.register_plan(to_plan(plan));
           })
      .def("set_launcher",
           [](NativeEngine& e, py::function fn) {
             e.set_launcher([fn](uint64_t run, int step, int branch,
                                 uint32_t attempt, const std::string& engram,
                                 const std::string& step_name,
                                 const JValue& input) {
               py::gil_scoped_acquire g;
               fn(run, step, branch, attempt, engram, step_name,
                  to_python(input));
             });
           })
      .def("start", &NativeEngine::start,
           py::call_guard<py::gil_scoped_release>())
      .def("stop", &NativeEngine::stop,
           py::call_guard<py::gil_scoped_release>())
      .def(
          "submit",
          [](NativeEngine& e, int plan_id, const py::object& inputs) {
            JValue v = to_jvalue(inputs);
            py::gil_scoped_release r;
            return e.submit(plan_id, std::move(v));
          },
          py::arg("plan_id"), py::arg("inputs") = py::none())
      .def(
          "complete_engram",
          [](NativeEngine& e, uint64_t run, int step, int branch,
             uint32_t attempt, int exit_code, const py::object& output,
             const std::string& error) {
            JValue v = to_jvalue(output);
            py::gil_scoped_release r;
            e.complete_engram(run, step, branch, attempt, exit_code,
                              std::move(v), error);
          },
          py::arg("run"), py::arg("step"), py::arg("branch"),
          py::arg("attempt"), py::arg("exit_code"),
          py::arg("output") = py::none(), py::arg("error") = "")
      .def("set_native_lane",
           [](NativeEngine& e, py::capsule cap) {
             e.set_native_lane(
                 reinterpret_cast<const NativeLane*>(cap.get_pointer()));
           })
      .def("set_devices",
           [](NativeEngine& e, std::vector<int> devs) {
             e.set_devices(std::move(devs));
           })
      .def("decide_gate", &NativeEngine::decide_gate,
           py::call_guard<py::gil_scoped_release>())
      .def("cancel", &NativeEngine::cancel,
           py::call_guard<py::gil_scoped_release>())
      .def("wait", &NativeEngine::wait,
           py::call_guard<py::gil_scoped_release>(), py::arg("run"),
           py::arg("timeout") = 0.0)
      .def("run_status",
           [](NativeEngine& e, uint64_t run) {
             JValue v;
             {
               py::gil_scoped_release r;
               v = e.run_status(run);
             }
             return to_python(v);
           })
      .def("run_count", &NativeEngine::run_count,
           py::call_guard<py::gil_scoped_release>())
      .def("gc_run", &NativeEngine::gc_run,
           py::call_guard<py::gil_scoped_release>());

  m.def("set_expr_hydrator", [](py::object fn) {
    // the callable lives in a deliberately LEAKED holder: a py::object
    // captured inside the static std::function would be decref'd during
    // static teardown, after the interpreter is gone (segfault at exit)
    static py::object* holder = new py::object();
    *holder = fn;  // GIL held in a pybind def — safe decref of the old one
    if (fn.is_none()) {
      expr_hydrator() = nullptr;
      return;
    }
    py::object* h = holder;
    expr_hydrator() = [h](const JValue& marker) -> JValue {
      py::gil_scoped_acquire g;
      return to_jvalue((*h)(to_python(marker)));
    };
  });

  m.def("eval_expression", [](const py::handle& ast, const py::dict& scope) {
    ExprPtr e = to_expr(ast);
    JValue sv = to_jvalue(scope);
    return to_python(eval_expr(*e, sv.as_object()));
  });
  m.attr("__version__") = "0.1.0";
}
