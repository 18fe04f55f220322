// Expression VM for the native DAG core.
//
// Executes the AST produced by bobrapet_amd.templating.parser (the Python
// compiler ships the nested-tuple AST; the binding layer converts it to
// ExprNode trees once per compiled story).  Semantics mirror
// templating/evaluator.py: null-propagating member access, truthiness,
// '+' string concat, a small function library.
#pragma once

#include <cmath>
#include <functional>
#include <memory>
#include <stdexcept>

#include "jvalue.h"

namespace bobraccel {

// Optional offloaded-payload hydrator (set by the Python side): when an
// expression CONSUMES a `$storageRef` marker (comparison, arithmetic,
// truthiness, function argument), the value is materialized through the
// storage layer; pure pass-through (member access / template splicing)
// keeps moving markers so the loop never copies payload bytes.
inline std::function<JValue(const JValue&)>& expr_hydrator() {
  static std::function<JValue(const JValue&)> h;
  return h;
}

inline bool is_storage_marker(const JValue& v) {
  return v.is_object() && v.as_object().count("$storageRef") > 0;
}

inline JValue materialize(JValue v) {
  if (is_storage_marker(v)) {
    auto& h = expr_hydrator();
    if (h) return h(v);
  }
  return v;
}

enum class Op {
  Const,
  Var,
  Get,     // a = child0, key = str
  Index,   // child0[child1]
  And,
  Or,
  Not,
  Cmp,     // str = operator
  Bin,     // str = operator
  Neg,
  Cond,    // child0 ? child1 : child2
  List,
  Map,     // keys in strs, values in children
  Call,    // str = fn name, args = children
  Method,  // child0.str(children[1..])
};

struct ExprNode {
  Op op;
  JValue constant;
  std::string str;
  std::vector<std::string> strs;
  std::vector<std::shared_ptr<ExprNode>> children;
};

using ExprPtr = std::shared_ptr<ExprNode>;

class ExprError : public std::runtime_error {
 public:
  using std::runtime_error::runtime_error;
};

inline JValue eval_expr(const ExprNode& n, const JObject& scope);

inline JValue call_fn(const std::string& name, std::vector<JValue>& args) {
  auto arity = [&](size_t n) {
    if (args.size() < n) throw ExprError("function " + name + ": missing args");
  };
  if (name == "size" || name == "len") {
    arity(1);
    return (int64_t)args[0].size();
  }
  if (name == "has") {
    arity(1);
    if (args.size() == 1) return !args[0].is_null();
    return args[0].is_object() &&
           args[0].as_object().count(args[1].to_string()) > 0;
  }
  if (name == "string") {
    arity(1);
    return args[0].to_string();
  }
  if (name == "int") {
    arity(1);
    if (args[0].is_string()) return (int64_t)std::stoll(args[0].as_string());
    return (int64_t)args[0].as_double();
  }
  if (name == "float") {
    arity(1);
    if (args[0].is_string()) return std::stod(args[0].as_string());
    return args[0].as_double();
  }
  if (name == "bool") {
    arity(1);
    return args[0].truthy();
  }
  if (name == "abs") {
    arity(1);
    if (args[0].is_int()) return (int64_t)std::llabs(args[0].as_int());
    return std::fabs(args[0].as_double());
  }
  if (name == "min" || name == "max") {
    arity(1);
    const JArray* items;
    JArray tmp;
    if (args.size() == 1 && args[0].is_array()) {
      items = &args[0].as_array();
    } else {
      tmp = JArray(args.begin(), args.end());
      items = &tmp;
    }
    if (items->empty()) throw ExprError(name + "() of empty sequence");
    JValue best = (*items)[0];
    for (const auto& v : *items) {
      bool lt = v.as_double() < best.as_double();
      if ((name == "min") == lt) best = v;
    }
    return best;
  }
  if (name == "floor") {
    arity(1);
    return (int64_t)std::floor(args[0].as_double());
  }
  if (name == "ceil") {
    arity(1);
    return (int64_t)std::ceil(args[0].as_double());
  }
  if (name == "round") {
    arity(1);
    return (int64_t)std::llround(args[0].as_double());
  }
  if (name == "coalesce") {
    for (auto& a : args)
      if (!a.is_null()) return a;
    return JValue();
  }
  if (name == "default") {
    arity(2);
    return args[0].is_null() ? args[1] : args[0];
  }
  if (name == "contains") {
    arity(2);
    if (args[0].is_string())
      return args[0].as_string().find(args[1].to_string()) !=
             std::string::npos;
    if (args[0].is_array()) {
      for (const auto& v : args[0].as_array())
        if (v.equals(args[1])) return true;
      return false;
    }
    if (args[0].is_object())
      return args[0].as_object().count(args[1].to_string()) > 0;
    return false;
  }
  if (name == "startsWith" || name == "endsWith") {
    arity(2);
    const std::string s = args[0].to_string(), p = args[1].to_string();
    if (p.size() > s.size()) return false;
    if (name == "startsWith") return s.compare(0, p.size(), p) == 0;
    return s.compare(s.size() - p.size(), p.size(), p) == 0;
  }
  if (name == "lower" || name == "upper") {
    arity(1);
    std::string s = args[0].to_string();
    for (auto& c : s)
      c = name == "lower" ? (char)tolower(c) : (char)toupper(c);
    return s;
  }
  if (name == "trim") {
    arity(1);
    std::string s = args[0].to_string();
    size_t a = s.find_first_not_of(" \t\n\r");
    size_t b = s.find_last_not_of(" \t\n\r");
    if (a == std::string::npos) return std::string();
    return s.substr(a, b - a + 1);
  }
  if (name == "join") {
    arity(1);
    std::string sep = args.size() > 1 ? args[1].to_string() : "";
    std::string out;
    bool first = true;
    if (args[0].is_array())
      for (const auto& v : args[0].as_array()) {
        if (!first) out += sep;
        first = false;
        out += v.to_string();
      }
    return out;
  }
  if (name == "keys") {
    arity(1);
    JArray out;
    if (args[0].is_object())
      for (const auto& [k, _] : args[0].as_object()) out.push_back(k);
    return out;
  }
  if (name == "range") {
    arity(1);
    int64_t start = 0, stop, step = 1;
    if (args.size() == 1) stop = args[0].as_int();
    else {
      start = args[0].as_int();
      stop = args[1].as_int();
      if (args.size() > 2) step = args[2].as_int();
    }
    JArray out;
    for (int64_t i = start; step > 0 ? i < stop : i > stop; i += step)
      out.push_back(i);
    return out;
  }
  throw ExprError("unknown function " + name +
                  " (now()/uuid() are blocked: deterministic mode)");
}

inline JValue cmp_values(const std::string& op, const JValue& a,
                         const JValue& b) {
  if (op == "==") return a.equals(b);
  if (op == "!=") return !a.equals(b);
  if (op == "in") {
    if (b.is_array()) {
      for (const auto& v : b.as_array())
        if (v.equals(a)) return true;
      return false;
    }
    if (b.is_object()) return b.as_object().count(a.to_string()) > 0;
    if (b.is_string())
      return b.as_string().find(a.to_string()) != std::string::npos;
    return false;
  }
  if (a.is_null() || b.is_null()) return false;
  double x, y;
  if (a.is_number() && b.is_number()) {
    x = a.as_double();
    y = b.as_double();
  } else if (a.is_string() && b.is_string()) {
    int c = a.as_string().compare(b.as_string());
    x = (double)c;
    y = 0.0;
  } else {
    return false;
  }
  if (op == "<") return x < y;
  if (op == "<=") return x <= y;
  if (op == ">") return x > y;
  if (op == ">=") return x >= y;
  throw ExprError("unknown comparison " + op);
}

inline JValue bin_values(const std::string& op, const JValue& a,
                         const JValue& b) {
  if (op == "+") {
    if (a.is_string() || b.is_string()) return a.to_string() + b.to_string();
    if (a.is_array() && b.is_array()) {
      JArray out = a.as_array();
      for (const auto& v : b.as_array()) out.push_back(v);
      return out;
    }
    if (a.is_null()) return b;
    if (b.is_null()) return a;
    if (a.is_int() && b.is_int()) return a.as_int() + b.as_int();
    return a.as_double() + b.as_double();
  }
  if (a.is_null() || b.is_null()) return JValue();
  if (op == "-") {
    if (a.is_int() && b.is_int()) return a.as_int() - b.as_int();
    return a.as_double() - b.as_double();
  }
  if (op == "*") {
    if (a.is_int() && b.is_int()) return a.as_int() * b.as_int();
    return a.as_double() * b.as_double();
  }
  if (op == "/") {
    double d = b.as_double();
    if (d == 0.0) throw ExprError("division by zero");
    return a.as_double() / d;
  }
  if (op == "%") {
    int64_t d = b.as_int();
    if (d == 0) throw ExprError("modulo by zero");
    return a.as_int() % d;
  }
  throw ExprError("unknown operator " + op);
}

inline JValue call_method(const std::string& name, const JValue& obj,
                          std::vector<JValue>& args) {
  std::vector<JValue> all;
  all.push_back(obj);
  for (auto& a : args) all.push_back(a);
  if (name == "get") {
    JValue v = obj.get(args.empty() ? "" : args[0].to_string());
    if (v.is_null() && args.size() > 1) return args[1];
    return v;
  }
  return call_fn(name, all);
}

inline JValue eval_expr(const ExprNode& n, const JObject& scope) {
  switch (n.op) {
    case Op::Const:
      return n.constant;
    case Op::Var: {
      auto it = scope.find(n.str);
      return it != scope.end() ? it->second : JValue();
    }
    case Op::Get:
      return eval_expr(*n.children[0], scope).get(n.str);
    case Op::Index: {
      JValue obj = eval_expr(*n.children[0], scope);
      JValue key = eval_expr(*n.children[1], scope);
      if (key.is_number()) return obj.index(key.as_int());
      if (key.is_string()) return obj.get(key.as_string());
      return JValue();
    }
    case Op::And:
      if (!materialize(eval_expr(*n.children[0], scope)).truthy()) return false;
      return materialize(eval_expr(*n.children[1], scope)).truthy();
    case Op::Or:
      if (materialize(eval_expr(*n.children[0], scope)).truthy()) return true;
      return materialize(eval_expr(*n.children[1], scope)).truthy();
    case Op::Not:
      return !materialize(eval_expr(*n.children[0], scope)).truthy();
    case Op::Cmp:
      return cmp_values(n.str, materialize(eval_expr(*n.children[0], scope)),
                        materialize(eval_expr(*n.children[1], scope)));
    case Op::Bin:
      return bin_values(n.str, materialize(eval_expr(*n.children[0], scope)),
                        materialize(eval_expr(*n.children[1], scope)));
    case Op::Neg: {
      JValue v = materialize(eval_expr(*n.children[0], scope));
      if (v.is_int()) return -v.as_int();
      if (v.is_double()) return -v.as_double();
      return JValue();
    }
    case Op::Cond:
      return materialize(eval_expr(*n.children[0], scope)).truthy()
                 ? eval_expr(*n.children[1], scope)
                 : eval_expr(*n.children[2], scope);
    case Op::List: {
      JArray out;
      for (const auto& c : n.children) out.push_back(eval_expr(*c, scope));
      return out;
    }
    case Op::Map: {
      JObject out;
      for (size_t i = 0; i < n.children.size(); ++i)
        out[n.strs[i]] = eval_expr(*n.children[i], scope);
      return out;
    }
    case Op::Call: {
      std::vector<JValue> args;
      for (const auto& c : n.children)
        args.push_back(materialize(eval_expr(*c, scope)));
      return call_fn(n.str, args);
    }
    case Op::Method: {
      JValue obj = materialize(eval_expr(*n.children[0], scope));
      std::vector<JValue> args;
      for (size_t i = 1; i < n.children.size(); ++i)
        args.push_back(materialize(eval_expr(*n.children[i], scope)));
      return call_method(n.str, obj, args);
    }
  }
  throw ExprError("bad expression node");
}

// ---------------------------------------------------------------------------
// Templates: a JValue tree where string leaves may be {{ expr }} templates.
// Compiled form: TNode = literal | single-expr | parts | array | object.
// ---------------------------------------------------------------------------

struct TNode {
  enum Kind { Literal, Single, Parts, Arr, Obj } kind = Literal;
  JValue literal;
  ExprPtr expr;                                   // Single
  std::vector<std::pair<std::string, ExprPtr>> parts;  // Parts: lit + opt expr
  std::vector<std::shared_ptr<TNode>> items;      // Arr
  std::vector<std::pair<std::string, std::shared_ptr<TNode>>> fields;  // Obj
};

using TNodePtr = std::shared_ptr<TNode>;

inline JValue eval_template(const TNode& t, const JObject& scope) {
  switch (t.kind) {
    case TNode::Literal:
      return t.literal;
    case TNode::Single:
      return eval_expr(*t.expr, scope);
    case TNode::Parts: {
      std::string out;
      for (const auto& [lit, e] : t.parts) {
        out += lit;
        // splicing into a string CONSUMES the value: hydrate markers so
        // "id-{{ steps.a.output.big.field }}" interpolates the real value
        if (e) out += materialize(eval_expr(*e, scope)).to_string();
      }
      return out;
    }
    case TNode::Arr: {
      JArray out;
      for (const auto& it : t.items) out.push_back(eval_template(*it, scope));
      return out;
    }
    case TNode::Obj: {
      JObject out;
      for (const auto& [k, v] : t.fields) out[k] = eval_template(*v, scope);
      return out;
    }
  }
  return JValue();
}

}  // namespace bobraccel
