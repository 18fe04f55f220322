// JValue: the JSON-like value type of the native DAG core (bobraccel).
//
// Role: payloads, template scopes and step outputs inside the C++ engine.
// Opaque handles carry Python objects (e.g. torch tensors) through the
// engine untouched — payload data itself stays in HBM; the control plane
// only moves references.
#pragma once

#include <cstdint>
#include <map>
#include <memory>
#include <string>
#include <variant>
#include <vector>

namespace bobraccel {

class JValue;
using JArray = std::vector<JValue>;
using JObject = std::map<std::string, JValue>;

// Opaque reference to a host-language object (refcount managed by the
// binding layer via the deleter).
struct Opaque {
  void* ptr = nullptr;
  void (*incref)(void*) = nullptr;
  void (*decref)(void*) = nullptr;

  Opaque() = default;
  Opaque(void* p, void (*inc)(void*), void (*dec)(void*))
      : ptr(p), incref(inc), decref(dec) {
    if (ptr && incref) incref(ptr);
  }
  Opaque(const Opaque& o) : ptr(o.ptr), incref(o.incref), decref(o.decref) {
    if (ptr && incref) incref(ptr);
  }
  Opaque& operator=(const Opaque& o) {
    if (this == &o) return *this;
    if (ptr && decref) decref(ptr);
    ptr = o.ptr;
    incref = o.incref;
    decref = o.decref;
    if (ptr && incref) incref(ptr);
    return *this;
  }
  ~Opaque() {
    if (ptr && decref) decref(ptr);
  }
  bool operator==(const Opaque& o) const { return ptr == o.ptr; }
  bool operator<(const Opaque& o) const { return ptr < o.ptr; }
};

class JValue {
 public:
  using Storage = std::variant<std::monostate, bool, int64_t, double,
                               std::string, std::shared_ptr<JArray>,
                               std::shared_ptr<JObject>, Opaque>;

  JValue() : v_() {}
  JValue(std::nullptr_t) : v_() {}
  JValue(bool b) : v_(b) {}
  JValue(int i) : v_((int64_t)i) {}
  JValue(int64_t i) : v_(i) {}
  JValue(double d) : v_(d) {}
  JValue(const char* s) : v_(std::string(s)) {}
  JValue(std::string s) : v_(std::move(s)) {}
  JValue(JArray a) : v_(std::make_shared<JArray>(std::move(a))) {}
  JValue(JObject o) : v_(std::make_shared<JObject>(std::move(o))) {}
  JValue(Opaque o) : v_(std::move(o)) {}

  bool is_null() const { return std::holds_alternative<std::monostate>(v_); }
  bool is_bool() const { return std::holds_alternative<bool>(v_); }
  bool is_int() const { return std::holds_alternative<int64_t>(v_); }
  bool is_double() const { return std::holds_alternative<double>(v_); }
  bool is_number() const { return is_int() || is_double(); }
  bool is_string() const { return std::holds_alternative<std::string>(v_); }
  bool is_array() const {
    return std::holds_alternative<std::shared_ptr<JArray>>(v_);
  }
  bool is_object() const {
    return std::holds_alternative<std::shared_ptr<JObject>>(v_);
  }
  bool is_opaque() const { return std::holds_alternative<Opaque>(v_); }

  bool as_bool() const { return std::get<bool>(v_); }
  int64_t as_int() const { return std::get<int64_t>(v_); }
  double as_double() const {
    if (is_int()) return (double)std::get<int64_t>(v_);
    return std::get<double>(v_);
  }
  const std::string& as_string() const { return std::get<std::string>(v_); }
  JArray& as_array() { return *std::get<std::shared_ptr<JArray>>(v_); }
  const JArray& as_array() const {
    return *std::get<std::shared_ptr<JArray>>(v_);
  }
  JObject& as_object() { return *std::get<std::shared_ptr<JObject>>(v_); }
  const JObject& as_object() const {
    return *std::get<std::shared_ptr<JObject>>(v_);
  }
  const Opaque& as_opaque() const { return std::get<Opaque>(v_); }

  // truthiness: null/false/0/""/empty containers are false
  bool truthy() const {
    if (is_null()) return false;
    if (is_bool()) return as_bool();
    if (is_int()) return as_int() != 0;
    if (is_double()) return as_double() != 0.0;
    if (is_string()) return !as_string().empty();
    if (is_array()) return !as_array().empty();
    if (is_object()) return !as_object().empty();
    return true;  // opaque
  }

  bool equals(const JValue& o) const {
    if (is_number() && o.is_number()) return as_double() == o.as_double();
    if (v_.index() != o.v_.index()) return false;
    if (is_null()) return true;
    if (is_bool()) return as_bool() == o.as_bool();
    if (is_string()) return as_string() == o.as_string();
    if (is_array()) {
      const auto &a = as_array(), &b = o.as_array();
      if (a.size() != b.size()) return false;
      for (size_t i = 0; i < a.size(); ++i)
        if (!a[i].equals(b[i])) return false;
      return true;
    }
    if (is_object()) {
      const auto &a = as_object(), &b = o.as_object();
      if (a.size() != b.size()) return false;
      for (const auto& [k, va] : a) {
        auto it = b.find(k);
        if (it == b.end() || !va.equals(it->second)) return false;
      }
      return true;
    }
    if (is_opaque()) return as_opaque() == o.as_opaque();
    return false;
  }

  // member access: null-propagating (missing → null)
  JValue get(const std::string& key) const {
    if (is_object()) {
      const auto& obj = as_object();
      auto it = obj.find(key);
      if (it != obj.end()) return it->second;
      // indexing THROUGH an offloaded payload: derive a sub-path marker
      // instead of mis-reading the marker dict.  The storage layer's ref
      // path DSL (storage/manager.py path.go parity) resolves it when the
      // value is hydrated on the worker — the engine loop never touches
      // payload bytes (offloaded-data discipline, dag.go 3-way policy).
      auto ref = obj.find("$storageRef");
      if (ref != obj.end() && ref->second.is_object()) {
        JObject inner = ref->second.as_object();
        auto pit = inner.find("path");
        std::string base =
            (pit != inner.end() && pit->second.is_string()) ? pit->second.as_string() : "";
        inner["path"] = base.empty() ? key : base + "." + key;
        JObject marker;
        marker["$storageRef"] = std::move(inner);
        return JValue(std::move(marker));
      }
      // alias tolerance: '_' in template identifiers ↔ '-' in step names
      if (key.find('_') != std::string::npos) {
        std::string alt = key;
        for (auto& c : alt)
          if (c == '_') c = '-';
        it = obj.find(alt);
        if (it != obj.end()) return it->second;
      }
    }
    return JValue();
  }

  JValue index(int64_t i) const {
    if (is_array()) {
      const auto& a = as_array();
      if (i < 0) i += (int64_t)a.size();
      if (i >= 0 && i < (int64_t)a.size()) return a[(size_t)i];
    }
    if (is_object()) {
      const auto& obj = as_object();
      auto ref = obj.find("$storageRef");
      if (ref != obj.end() && ref->second.is_object()) {
        JObject inner = ref->second.as_object();
        auto pit = inner.find("path");
        std::string base =
            (pit != inner.end() && pit->second.is_string()) ? pit->second.as_string() : "";
        std::string key = "[" + std::to_string(i) + "]";
        inner["path"] = base.empty() ? key : base + key;
        JObject marker;
        marker["$storageRef"] = std::move(inner);
        return JValue(std::move(marker));
      }
    }
    return JValue();
  }

  std::string to_string() const;  // for string interpolation
  size_t size() const {
    if (is_string()) return as_string().size();
    if (is_array()) return as_array().size();
    if (is_object()) return as_object().size();
    return 0;
  }

  const Storage& storage() const { return v_; }

 private:
  Storage v_;
};

inline std::string JValue::to_string() const {
  if (is_null()) return "";
  if (is_bool()) return as_bool() ? "true" : "false";
  if (is_int()) return std::to_string(as_int());
  if (is_double()) {
    double d = as_double();
    if (d == (int64_t)d && d < 1e15 && d > -1e15)
      return std::to_string((int64_t)d);
    char buf[32];
    snprintf(buf, sizeof(buf), "%g", d);
    return buf;
  }
  if (is_string()) return as_string();
  if (is_array()) {
    std::string out = "[";
    bool first = true;
    for (const auto& v : as_array()) {
      if (!first) out += ",";
      first = false;
      out += v.is_string() ? ("\"" + v.as_string() + "\"") : v.to_string();
    }
    return out + "]";
  }
  if (is_object()) {
    std::string out = "{";
    bool first = true;
    for (const auto& [k, v] : as_object()) {
      if (!first) out += ",";
      first = false;
      out += "\"" + k + "\":";
      out += v.is_string() ? ("\"" + v.as_string() + "\"") : v.to_string();
    }
    return out + "}";
  }
  return "<opaque>";
}

}  // namespace bobraccel
