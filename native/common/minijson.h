// Minimal JSON DOM (parse + serialize), order- and number-preserving.
//
// Used by the OCI runtime wrapper to edit a container's config.json: every
// field we don't understand must round-trip untouched, so numbers keep
// their raw spelling and object key order is preserved.

#pragma once

#include <memory>
#include <string>
#include <utility>
#include <vector>

namespace k3samd {

struct JValue;
using JPtr = std::shared_ptr<JValue>;

struct JValue {
  enum Type { kNull, kBool, kNumber, kString, kArray, kObject };
  Type type = kNull;
  bool b = false;
  std::string num;  // raw number text
  std::string str;
  std::vector<JPtr> arr;
  std::vector<std::pair<std::string, JPtr>> obj;

  static JPtr make_null() { return std::make_shared<JValue>(); }
  static JPtr make_bool(bool v) {
    auto j = std::make_shared<JValue>();
    j->type = kBool;
    j->b = v;
    return j;
  }
  static JPtr make_int(long long v) {
    auto j = std::make_shared<JValue>();
    j->type = kNumber;
    j->num = std::to_string(v);
    return j;
  }
  static JPtr make_str(std::string v) {
    auto j = std::make_shared<JValue>();
    j->type = kString;
    j->str = std::move(v);
    return j;
  }
  static JPtr make_arr() {
    auto j = std::make_shared<JValue>();
    j->type = kArray;
    return j;
  }
  static JPtr make_obj() {
    auto j = std::make_shared<JValue>();
    j->type = kObject;
    return j;
  }

  // object helpers
  JPtr get(const std::string& key) const {
    if (type != kObject) return nullptr;
    for (auto& [k, v] : obj)
      if (k == key) return v;
    return nullptr;
  }
  void set(const std::string& key, JPtr v) {
    for (auto& [k, val] : obj)
      if (k == key) {
        val = std::move(v);
        return;
      }
    obj.emplace_back(key, std::move(v));
  }
  // get-or-create nested object/array
  JPtr ensure_obj(const std::string& key) {
    JPtr v = get(key);
    if (!v || v->type != kObject) {
      v = make_obj();
      set(key, v);
    }
    return v;
  }
  JPtr ensure_arr(const std::string& key) {
    JPtr v = get(key);
    if (!v || v->type != kArray) {
      v = make_arr();
      set(key, v);
    }
    return v;
  }
  long long as_int(long long dflt = 0) const {
    if (type != kNumber) return dflt;
    try {
      return std::stoll(num);
    } catch (...) {
      return dflt;
    }
  }
};

// Throws std::runtime_error on malformed input.
JPtr json_parse(const std::string& text);
std::string json_serialize(const JPtr& v, int indent = 0);

}  // namespace k3samd
