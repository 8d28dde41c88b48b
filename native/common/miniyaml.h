// Minimal YAML subset parser: block maps, block lists, scalars, comments.
//
// Covers the device-plugin config schema the reference exposes through Helm
// (/root/reference/values.yaml:6-18: version / flags.migStrategy /
// sharing.timeSlicing.{renameByDefault,failRequestsGreaterThanOne,
// resources[].name/replicas}) and similar small config files. Not a general
// YAML implementation (no anchors, no flow collections, no multi-line
// scalars) — unsupported syntax fails loudly.

#pragma once

#include <cstdint>
#include <memory>
#include <string>
#include <utility>
#include <vector>

namespace k3samd {

struct YNode {
  enum Type { kNull, kScalar, kMap, kList };
  Type type = kNull;
  std::string scalar;
  std::vector<std::pair<std::string, YNode>> map;
  std::vector<YNode> list;

  const YNode* get(const std::string& key) const {
    if (type != kMap) return nullptr;
    for (auto& [k, v] : map)
      if (k == key) return &v;
    return nullptr;
  }
  // dotted-path convenience: get_path("sharing.timeSlicing.resources")
  const YNode* get_path(const std::string& dotted) const;

  std::string as_str(const std::string& dflt = "") const {
    return type == kScalar ? scalar : dflt;
  }
  int64_t as_int(int64_t dflt = 0) const;
  bool as_bool(bool dflt = false) const;
};

// Throws std::runtime_error on malformed input.
YNode yaml_parse(const std::string& text);

}  // namespace k3samd
