// Minimal JSON emission helper (no third-party deps in this image).
#pragma once

#include <cstdint>
#include <sstream>
#include <string>

namespace k3samd {

inline std::string json_escape(const std::string& s) {
  std::string out;
  out.reserve(s.size() + 8);
  for (char c : s) {
    switch (c) {
      case '"': out += "\\\""; break;
      case '\\': out += "\\\\"; break;
      case '\n': out += "\\n"; break;
      case '\r': out += "\\r"; break;
      case '\t': out += "\\t"; break;
      default:
        if ((unsigned char)c < 0x20) {
          char buf[8];
          std::snprintf(buf, sizeof(buf), "\\u%04x", c);
          out += buf;
        } else {
          out += c;
        }
    }
  }
  return out;
}

// Tiny streaming writer: caller is responsible for well-formed nesting.
class JsonWriter {
 public:
  std::string str() const { return ss_.str(); }

  JsonWriter& raw(const std::string& s) { ss_ << s; return *this; }
  JsonWriter& key(const std::string& k) {
    comma();
    ss_ << '"' << json_escape(k) << "\":";
    fresh_ = true;
    return *this;
  }
  JsonWriter& value(const std::string& v) {
    comma();
    ss_ << '"' << json_escape(v) << '"';
    return *this;
  }
  JsonWriter& value(const char* v) { return value(std::string(v)); }
  JsonWriter& value(uint64_t v) { comma(); ss_ << v; return *this; }
  JsonWriter& value(int64_t v) { comma(); ss_ << v; return *this; }
  JsonWriter& value(int v) { comma(); ss_ << v; return *this; }
  JsonWriter& value(unsigned v) { comma(); ss_ << v; return *this; }
  JsonWriter& value(double v) { comma(); ss_ << v; return *this; }
  JsonWriter& value(bool v) { comma(); ss_ << (v ? "true" : "false"); return *this; }
  JsonWriter& begin_obj() { comma(); ss_ << '{'; fresh_ = true; return *this; }
  JsonWriter& end_obj() { ss_ << '}'; fresh_ = false; return *this; }
  JsonWriter& begin_arr() { comma(); ss_ << '['; fresh_ = true; return *this; }
  JsonWriter& end_arr() { ss_ << ']'; fresh_ = false; return *this; }

 private:
  void comma() {
    if (!fresh_) ss_ << ',';
    fresh_ = false;
  }
  std::stringstream ss_;
  bool fresh_ = true;
};

}  // namespace k3samd
