// Minimal protobuf wire-format encode/decode (header-only).
//
// The kubelet DevicePlugin v1beta1 messages are small and flat, so the
// plugin encodes/decodes them directly at the wire level instead of
// depending on a protobuf code generator (none is available in this image;
// the Python test suite mirrors these helpers and grpcio round-trips them).

#pragma once

#include <cstdint>
#include <string>
#include <string_view>

namespace k3samd::pb {

enum Wire { kVarint = 0, kFixed64 = 1, kLenDelim = 2, kFixed32 = 5 };

inline void put_varint(std::string& out, uint64_t v) {
  while (v >= 0x80) {
    out.push_back((char)((v & 0x7f) | 0x80));
    v >>= 7;
  }
  out.push_back((char)v);
}

inline void put_tag(std::string& out, int field, int wire) {
  put_varint(out, ((uint64_t)field << 3) | (uint64_t)wire);
}

inline void put_bytes(std::string& out, int field, std::string_view payload) {
  put_tag(out, field, kLenDelim);
  put_varint(out, payload.size());
  out.append(payload.data(), payload.size());
}

inline void put_string(std::string& out, int field, std::string_view s) {
  put_bytes(out, field, s);
}

inline void put_uint(std::string& out, int field, uint64_t v) {
  put_tag(out, field, kVarint);
  put_varint(out, v);
}

inline void put_bool(std::string& out, int field, bool b) {
  // proto3 default: omit false
  if (b) put_uint(out, field, 1);
}

// map<string,string> entry: submessage {1: key, 2: value}
inline void put_map_entry(std::string& out, int field, std::string_view k,
                          std::string_view v) {
  std::string entry;
  put_string(entry, 1, k);
  put_string(entry, 2, v);
  put_bytes(out, field, entry);
}

class Reader {
 public:
  Reader(std::string_view buf)
      : p_((const uint8_t*)buf.data()), end_(p_ + buf.size()) {}

  bool ok() const { return ok_; }
  bool done() const { return p_ >= end_; }

  // Advance to the next field. Returns false at end or on malformed input
  // (check ok()). For kVarint fields `varint` is set; for kLenDelim `data`.
  bool next(int& field, int& wire, uint64_t& varint, std::string_view& data) {
    if (done()) return false;
    uint64_t tag;
    if (!read_varint(tag)) return fail();
    field = (int)(tag >> 3);
    wire = (int)(tag & 7);
    switch (wire) {
      case kVarint:
        return read_varint(varint) ? true : fail();
      case kLenDelim: {
        uint64_t len;
        if (!read_varint(len) || (uint64_t)(end_ - p_) < len) return fail();
        data = std::string_view((const char*)p_, len);
        p_ += len;
        return true;
      }
      case kFixed64:
        if (end_ - p_ < 8) return fail();
        varint = 0;
        for (int i = 7; i >= 0; --i) varint = (varint << 8) | p_[i];
        p_ += 8;
        return true;
      case kFixed32:
        if (end_ - p_ < 4) return fail();
        varint = 0;
        for (int i = 3; i >= 0; --i) varint = (varint << 8) | p_[i];
        p_ += 4;
        return true;
      default:
        return fail();
    }
  }

 private:
  bool read_varint(uint64_t& v) {
    v = 0;
    int shift = 0;
    while (p_ < end_ && shift < 64) {
      uint8_t b = *p_++;
      v |= (uint64_t)(b & 0x7f) << shift;
      if (!(b & 0x80)) return true;
      shift += 7;
    }
    return false;
  }
  bool fail() {
    ok_ = false;
    return false;
  }
  const uint8_t* p_;
  const uint8_t* end_;
  bool ok_ = true;
};

// parse a map<string,string> entry submessage
inline bool parse_map_entry(std::string_view buf, std::string& k,
                            std::string& v) {
  Reader r(buf);
  int f, w;
  uint64_t u;
  std::string_view d;
  while (r.next(f, w, u, d)) {
    if (f == 1 && w == kLenDelim) k.assign(d);
    if (f == 2 && w == kLenDelim) v.assign(d);
  }
  return r.ok();
}

}  // namespace k3samd::pb
