#include "hpack.h"

#include <unordered_map>

#include "hpack_huffman_table.h"

namespace k3samd {

namespace {

// RFC 7541 Appendix A static table (1-based, 61 entries).
const Header kStatic[] = {
    {"", ""},  // index 0 unused
    {":authority", ""},
    {":method", "GET"},
    {":method", "POST"},
    {":path", "/"},
    {":path", "/index.html"},
    {":scheme", "http"},
    {":scheme", "https"},
    {":status", "200"},
    {":status", "204"},
    {":status", "206"},
    {":status", "304"},
    {":status", "400"},
    {":status", "404"},
    {":status", "500"},
    {"accept-charset", ""},
    {"accept-encoding", "gzip, deflate"},
    {"accept-language", ""},
    {"accept-ranges", ""},
    {"accept", ""},
    {"access-control-allow-origin", ""},
    {"age", ""},
    {"allow", ""},
    {"authorization", ""},
    {"cache-control", ""},
    {"content-disposition", ""},
    {"content-encoding", ""},
    {"content-language", ""},
    {"content-length", ""},
    {"content-location", ""},
    {"content-range", ""},
    {"content-type", ""},
    {"cookie", ""},
    {"date", ""},
    {"etag", ""},
    {"expect", ""},
    {"expires", ""},
    {"from", ""},
    {"host", ""},
    {"if-match", ""},
    {"if-modified-since", ""},
    {"if-none-match", ""},
    {"if-range", ""},
    {"if-unmodified-since", ""},
    {"last-modified", ""},
    {"link", ""},
    {"location", ""},
    {"max-forwards", ""},
    {"proxy-authenticate", ""},
    {"proxy-authorization", ""},
    {"range", ""},
    {"referer", ""},
    {"refresh", ""},
    {"retry-after", ""},
    {"server", ""},
    {"set-cookie", ""},
    {"strict-transport-security", ""},
    {"transfer-encoding", ""},
    {"user-agent", ""},
    {"vary", ""},
    {"via", ""},
    {"www-authenticate", ""},
};
constexpr size_t kStaticCount = 61;

// Huffman decode map: key = (bits << 32) | code.
const std::unordered_map<uint64_t, int>& huffman_map() {
  static const std::unordered_map<uint64_t, int> m = [] {
    std::unordered_map<uint64_t, int> mm;
    for (int sym = 0; sym < 257; ++sym) {
      const auto& hc = kHpackHuffman[sym];
      mm.emplace(((uint64_t)hc.bits << 32) | hc.code, sym);
    }
    return mm;
  }();
  return m;
}

struct BitReader {
  const uint8_t* p;
  const uint8_t* end;

  bool read_int(int prefix_bits, uint64_t& v) {
    if (p >= end) return false;
    uint8_t mask = (uint8_t)((1u << prefix_bits) - 1);
    v = *p++ & mask;
    if (v < mask) return true;
    int shift = 0;
    while (p < end) {
      uint8_t b = *p++;
      v += (uint64_t)(b & 0x7f) << shift;
      if (!(b & 0x80)) return true;
      shift += 7;
      if (shift > 62) return false;
    }
    return false;
  }

  bool read_string(std::string& out) {
    if (p >= end) return false;
    bool huff = (*p & 0x80) != 0;
    uint64_t len;
    if (!read_int(7, len)) return false;
    if ((uint64_t)(end - p) < len) return false;
    std::string_view raw((const char*)p, len);
    p += len;
    if (!huff) {
      out.assign(raw);
      return true;
    }
    return hpack_huffman_decode(raw, out);
  }
};

}  // namespace

bool hpack_huffman_decode(std::string_view in, std::string& out) {
  const auto& map = huffman_map();
  out.clear();
  uint32_t code = 0;
  int bits = 0;
  for (unsigned char byte : in) {
    for (int i = 7; i >= 0; --i) {
      code = (code << 1) | ((byte >> i) & 1);
      ++bits;
      auto it = map.find(((uint64_t)bits << 32) | code);
      if (it != map.end()) {
        if (it->second == 256) return false;  // EOS in data is an error
        out.push_back((char)it->second);
        code = 0;
        bits = 0;
      } else if (bits > 30) {
        return false;
      }
    }
  }
  // remaining bits must be a prefix of EOS (all ones), < 8 bits
  if (bits >= 8) return false;
  if (code != (uint32_t)((1u << bits) - 1)) return false;
  return true;
}

bool HpackDecoder::lookup(uint64_t index, Header& h) const {
  if (index == 0) return false;
  if (index <= kStaticCount) {
    h = kStatic[index];
    return true;
  }
  size_t di = index - kStaticCount - 1;
  if (di >= dynamic_.size()) return false;
  h = dynamic_[di];
  return true;
}

void HpackDecoder::add_dynamic(const std::string& name,
                               const std::string& value) {
  dynamic_.emplace_front(name, value);
  dyn_size_ += name.size() + value.size() + 32;
  evict();
}

void HpackDecoder::evict() {
  while (dyn_size_ > max_size_ && !dynamic_.empty()) {
    dyn_size_ -= dynamic_.back().first.size() + dynamic_.back().second.size() + 32;
    dynamic_.pop_back();
  }
}

bool HpackDecoder::decode(std::string_view block, std::vector<Header>& out) {
  // header-list size guard (HTTP/2 SETTINGS_MAX_HEADER_LIST_SIZE analog):
  // a malicious peer must not expand a small block into unbounded memory
  constexpr size_t kMaxHeaderListBytes = 1 << 20;
  size_t total = 0;
  BitReader r{(const uint8_t*)block.data(),
              (const uint8_t*)block.data() + block.size()};
  while (r.p < r.end) {
    if (total > kMaxHeaderListBytes) return false;
    uint8_t b = *r.p;
    if (b & 0x80) {  // indexed header field
      uint64_t idx;
      if (!r.read_int(7, idx)) return false;
      Header h;
      if (!lookup(idx, h)) return false;
      total += h.first.size() + h.second.size() + 32;
      out.push_back(h);
    } else if (b & 0x40) {  // literal with incremental indexing
      uint64_t idx;
      if (!r.read_int(6, idx)) return false;
      Header h;
      if (idx) {
        if (!lookup(idx, h)) return false;
      } else if (!r.read_string(h.first)) {
        return false;
      }
      if (!r.read_string(h.second)) return false;
      add_dynamic(h.first, h.second);
      total += h.first.size() + h.second.size() + 32;
      out.push_back(h);
    } else if (b & 0x20) {  // dynamic table size update
      uint64_t sz;
      if (!r.read_int(5, sz)) return false;
      if (sz > max_size_limit_) return false;
      max_size_ = sz;
      evict();
    } else {  // literal without indexing (0x00) / never indexed (0x10)
      uint64_t idx;
      if (!r.read_int(4, idx)) return false;
      Header h;
      if (idx) {
        if (!lookup(idx, h)) return false;
      } else if (!r.read_string(h.first)) {
        return false;
      }
      if (!r.read_string(h.second)) return false;
      total += h.first.size() + h.second.size() + 32;
      out.push_back(h);
    }
  }
  return true;
}

namespace {

void encode_int(std::string& out, uint64_t v, int prefix_bits,
                uint8_t pattern) {
  uint8_t mask = (uint8_t)((1u << prefix_bits) - 1);
  if (v < mask) {
    out.push_back((char)(pattern | v));
    return;
  }
  out.push_back((char)(pattern | mask));
  v -= mask;
  while (v >= 0x80) {
    out.push_back((char)((v & 0x7f) | 0x80));
    v >>= 7;
  }
  out.push_back((char)v);
}

void encode_literal_string(std::string& out, std::string_view s) {
  encode_int(out, s.size(), 7, 0x00);  // H=0 (no huffman)
  out.append(s.data(), s.size());
}

}  // namespace

void HpackEncoder::encode(const std::vector<Header>& headers,
                          std::string& out) {
  for (const auto& [name, value] : headers) {
    // exact static-table match -> indexed field
    size_t name_idx = 0;
    bool emitted = false;
    for (size_t i = 1; i <= kStaticCount; ++i) {
      if (kStatic[i].first == name) {
        if (name_idx == 0) name_idx = i;
        if (kStatic[i].second == value) {
          encode_int(out, i, 7, 0x80);
          emitted = true;
          break;
        }
      }
    }
    if (emitted) continue;
    // literal without indexing; name by index if the static table has it
    encode_int(out, name_idx, 4, 0x00);
    if (name_idx == 0) encode_literal_string(out, name);
    encode_literal_string(out, value);
  }
}

}  // namespace k3samd
