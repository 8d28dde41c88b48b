// HPACK (RFC 7541) header compression — decoder + minimal encoder.
//
// Decoder is complete (static+dynamic tables, Huffman, size updates) since
// peers (kubelet's grpc-go, test grpcio) use all of it. The encoder emits
// static-table indexed fields where possible and literals-without-indexing
// otherwise (no Huffman, no dynamic table) — always legal HPACK.

#pragma once

#include <cstdint>
#include <deque>
#include <string>
#include <string_view>
#include <utility>
#include <vector>

namespace k3samd {

using Header = std::pair<std::string, std::string>;

class HpackDecoder {
 public:
  HpackDecoder() = default;

  // Decode one complete header block. Returns false on malformed input.
  bool decode(std::string_view block, std::vector<Header>& out);

  void set_max_table_size(size_t n) { max_size_limit_ = n; }

 private:
  bool lookup(uint64_t index, Header& h) const;
  void add_dynamic(const std::string& name, const std::string& value);
  void evict();

  std::deque<Header> dynamic_;  // front = most recent
  size_t dyn_size_ = 0;
  size_t max_size_ = 4096;        // current table size (peer-controlled)
  size_t max_size_limit_ = 65536; // our cap on peer updates
};

class HpackEncoder {
 public:
  // Append the encoding of `h` to `out`.
  static void encode(const std::vector<Header>& headers, std::string& out);
};

// Huffman decode (RFC 7541 App B). Returns false on invalid padding/code.
bool hpack_huffman_decode(std::string_view in, std::string& out);

}  // namespace k3samd
