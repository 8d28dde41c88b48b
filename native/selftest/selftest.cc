// k3samd-selftest — unit checks for the dependency-free protocol internals
// (HPACK incl. RFC 7541 Appendix C vectors, protobuf wire helpers,
// miniyaml, minijson). Run by tests/test_native_selftest.py; exits non-zero
// on the first failure.

#include <cstdio>
#include <cstring>
#include <iostream>
#include <iterator>
#include <string>
#include <vector>

#include "../common/minijson.h"
#include "../common/miniyaml.h"
#include "../deviceplugin/dp_messages.h"
#include "../grpc/hpack.h"
#include "../grpc/proto.h"

#define CHECK(cond)                                                          \
  do {                                                                       \
    if (!(cond)) {                                                           \
      std::fprintf(stderr, "FAIL %s:%d: %s\n", __FILE__, __LINE__, #cond);   \
      return 1;                                                              \
    }                                                                        \
  } while (0)

namespace {

std::string unhex(const char* h) {
  std::string out;
  for (size_t i = 0; h[i] && h[i + 1]; i += 2) {
    auto nib = [](char c) -> int {
      if (c >= '0' && c <= '9') return c - '0';
      if (c >= 'a' && c <= 'f') return c - 'a' + 10;
      return c - 'A' + 10;
    };
    out.push_back((char)((nib(h[i]) << 4) | nib(h[i + 1])));
  }
  return out;
}

}  // namespace

int test_huffman() {
  // RFC 7541 C.4.1 / C.6.1 public vectors
  std::string out;
  CHECK(k3samd::hpack_huffman_decode(unhex("f1e3c2e5f23a6ba0ab90f4ff"), out));
  CHECK(out == "www.example.com");
  CHECK(k3samd::hpack_huffman_decode(unhex("a8eb10649cbf"), out));
  CHECK(out == "no-cache");
  CHECK(k3samd::hpack_huffman_decode(unhex("25a849e95ba97d7f"), out));
  CHECK(out == "custom-key");
  CHECK(k3samd::hpack_huffman_decode(unhex("25a849e95bb8e8b4bf"), out));
  CHECK(out == "custom-value");
  CHECK(k3samd::hpack_huffman_decode(unhex("6402"), out));  // C.6.1 ":status 302"
  CHECK(out == "302");
  // invalid: EOS-in-data must fail (30 one-bits = EOS code)
  CHECK(!k3samd::hpack_huffman_decode(unhex("ffffffff"), out));
  return 0;
}

int test_hpack_decode_rfc_c4() {
  // RFC 7541 C.4: three huffman-coded requests on one connection,
  // exercising incremental indexing + dynamic table references.
  k3samd::HpackDecoder dec;
  std::vector<k3samd::Header> h;

  CHECK(dec.decode(unhex("828684418cf1e3c2e5f23a6ba0ab90f4ff"), h));
  CHECK(h.size() == 4);
  CHECK(h[0].first == ":method" && h[0].second == "GET");
  CHECK(h[1].first == ":scheme" && h[1].second == "http");
  CHECK(h[2].first == ":path" && h[2].second == "/");
  CHECK(h[3].first == ":authority" && h[3].second == "www.example.com");

  h.clear();
  CHECK(dec.decode(unhex("828684be5886a8eb10649cbf"), h));
  CHECK(h.size() == 5);
  CHECK(h[3].first == ":authority" && h[3].second == "www.example.com");
  CHECK(h[4].first == "cache-control" && h[4].second == "no-cache");

  h.clear();
  CHECK(dec.decode(
      unhex("828785bf408825a849e95ba97d7f8925a849e95bb8e8b4bf"), h));
  CHECK(h.size() == 5);
  CHECK(h[1].first == ":scheme" && h[1].second == "https");
  CHECK(h[2].first == ":path" && h[2].second == "/index.html");
  CHECK(h[4].first == "custom-key" && h[4].second == "custom-value");
  return 0;
}

int test_hpack_encoder_roundtrip() {
  std::vector<k3samd::Header> in = {
      {":method", "POST"},
      {":scheme", "http"},
      {":path", "/v1beta1.DevicePlugin/ListAndWatch"},
      {":authority", "localhost"},
      {"content-type", "application/grpc"},
      {"te", "trailers"},
      {"grpc-status", "0"},
      {"x-custom", "value with spaces"},
  };
  std::string block;
  k3samd::HpackEncoder::encode(in, block);
  k3samd::HpackDecoder dec;
  std::vector<k3samd::Header> out;
  CHECK(dec.decode(block, out));
  CHECK(out == in);
  return 0;
}

int test_proto() {
  // varint boundaries
  std::string b;
  k3samd::pb::put_varint(b, 0);
  k3samd::pb::put_varint(b, 127);
  k3samd::pb::put_varint(b, 128);
  k3samd::pb::put_varint(b, 300);
  k3samd::pb::put_varint(b, (uint64_t)1 << 62);
  CHECK((uint8_t)b[0] == 0 && (uint8_t)b[1] == 127);
  CHECK((uint8_t)b[2] == 0x80 && (uint8_t)b[3] == 0x01);

  // RegisterRequest round trip
  k3samd::dp::RegisterRequest rr;
  rr.endpoint = "amd-gpu.sock";
  rr.resource_name = "amd.com/gpu";
  std::string enc = rr.encode();
  k3samd::dp::RegisterRequest back;
  CHECK(k3samd::dp::RegisterRequest::decode(enc, back));
  CHECK(back.version == "v1beta1");
  CHECK(back.endpoint == "amd-gpu.sock");
  CHECK(back.resource_name == "amd.com/gpu");

  // AllocateRequest decode of a hand-built message
  std::string msg;
  {
    std::string cr;
    k3samd::pb::put_string(cr, 1, "id-a");
    k3samd::pb::put_string(cr, 1, "id-b");
    k3samd::pb::put_bytes(msg, 1, cr);
    cr.clear();
    k3samd::pb::put_string(cr, 1, "id-c");
    k3samd::pb::put_bytes(msg, 1, cr);
  }
  std::vector<std::vector<std::string>> reqs;
  CHECK(k3samd::dp::decode_allocate_request(msg, reqs));
  CHECK(reqs.size() == 2 && reqs[0].size() == 2 && reqs[1][0] == "id-c");

  // map entry
  std::string me;
  k3samd::pb::put_map_entry(me, 1, "k", "v");
  k3samd::pb::Reader r(me);
  int f, w;
  uint64_t u;
  std::string_view d;
  CHECK(r.next(f, w, u, d) && f == 1 && w == 2);
  std::string k, v;
  CHECK(k3samd::pb::parse_map_entry(d, k, v) && k == "k" && v == "v");

  // truncated input must fail, not crash
  std::string bad = enc.substr(0, enc.size() - 3);
  k3samd::dp::RegisterRequest rb;
  k3samd::dp::RegisterRequest::decode(bad, rb);  // ok() false or partial
  return 0;
}

int test_miniyaml() {
  auto root = k3samd::yaml_parse(
      "version: v1            # comment\n"
      "flags:\n"
      "  migStrategy: none\n"
      "sharing:\n"
      "  timeSlicing:\n"
      "    renameByDefault: false\n"
      "    resources:\n"
      "    - name: amd.com/gpu\n"
      "      replicas: 4\n"
      "    - name: other/res\n"
      "      replicas: 2\n"
      "list2:\n"
      "  - plain\n"
      "  - \"quoted\"\n");
  CHECK(root.get("version")->as_str() == "v1");
  CHECK(root.get_path("flags.migStrategy")->as_str() == "none");
  auto* rs = root.get_path("sharing.timeSlicing.resources");
  CHECK(rs && rs->type == k3samd::YNode::kList && rs->list.size() == 2);
  CHECK(rs->list[0].get("replicas")->as_int() == 4);
  CHECK(rs->list[1].get("name")->as_str() == "other/res");
  auto* l2 = root.get("list2");
  CHECK(l2->list.size() == 2 && l2->list[0].scalar == "plain" &&
        l2->list[1].scalar == "quoted");

  bool threw = false;
  try {
    k3samd::yaml_parse("\tkey: tabs-not-allowed\n");
  } catch (...) {
    threw = true;
  }
  CHECK(threw);
  return 0;
}

int test_miniyaml_edges() {
  // deeper-indented list under a key
  auto a = k3samd::yaml_parse("k:\n    - x\n    - y\n");
  CHECK(a.get("k")->list.size() == 2);
  // empty values next to filled ones
  auto b = k3samd::yaml_parse("a:\nb: v\n");
  CHECK(b.get("a")->type == k3samd::YNode::kNull);
  CHECK(b.get("b")->as_str() == "v");
  // quoted scalar keeps inner colon-space
  auto c = k3samd::yaml_parse("k: \"a: b\"\n");
  CHECK(c.get("k")->as_str() == "a: b");
  // document marker + comment-only lines
  auto d = k3samd::yaml_parse("---\n# only a comment\nk: 1\n");
  CHECK(d.get("k")->as_int() == 1);
  // as_bool variants
  auto e = k3samd::yaml_parse("a: True\nb: OFF\nc: weird\n");
  CHECK(e.get("a")->as_bool(false) == true);
  CHECK(e.get("b")->as_bool(true) == false);
  CHECK(e.get("c")->as_bool(true) == true);  // falls back to default
  return 0;
}

int test_minijson_edges() {
  // deeply nested arrays round-trip
  auto v = k3samd::json_parse("[[[[1,2],[3]],[]],null]");
  CHECK(v->arr.size() == 2);
  CHECK(v->arr[0]->arr[0]->arr[0]->arr[1]->as_int() == 2);
  std::string ser = k3samd::json_serialize(v);
  auto v2 = k3samd::json_parse(ser);
  CHECK(v2->arr[0]->arr[0]->arr[1]->arr[0]->as_int() == 3);
  // duplicate keys are preserved in order; get() returns the first
  auto d = k3samd::json_parse("{\"k\": 1, \"k\": 2}");
  CHECK(d->get("k")->as_int() == 1 && d->obj.size() == 2);
  // pretty serialization parses back
  std::string pretty = k3samd::json_serialize(v, 2);
  CHECK(k3samd::json_parse(pretty)->arr.size() == 2);
  return 0;
}

int test_minijson() {
  const char* text =
      "{\"a\": [1, 2.5, -3e2], \"s\": \"q\\\"uote\\n\", \"n\": null, "
      "\"b\": true, \"o\": {\"x\": 1}}";
  auto v = k3samd::json_parse(text);
  CHECK(v->get("a")->arr.size() == 3);
  CHECK(v->get("a")->arr[1]->num == "2.5");    // raw number preserved
  CHECK(v->get("a")->arr[2]->num == "-3e2");
  CHECK(v->get("s")->str == "q\"uote\n");
  CHECK(v->get("b")->b == true);
  // round trip: parse(serialize(x)) == structurally x
  std::string ser = k3samd::json_serialize(v);
  auto v2 = k3samd::json_parse(ser);
  CHECK(v2->get("a")->arr[2]->num == "-3e2");
  CHECK(v2->get("o")->get("x")->as_int() == 1);
  // \u escape
  auto u = k3samd::json_parse("\"\\u00e9\"");
  CHECK(u->str == "\xc3\xa9");
  // surrogate pair combines into ONE 4-byte UTF-8 code point (U+1F600),
  // not two 3-byte CESU-8 halves runc's parser would reject
  auto emoji = k3samd::json_parse("\"\\ud83d\\ude00\"");
  CHECK(emoji->str == "\xf0\x9f\x98\x80");
  // and the raw UTF-8 round-trips through serialize -> parse untouched
  CHECK(k3samd::json_parse(k3samd::json_serialize(emoji))->str ==
        "\xf0\x9f\x98\x80");
  // unpaired high surrogate stays lenient (no throw)
  CHECK(!k3samd::json_parse("\"\\ud83d x\"")->str.empty());
  bool threw = false;
  try {
    k3samd::json_parse("{\"unterminated\": ");
  } catch (...) {
    threw = true;
  }
  CHECK(threw);
  return 0;
}

// --fuzz <mode>: read stdin, feed it to a parser, exit 0 unless the parser
// crashes/throws-unexpectedly (drives the hypothesis tests in
// tests/test_fuzz.py; parsers must reject, never die, on malformed input).
int fuzz_main(const char* mode) {
  std::string input((std::istreambuf_iterator<char>(std::cin)),
                    std::istreambuf_iterator<char>());
  if (!std::strcmp(mode, "hpack")) {
    k3samd::HpackDecoder dec;
    std::vector<k3samd::Header> out;
    (void)dec.decode(input, out);
    return 0;
  }
  if (!std::strcmp(mode, "hpack-dump")) {
    // correctness oracle for encoder interop fuzzing: input is repeated
    // [u32be len][header block]; all blocks decode through ONE decoder
    // (dynamic-table state persists, as on a real connection); output is
    // one "name\tvalue" line per header, "---" between blocks, "!err" on
    // a decode failure.
    k3samd::HpackDecoder dec;
    size_t pos = 0;
    while (pos + 4 <= input.size()) {
      uint32_t len = ((uint32_t)(uint8_t)input[pos] << 24) |
                     ((uint32_t)(uint8_t)input[pos + 1] << 16) |
                     ((uint32_t)(uint8_t)input[pos + 2] << 8) |
                     (uint8_t)input[pos + 3];
      pos += 4;
      if (pos + len > input.size()) break;
      std::vector<k3samd::Header> out;
      if (!dec.decode(input.substr(pos, len), out)) {
        std::printf("!err\n");
        return 1;
      }
      for (auto& [n, v] : out) std::printf("%s\t%s\n", n.c_str(), v.c_str());
      std::printf("---\n");
      pos += len;
    }
    return 0;
  }
  if (!std::strcmp(mode, "huffman")) {
    std::string out;
    (void)k3samd::hpack_huffman_decode(input, out);
    return 0;
  }
  if (!std::strcmp(mode, "proto")) {
    std::vector<std::vector<std::string>> a;
    (void)k3samd::dp::decode_allocate_request(input, a);
    std::vector<k3samd::dp::PreferredRequest> p;
    (void)k3samd::dp::decode_preferred_request(input, p);
    k3samd::dp::RegisterRequest r;
    (void)k3samd::dp::RegisterRequest::decode(input, r);
    return 0;
  }
  if (!std::strcmp(mode, "yaml")) {
    try {
      (void)k3samd::yaml_parse(input);
    } catch (const std::exception&) {
    }
    return 0;
  }
  if (!std::strcmp(mode, "json")) {
    try {
      (void)k3samd::json_parse(input);
    } catch (const std::exception&) {
    }
    return 0;
  }
  std::fprintf(stderr, "unknown fuzz mode %s\n", mode);
  return 2;
}

int main(int argc, char** argv) {
  if (argc >= 3 && !std::strcmp(argv[1], "--fuzz")) return fuzz_main(argv[2]);
  int rc = 0;
  rc |= test_huffman();
  rc |= test_hpack_decode_rfc_c4();
  rc |= test_hpack_encoder_roundtrip();
  rc |= test_proto();
  rc |= test_miniyaml();
  rc |= test_miniyaml_edges();
  rc |= test_minijson();
  rc |= test_minijson_edges();
  if (rc == 0) std::printf("k3samd-selftest: all checks passed\n");
  return rc;
}
