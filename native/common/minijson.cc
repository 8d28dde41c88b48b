#include "minijson.h"

#include <cctype>
#include <cstdio>
#include <stdexcept>

#include "json_writer.h"  // json_escape

namespace k3samd {

namespace {

struct Parser {
  const char* p;
  const char* end;

  [[noreturn]] void fail(const char* msg) {
    throw std::runtime_error(std::string("minijson: ") + msg);
  }

  void skip_ws() {
    while (p < end && (*p == ' ' || *p == '\t' || *p == '\n' || *p == '\r'))
      ++p;
  }

  char peek() {
    skip_ws();
    if (p >= end) fail("unexpected end");
    return *p;
  }

  void expect(char c) {
    if (peek() != c) fail("unexpected character");
    ++p;
  }

  bool try_consume(char c) {
    skip_ws();
    if (p < end && *p == c) {
      ++p;
      return true;
    }
    return false;
  }

  std::string parse_string_raw() {
    expect('"');
    std::string out;
    while (p < end && *p != '"') {
      char c = *p++;
      if (c == '\\') {
        if (p >= end) fail("bad escape");
        char e = *p++;
        switch (e) {
          case '"': out += '"'; break;
          case '\\': out += '\\'; break;
          case '/': out += '/'; break;
          case 'b': out += '\b'; break;
          case 'f': out += '\f'; break;
          case 'n': out += '\n'; break;
          case 'r': out += '\r'; break;
          case 't': out += '\t'; break;
          case 'u': {
            if (end - p < 4) fail("bad \\u escape");
            unsigned cp = 0;
            for (int i = 0; i < 4; ++i) {
              char h = *p++;
              cp <<= 4;
              if (h >= '0' && h <= '9') cp |= (unsigned)(h - '0');
              else if (h >= 'a' && h <= 'f') cp |= (unsigned)(h - 'a' + 10);
              else if (h >= 'A' && h <= 'F') cp |= (unsigned)(h - 'A' + 10);
              else fail("bad \\u digit");
            }
            // A high surrogate must combine with the following \uDC00..
            // \uDFFF escape into one code point — encoding the halves
            // separately produces CESU-8, which runc's JSON parser can
            // reject (non-BMP chars, e.g. emoji in env values).
            if (cp >= 0xD800 && cp <= 0xDBFF) {
              if (end - p >= 6 && p[0] == '\\' && p[1] == 'u') {
                unsigned lo = 0;
                bool ok = true;
                for (int i = 2; i < 6; ++i) {
                  char h = p[i];
                  lo <<= 4;
                  if (h >= '0' && h <= '9') lo |= (unsigned)(h - '0');
                  else if (h >= 'a' && h <= 'f') lo |= (unsigned)(h - 'a' + 10);
                  else if (h >= 'A' && h <= 'F') lo |= (unsigned)(h - 'A' + 10);
                  else { ok = false; break; }
                }
                if (ok && lo >= 0xDC00 && lo <= 0xDFFF) {
                  p += 6;
                  cp = 0x10000 + ((cp - 0xD800) << 10) + (lo - 0xDC00);
                }
              }
              // unpaired surrogate falls through and is encoded as-is
              // (matches lenient JSON parsers; round-trips our output)
            }
            if (cp < 0x80) {
              out += (char)cp;
            } else if (cp < 0x800) {
              out += (char)(0xC0 | (cp >> 6));
              out += (char)(0x80 | (cp & 0x3F));
            } else if (cp < 0x10000) {
              out += (char)(0xE0 | (cp >> 12));
              out += (char)(0x80 | ((cp >> 6) & 0x3F));
              out += (char)(0x80 | (cp & 0x3F));
            } else {
              out += (char)(0xF0 | (cp >> 18));
              out += (char)(0x80 | ((cp >> 12) & 0x3F));
              out += (char)(0x80 | ((cp >> 6) & 0x3F));
              out += (char)(0x80 | (cp & 0x3F));
            }
            break;
          }
          default: fail("bad escape char");
        }
      } else {
        out += c;
      }
    }
    if (p >= end) fail("unterminated string");
    ++p;  // closing quote
    return out;
  }

  JPtr parse_value() {
    char c = peek();
    if (c == '{') {
      ++p;
      auto j = JValue::make_obj();
      skip_ws();
      if (try_consume('}')) return j;
      for (;;) {
        skip_ws();
        std::string key = parse_string_raw();
        expect(':');
        j->obj.emplace_back(std::move(key), parse_value());
        skip_ws();
        if (try_consume(',')) continue;
        expect('}');
        return j;
      }
    }
    if (c == '[') {
      ++p;
      auto j = JValue::make_arr();
      skip_ws();
      if (try_consume(']')) return j;
      for (;;) {
        j->arr.push_back(parse_value());
        skip_ws();
        if (try_consume(',')) continue;
        expect(']');
        return j;
      }
    }
    if (c == '"') {
      auto j = JValue::make_str(parse_string_raw());
      return j;
    }
    if (c == 't') {
      if (end - p < 4 || std::string(p, 4) != "true") fail("bad literal");
      p += 4;
      return JValue::make_bool(true);
    }
    if (c == 'f') {
      if (end - p < 5 || std::string(p, 5) != "false") fail("bad literal");
      p += 5;
      return JValue::make_bool(false);
    }
    if (c == 'n') {
      if (end - p < 4 || std::string(p, 4) != "null") fail("bad literal");
      p += 4;
      return JValue::make_null();
    }
    // number: keep raw spelling
    const char* start = p;
    if (*p == '-') ++p;
    while (p < end && (std::isdigit((unsigned char)*p) || *p == '.' ||
                       *p == 'e' || *p == 'E' || *p == '+' || *p == '-'))
      ++p;
    if (p == start) fail("unexpected token");
    auto j = std::make_shared<JValue>();
    j->type = JValue::kNumber;
    j->num.assign(start, p);
    return j;
  }
};

void serialize(const JPtr& v, std::string& out, int indent, int depth) {
  auto pad = [&](int d) {
    if (indent > 0) {
      out += '\n';
      out.append((size_t)(indent * d), ' ');
    }
  };
  if (!v) {
    out += "null";
    return;
  }
  switch (v->type) {
    case JValue::kNull: out += "null"; break;
    case JValue::kBool: out += v->b ? "true" : "false"; break;
    case JValue::kNumber: out += v->num; break;
    case JValue::kString:
      out += '"';
      out += json_escape(v->str);
      out += '"';
      break;
    case JValue::kArray: {
      out += '[';
      for (size_t i = 0; i < v->arr.size(); ++i) {
        if (i) out += ',';
        pad(depth + 1);
        serialize(v->arr[i], out, indent, depth + 1);
      }
      if (!v->arr.empty()) pad(depth);
      out += ']';
      break;
    }
    case JValue::kObject: {
      out += '{';
      for (size_t i = 0; i < v->obj.size(); ++i) {
        if (i) out += ',';
        pad(depth + 1);
        out += '"';
        out += json_escape(v->obj[i].first);
        out += "\":";
        if (indent > 0) out += ' ';
        serialize(v->obj[i].second, out, indent, depth + 1);
      }
      if (!v->obj.empty()) pad(depth);
      out += '}';
      break;
    }
  }
}

}  // namespace

JPtr json_parse(const std::string& text) {
  Parser pr{text.data(), text.data() + text.size()};
  JPtr v = pr.parse_value();
  pr.skip_ws();
  if (pr.p != pr.end) throw std::runtime_error("minijson: trailing data");
  return v;
}

std::string json_serialize(const JPtr& v, int indent) {
  std::string out;
  serialize(v, out, indent, 0);
  if (indent > 0) out += '\n';
  return out;
}

}  // namespace k3samd
