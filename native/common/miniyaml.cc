#include "miniyaml.h"

#include <algorithm>
#include <stdexcept>

namespace k3samd {

namespace {

struct Line {
  int indent;
  std::string text;  // content without indentation
};

std::string strip_comment(const std::string& s) {
  bool in_sq = false, in_dq = false;
  for (size_t i = 0; i < s.size(); ++i) {
    char c = s[i];
    if (c == '\'' && !in_dq) in_sq = !in_sq;
    else if (c == '"' && !in_sq) in_dq = !in_dq;
    else if (c == '#' && !in_sq && !in_dq &&
             (i == 0 || s[i - 1] == ' ' || s[i - 1] == '\t'))
      return s.substr(0, i);
  }
  return s;
}

std::string rstrip(std::string s) {
  while (!s.empty() && (s.back() == ' ' || s.back() == '\t' || s.back() == '\r'))
    s.pop_back();
  return s;
}

std::string unquote(std::string s) {
  if (s.size() >= 2 && ((s.front() == '"' && s.back() == '"') ||
                        (s.front() == '\'' && s.back() == '\'')))
    return s.substr(1, s.size() - 2);
  return s;
}

std::vector<Line> to_lines(const std::string& text) {
  std::vector<Line> out;
  size_t pos = 0;
  while (pos <= text.size()) {
    size_t nl = text.find('\n', pos);
    std::string raw = text.substr(pos, nl == std::string::npos ? std::string::npos
                                                               : nl - pos);
    pos = nl == std::string::npos ? text.size() + 1 : nl + 1;
    raw = rstrip(strip_comment(raw));
    if (raw.empty()) continue;
    if (raw == "---") continue;  // document marker
    int ind = 0;
    while (ind < (int)raw.size() && raw[ind] == ' ') ++ind;
    if (raw[ind] == '\t')
      throw std::runtime_error("miniyaml: tabs not supported");
    std::string content = raw.substr(ind);
    out.push_back({ind, content});
  }
  return out;
}

// Parse a block starting at lines[i] whose items share lines[i].indent.
YNode parse_block(const std::vector<Line>& lines, size_t& i);

// Parse the value of a map key / list item given the first inline text and
// the indent threshold for a nested block.
YNode parse_value(const std::string& inline_text,
                  const std::vector<Line>& lines, size_t& i,
                  int parent_indent) {
  if (!inline_text.empty()) {
    YNode n;
    n.type = YNode::kScalar;
    n.scalar = unquote(inline_text);
    return n;
  }
  if (i < lines.size()) {
    bool is_item =
        lines[i].text.rfind("- ", 0) == 0 || lines[i].text == "-";
    // nested block: deeper indent, or a list at the parent key's own indent
    // (both are valid YAML for `key:` followed by `- item`)
    if (lines[i].indent > parent_indent ||
        (is_item && lines[i].indent == parent_indent)) {
      return parse_block(lines, i);
    }
  }
  YNode n;  // empty value
  n.type = YNode::kNull;
  return n;
}

YNode parse_block(const std::vector<Line>& lines, size_t& i) {
  const int indent = lines[i].indent;
  const bool is_list = lines[i].text.rfind("- ", 0) == 0 || lines[i].text == "-";
  YNode node;
  node.type = is_list ? YNode::kList : YNode::kMap;

  // For list items with inline content ("- key: v"), we re-parse the item
  // content as a virtual line at indent+2 plus the following deeper lines.
  std::vector<Line> scratch;

  while (i < lines.size()) {
    if (lines[i].indent < indent) break;
    if (lines[i].indent > indent)
      throw std::runtime_error("miniyaml: bad indentation at '" +
                               lines[i].text + "'");
    const std::string& t = lines[i].text;
    bool line_is_item = t.rfind("- ", 0) == 0 || t == "-";
    // a kind change at the same indent ends this block (e.g. a list written
    // at its parent map's indent, followed by the map's next key)
    if (line_is_item != is_list) break;

    if (is_list) {
      std::string rest = t == "-" ? "" : t.substr(2);
      ++i;
      if (rest.empty()) {
        node.list.push_back(parse_value("", lines, i, indent));
      } else {
        // collect nested lines of this item, then parse the virtual block
        scratch.clear();
        scratch.push_back({indent + 2, rest});
        while (i < lines.size() && lines[i].indent > indent) {
          scratch.push_back(lines[i]);
          ++i;
        }
        size_t j = 0;
        // if the inline rest is a plain scalar with no ':', treat as scalar
        if (scratch.size() == 1 && rest.find(": ") == std::string::npos &&
            rest.back() != ':') {
          YNode s;
          s.type = YNode::kScalar;
          s.scalar = unquote(rest);
          node.list.push_back(s);
        } else {
          node.list.push_back(parse_block(scratch, j));
        }
      }
    } else {
      size_t colon = std::string::npos;
      // find "key:" — first ':' that ends the key (followed by space or EOL)
      for (size_t p = 0; p < t.size(); ++p) {
        if (t[p] == ':' && (p + 1 == t.size() || t[p + 1] == ' ')) {
          colon = p;
          break;
        }
      }
      if (colon == std::string::npos)
        throw std::runtime_error("miniyaml: expected 'key:' in '" + t + "'");
      std::string key = unquote(rstrip(t.substr(0, colon)));
      std::string rest = colon + 1 < t.size() ? t.substr(colon + 2) : "";
      // trim leading spaces of rest
      size_t b = rest.find_first_not_of(' ');
      rest = b == std::string::npos ? "" : rest.substr(b);
      ++i;
      node.map.emplace_back(key, parse_value(rest, lines, i, indent));
    }
  }
  return node;
}

}  // namespace

const YNode* YNode::get_path(const std::string& dotted) const {
  const YNode* cur = this;
  size_t pos = 0;
  while (cur && pos <= dotted.size()) {
    size_t dot = dotted.find('.', pos);
    std::string key = dotted.substr(pos, dot == std::string::npos
                                             ? std::string::npos
                                             : dot - pos);
    cur = cur->get(key);
    if (dot == std::string::npos) break;
    pos = dot + 1;
  }
  return cur;
}

int64_t YNode::as_int(int64_t dflt) const {
  if (type != kScalar) return dflt;
  try {
    return std::stoll(scalar);
  } catch (...) {
    return dflt;
  }
}

bool YNode::as_bool(bool dflt) const {
  if (type != kScalar) return dflt;
  std::string s = scalar;
  std::transform(s.begin(), s.end(), s.begin(), ::tolower);
  if (s == "true" || s == "yes" || s == "on" || s == "1") return true;
  if (s == "false" || s == "no" || s == "off" || s == "0") return false;
  return dflt;
}

YNode yaml_parse(const std::string& text) {
  auto lines = to_lines(text);
  if (lines.empty()) {
    YNode n;
    n.type = YNode::kNull;
    return n;
  }
  size_t i = 0;
  YNode root = parse_block(lines, i);
  if (i != lines.size())
    throw std::runtime_error("miniyaml: trailing content at '" +
                             lines[i].text + "'");
  return root;
}

}  // namespace k3samd
