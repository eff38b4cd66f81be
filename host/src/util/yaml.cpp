// Block-style YAML subset parser -> hs::Json.
//
// Covers the reference's config dialect (config/*.yaml in the reference
// tree): indentation-nested maps, "- " sequences, quoted/plain scalars,
// '#' comments, ${ENV_VAR} expansion, booleans/numbers/null inference.
// Anchors, multi-line scalars and flow collections beyond [] / {} one-line
// forms are intentionally out of scope.
#include "yaml.h"

#include <cstdlib>

namespace hs {

namespace {

struct Line {
  int indent;
  std::string text;  // content without indent / comment
};

bool is_blank_or_comment(const std::string& s) {
  for (char c : s) {
    if (c == '#') return true;
    if (c != ' ' && c != '\t' && c != '\r') return false;
  }
  return true;
}

std::string strip_comment(const std::string& s) {
  // comment = '#' not inside quotes
  bool sq = false, dq = false;
  for (size_t i = 0; i < s.size(); ++i) {
    char c = s[i];
    if (c == '\'' && !dq) sq = !sq;
    else if (c == '"' && !sq) dq = !dq;
    else if (c == '#' && !sq && !dq && (i == 0 || s[i - 1] == ' '))
      return s.substr(0, i);
  }
  return s;
}

std::string trim(const std::string& s) {
  size_t b = s.find_first_not_of(" \t\r");
  if (b == std::string::npos) return "";
  size_t e = s.find_last_not_of(" \t\r");
  return s.substr(b, e - b + 1);
}

std::string expand_env(const std::string& s) {
  std::string out;
  for (size_t i = 0; i < s.size();) {
    if (s[i] == '$' && i + 1 < s.size() && s[i + 1] == '{') {
      size_t close = s.find('}', i + 2);
      if (close != std::string::npos) {
        std::string name = s.substr(i + 2, close - i - 2);
        std::string dflt;
        size_t dash = name.find(":-");
        if (dash != std::string::npos) {
          dflt = name.substr(dash + 2);
          name = name.substr(0, dash);
        }
        const char* v = getenv(name.c_str());
        out += v ? v : dflt;
        i = close + 1;
        continue;
      }
    }
    out += s[i++];
  }
  return out;
}

Json scalar(const std::string& raw) {
  std::string s = trim(raw);
  if (s.empty() || s == "~" || s == "null") return Json(nullptr);
  if (s.size() >= 2 && (s.front() == '"' || s.front() == '\'') &&
      s.back() == s.front())
    return Json(expand_env(s.substr(1, s.size() - 2)));
  if (s == "true" || s == "True") return Json(true);
  if (s == "false" || s == "False") return Json(false);
  // flow collections (one line)
  if (s.front() == '[' || s.front() == '{') {
    try {
      // YAML flow syntax is close enough to JSON for config use once
      // unquoted tokens are quoted; try JSON parse first
      return Json::parse(s);
    } catch (...) {
      if (s.front() == '[' && s.back() == ']') {
        JsonArray a;
        std::string body = s.substr(1, s.size() - 2);
        size_t start = 0;
        while (start < body.size()) {
          size_t comma = body.find(',', start);
          std::string item = body.substr(
              start, comma == std::string::npos ? std::string::npos
                                                : comma - start);
          if (!trim(item).empty()) a.push_back(scalar(item));
          if (comma == std::string::npos) break;
          start = comma + 1;
        }
        return Json(std::move(a));
      }
      return Json(expand_env(s));
    }
  }
  char* e = nullptr;
  std::string es = expand_env(s);
  double d = strtod(es.c_str(), &e);
  if (e && *e == '\0' && e != es.c_str()) return Json(d);
  return Json(es);
}

struct YamlParser {
  std::vector<Line> lines;
  size_t pos = 0;

  Json parse_block(int indent) {
    if (pos >= lines.size()) return Json(nullptr);
    const Line& first = lines[pos];
    if (first.indent < indent) return Json(nullptr);
    if (first.text.rfind("- ", 0) == 0 || first.text == "-")
      return parse_seq(first.indent);
    return parse_map(first.indent);
  }

  Json parse_seq(int indent) {
    JsonArray arr;
    while (pos < lines.size()) {
      Line& ln = lines[pos];
      if (ln.indent != indent || !(ln.text.rfind("- ", 0) == 0 ||
                                   ln.text == "-"))
        break;
      std::string rest = ln.text == "-" ? "" : trim(ln.text.substr(2));
      if (rest.empty()) {
        ++pos;
        arr.push_back(parse_block(indent + 1));
      } else if (rest.find(':') != std::string::npos &&
                 !looks_scalar_with_colon(rest)) {
        // inline first key of a nested map: rewrite as a map line
        lines[pos].indent = indent + 2;
        lines[pos].text = rest;
        arr.push_back(parse_map(indent + 2));
      } else {
        ++pos;
        arr.push_back(scalar(rest));
      }
    }
    return Json(std::move(arr));
  }

  static bool looks_scalar_with_colon(const std::string& s) {
    // "key: value" vs scalar like "127.0.0.1:8087" — YAML requires a space
    // after ':' for mappings
    size_t c = s.find(':');
    while (c != std::string::npos) {
      if (c + 1 == s.size() || s[c + 1] == ' ') return false;
      c = s.find(':', c + 1);
    }
    return true;
  }

  Json parse_map(int indent) {
    JsonObject obj;
    while (pos < lines.size()) {
      const Line& ln = lines[pos];
      if (ln.indent != indent) break;
      if (ln.text.rfind("- ", 0) == 0) break;
      size_t colon = find_key_colon(ln.text);
      if (colon == std::string::npos)
        throw std::runtime_error("yaml: expected 'key:' at line '" +
                                 ln.text + "'");
      std::string key = trim(ln.text.substr(0, colon));
      if (key.size() >= 2 && (key.front() == '"' || key.front() == '\''))
        key = key.substr(1, key.size() - 2);
      std::string rest = trim(ln.text.substr(colon + 1));
      ++pos;
      if (rest.empty()) {
        if (pos < lines.size() && lines[pos].indent > indent)
          obj[key] = parse_block(lines[pos].indent);
        else
          obj[key] = Json(nullptr);
      } else {
        obj[key] = scalar(rest);
      }
    }
    return Json(std::move(obj));
  }

  static size_t find_key_colon(const std::string& s) {
    bool sq = false, dq = false;
    for (size_t i = 0; i < s.size(); ++i) {
      char c = s[i];
      if (c == '\'' && !dq) sq = !sq;
      else if (c == '"' && !sq) dq = !dq;
      else if (c == ':' && !sq && !dq &&
               (i + 1 == s.size() || s[i + 1] == ' '))
        return i;
    }
    return std::string::npos;
  }
};

}  // namespace

Json yaml_parse(const std::string& text) {
  YamlParser yp;
  size_t start = 0;
  while (start <= text.size()) {
    size_t nl = text.find('\n', start);
    std::string raw = text.substr(
        start, nl == std::string::npos ? std::string::npos : nl - start);
    if (!is_blank_or_comment(raw)) {
      std::string noc = strip_comment(raw);
      size_t ind = noc.find_first_not_of(' ');
      yp.lines.push_back({(int)ind, trim(noc)});
    }
    if (nl == std::string::npos) break;
    start = nl + 1;
  }
  if (yp.lines.empty()) return Json::object();
  Json j = yp.parse_block(yp.lines[0].indent);
  if (yp.pos != yp.lines.size())
    throw std::runtime_error("yaml: trailing content at '" +
                             yp.lines[yp.pos].text + "'");
  return j;
}

}  // namespace hs
