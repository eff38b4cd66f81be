#include "json.h"

#include <cmath>
#include <cstdio>
#include <cstring>

namespace hs {

namespace {

struct Parser {
  const char* p;
  const char* end;

  [[noreturn]] void fail(const std::string& msg) {
    throw std::runtime_error("json parse error: " + msg);
  }
  void ws() {
    while (p < end && (*p == ' ' || *p == '\t' || *p == '\n' || *p == '\r'))
      ++p;
  }
  char peek() {
    ws();
    if (p >= end) fail("unexpected end");
    return *p;
  }
  void expect(char c) {
    if (peek() != c) fail(std::string("expected '") + c + "'");
    ++p;
  }
  bool consume(char c) {
    if (p < end && peek() == c) { ++p; return true; }
    return false;
  }

  Json value() {
    switch (peek()) {
      case '{': return object();
      case '[': return array();
      case '"': return Json(string());
      case 't': lit("true"); return Json(true);
      case 'f': lit("false"); return Json(false);
      case 'n': lit("null"); return Json(nullptr);
      default: return number();
    }
  }
  void lit(const char* s) {
    size_t n = strlen(s);
    if (size_t(end - p) < n || strncmp(p, s, n) != 0) fail("bad literal");
    p += n;
  }
  Json number() {
    char* e = nullptr;
    double d = strtod(p, &e);
    if (e == p) fail("bad number");
    p = e;
    return Json(d);
  }
  std::string string() {
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
            if (end - p < 4) fail("bad \\u");
            unsigned cp = 0;
            for (int i = 0; i < 4; ++i) {
              char h = *p++;
              cp <<= 4;
              if (h >= '0' && h <= '9') cp |= h - '0';
              else if (h >= 'a' && h <= 'f') cp |= h - 'a' + 10;
              else if (h >= 'A' && h <= 'F') cp |= h - 'A' + 10;
              else fail("bad \\u");
            }
            // utf-8 encode (surrogate pairs folded to replacement)
            if (cp < 0x80) out += char(cp);
            else if (cp < 0x800) {
              out += char(0xC0 | (cp >> 6));
              out += char(0x80 | (cp & 0x3F));
            } else {
              out += char(0xE0 | (cp >> 12));
              out += char(0x80 | ((cp >> 6) & 0x3F));
              out += char(0x80 | (cp & 0x3F));
            }
            break;
          }
          default: fail("bad escape");
        }
      } else {
        out += c;
      }
    }
    expect('"');
    return out;
  }
  Json array() {
    expect('[');
    JsonArray a;
    if (consume(']')) return Json(std::move(a));
    while (true) {
      a.push_back(value());
      if (consume(']')) break;
      expect(',');
    }
    return Json(std::move(a));
  }
  Json object() {
    expect('{');
    JsonObject o;
    if (consume('}')) return Json(std::move(o));
    while (true) {
      std::string k = string();
      expect(':');
      o[k] = value();
      if (consume('}')) break;
      expect(',');
    }
    return Json(std::move(o));
  }
};

void escape_to(std::string& out, const std::string& s) {
  out += '"';
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
          snprintf(buf, sizeof buf, "\\u%04x", c);
          out += buf;
        } else {
          out += c;
        }
    }
  }
  out += '"';
}

void dump_to(std::string& out, const Json& j, int indent, int depth) {
  auto pad = [&](int d) {
    if (indent >= 0) {
      out += '\n';
      out.append(size_t(indent) * d, ' ');
    }
  };
  if (j.is_null()) { out += "null"; return; }
  if (j.is_bool()) { out += j.as_bool() ? "true" : "false"; return; }
  if (j.is_number()) {
    double d = j.as_number();
    if (std::isfinite(d) && d == std::floor(d) && std::fabs(d) < 1e15) {
      char buf[32];
      snprintf(buf, sizeof buf, "%lld", (long long)d);
      out += buf;
    } else {
      char buf[32];
      snprintf(buf, sizeof buf, "%.17g", d);
      out += buf;
    }
    return;
  }
  if (j.is_string()) { escape_to(out, j.as_string()); return; }
  if (j.is_array()) {
    out += '[';
    bool first = true;
    for (const auto& e : j.arr()) {
      if (!first) out += ',';
      first = false;
      pad(depth + 1);
      dump_to(out, e, indent, depth + 1);
    }
    if (!first) pad(depth);
    out += ']';
    return;
  }
  out += '{';
  bool first = true;
  for (const auto& [k, v] : j.obj()) {
    if (!first) out += ',';
    first = false;
    pad(depth + 1);
    escape_to(out, k);
    out += indent >= 0 ? ": " : ":";
    dump_to(out, v, indent, depth + 1);
  }
  if (!first) pad(depth);
  out += '}';
}

}  // namespace

std::string Json::dump(int indent) const {
  std::string out;
  dump_to(out, *this, indent, 0);
  return out;
}

Json Json::parse(const std::string& text) {
  Parser ps{text.data(), text.data() + text.size()};
  Json j = ps.value();
  ps.ws();
  if (ps.p != ps.end) ps.fail("trailing data");
  return j;
}

const Json& Json::path(const std::string& dotted) const {
  static const Json null_json;
  const Json* cur = this;
  size_t start = 0;
  while (start <= dotted.size()) {
    size_t dot = dotted.find('.', start);
    std::string key = dotted.substr(
        start, dot == std::string::npos ? std::string::npos : dot - start);
    if (!cur->is_object()) return null_json;
    auto it = cur->obj().find(key);
    if (it == cur->obj().end()) return null_json;
    cur = &it->second;
    if (dot == std::string::npos) break;
    start = dot + 1;
  }
  return *cur;
}

void Json::merge_from(const Json& other) {
  if (!is_object() || !other.is_object()) {
    *this = other;
    return;
  }
  for (const auto& [k, v] : other.obj()) {
    auto it = obj().find(k);
    if (it != obj().end() && it->second.is_object() && v.is_object())
      it->second.merge_from(v);
    else
      obj()[k] = v;
  }
}

}  // namespace hs
