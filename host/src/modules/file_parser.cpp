#include "file_parser.h"

#include <limits.h>
#include <stdlib.h>

#include <fstream>
#include <sstream>

#include "../http/client.h"
#include "system_modules.h"

namespace hs {

namespace {

std::string ext_of(const std::string& name) {
  size_t dot = name.rfind('.');
  if (dot == std::string::npos) return "";
  std::string e = name.substr(dot + 1);
  for (auto& c : e) c = (char)tolower((unsigned char)c);
  return e;
}

// ------------------------------------------------------------ plain text
class PlainBackend : public FileParserBackend {
 public:
  std::string id() const override { return "plain"; }
  bool can_parse(const std::string& e) const override {
    return e == "txt" || e == "md" || e == "markdown" || e == "log" ||
           e == "" || e == "text";
  }
  std::string parse_text(const std::string& b) const override { return b; }
  std::string parse_markdown(const std::string& b) const override {
    return b;
  }
};

// ------------------------------------------------------------------ html
// Tag-stripping extractor with structural markdown mapping for headings,
// list items, paragraphs, line breaks and <a href>.
class HtmlBackend : public FileParserBackend {
 public:
  std::string id() const override { return "html"; }
  bool can_parse(const std::string& e) const override {
    return e == "html" || e == "htm" || e == "xhtml";
  }
  std::string parse_text(const std::string& b) const override {
    return extract(b, false);
  }
  std::string parse_markdown(const std::string& b) const override {
    return extract(b, true);
  }

 private:
  static std::string decode_entities(const std::string& s) {
    std::string out;
    for (size_t i = 0; i < s.size();) {
      if (s[i] == '&') {
        size_t sc = s.find(';', i);
        if (sc != std::string::npos && sc - i <= 6) {
          std::string e = s.substr(i + 1, sc - i - 1);
          const char* rep = nullptr;
          if (e == "amp") rep = "&";
          else if (e == "lt") rep = "<";
          else if (e == "gt") rep = ">";
          else if (e == "quot") rep = "\"";
          else if (e == "apos") rep = "'";
          else if (e == "nbsp") rep = " ";
          if (rep) {
            out += rep;
            i = sc + 1;
            continue;
          }
        }
      }
      out += s[i++];
    }
    return out;
  }

  static std::string extract(const std::string& b, bool md) {
    std::string out;
    size_t i = 0;
    bool skip = false;          // inside <script>/<style>
    std::string href;
    while (i < b.size()) {
      if (b[i] == '<') {
        size_t end = b.find('>', i);
        if (end == std::string::npos) break;
        std::string tag = b.substr(i + 1, end - i - 1);
        std::string low;
        for (char c : tag) low += (char)tolower((unsigned char)c);
        auto starts = [&](const char* p) {
          return low.rfind(p, 0) == 0;
        };
        if (starts("script") || starts("style")) skip = true;
        else if (starts("/script") || starts("/style")) skip = false;
        else if (!skip) {
          if (md && low.size() >= 2 && low[0] == 'h' && low[1] >= '1' &&
              low[1] <= '6' && (low.size() == 2 || low[2] == ' '))
            out += "\n" + std::string((size_t)(low[1] - '0'), '#') + " ";
          else if (starts("/h") && md) out += "\n";
          else if (starts("li")) out += md ? "\n- " : "\n";
          else if (starts("br") || starts("/p") || starts("/div") ||
                   starts("/tr") || starts("/li"))
            out += "\n";
          else if (md && starts("a ")) {
            size_t h = low.find("href=");
            if (h != std::string::npos) {
              char q = tag[h + 5];
              size_t e2 = tag.find(q, h + 6);
              if ((q == '"' || q == '\'') && e2 != std::string::npos)
                href = tag.substr(h + 6, e2 - h - 6);
              out += "[";
            }
          } else if (md && starts("/a") && !href.empty()) {
            out += "](" + href + ")";
            href.clear();
          } else if (md && (low == "b" || starts("b ") ||
                            starts("strong") || low == "/b" ||
                            starts("/strong")))
            out += "**";
          else if (md && (low == "i" || starts("i ") || starts("em") ||
                          low == "/i" || starts("/em")))
            out += "*";
        }
        i = end + 1;
      } else if (skip) {
        ++i;
      } else {
        out += b[i++];
      }
    }
    // collapse whitespace runs but keep newlines
    std::string clean;
    int nl = 0;
    bool sp = false;
    for (char c : decode_entities(out)) {
      if (c == '\n') {
        if (nl < 2) clean += '\n';
        nl++;
        sp = false;
      } else if (isspace((unsigned char)c)) {
        sp = true;
      } else {
        if (sp && !clean.empty() && clean.back() != '\n') clean += ' ';
        sp = false;
        nl = 0;
        clean += c;
      }
    }
    return clean;
  }
};

// ------------------------------------------------------------------- csv
class CsvBackend : public FileParserBackend {
 public:
  std::string id() const override { return "csv"; }
  bool can_parse(const std::string& e) const override {
    return e == "csv" || e == "tsv";
  }
  std::string parse_text(const std::string& b) const override { return b; }
  std::string parse_markdown(const std::string& b) const override {
    std::istringstream in(b);
    std::string line, out;
    int row = 0;
    const char sep = b.find('\t') != std::string::npos ? '\t' : ',';
    while (std::getline(in, line)) {
      if (!line.empty() && line.back() == '\r') line.pop_back();
      std::string cells = "|";
      int ncell = 0;
      std::string cur;
      bool inq = false;
      for (size_t i = 0; i <= line.size(); ++i) {
        char c = i < line.size() ? line[i] : sep;
        if (inq) {
          if (c == '"') inq = false;
          else cur += c;
        } else if (c == '"') inq = true;
        else if (c == sep && i <= line.size()) {
          cells += " " + cur + " |";
          ncell++;
          cur.clear();
        } else cur += c;
      }
      out += cells + "\n";
      if (row++ == 0) {
        out += "|";
        for (int i = 0; i < ncell; ++i) out += "---|";
        out += "\n";
      }
    }
    return out;
  }
};

// ------------------------------------------------------------------ json
class JsonBackend : public FileParserBackend {
 public:
  std::string id() const override { return "json"; }
  bool can_parse(const std::string& e) const override {
    return e == "json" || e == "yaml" || e == "yml";
  }
  std::string parse_text(const std::string& b) const override { return b; }
  std::string parse_markdown(const std::string& b) const override {
    return "```json\n" + b + "\n```\n";
  }
};

// naive multipart/form-data: returns (filename, bytes) of the first part
std::optional<std::pair<std::string, std::string>> parse_multipart(
    const std::string& content_type, const std::string& body) {
  size_t bp = content_type.find("boundary=");
  if (bp == std::string::npos) return std::nullopt;
  std::string boundary = "--" + content_type.substr(bp + 9);
  if (!boundary.empty() && boundary.back() == '"') {
    boundary.pop_back();
    boundary.erase(2, 1);
  }
  size_t start = body.find(boundary);
  if (start == std::string::npos) return std::nullopt;
  start = body.find("\r\n\r\n", start);
  if (start == std::string::npos) return std::nullopt;
  size_t hdr_start = body.find(boundary) + boundary.size();
  std::string headers = body.substr(hdr_start, start - hdr_start);
  std::string filename = "upload.txt";
  size_t fp = headers.find("filename=\"");
  if (fp != std::string::npos) {
    size_t fe = headers.find('"', fp + 10);
    if (fe != std::string::npos)
      filename = headers.substr(fp + 10, fe - fp - 10);
  }
  start += 4;
  size_t end = body.find(boundary, start);
  if (end == std::string::npos) end = body.size();
  else if (end >= 2 && body[end - 2] == '\r') end -= 2;
  return std::make_pair(filename, body.substr(start, end - start));
}

}  // namespace

void FileParserModule::init(ModuleCtx& ctx) {
  hub_ = ctx.hub;
  backends_.push_back(std::make_unique<PlainBackend>());
  backends_.push_back(std::make_unique<HtmlBackend>());
  backends_.push_back(std::make_unique<CsvBackend>());
  backends_.push_back(std::make_unique<JsonBackend>());
  add_document_backends(backends_);
  const Json& roots = ctx.config.at("allowed_roots");
  if (roots.is_array())
    for (auto& r : roots.arr()) allowed_roots_.push_back(r.as_string());
  // remote_backends: [{module: "<oop module name>", extensions: [..]}]
  // (reference pattern: file-parser as the plugin HOST — a parser can
  // live out-of-process and be reached via the directory, like the
  // calculator/calculator-gateway OoP pair)
  const Json& rbs = ctx.config.at("remote_backends");
  if (rbs.is_array())
    for (auto& rb : rbs.arr()) {
      const std::string mod = rb.at("module").as_string();
      if (rb.at("extensions").is_array())
        for (auto& e : rb.at("extensions").arr())
          remote_ext_[e.as_string()] = mod;
    }
}

const FileParserBackend* FileParserModule::backend_for(
    const std::string& ext) const {
  for (auto& b : backends_)
    if (b->can_parse(ext)) return b.get();
  return nullptr;
}

void FileParserModule::parse_and_respond(const std::string& filename,
                                         const std::string& bytes,
                                         bool markdown, ResponseWriter& w) {
  const std::string ext = ext_of(filename);
  // out-of-process backend: resolve the child via the hub
  // DirectoryClient and forward the raw bytes (binary-safe body;
  // filename/format travel as headers)
  auto rit = remote_ext_.find(ext);
  if (rit != remote_ext_.end()) {
    auto dir = hub_ ? hub_->get<DirectoryClient>("module-orchestrator")
                    : nullptr;
    const std::string ep = dir ? dir->resolve(rit->second) : "";
    if (ep.empty())
      throw Problem{503, "Service Unavailable", "about:blank",
                    "no live instance of parser module '" + rit->second +
                        "'", "provider_error"};
    std::string host;
    int port = 80;
    const size_t hs = ep.find("://");
    std::string rest = hs == std::string::npos ? ep : ep.substr(hs + 3);
    const size_t cp = rest.find(':');
    host = rest.substr(0, cp == std::string::npos ? rest.size() : cp);
    if (cp != std::string::npos) port = atoi(rest.c_str() + cp + 1);
    auto r = http_request(host, port, "POST", "/parse",
                          {{"x-filename", filename},
                           {"x-markdown", markdown ? "1" : "0"},
                           {"content-type", "application/octet-stream"}},
                          bytes, 5000);
    if (!r || r->status != 200)
      throw Problem{502, "Bad Gateway", "about:blank",
                    "remote parser failed", "provider_error"};
    w.respond(200, "application/json", r->body);
    return;
  }
  const FileParserBackend* be = backend_for(ext);
  if (!be)
    throw Problem{415, "Unsupported Media Type", "about:blank",
                  "no parser for extension '" + ext + "'",
                  "unsupported_format"};
  Json out = Json::object();
  out["filename"] = filename;
  out["backend"] = be->id();
  out["format"] = markdown ? "markdown" : "text";
  out["content"] =
      markdown ? be->parse_markdown(bytes) : be->parse_text(bytes);
  w.respond(200, "application/json", out.dump());
}

void FileParserModule::register_rest(ModuleCtx& ctx, RestRegistry& rest) {
  {
    OperationSpec op;
    op.method = "GET";
    op.path = "/file-parser/v1/info";
    op.operation_id = "parser_info";
    op.summary = "Available parser backends";
    op.authenticated = true;
    op.tags = {"file-parser"};
    rest.register_op(op, [this](HttpRequest& rq, ResponseWriter& w) {
      Json bs = Json::array();
      for (auto& b : backends_) bs.push_back(b->id());
      Json out = Json::object();
      out["backends"] = bs;
      Json exts = Json::array();
      for (const char* e : {"txt", "md", "html", "htm", "csv", "tsv",
                            "json", "yaml", "docx", "xlsx", "pptx",
                            "pdf"})
        exts.push_back(e);
      out["extensions"] = exts;
      w.respond(200, "application/json", out.dump());
    });
  }
  for (bool md : {false, true}) {
    OperationSpec op;
    op.method = "POST";
    op.path = std::string("/file-parser/v1/parse-local") +
              (md ? "/markdown" : "");
    op.operation_id = std::string("parse_local") + (md ? "_markdown" : "");
    op.summary = "Parse a server-local file";
    op.authenticated = true;
    op.allowed_content_types = {"application/json"};
    op.tags = {"file-parser"};
    rest.register_op(op, [this, md](HttpRequest& rq, ResponseWriter& w) {
      Json body;
      try { body = Json::parse(rq.body); }
      catch (...) { throw Problem::bad_request("invalid JSON body"); }
      const std::string path = body.at("path").as_string();
      if (path.empty())
        throw Problem::bad_request("'path' is required");
      if (path.find("..") != std::string::npos)
        throw Problem::forbidden("path traversal rejected");
      // default-deny: parse-local is only usable when the operator has
      // configured allowed_roots; both the root and the request path are
      // canonicalised (symlinks resolved) and the containment check must
      // end at a path-separator boundary ('/data' must not admit
      // '/database-secrets')
      if (allowed_roots_.empty())
        throw Problem::forbidden(
            "parse-local disabled: no allowed_roots configured");
      char canon_buf[PATH_MAX];
      if (realpath(path.c_str(), canon_buf) == nullptr)
        throw Problem::not_found("cannot open " + path);
      const std::string canon(canon_buf);
      bool ok = false;
      for (auto& r : allowed_roots_) {
        char root_buf[PATH_MAX];
        if (realpath(r.c_str(), root_buf) == nullptr) continue;
        std::string root(root_buf);
        if (canon == root ||
            (canon.size() > root.size() &&
             canon.compare(0, root.size(), root) == 0 &&
             canon[root.size()] == '/'))
          ok = true;
      }
      if (!ok) throw Problem::forbidden("path outside allowed roots");
      std::ifstream f(canon, std::ios::binary);
      if (!f) throw Problem::not_found("cannot open " + path);
      std::stringstream ss;
      ss << f.rdbuf();
      parse_and_respond(path, ss.str(), md, w);
    });
  }
  for (bool md : {false, true}) {
    OperationSpec op;
    op.method = "POST";
    op.path =
        std::string("/file-parser/v1/upload") + (md ? "/markdown" : "");
    op.operation_id = std::string("upload") + (md ? "_markdown" : "");
    op.summary = "Parse an uploaded file (multipart/form-data or raw)";
    op.authenticated = true;
    op.tags = {"file-parser"};
    rest.register_op(op, [this, md](HttpRequest& rq, ResponseWriter& w) {
      const std::string ct = rq.header("content-type");
      std::string filename, bytes;
      if (ct.rfind("multipart/form-data", 0) == 0) {
        auto part = parse_multipart(ct, rq.body);
        if (!part)
          throw Problem::bad_request("malformed multipart body");
        filename = part->first;
        bytes = part->second;
      } else {
        auto it = rq.query.find("filename");
        filename = it != rq.query.end() ? it->second : "upload.txt";
        bytes = rq.body;
      }
      parse_and_respond(filename, bytes, md, w);
    });
  }
}

}  // namespace hs
