// Real document backends for file-parser: docx / xlsx / pptx (OOXML =
// ZIP + XML) and PDF (FlateDecode content streams), matching the
// reference's embedded parser set (modules/file-parser/src/infra/
// parsers/{docx,xlsx,pptx,pdf}.rs — the rebuild decodes the formats
// directly: a minimal central-directory ZIP reader over zlib inflate,
// a tag-level XML text extractor, and a PDF content-stream scanner).
#include <zlib.h>

#include <algorithm>
#include <cstring>
#include <map>
#include <sstream>

#include "file_parser.h"

namespace hs {

namespace {

// --------------------------------------------------------------- inflate

std::string inflate_raw(const std::string& in, bool raw,
                        size_t max_out = 64u << 20) {
  z_stream st{};
  if (inflateInit2(&st, raw ? -15 : 15 + 32) != Z_OK) return "";
  std::string out;
  out.resize(std::min<size_t>(std::max<size_t>(in.size() * 4, 4096),
                              max_out));
  st.next_in = (Bytef*)in.data();
  st.avail_in = (uInt)in.size();
  size_t off = 0;
  int rc = Z_OK;
  while (rc == Z_OK) {
    if (off == out.size()) {
      if (out.size() >= max_out) break;
      out.resize(std::min(out.size() * 2, max_out));
    }
    st.next_out = (Bytef*)out.data() + off;
    st.avail_out = (uInt)(out.size() - off);
    rc = inflate(&st, Z_NO_FLUSH);
    off = out.size() - st.avail_out;
    if (rc == Z_STREAM_END) break;
    if (rc != Z_OK) { inflateEnd(&st); return ""; }
  }
  inflateEnd(&st);
  out.resize(off);
  return out;
}

// --------------------------------------------------------------- ZIP

uint32_t rd32(const unsigned char* p) {
  return p[0] | (p[1] << 8) | (uint32_t(p[2]) << 16) |
         (uint32_t(p[3]) << 24);
}
uint16_t rd16(const unsigned char* p) { return p[0] | (p[1] << 8); }

// central-directory ZIP extraction (stored + deflate entries only)
std::map<std::string, std::string> zip_extract(const std::string& z) {
  std::map<std::string, std::string> out;
  const unsigned char* d = (const unsigned char*)z.data();
  const size_t n = z.size();
  if (n < 22) return out;
  // find EOCD (0x06054b50) scanning back over the comment
  size_t eocd = std::string::npos;
  const size_t lo = n >= 22 + 65535 ? n - 22 - 65535 : 0;
  for (size_t i = n - 22 + 1; i-- > lo;) {
    if (rd32(d + i) == 0x06054b50) { eocd = i; break; }
  }
  if (eocd == std::string::npos) return out;
  uint16_t count = rd16(d + eocd + 10);
  uint32_t cd_off = rd32(d + eocd + 16);
  size_t p = cd_off;
  for (int i = 0; i < count && p + 46 <= n; ++i) {
    if (rd32(d + p) != 0x02014b50) break;
    const uint16_t method = rd16(d + p + 10);
    const uint32_t csize = rd32(d + p + 20);
    const uint32_t usize = rd32(d + p + 24);
    const uint16_t nlen = rd16(d + p + 28);
    const uint16_t xlen = rd16(d + p + 30);
    const uint16_t clen = rd16(d + p + 32);
    const uint32_t lho = rd32(d + p + 42);
    if (p + 46 + nlen > n) break;
    std::string name((const char*)d + p + 46, nlen);
    p += 46 + nlen + xlen + clen;
    if (lho + 30 > n || rd32(d + lho) != 0x04034b50) continue;
    const uint16_t lnlen = rd16(d + lho + 26);
    const uint16_t lxlen = rd16(d + lho + 28);
    const size_t data_off = lho + 30 + lnlen + lxlen;
    if (data_off + csize > n) continue;
    std::string comp((const char*)d + data_off, csize);
    if (method == 0) {
      out[name] = comp;
    } else if (method == 8) {
      std::string u = inflate_raw(comp, true);
      if (!u.empty() || usize == 0) out[name] = std::move(u);
    }
  }
  return out;
}

// --------------------------------------------------------------- XML

std::string xml_unescape(const std::string& s) {
  std::string out;
  out.reserve(s.size());
  for (size_t i = 0; i < s.size(); ++i) {
    if (s[i] != '&') { out += s[i]; continue; }
    if (s.compare(i, 5, "&amp;") == 0) { out += '&'; i += 4; }
    else if (s.compare(i, 4, "&lt;") == 0) { out += '<'; i += 3; }
    else if (s.compare(i, 4, "&gt;") == 0) { out += '>'; i += 3; }
    else if (s.compare(i, 6, "&quot;") == 0) { out += '"'; i += 5; }
    else if (s.compare(i, 6, "&apos;") == 0) { out += '\''; i += 5; }
    else out += s[i];
  }
  return out;
}

// concatenated text of every <tag ...>text</tag> inside `scope`
std::string xml_text_of(const std::string& scope, const std::string& tag) {
  std::string out;
  const std::string open = "<" + tag;
  const std::string close = "</" + tag + ">";
  size_t p = 0;
  while ((p = scope.find(open, p)) != std::string::npos) {
    const char after = p + open.size() < scope.size()
                           ? scope[p + open.size()] : 0;
    if (after != '>' && after != ' ' && after != '/') { p += open.size(); continue; }
    size_t gt = scope.find('>', p);
    if (gt == std::string::npos) break;
    if (scope[gt - 1] == '/') { p = gt + 1; continue; }   // self-closing
    size_t end = scope.find(close, gt);
    if (end == std::string::npos) break;
    out += xml_unescape(scope.substr(gt + 1, end - gt - 1));
    p = end + close.size();
  }
  return out;
}

// every top-level <tag>...</tag> block (including attributes form)
std::vector<std::string> xml_blocks(const std::string& scope,
                                    const std::string& tag) {
  std::vector<std::string> out;
  const std::string open = "<" + tag;
  const std::string close = "</" + tag + ">";
  size_t p = 0;
  while ((p = scope.find(open, p)) != std::string::npos) {
    const char after = p + open.size() < scope.size()
                           ? scope[p + open.size()] : 0;
    if (after != '>' && after != ' ' && after != '/') { p += open.size(); continue; }
    size_t gt = scope.find('>', p);
    if (gt == std::string::npos) break;
    if (scope[gt - 1] == '/') { p = gt + 1; continue; }
    size_t end = scope.find(close, gt);
    if (end == std::string::npos) break;
    out.push_back(scope.substr(p, end + close.size() - p));
    p = end + close.size();
  }
  return out;
}

// --------------------------------------------------------------- docx

class DocxBackend : public FileParserBackend {
 public:
  std::string id() const override { return "docx"; }
  bool can_parse(const std::string& ext) const override {
    return ext == "docx";
  }
  std::string parse_text(const std::string& bytes) const override {
    return render(bytes, false);
  }
  std::string parse_markdown(const std::string& bytes) const override {
    return render(bytes, true);
  }

 private:
  static std::string render(const std::string& bytes, bool md) {
    auto files = zip_extract(bytes);
    auto it = files.find("word/document.xml");
    if (it == files.end())
      throw Problem::bad_request("not a docx (word/document.xml missing)");
    std::string out;
    for (auto& para : xml_blocks(it->second, "w:p")) {
      std::string text = xml_text_of(para, "w:t");
      if (text.empty()) continue;
      if (md) {
        // heading level from <w:pStyle w:val="HeadingN"/>
        size_t hs = para.find("w:pStyle");
        int level = 0;
        if (hs != std::string::npos) {
          size_t hv = para.find("Heading", hs);
          if (hv != std::string::npos &&
              isdigit((unsigned char)para[hv + 7]))
            level = para[hv + 7] - '0';
        }
        const bool bullet = para.find("<w:numPr>") != std::string::npos;
        if (level > 0)
          out += std::string(std::min(level, 6), '#') + " " + text;
        else if (bullet)
          out += "- " + text;
        else
          out += text;
      } else {
        out += text;
      }
      out += "\n";
      if (md) out += "\n";
    }
    return out;
  }
};

// --------------------------------------------------------------- xlsx

class XlsxBackend : public FileParserBackend {
 public:
  std::string id() const override { return "xlsx"; }
  bool can_parse(const std::string& ext) const override {
    return ext == "xlsx";
  }
  std::string parse_text(const std::string& bytes) const override {
    return render(bytes, false);
  }
  std::string parse_markdown(const std::string& bytes) const override {
    return render(bytes, true);
  }

 private:
  static std::string render(const std::string& bytes, bool md) {
    auto files = zip_extract(bytes);
    std::vector<std::string> shared;
    auto ss = files.find("xl/sharedStrings.xml");
    if (ss != files.end())
      for (auto& si : xml_blocks(ss->second, "si"))
        shared.push_back(xml_text_of(si, "t"));
    std::string out;
    for (int sheet = 1;; ++sheet) {
      auto it = files.find("xl/worksheets/sheet" + std::to_string(sheet) +
                           ".xml");
      if (it == files.end()) {
        if (sheet == 1)
          throw Problem::bad_request("not an xlsx (no worksheets)");
        break;
      }
      if (sheet > 1) out += "\n";
      bool first_row = true;
      for (auto& row : xml_blocks(it->second, "row")) {
        std::vector<std::string> cells;
        for (auto& c : xml_blocks(row, "c")) {
          std::string v = xml_text_of(c, "v");
          if (c.find("t=\"s\"") != std::string::npos) {
            size_t idx = (size_t)atol(v.c_str());
            v = idx < shared.size() ? shared[idx] : "";
          } else if (c.find("t=\"inlineStr\"") != std::string::npos) {
            v = xml_text_of(c, "t");
          }
          cells.push_back(v);
        }
        if (cells.empty()) continue;
        if (md) {
          out += "|";
          for (auto& c : cells) out += " " + c + " |";
          out += "\n";
          if (first_row) {
            out += "|";
            for (size_t i = 0; i < cells.size(); ++i) out += " --- |";
            out += "\n";
          }
        } else {
          for (size_t i = 0; i < cells.size(); ++i)
            out += (i ? "\t" : "") + cells[i];
          out += "\n";
        }
        first_row = false;
      }
    }
    return out;
  }
};

// --------------------------------------------------------------- pptx

class PptxBackend : public FileParserBackend {
 public:
  std::string id() const override { return "pptx"; }
  bool can_parse(const std::string& ext) const override {
    return ext == "pptx";
  }
  std::string parse_text(const std::string& bytes) const override {
    return render(bytes, false);
  }
  std::string parse_markdown(const std::string& bytes) const override {
    return render(bytes, true);
  }

 private:
  static std::string render(const std::string& bytes, bool md) {
    auto files = zip_extract(bytes);
    std::string out;
    bool any = false;
    for (int slide = 1;; ++slide) {
      auto it = files.find("ppt/slides/slide" + std::to_string(slide) +
                           ".xml");
      if (it == files.end()) break;
      any = true;
      if (md)
        out += "## Slide " + std::to_string(slide) + "\n\n";
      for (auto& para : xml_blocks(it->second, "a:p")) {
        std::string text = xml_text_of(para, "a:t");
        if (text.empty()) continue;
        out += (md ? "- " : "") + text + "\n";
      }
      if (md) out += "\n";
    }
    if (!any) throw Problem::bad_request("not a pptx (no slides)");
    return out;
  }
};

// --------------------------------------------------------------- pdf

class PdfBackend : public FileParserBackend {
 public:
  std::string id() const override { return "pdf"; }
  bool can_parse(const std::string& ext) const override {
    return ext == "pdf";
  }
  std::string parse_text(const std::string& bytes) const override {
    return render(bytes);
  }
  std::string parse_markdown(const std::string& bytes) const override {
    return render(bytes);
  }

 private:
  // text from content-stream show operators: (..)Tj, (..)' and
  // [(..) -n (..)]TJ, newline on Td/TD/T*; handles \-escapes + octal
  static void extract_ops(const std::string& cs, std::string* out) {
    for (size_t i = 0; i < cs.size(); ++i) {
      if (cs[i] == '(') {
        std::string lit;
        int depth = 1;
        ++i;
        while (i < cs.size() && depth > 0) {
          char ch = cs[i];
          if (ch == '\\' && i + 1 < cs.size()) {
            char e = cs[++i];
            switch (e) {
              case 'n': lit += '\n'; break;
              case 't': lit += '\t'; break;
              case 'r': lit += '\r'; break;
              case '(': lit += '('; break;
              case ')': lit += ')'; break;
              case '\\': lit += '\\'; break;
              default:
                if (e >= '0' && e <= '7') {
                  int v = e - '0';
                  for (int k = 0; k < 2 && i + 1 < cs.size() &&
                                  cs[i + 1] >= '0' && cs[i + 1] <= '7';
                       ++k)
                    v = v * 8 + (cs[++i] - '0');
                  lit += (char)v;
                } else {
                  lit += e;
                }
            }
          } else if (ch == '(') {
            ++depth;
            lit += ch;
          } else if (ch == ')') {
            if (--depth > 0) lit += ch;
          } else {
            lit += ch;
          }
          ++i;
        }
        --i;
        *out += lit;
      } else if (cs.compare(i, 2, "Td") == 0 ||
                 cs.compare(i, 2, "TD") == 0 ||
                 cs.compare(i, 2, "T*") == 0) {
        if (!out->empty() && out->back() != '\n') *out += '\n';
        ++i;
      }
    }
  }

  static std::string render(const std::string& bytes) {
    if (bytes.compare(0, 5, "%PDF-") != 0)
      throw Problem::bad_request("not a PDF");
    std::string out;
    size_t p = 0;
    while ((p = bytes.find("stream", p)) != std::string::npos) {
      // dict immediately before this stream keyword
      size_t dict0 = bytes.rfind("<<", p);
      const std::string dict =
          dict0 == std::string::npos ? "" : bytes.substr(dict0, p - dict0);
      size_t data0 = p + 6;
      if (data0 < bytes.size() && bytes[data0] == '\r') ++data0;
      if (data0 < bytes.size() && bytes[data0] == '\n') ++data0;
      size_t dend = bytes.find("endstream", data0);
      if (dend == std::string::npos) break;
      size_t dlen = dend - data0;
      while (dlen > 0 && (bytes[data0 + dlen - 1] == '\n' ||
                          bytes[data0 + dlen - 1] == '\r'))
        --dlen;
      std::string data = bytes.substr(data0, dlen);
      if (dict.find("/FlateDecode") != std::string::npos)
        data = inflate_raw(data, false);
      // only content-like streams (text operators present)
      if (data.find("BT") != std::string::npos &&
          data.find("ET") != std::string::npos)
        extract_ops(data, &out);
      p = dend + 9;
    }
    return out;
  }
};

}  // namespace

void add_document_backends(
    std::vector<std::unique_ptr<FileParserBackend>>& backends) {
  backends.push_back(std::make_unique<DocxBackend>());
  backends.push_back(std::make_unique<XlsxBackend>());
  backends.push_back(std::make_unique<PptxBackend>());
  backends.push_back(std::make_unique<PdfBackend>());
}

}  // namespace hs
