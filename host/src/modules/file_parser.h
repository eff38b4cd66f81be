// file-parser — document -> text/markdown extraction; the reference's
// canonical plugin-host business module (modules/file-parser: backend
// trait FileParserBackend src/domain/parser.rs:9, embedded parsers under
// src/infra/parsers/, REST /file-parser/v1/{info,parse-local,upload}
// [+ /markdown]).
#pragma once

#include "../modkit/modkit.h"

namespace hs {

// the backend trait: one embedded parser per format family
struct FileParserBackend {
  virtual ~FileParserBackend() = default;
  virtual std::string id() const = 0;
  virtual bool can_parse(const std::string& ext) const = 0;
  virtual std::string parse_text(const std::string& bytes) const = 0;
  virtual std::string parse_markdown(const std::string& bytes) const = 0;
};

// docx/xlsx/pptx/pdf backends (file_parser_docs.cpp)
void add_document_backends(
    std::vector<std::unique_ptr<FileParserBackend>>& backends);

class FileParserModule : public Module {
 public:
  std::string name() const override { return "file-parser"; }
  void init(ModuleCtx& ctx) override;
  void register_rest(ModuleCtx& ctx, RestRegistry& rest) override;

 private:
  const FileParserBackend* backend_for(const std::string& ext) const;
  void parse_and_respond(const std::string& filename,
                         const std::string& bytes, bool markdown,
                         ResponseWriter& w);
  std::vector<std::unique_ptr<FileParserBackend>> backends_;
  std::vector<std::string> allowed_roots_;   // parse-local path allow-list
  // extension -> OoP module name (config remote_backends[]; requests for
  // these extensions forward to the directory-resolved child process)
  std::map<std::string, std::string> remote_ext_;
  ClientHub* hub_ = nullptr;
};

}  // namespace hs
