// Minimal JSON-Schema validation shared by the REST layer
// (OperationSpec.request_schema enforcement — the runtime analog of the
// reference's typed OperationBuilder request bodies) and
// serverless-runtime io_schema (ADR:131-185).  Subset: type (string or
// list), required, properties, additionalProperties:false, items, enum,
// const, minimum/maximum, minLength/maxLength; `$ref` passes.
#pragma once

#include <string>

#include "../util/json.h"

namespace hs {

// "" when valid; else a human-readable pointer + reason
std::string json_schema_err(const Json& schema, const Json& value,
                            const std::string& path);

}  // namespace hs
