#pragma once

#include "json.h"

namespace hs {

// Parse a block-style YAML document (the reference config dialect) into a
// Json value.  ${ENV} / ${ENV:-default} are expanded in scalars.
Json yaml_parse(const std::string& text);

}  // namespace hs
