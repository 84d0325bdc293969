// Stand-in for the reference's CMake-generated environment.cc
// (kaminpar-common/environment.cc.in): version strings only.
#include "kaminpar-common/environment.h"
namespace kaminpar {
const std::string_view Environment::GIT_SHA1 = "ref-oracle";
const std::string_view Environment::GIT_MODIFIED_FILES = "";
const std::string_view Environment::HOSTNAME = "container";
} // namespace kaminpar
