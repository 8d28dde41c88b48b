// Shared version reporting for the k3samd CLIs.
#pragma once
#include <cstdio>
#include <cstring>

#ifndef K3SAMD_VERSION
#define K3SAMD_VERSION "dev"
#endif

namespace k3samd {
// Returns true (and prints) when argv contains --version.
inline bool handle_version_flag(int argc, char** argv, const char* tool) {
  for (int i = 1; i < argc; ++i) {
    if (!std::strcmp(argv[i], "--version")) {
      std::printf("%s %s\n", tool, K3SAMD_VERSION);
      return true;
    }
  }
  return false;
}
}  // namespace k3samd
