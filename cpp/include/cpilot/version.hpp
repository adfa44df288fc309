// Version stamping (reference: version/version.go + makefile -ldflags).
// Overridable at build time with -DCPILOT_VERSION / -DCPILOT_GITHASH.
#pragma once

namespace cpilot {

#ifndef CPILOT_VERSION
#define CPILOT_VERSION "3.10.0-amd"
#endif
#ifndef CPILOT_GITHASH
#define CPILOT_GITHASH ""
#endif

constexpr const char* kVersion = CPILOT_VERSION;
constexpr const char* kGitHash = CPILOT_GITHASH;

}  // namespace cpilot
