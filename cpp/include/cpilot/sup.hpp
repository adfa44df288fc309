// PID-1 fork/reaper shim (reference: sup/sup.go).
#pragma once

namespace cpilot {

// Fork a worker and reap/forward as PID 1. Returns the worker's exit code
// in the parent; returns -1 in the worker child (caller continues running
// the normal main path).
int supRun(int argc, char** argv);

}  // namespace cpilot
