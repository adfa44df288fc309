#!/bin/bash
# Static-analysis gate (reference parity: makefile:139-141 scripts/lint.sh).
# No clang-tidy/cppcheck exists in this image; the gate uses the Clang
# Static Analyzer (ROCm LLVM) plus a -Werror strict-warnings compile.
set -euo pipefail
cd "$(dirname "$0")/.."

CLANG=/opt/rocm/lib/llvm/bin/clang++
SRCS=(cpp/src/*.cpp)
INC=(-Icpp/include)
STD=(-std=c++17)

fail=0

echo "== clang static analyzer =="
for f in "${SRCS[@]}"; do
  out=$("$CLANG" --analyze "${STD[@]}" "${INC[@]}" \
        --analyzer-output text "$f" -o /dev/null 2>&1 || true)
  if [[ -n "$out" ]]; then
    echo "$out"
    fail=1
  fi
done

echo "== strict warnings (g++ -Werror) =="
for f in "${SRCS[@]}"; do
  if ! g++ -fsyntax-only "${STD[@]}" "${INC[@]}" \
       -Wall -Wextra -Wshadow -Wnon-virtual-dtor \
       -Wno-unused-parameter -Werror "$f" 2>/tmp/lint_warn.$$; then
    cat /tmp/lint_warn.$$
    fail=1
  fi
done
rm -f /tmp/lint_warn.$$

if [[ $fail -ne 0 ]]; then
  echo "LINT FAILED"
  exit 1
fi
echo "lint clean"
