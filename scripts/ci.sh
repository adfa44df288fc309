#!/bin/sh
# Full local CI: build, native unit tests, pytest suite, benchmark
# sanity. Mirrors the reference's lint -> build -> unit -> integration
# pipeline (.travis.yml:22-26) without docker.
set -e
cd "$(dirname "$0")/.."
./scripts/lint.sh
make build
./bin/cpilot_unittests
python3 -m pytest tests/ -q -m "not gpu"
python3 bench.py --steps 5 --warmup 2 --jobs 20 --watches 5
