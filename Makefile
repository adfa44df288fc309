# Convenience wrapper around the CMake build (reference: makefile).
BUILD_DIR := build

.PHONY: all build test unit integration bench clean

all: build

build:
	cmake -S . -B $(BUILD_DIR) -G Ninja -DCMAKE_BUILD_TYPE=Release
	ninja -C $(BUILD_DIR)

unit: build
	./bin/cpilot_unittests

integration: build
	python3 -m pytest tests/ -q -m "not gpu"

test: unit integration

bench: build
	python3 bench.py --steps 30 --warmup 5

clean:
	rm -rf $(BUILD_DIR) bin containerpilot_amd/_native.so

# Sanitizer builds (evidence in profiles/sanitizers.md)
# sanitizer binaries stay inside their build dirs (never bin/: that
# ships to the GPU box and must only hold release builds)
tsan:
	cmake -S . -B build-tsan -G Ninja -DCMAKE_BUILD_TYPE=RelWithDebInfo \
	  -DCPILOT_BIN_IN_TREE=OFF \
	  -DCMAKE_CXX_FLAGS="-fsanitize=thread -g -O1"
	ninja -C build-tsan cpilot_unittests containerpilot cpilot-spawn-helper
	CPILOT_SPAWN_HELPER=build-tsan/bin/cpilot-spawn-helper \
	  ./build-tsan/bin/cpilot_unittests

asan:
	cmake -S . -B build-asan -G Ninja -DCMAKE_BUILD_TYPE=RelWithDebInfo \
	  -DCPILOT_BIN_IN_TREE=OFF \
	  -DCMAKE_CXX_FLAGS="-fsanitize=address,undefined -g -O1"
	ninja -C build-asan cpilot_unittests containerpilot cpilot-spawn-helper
	CPILOT_SPAWN_HELPER=build-asan/bin/cpilot-spawn-helper \
	  ./build-asan/bin/cpilot_unittests

soak: build
	python3 scripts/soak.py 300

capacity: build
	python3 scripts/capacity.py

VERSION := 3.10.0-amd
release: build
	mkdir -p release
	tar -czf release/containerpilot-$(VERSION).tar.gz -C bin containerpilot cpilot-spawn-helper
	cd release && sha1sum containerpilot-$(VERSION).tar.gz > containerpilot-$(VERSION).tar.gz.sha1
