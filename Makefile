# Convenience driver (parity with the reference's Makefile: build / test /
# debug-sanitizer targets; CMake+Ninja is the primary build system).
.PHONY: all test gpu-test asan bench profile clean

all:
	python3 __graft_entry__.py build

test: all
	python3 -m pytest tests/ -x -q -m "not gpu"

gpu-test: all
	python3 -m pytest tests/ -x -q -m gpu

asan:
	bash ci/asan_test.sh

bench: all
	python3 bench.py --steps 3 --warmup 1

profile: all
	bash tools/profile.sh trace

clean:
	rm -rf build build-asan
