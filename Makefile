# horovod_amd — common targets
PY ?= python

.PHONY: build test test-gpu bench clean

build:
	$(PY) build.py

test: build
	$(PY) -m pytest tests/ -q -m "not gpu"

test-gpu: build
	$(PY) -m pytest tests/ -q -m gpu

bench: build
	$(PY) bench.py --steps 20 --warmup 10

clean:
	rm -rf build horovod_amd/_core.so
