# Convenience targets (reference has Makefile-driven builds; this build
# is driven by build_native.py + pytest).

PY ?= python3

.PHONY: all build test test-gpu bench soak clean image

all: build

build:
	$(PY) build_native.py

test: build
	$(PY) -m pytest tests/ -q -m "not gpu"

test-gpu: build
	$(PY) -m pytest tests/ -q -m gpu

bench: build
	$(PY) bench.py

soak: build
	$(PY) tools/soak.py 240 1

image:
	docker build -f deploy/Dockerfile -t parca-agent-amd .

clean:
	rm -f parca_agent_amd/native/*.so parca_agent_amd/native/perl_offsets.json
