# Build/test entry points (reference Makefile:1-13 analogue).
TAG ?= elastic-gpu-scheduler-amd:latest

.PHONY: native test test-gpu bench e2e image clean

native:
	python3 build_native.py

test: native
	python3 -m pytest tests/ -q -m "not gpu"

test-gpu: native
	python3 -m pytest tests/ -q -m gpu

bench: native
	python3 bench.py --steps 20 --warmup 3

# Wire-strict end-to-end: mTLS control plane + native HTTPS + the
# real-wire config-1 measurement (apiserver as its own process).
e2e: native
	python3 -m pytest tests/test_strict_apiserver_e2e.py tests/test_native_tls.py -q
	python3 benchmarks/e2e_real_wire.py --steps 2 --warmup 1 --batch 50

image:
	docker build -t $(TAG) .

clean:
	rm -f elastic_gpu_scheduler_amd/*.so
	find . -name __pycache__ -type d -exec rm -rf {} +

# Race-detection harness: the allocator core under ThreadSanitizer
# (the reference ships no race detection at all — SURVEY.md §5).
tsan-stress:
	g++ -O1 -g -std=c++17 -fsanitize=thread -pthread \
	  -Ielastic_gpu_scheduler_amd/csrc/core \
	  elastic_gpu_scheduler_amd/csrc/stress/stress_main.cc \
	  -o build_tsan_stress && ./build_tsan_stress
