# Build/test entry points (reference Makefile:1-13 analogue).
TAG ?= elastic-gpu-scheduler-amd:latest

.PHONY: native test test-gpu bench image clean

native:
	python3 build_native.py

test: native
	python3 -m pytest tests/ -q -m "not gpu"

test-gpu: native
	python3 -m pytest tests/ -q -m gpu

bench: native
	python3 bench.py --steps 20 --warmup 3

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
