PY ?= python

.PHONY: native test test-gpu bench image clean

native:
	$(PY) -m elastic_gpu_agent_amd.native.build

test: native
	$(PY) -m pytest tests/ -x -q -m "not gpu"

test-gpu: native
	$(PY) -m pytest tests/ -q -m gpu

bench: native
	$(PY) bench.py --steps 20 --warmup 5

image:
	docker build -t elastic-gpu-agent-amd:latest .

clean:
	rm -f elastic_gpu_agent_amd/*.so bin/egpu-hook
	find . -name __pycache__ -type d -exec rm -rf {} +
