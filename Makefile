# Convenience targets (the repo runs in-tree by design: the built
# extension lives inside the package so it ships with any snapshot).

.PHONY: build test test-gpu bench serve clean

build:
	python csrc/build.py

test:
	python -m pytest tests -q -m "not gpu"

test-gpu:
	python -m pytest tests -q -m gpu

bench:
	python bench.py --gpus 1 --steps 20 --warmup 5

serve:
	python examples/serve.py --demo --continuous

clean:
	rm -rf csrc/.hip_build lstm_tensorspark_amd/_lstm_hip.so \
	       .pytest_cache $(shell find . -name __pycache__ -type d)
