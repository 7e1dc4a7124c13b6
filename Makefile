# Convenience targets (CI-style; reference keeps shell runners in scripts/)
.PHONY: build test test-gpu bench lint

build:
	python setup.py build_ext --inplace

test:
	python -m pytest tests -q -m "not gpu"

test-gpu:
	python -m pytest tests -q -m gpu

bench:
	python bench.py --gpus 1 --steps 60 --warmup 10
