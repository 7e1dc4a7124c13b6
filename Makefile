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

lint:
	python -m compileall -q glt_amd tools bench.py __graft_entry__.py
	python tools/check_imports.py
