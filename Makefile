# Convenience targets (the canonical build is setup.py / __graft_entry__.build)
.PHONY: build test gpu-test bench asan soak clean

build:
	python setup.py build_ext --inplace

test: build
	python -m pytest tests -q -m "not gpu"

gpu-test: build
	python -m pytest tests -q -m gpu

bench: build
	python bench.py --steps 10 --warmup 3

asan:
	bash scripts/asan.sh

soak:
	python scripts/soak.py

clean:
	rm -rf build bin splatt_amd/_C*.so splatt_amd/__pycache__ .pytest_cache
