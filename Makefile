# Convenience targets; the canonical build entry is __graft_entry__.build().
all:
	python3 __graft_entry__.py build

test:
	python3 -m pytest tests -q -m "not gpu"

test-gpu:
	python3 -m pytest tests -q -m gpu

bench:
	python3 bench.py --gpus 1 --steps 20 --warmup 5

clean:
	rm -rf build harness/build module/shim/build
	find . -name '*.so' -path './rocnrdma_amd/*' -delete

.PHONY: all test test-gpu bench clean
