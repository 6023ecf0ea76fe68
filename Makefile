# d9d_amd build + test targets (mirrors the reference's local/distributed split)

.PHONY: build test test-local test-distributed test-gpu bench

build:
	python -c "import __graft_entry__ as g; g.build()"

test-local:
	python -m pytest tests -q -m "not gpu and not distributed"

test-distributed:
	python -m pytest tests -q -m "distributed and not gpu"

test: 
	python -m pytest tests -q -m "not gpu"

test-gpu:
	python -m pytest tests -q -m gpu

bench:
	python bench.py --steps 8 --warmup 3
