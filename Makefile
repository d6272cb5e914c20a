.PHONY: test test-gpu lint build bench

test:
	python -m pytest tests/ -x -q -m "not gpu"

test-gpu:
	python -m pytest tests/ -x -q -m gpu

lint:
	@command -v ruff >/dev/null 2>&1 && ruff check prime_amd tools tests bench.py \
	  || echo "ruff not installed; running offline lint approximation"
	python -m pytest tests/test_lint.py -q

build:
	python -c "from prime_amd.ops.build import build; build(force=True)"

bench:
	python bench.py --steps 10 --warmup 3
