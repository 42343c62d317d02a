PYTHON ?= python

build:
	$(PYTHON) -m flashy_amd.ops.build

tests:
	$(PYTHON) -m pytest tests -q -m "not gpu"

tests_gpu:
	$(PYTHON) -m pytest tests -q -m gpu

bench:
	$(PYTHON) bench.py --steps 50 --warmup 20

linter:
	$(PYTHON) -m flake8 flashy_amd tests examples bench.py || true

docs:
	$(PYTHON) -m pdoc flashy_amd -o docs/  # pdoc not in the offline image; CI target

.PHONY: build tests tests_gpu bench linter docs
