# Developer targets (reference Makefile analog).
PYTHON ?= python

.PHONY: build test test-gpu bench soak lint clean

build:
	PYTORCH_ROCM_ARCH=gfx950 $(PYTHON) setup.py build_ext --inplace

test:
	$(PYTHON) -m pytest tests -q -m "not gpu"

test-gpu:
	$(PYTHON) -m pytest tests -q -m gpu

bench:
	$(PYTHON) bench.py --steps 100

soak:
	$(PYTHON) bench.py --steps 500 --warmup 30

lint:
	$(PYTHON) -m compileall -q persia_amd tests examples tools

clean:
	rm -rf build persia_amd/*.so persia_amd/__pycache__
