# sutro-amd build & test targets

PYTHON ?= python

.PHONY: build test test-gpu bench lint clean

build:
	PYTORCH_ROCM_ARCH=gfx950 $(PYTHON) setup.py build_ext --inplace

test:
	$(PYTHON) -m pytest tests -q -m "not gpu"

test-gpu:
	$(PYTHON) -m pytest tests -q -m gpu

bench:
	$(PYTHON) bench.py --steps 16 --warmup 8

lint:
	$(PYTHON) -m compileall -q sutro_amd tests bench.py setup.py

clean:
	rm -rf build sutro_amd/*.so csrc/*_hip.hip
