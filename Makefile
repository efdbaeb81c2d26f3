# Convenience targets (reference Makefile analog).
.PHONY: build test test-gpu bench lint clean

build:
	python setup.py build_ext --inplace

test: build
	sh test.sh

test-gpu: build
	sh test_gpu.sh

bench: build
	python bench.py

clean:
	rm -rf build fiber_amd/*.so
