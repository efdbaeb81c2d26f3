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

tsan:  ## ThreadSanitizer run over the transport engine (no GPU needed)
	g++ -std=c++17 -O1 -g -fsanitize=thread -DFAM_NO_PYBIND \
	  fiber_amd/csrc/tsan_harness.cpp -o /tmp/fam_tsan -lpthread -lrt
	/tmp/fam_tsan

asan:  ## Address+UB sanitizer run over the transport engine
	g++ -std=c++17 -O1 -g -fsanitize=address,undefined -DFAM_NO_PYBIND \
	  fiber_amd/csrc/tsan_harness.cpp -o /tmp/fam_asan -lpthread -lrt
	/tmp/fam_asan
