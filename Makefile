# Convenience wrapper; the real build lives in native/Makefile.
all: native

native:
	$(MAKE) -C native all

tsan:
	$(MAKE) -C native tsan

test:
	python -m pytest tests -q -m "not gpu"

test-gpu:
	python -m pytest tests -q -m gpu

bench:
	python bench.py --gpus 1 --steps 50 --warmup 10

clean:
	$(MAKE) -C native clean

.PHONY: all native tsan test test-gpu bench clean
