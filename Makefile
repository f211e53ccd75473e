# Convenience targets (PYTORCH_ROCM_ARCH=gfx950 is set by setup.py)
.PHONY: build test test-gpu bench smoke clean

build:
	python setup.py build_ext --inplace

test:
	python -m pytest tests/ -q -m "not gpu"

test-gpu:
	python -m pytest tests/ -q -m gpu

bench:
	python bench.py --steps 30 --warmup 8

smoke:
	python -c "import __graft_entry__ as g; g.build(); g.smoke()"

clean:
	rm -rf build g2vec_amd/_C*.so
