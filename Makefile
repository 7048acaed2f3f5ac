# Developer convenience targets (the build is plain setuptools + hipcc).

.PHONY: build test test-gpu bench demo clean

build:
	PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

test:
	python -m pytest tests -q -m "not gpu"

test-gpu:
	python -m pytest tests -q -m gpu

bench:
	python bench.py

demo:
	python examples/make_example.py --records 1000 --levels 1 --out /tmp/dblink_demo
	python -m dblink_amd /tmp/dblink_demo/project.conf

clean:
	rm -rf build dblink_amd/__pycache__ dblink_amd/*/__pycache__ tests/__pycache__
