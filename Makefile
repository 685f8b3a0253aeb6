# Reference parity (Makefile:2-3 ran `mpirun -n 2 py.test -s`); here the
# 2-process world runs inside pytest via torch.multiprocessing + gloo.
.PHONY: build test test-gpu bench
build:
	PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
test:
	python -m pytest tests -q -m "not gpu"
test-gpu:
	python -m pytest tests -q -m gpu
bench:
	python bench.py
