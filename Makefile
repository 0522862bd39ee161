# Test/build targets mirroring the reference workflow.
PY ?= python

build:
	$(PY) setup.py build_ext --inplace
	PYTORCH_ROCM_ARCH=gfx950 $(PY) setup_ops.py

unittest:
	$(PY) -m pytest tests/ -q -m "not gpu and not benchmark"

cudatest:
	$(PY) -m pytest tests/ -q -m gpu

benchmark:
	$(PY) -m pytest tests/ -q -m benchmark
	$(PY) bench.py --steps 5 --warmup 2
	$(PY) bench.py --workload impala --steps 10 --warmup 3

algotest:
	$(PY) -m pytest tests/test_policy_breadth.py tests/test_entries.py tests/test_marl.py \
	  tests/test_offline.py tests/test_misc_policies.py tests/test_mbpolicy.py \
	  tests/test_dreamer.py tests/test_diffusion.py -q

.PHONY: build unittest cudatest benchmark algotest
