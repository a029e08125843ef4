# Developer entry points (the driver uses __graft_entry__.py and bench.py).
PY ?= python

.PHONY: build test test-gpu bench serve clean

build:           ## compile the gfx950 HIP extension in-tree
	PYTORCH_ROCM_ARCH=gfx950 $(PY) setup.py build_ext --inplace

test:            ## CPU test suite (runs anywhere)
	$(PY) -m pytest tests/ -q -m "not gpu"

test-gpu:        ## GPU suite (run on an MI355X box)
	$(PY) -m pytest tests/ -q -m gpu

bench:           ## single-GPU headline benchmark
	$(PY) bench.py

serve:           ## synthetic-weights OpenAI server on :3000
	$(PY) -m parallax_amd.cli serve --model deepseek-r1-distill-llama-8b

clean:
	rm -rf build parallax_amd/ops/_C*.so parallax_amd/ops/csrc/*_hip.hip
