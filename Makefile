# bee2bee-amd — common targets (see docs/DEPLOYMENT.md for the full story)

PY ?= python

.PHONY: build test test-gpu bench serve doctor clean

build:          ## compile the gfx950 HIP extension in-tree
	PYTORCH_ROCM_ARCH=gfx950 $(PY) setup.py build_ext --inplace

build-debug:    ## assert-laden kernel build (BB_KASSERT traps)
	BEE2BEE_DEBUG_KERNELS=1 PYTORCH_ROCM_ARCH=gfx950 $(PY) setup.py build_ext --inplace

test:           ## CPU suite (no GPU required)
	$(PY) -m pytest tests/ -q -m "not gpu"

test-gpu:       ## kernel numerics + engine + mesh integration on an MI355X
	$(PY) -m pytest tests/ -q -m gpu

bench:          ## flagship decode benchmark (one JSON line)
	$(PY) bench.py --gpus 1 --steps 32 --warmup 8

preflight:      ## scale-run preflight (RCCL self-test, env, rank wiring)
	$(PY) bench.py --preflight --gpus 8

serve:          ## serve a model on the native engine + mesh + API
	$(PY) -m bee2bee_amd serve-native --model llama3-8b --api-port 8000

doctor:         ## environment diagnosis
	$(PY) -m bee2bee_amd doctor

clean:
	rm -rf build bee2bee_amd/ops/*.so .pytest_cache
	find . -name __pycache__ -type d -prune -exec rm -rf {} +
