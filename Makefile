# Build/test entry points (the reference Makefile's roles, re-targeted).
PYTHON ?= python3
export PYTORCH_ROCM_ARCH ?= gfx950

.PHONY: all build test test_gpu bench manifest wheel clean

all: build

build:            ## compile the gfx950 HIP extension in-tree
	$(PYTHON) -c "import __graft_entry__ as g; g.build()"

test:             ## CPU test tiers (unit + integration + examples)
	$(PYTHON) -m pytest tests/ -x -q -m "not gpu"

test_gpu:         ## GPU tiers (numerics vs fp32 reference, models, bench path)
	$(PYTHON) -m pytest tests/ -x -q -m gpu

bench:            ## 1-GPU flagship benchmark
	$(PYTHON) bench.py --gpus 1 --steps 30 --warmup 10

manifest:         ## regenerate the CRD + deploy/v2beta1/mpi-operator.yaml
	$(PYTHON) hack/gen_crd.py
	hack/generate-manifest.sh

wheel:            ## build the installable wheel (amdrun image ingredient)
	$(PYTHON) -m pip wheel --no-deps --no-build-isolation -w dist .

clean:
	rm -rf build dist *.egg-info mpi_operator_amd/ops/*.so mpi_operator_amd/ops/csrc/*.o
