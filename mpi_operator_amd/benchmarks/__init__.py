"""In-image benchmark entry points (`python -m mpi_operator_amd.benchmarks.resnet`)
— what the resnet-benchmarks MPIJob example runs under amdrun. The repo-root
bench.py (the driver contract) shares this code path."""
