# mpioperator-amd/amdrun — launcher/worker image with the stack's own boot
# plane (amdrun + per-host agent) and the PyTorch-ROCm + HIP-kernel training
# runtime. Fills the role of the reference's openmpi/intel/mpich flavor
# images: there is exactly ONE flavor here because the data plane is always
# RCCL over xGMI and the boot plane is always ssh+amdrun.
FROM mpioperator-amd/base
# PyTorch-ROCm wheel for gfx950 + the framework itself
RUN python3 -m pip install --no-cache-dir torch --index-url https://download.pytorch.org/whl/rocm7.0
COPY dist/mpi_operator_amd-*.whl /tmp/
RUN python3 -m pip install --no-cache-dir /tmp/mpi_operator_amd-*.whl && rm /tmp/*.whl
# amdrun / amdrun-agent entry points land on PATH via the wheel
