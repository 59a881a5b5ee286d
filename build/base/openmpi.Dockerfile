# mpioperator-amd/openmpi — OpenMPI flavor of the worker image: lets real
# `mpirun` MPIJobs (launcher command mpirun ... ) run against the SAME boot
# plane the controller renders (OMPI_MCA_orte_default_hostfile env +
# "host slots=N" hostfile dialect; reference build/base/openmpi.Dockerfile
# role, controller env block builders.py:250-268). The stack's own amdrun
# remains the default; this image proves the OpenMPI dialect is real.
ARG BASE_LABEL=latest
FROM mpioperator-amd/base:${BASE_LABEL}

RUN apt-get update \
    && apt-get install -y --no-install-recommends openmpi-bin \
    && rm -rf /var/lib/apt/lists/*
