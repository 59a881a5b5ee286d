# mpioperator-amd/mpich — MPICH flavor of the worker image (hydra launcher;
# "host:N" hostfile dialect via HYDRA_HOST_FILE; reference
# build/base/mpich.Dockerfile role, controller env block builders.py).
ARG BASE_LABEL=latest
FROM mpioperator-amd/base:${BASE_LABEL}

RUN apt-get update \
    && apt-get install -y --no-install-recommends mpich \
    && rm -rf /var/lib/apt/lists/*
