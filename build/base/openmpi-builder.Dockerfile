# Builder for OpenMPI workloads (mpic++ for examples/v2beta1/pi's MPI build;
# reference build/base/openmpi-builder.Dockerfile role).
FROM ubuntu:22.04 AS builder

RUN apt-get update \
    && apt-get install -y --no-install-recommends g++ make libopenmpi-dev \
    && rm -rf /var/lib/apt/lists/*
