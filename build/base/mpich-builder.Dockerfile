# Builder for MPICH workloads (reference build/base/mpich-builder.Dockerfile role).
FROM ubuntu:22.04 AS builder

RUN apt-get update \
    && apt-get install -y --no-install-recommends g++ make libmpich-dev \
    && rm -rf /var/lib/apt/lists/*
