# mpioperator-amd/intel — Intel MPI flavor of the worker image (mpiexec +
# I_MPI_HYDRA_HOST_FILE; "host:N" hostfile dialect; reference
# build/base/intel.Dockerfile role). Mirrors the reference's oneAPI apt
# install; requires network access to apt.repos.intel.com at build time.
ARG BASE_LABEL=latest
FROM mpioperator-amd/base:${BASE_LABEL}

RUN apt-get update \
    && apt-get install -y --no-install-recommends gnupg2 ca-certificates apt-transport-https wget \
    && wget -qO /tmp/key.PUB https://apt.repos.intel.com/intel-gpg-keys/GPG-PUB-KEY-INTEL-SW-PRODUCTS.PUB \
    && gpg --dearmor -o /usr/share/keyrings/oneapi-archive-keyring.gpg /tmp/key.PUB \
    && rm /tmp/key.PUB \
    && echo "deb [signed-by=/usr/share/keyrings/oneapi-archive-keyring.gpg trusted=yes] https://apt.repos.intel.com/oneapi all main" > /etc/apt/sources.list.d/oneAPI.list \
    && apt-get update \
    && apt-get install -y --no-install-recommends intel-oneapi-mpi-2021.13 \
    && apt-get remove -y gnupg2 apt-transport-https wget \
    && apt-get autoremove -y \
    && rm -rf /var/lib/apt/lists/*
