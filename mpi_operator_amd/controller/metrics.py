"""Prometheus metrics — parity with reference mpi_job_controller.go:125-140
and cmd/mpi-operator/app/server.go:72-77."""
from __future__ import annotations

try:
    from prometheus_client import Counter, Gauge

    jobs_created_total = Counter("mpi_operator_jobs_created_total",
                                 "Counts number of MPI jobs created")
    jobs_successful_total = Counter("mpi_operator_jobs_successful_total",
                                    "Counts number of MPI jobs successful")
    jobs_failed_total = Counter("mpi_operator_jobs_failed_total",
                                "Counts number of MPI jobs failed")
    job_info = Gauge("mpi_operator_job_info", "Information about MPIJob",
                     ["launcher", "namespace"])
    is_leader = Gauge("mpi_operator_is_leader", "Is this client the leader of this mpi-operator client set?")
    HAVE_PROMETHEUS = True
except Exception:  # pragma: no cover - prometheus_client is in the image
    class _Nop:
        def inc(self, *_): pass
        def set(self, *_): pass
        def labels(self, *_, **__): return self

    jobs_created_total = jobs_successful_total = jobs_failed_total = _Nop()
    job_info = is_leader = _Nop()
    HAVE_PROMETHEUS = False
