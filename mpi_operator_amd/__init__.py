"""mpi_operator_amd — an MI355X-native MPI-job training stack.

A from-scratch rebuild of the capabilities of kubeflow/mpi-operator
(reference: /root/reference) designed MI355X-first:

- ``controller/``: the MPIJob v2beta1 control plane (CRD types, defaulting,
  validation, reconciler, gang-scheduling, operator shell) — behavioral
  parity with the reference's Go controller
  (reference: pkg/controller/mpi_job_controller.go).
- ``runtime/``: the launcher plane — hostfile parsing and an mpirun-style
  SSH/local rank launcher (``amdrun``) replacing OpenMPI's orted boot path
  (reference: build/base/, mpi_job_controller.go:181-215).
- ``parallel/``: the workload data plane — a Horovod-equivalent
  (init/rank/size, DistributedOptimizer with bucketized allreduce overlapped
  with backward, broadcast, elastic re-formation) over RCCL/xGMI via
  torch.distributed (reference: examples/v2beta1/horovod/tensorflow_mnist.py).
- ``ops/``: hand-written CDNA4 (gfx950) HIP kernels for the training hot path
  (implicit-GEMM MFMA conv, fused BN+ReLU, fused SGD-momentum, pooling,
  softmax-CE) (reference: the external tf_cnn_benchmarks/cuDNN images,
  examples/v2beta1/tensorflow-benchmarks/).
- ``models/``: ResNet-50/101, BERT-Large and the MNIST CNN of the Horovod
  example, built on ``ops``.
"""

__version__ = "0.1.0"
