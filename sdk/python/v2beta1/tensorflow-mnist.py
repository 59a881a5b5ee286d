"""SDK usage example — builds the mnist MPIJob with the typed models and
submits it via the kubernetes CustomObjectsApi (the same flow as the
reference's sdk/python/v2beta1/tensorflow-mnist.py, against the stack's
amdrun image instead of a Horovod image)."""
from kubernetes import client, config

from mpijob import (V2beta1MPIJob, V2beta1MPIJobSpec, V2beta1ReplicaSpec,
                    V2beta1RunPolicy)

GROUP, VERSION, PLURAL, NAMESPACE = "kubeflow.org", "v2beta1", "mpijobs", "default"


def make_job() -> V2beta1MPIJob:
    launcher = V2beta1ReplicaSpec(
        replicas=1,
        template={"spec": {"containers": [{
            "image": "mpioperator-amd/amdrun",
            "name": "mnist-launcher",
            "command": ["amdrun"],
            "args": ["-np", "2", "--hostfile", "/etc/mpi/hostfile", "--",
                     "python3", "/examples/train_mnist.py"],
        }]}})
    worker = V2beta1ReplicaSpec(
        replicas=2,
        template={"spec": {"containers": [{
            "image": "mpioperator-amd/amdrun",
            "name": "mnist-worker",
            "resources": {"limits": {"amd.com/gpu": 1}},
        }]}})
    return V2beta1MPIJob(
        api_version=f"{GROUP}/{VERSION}",
        kind="MPIJob",
        metadata={"name": "mnist", "namespace": NAMESPACE},
        spec=V2beta1MPIJobSpec(
            slots_per_worker=1,
            run_policy=V2beta1RunPolicy(clean_pod_policy="Running"),
            mpi_replica_specs={"Launcher": launcher, "Worker": worker},
        ),
    )


def main():
    config.load_kube_config()
    api = client.CustomObjectsApi()
    job = make_job()
    api.create_namespaced_custom_object(GROUP, VERSION, NAMESPACE, PLURAL, job.to_dict())
    print("created mpijob mnist")


if __name__ == "__main__":
    main()
