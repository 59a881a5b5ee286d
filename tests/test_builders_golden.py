"""Golden-object tests for the child-resource builders — the reference's
TestNewLauncherAndWorker/TestNewConfigMap analog (reference
mpi_job_controller_test.go:1582,2053): exact field-level expectations on
worker/launcher pod specs, per-implementation launcher env, SSH volume
modes, and the GPU-hiding env."""
import pytest

from mpi_operator_amd.controller import builders as b
from mpi_operator_amd.controller.api import constants as c
from mpi_operator_amd.controller.api import defaults


def mk_job(**spec):
    job = {
        "apiVersion": c.API_GROUP_VERSION,
        "kind": c.KIND,
        "metadata": {"name": "j", "namespace": "ns", "uid": "u1"},
        "spec": {
            "mpiReplicaSpecs": {
                "Launcher": {"replicas": 1, "template": {"spec": {"containers": [
                    {"name": "l", "image": "img", "command": ["amdrun"]}]}}},
                "Worker": {"replicas": 2, "template": {"spec": {"containers": [
                    {"name": "w", "image": "img"}]}}},
            },
        },
    }
    job["spec"].update(spec)
    defaults.set_defaults_mpijob(job)
    return job


def env_dict(container):
    return {e["name"]: e.get("value") for e in container.get("env", [])}


def test_worker_pod_golden():
    job = mk_job()
    pod = b.new_worker(job, 1)
    assert pod["metadata"]["name"] == "j-worker-1"
    spec = pod["spec"]
    assert spec["hostname"] == "j-worker-1"
    assert spec["subdomain"] == "j"  # headless Service name
    assert "j.ns.svc.cluster.local" in spec["dnsConfig"]["searches"]
    cont = spec["containers"][0]
    assert cont["command"] == ["/usr/sbin/sshd", "-De"]  # default cmd
    assert env_dict(cont)["K_MPI_JOB_ROLE"] == "worker"
    # SSH secret mounted; /root/.ssh gets 0600 default mode
    vols = {v["name"]: v for v in spec["volumes"]}
    assert vols[c.SSH_AUTH_VOLUME]["secret"]["secretName"] == "j-ssh"
    assert vols[c.SSH_AUTH_VOLUME]["secret"]["defaultMode"] == 0o600
    assert pod["metadata"]["ownerReferences"][0]["uid"] == "u1"
    assert pod["metadata"]["labels"][c.REPLICA_INDEX_LABEL] == "1"


def test_worker_custom_command_kept_and_nonroot_ssh_mode():
    job = mk_job(sshAuthMountPath="/home/mpiuser/.ssh")
    job["spec"]["mpiReplicaSpecs"]["Worker"]["template"]["spec"]["containers"][0][
        "command"] = ["/usr/sbin/sshd", "-De", "-f", "/home/mpiuser/.sshd_config"]
    pod = b.new_worker(job, 0)
    cont = pod["spec"]["containers"][0]
    assert cont["command"][-1] == "/home/mpiuser/.sshd_config"
    vols = {v["name"]: v for v in pod["spec"]["volumes"]}
    assert "defaultMode" not in vols[c.SSH_AUTH_VOLUME]["secret"]
    assert cont["volumeMounts"][-1]["mountPath"] == "/home/mpiuser/.ssh"


@pytest.mark.parametrize("impl,hostfile_env,slots_env", [
    ("OpenMPI", "OMPI_MCA_orte_default_hostfile", c.OPENMPI_SLOTS_ENV),
    ("Intel", "I_MPI_HYDRA_HOST_FILE", c.INTELMPI_SLOTS_ENV),
    ("MPICH", "HYDRA_HOST_FILE", None),
])
def test_launcher_env_per_mpi_impl(impl, hostfile_env, slots_env):
    job = mk_job(mpiImplementation=impl, slotsPerWorker=4)
    tmpl = b.new_launcher_pod_template(job)
    env = env_dict(tmpl["spec"]["containers"][0])
    assert env[hostfile_env] == f"{c.CONFIG_MOUNT_PATH}/{c.HOSTFILE_NAME}"
    if slots_env:
        assert env[slots_env] == "4"
    assert env["K_MPI_JOB_ROLE"] == "launcher"
    # non-worker launcher gets the node's GPUs hidden (ROCm env clearing)
    for name in c.AMD_DISABLE_GPU_ENV:
        assert env[name] == ""


def test_launcher_as_worker_keeps_gpus_and_gets_index_0():
    job = mk_job(runLauncherAsWorker=True)
    tmpl = b.new_launcher_pod_template(job)
    env = env_dict(tmpl["spec"]["containers"][0])
    for name in c.AMD_DISABLE_GPU_ENV:
        assert name not in env
    assert tmpl["metadata"]["labels"][c.REPLICA_INDEX_LABEL] == "0"
    # and worker indices shift by one (unique indices for Kueue TAS)
    pod = b.new_worker(job, 0)
    assert pod["metadata"]["labels"][c.REPLICA_INDEX_LABEL] == "1"


def test_exit_code_restart_policy_maps_to_never():
    job = mk_job()
    job["spec"]["mpiReplicaSpecs"]["Worker"]["restartPolicy"] = "ExitCode"
    pod = b.new_worker(job, 0)
    assert pod["spec"]["restartPolicy"] == "Never"


def test_config_volume_modes():
    job = mk_job()
    tmpl = b.new_launcher_pod_template(job)
    vols = {v["name"]: v for v in tmpl["spec"]["volumes"]}
    items = {i["path"]: i for i in vols[c.CONFIG_VOLUME_NAME]["configMap"]["items"]}
    assert items[c.HOSTFILE_NAME]["mode"] == 0o444
    assert items[c.DISCOVER_HOSTS_SCRIPT_NAME]["mode"] == 0o555
