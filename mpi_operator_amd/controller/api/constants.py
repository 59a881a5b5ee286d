"""MPIJob v2beta1 API constants — behavioral parity with the reference
(reference pkg/apis/kubeflow/v2beta1/constants.go:17-46,
pkg/controller/mpi_job_controller.go:72-118)."""

API_GROUP = "kubeflow.org"
API_VERSION = "v2beta1"
API_GROUP_VERSION = f"{API_GROUP}/{API_VERSION}"
KIND = "MPIJob"
PLURAL = "mpijobs"

ENV_KUBEFLOW_NAMESPACE = "KUBEFLOW_NAMESPACE"
OPERATOR_NAME = "mpi-operator"

# labels (training.kubeflow.org common labels)
REPLICA_INDEX_LABEL = "training.kubeflow.org/replica-index"
REPLICA_TYPE_LABEL = "training.kubeflow.org/replica-type"
OPERATOR_NAME_LABEL = "training.kubeflow.org/operator-name"
JOB_NAME_LABEL = "training.kubeflow.org/job-name"
JOB_ROLE_LABEL = "training.kubeflow.org/job-role"

# controller-internal constants (mpi_job_controller.go:72-96)
CONTROLLER_AGENT_NAME = "mpi-job-controller"
CONFIG_SUFFIX = "-config"
CONFIG_VOLUME_NAME = "mpi-job-config"
CONFIG_MOUNT_PATH = "/etc/mpi"
HOSTFILE_NAME = "hostfile"
DISCOVER_HOSTS_SCRIPT_NAME = "discover_hosts.sh"
SSH_AUTH_SECRET_SUFFIX = "-ssh"
SSH_AUTH_VOLUME = "ssh-auth"
ROOT_SSH_PATH = "/root/.ssh"
LAUNCHER = "launcher"
WORKER = "worker"
LAUNCHER_SUFFIX = "-launcher"
WORKER_SUFFIX = "-worker"
LABEL_GROUP_NAME = "group-name"
LABEL_MPI_JOB_NAME = "mpi-job-name"
LABEL_MPI_ROLE_TYPE = "mpi-job-role"
SSH_PUBLIC_KEY = "ssh-publickey"
SSH_PRIVATE_KEY_FILE = "id_rsa"
SSH_PUBLIC_KEY_FILE = "id_rsa.pub"
SSH_AUTHORIZED_KEYS_FILE = "authorized_keys"

# event reasons
ERR_RESOURCE_EXISTS = "ErrResourceExists"
MESSAGE_RESOURCE_EXISTS = 'Resource "%s" of Kind %q already exists and is not managed by MPIJob'
VALIDATION_ERROR = "ValidationError"
EVENT_MESSAGE_LIMIT = 1024

OPENMPI_SLOTS_ENV = "OMPI_MCA_orte_set_default_slots"
INTELMPI_SLOTS_ENV = "I_MPI_PERHOST"

# enums
MPI_IMPL_OPENMPI = "OpenMPI"
MPI_IMPL_INTEL = "Intel"
MPI_IMPL_MPICH = "MPICH"
MPI_IMPLEMENTATIONS = (MPI_IMPL_OPENMPI, MPI_IMPL_INTEL, MPI_IMPL_MPICH)

CLEAN_POD_POLICY_NONE = "None"
CLEAN_POD_POLICY_RUNNING = "Running"
CLEAN_POD_POLICY_ALL = "All"
CLEAN_POD_POLICIES = (CLEAN_POD_POLICY_NONE, CLEAN_POD_POLICY_RUNNING, CLEAN_POD_POLICY_ALL)

RESTART_POLICY_NEVER = "Never"
RESTART_POLICY_ON_FAILURE = "OnFailure"
RESTART_POLICY_ALWAYS = "Always"
RESTART_POLICY_EXIT_CODE = "ExitCode"
RESTART_POLICIES = (RESTART_POLICY_NEVER, RESTART_POLICY_ON_FAILURE,
                    RESTART_POLICY_ALWAYS, RESTART_POLICY_EXIT_CODE)

LAUNCHER_CREATION_AT_STARTUP = "AtStartup"
LAUNCHER_CREATION_WAIT_FOR_WORKERS_READY = "WaitForWorkersReady"

MPI_REPLICA_TYPE_LAUNCHER = "Launcher"
MPI_REPLICA_TYPE_WORKER = "Worker"

# JobCondition types (types.go:282-340)
JOB_CREATED = "Created"
JOB_RUNNING = "Running"
JOB_RESTARTING = "Restarting"
JOB_SUCCEEDED = "Succeeded"
JOB_SUSPENDED = "Suspended"
JOB_FAILED = "Failed"

KUBEFLOW_JOB_CONTROLLER = "kubeflow.org/mpi-operator"

# env plumbing for the workload (amd-native additions; reference used
# NVIDIA_VISIBLE_DEVICES clearing at mpi_job_controller.go:216-219)
AMD_DISABLE_GPU_ENV = ("ROCR_VISIBLE_DEVICES", "HIP_VISIBLE_DEVICES")
