"""Byte-compatible grove.io API contract constants.

Parity source: /root/reference/operator/api/common/labels.go:19-95,
/root/reference/operator/api/common/constants/constants.go:17-183 and
/root/reference/operator/internal/controller/podclique/components/pod/pod.go:69.
These are the kubectl-visible strings that must match the reference exactly;
everything else in this package is a fresh MI355X-native implementation.
"""

OPERATOR_NAME = "grove-operator"
GROUP = "grove.io"
API_VERSION = "grove.io/v1alpha1"
SCHEDULER_GROUP = "scheduler.grove.io"
SCHEDULER_API_VERSION = "scheduler.grove.io/v1alpha1"

# --- Kinds ---
KIND_PCS = "PodCliqueSet"
KIND_PCLQ = "PodClique"
KIND_PCSG = "PodCliqueScalingGroup"
KIND_CTB = "ClusterTopologyBinding"
KIND_PODGANG = "PodGang"

# --- Common labels (labels.go) ---
LABEL_APP_NAME = "app.kubernetes.io/name"
LABEL_MANAGED_BY = "app.kubernetes.io/managed-by"
LABEL_PART_OF = "app.kubernetes.io/part-of"
LABEL_MANAGED_BY_VALUE = "grove-operator"
LABEL_COMPONENT = "app.kubernetes.io/component"
LABEL_PODCLIQUE = "grove.io/podclique"
LABEL_PODGANG = "grove.io/podgang"
LABEL_BASE_PODGANG = "grove.io/base-podgang"
LABEL_PCS_REPLICA_INDEX = "grove.io/podcliqueset-replica-index"
LABEL_PCSG = "grove.io/podcliquescalinggroup"
LABEL_PCSG_REPLICA_INDEX = "grove.io/podcliquescalinggroup-replica-index"
LABEL_POD_INDEX = "grove.io/podclique-pod-index"
LABEL_POD_TEMPLATE_HASH = "grove.io/pod-template-hash"
LABEL_SCHEDULER_NAME = "grove.io/scheduler-name"

# --- Component name values for LABEL_COMPONENT ---
COMPONENT_HEADLESS_SERVICE = "pcs-headless-service"
COMPONENT_POD_ROLE = "pod-role"
COMPONENT_POD_ROLE_BINDING = "pod-role-binding"
COMPONENT_POD_SERVICE_ACCOUNT = "pod-service-account"
COMPONENT_SA_TOKEN_SECRET = "pod-sa-token-secret"
COMPONENT_PCSG = "pcs-podcliquescalinggroup"
COMPONENT_HPA = "pcs-hpa"
COMPONENT_PODGANG = "podgang"
COMPONENT_PCS_PODCLIQUE = "pcs-podclique"
COMPONENT_PCSG_PODCLIQUE = "pcsg-podclique"
COMPONENT_RESOURCE_CLAIM = "resource-claim"

# --- Finalizers ---
FINALIZER_PCS = "grove.io/podcliqueset.grove.io"
FINALIZER_PCLQ = "grove.io/podclique.grove.io"
FINALIZER_PCSG = "grove.io/podcliquescalinggroup.grove.io"

# --- Annotations ---
ANNOTATION_DISABLE_MANAGED_RESOURCE_PROTECTION = "grove.io/disable-managed-resource-protection"
ANNOTATION_RECONCILE_TRIGGER = "grove.io/reconcile-trigger"
ANNOTATION_TOPOLOGY_NAME = "grove.io/topology-name"
# MI355X analog of the reference's grove.io/mnnvl-group annotation (auto-mnnvl.md):
# cliques sharing a group name share one xGMI-domain ResourceClaim per PCS replica;
# propagates PCS -> PCSG -> PCLQ with lower levels overriding; "none" opts out.
ANNOTATION_XGMI_GROUP = "grove.io/xgmi-group"

# --- Pod scheduling gate (pod.go:69) ---
POD_GANG_SCHEDULING_GATE = "grove.io/podgang-pending-creation"

# --- Injected environment variables (constants.go:56-75) ---
ENV_PCS_NAME = "GROVE_PCS_NAME"
ENV_PCS_INDEX = "GROVE_PCS_INDEX"
ENV_PCLQ_NAME = "GROVE_PCLQ_NAME"
ENV_HEADLESS_SERVICE = "GROVE_HEADLESS_SERVICE"
ENV_POD_INDEX = "GROVE_PCLQ_POD_INDEX"
ENV_PCSG_NAME = "GROVE_PCSG_NAME"
ENV_PCSG_INDEX = "GROVE_PCSG_INDEX"
ENV_PCSG_TEMPLATE_NUM_PODS = "GROVE_PCSG_TEMPLATE_NUM_PODS"

# --- Events ---
EVENT_RECONCILING = "Reconciling"
EVENT_RECONCILED = "Reconciled"
EVENT_RECONCILE_ERROR = "ReconcileError"
EVENT_DELETING = "Deleting"
EVENT_DELETED = "Deleted"
EVENT_DELETE_ERROR = "DeleteError"

# --- Condition types ---
COND_MIN_AVAILABLE_BREACHED = "MinAvailableBreached"
COND_PODCLIQUE_SCHEDULED = "PodCliqueScheduled"
COND_GANG_TERMINATION_IN_PROGRESS = "GangTerminationInProgress"
COND_TOPOLOGY_LEVELS_UNAVAILABLE = "TopologyLevelsUnavailable"
COND_SCHEDULER_TOPOLOGY_DRIFT = "SchedulerTopologyDrift"

# --- Condition reasons ---
REASON_INSUFFICIENT_READY_PODS = "InsufficientReadyPods"
REASON_SUFFICIENT_READY_PODS = "SufficientReadyPods"
REASON_INSUFFICIENT_SCHEDULED_PODS = "InsufficientScheduledPods"
REASON_SUFFICIENT_SCHEDULED_PODS = "SufficientScheduledPods"
REASON_SCHEDULED_BELOW_MIN_AVAILABLE = "ScheduledReplicasBelowMinAvailable"
REASON_INSUFFICIENT_AVAILABLE_PCSG_REPLICAS = "InsufficientAvailablePodCliqueScalingGroupReplicas"
REASON_SUFFICIENT_AVAILABLE_PCSG_REPLICAS = "SufficientAvailablePodCliqueScalingGroupReplicas"
REASON_UPDATE_IN_PROGRESS = "UpdateInProgress"
REASON_GANG_TERMINATION_ACTIVE = "GangTerminationActive"
REASON_CLUSTER_TOPOLOGY_NOT_FOUND = "ClusterTopologyNotFound"
REASON_TOPOLOGY_LEVELS_UNAVAILABLE = "ClusterTopologyLevelsUnavailable"
REASON_ALL_TOPOLOGY_LEVELS_AVAILABLE = "AllClusterTopologyLevelsAvailable"
REASON_IN_SYNC = "InSync"
REASON_DRIFT = "Drift"
REASON_TOPOLOGY_NOT_FOUND = "TopologyNotFound"

# --- PodGang (scheduler.grove.io) condition types (podgang.go:152-171) ---
PODGANG_COND_SCHEDULED = "Scheduled"
PODGANG_COND_READY = "Ready"
PODGANG_COND_INITIALIZED = "Initialized"
PODGANG_COND_UNHEALTHY = "Unhealthy"
PODGANG_COND_DISRUPTION_TARGET = "DisruptionTarget"

# --- Startup types (podcliqueset.go:508) ---
STARTUP_ANY_ORDER = "CliqueStartupTypeAnyOrder"
STARTUP_IN_ORDER = "CliqueStartupTypeInOrder"
STARTUP_EXPLICIT = "CliqueStartupTypeExplicit"

# --- Update strategies ---
UPDATE_ROLLING_RECREATE = "RollingRecreate"
UPDATE_ON_DELETE = "OnDelete"

# --- Defaults (defaulting/podcliqueset.go:29 etc.) ---
DEFAULT_TERMINATION_DELAY_SECONDS = 4 * 3600
DEFAULT_TERMINATION_GRACE_SECONDS = 30

# --- AMD-native resource/topology constants (replaces nvidia.com/gpu, gpu.go:25) ---
AMD_GPU_RESOURCE = "amd.com/gpu"
MI355X_HBM_BYTES = 288 * 1024**3          # 288 GB HBM3E per MI355X
XGMI_PEER_LINKS = 7                        # point-to-point links per GPU in an 8-GPU hive
XGMI_LINK_GBPS = 153.0                     # ≈ per-link bandwidth, GB/s
NODE_LABEL_XGMI_HIVE = "topology.amd.com/xgmi-hive"
NODE_LABEL_GPU_COUNT = "topology.amd.com/gpu-count"
NODE_LABEL_GPU_PRODUCT = "topology.amd.com/gpu-product"

# Well-known topology domains (clustertopologybinding.go:140-155)
TOPOLOGY_DOMAINS = ("region", "zone", "datacenter", "block", "rack", "host", "numa")

# Scheduler names
SCHEDULER_DEFAULT = "default-scheduler"
SCHEDULER_AMD_GANG = "amd-gang-scheduler"
