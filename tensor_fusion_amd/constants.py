"""Shared constants: labels, annotations, paths, env names.

This is the public API surface users of the reference stack know
(reference: pkg/constants/constants.go:57-345, pkg/constants/env.go:100-216).
We keep the annotation/label *names* identical so workloads written for the
reference drop onto this stack unchanged; everything behind them is a
brand-new MI355X-native implementation.
"""

# ---------------------------------------------------------------- domain
Domain = "tensor-fusion.ai"
NodeUsedByTaintKey = f"{Domain}/used-by"  # legacy NoSchedule taint
TensorFusionSystemName = "tensor-fusion"
Finalizer = f"{Domain}/finalizer"  # worker pods: dealloc-before-delete
Version = "0.2.0"  # round-2 build

# ---------------------------------------------------------------- labels
LabelComponent = f"{Domain}/component"  # client | worker | hypervisor | operator
ComponentClient = "client"
ComponentWorker = "worker"
ComponentHypervisor = "hypervisor"
ComponentOperator = "operator"

LabelEnabled = f"{Domain}/enabled"  # "true" opts a pod in (webhook trigger)
LabelWorkload = f"{Domain}/workload"  # workload name on worker pods
LabelNode = f"{Domain}/node"  # GPUNode name
LabelPool = f"{Domain}/gpupool"  # pool name
LabelManaged = f"{Domain}/managed-by"

# ------------------------------------------------------------ annotations
# Requests / limits for the three first-class vGPU resources.
AnnoTflopsRequest = f"{Domain}/tflops-request"
AnnoTflopsLimit = f"{Domain}/tflops-limit"
AnnoVramRequest = f"{Domain}/vram-request"
AnnoVramLimit = f"{Domain}/vram-limit"
AnnoComputePercentRequest = f"{Domain}/compute-percent-request"
AnnoComputePercentLimit = f"{Domain}/compute-percent-limit"

AnnoGpuCount = f"{Domain}/gpu-count"  # 1..128 vGPUs for one pod
AnnoGpuModel = f"{Domain}/gpu-model"  # e.g. "MI355X"
AnnoGpuVendor = f"{Domain}/gpu-vendor"  # "AMD"
AnnoGpuIndices = f"{Domain}/gpu-indices"  # explicit device indices
AnnoQos = f"{Domain}/qos"  # low | medium | high | critical
AnnoIsolation = f"{Domain}/isolation"  # shared | soft | hard | partitioned
AnnoIsLocalGpu = f"{Domain}/is-local-gpu"  # "true": run on the GPU node itself
AnnoSidecarWorker = f"{Domain}/sidecar-worker"
AnnoInjectContainer = f"{Domain}/inject-container"
AnnoWorkloadProfile = f"{Domain}/workload-profile"
AnnoPartition = f"{Domain}/partition"  # request partitioned placement
AnnoPartitionId = f"{Domain}/partition-id"  # output: bound partition template id
AnnoAutoscale = f"{Domain}/autoscale"
AnnoHostPort = f"{Domain}/host-port"  # "auto" → port allocator assigns
AnnoHostPortAssigned = f"{Domain}/host-port-assigned"

# Gang scheduling (reference constants.go:188-204).
AnnoGangEnabled = f"{Domain}/gang-enabled"
AnnoGangMinMembers = f"{Domain}/gang-min-members"
AnnoGangTimeout = f"{Domain}/gang-timeout"
AnnoGangGroupKey = f"{Domain}/gang-group-key"

# Outputs written by the scheduler at PreBind (reference gpuresources.go:882).
AnnoGpuIds = f"{Domain}/gpu-ids"  # comma-joined device UUIDs
AnnoPodIndex = f"{Domain}/index"
AnnoEffectiveHardCuPercent = f"{Domain}/effective-hard-cu-percent"
AnnoContainerGpus = f"{Domain}/container-gpus"

# ----------------------------------------------------------------- QoS
QosLow = "low"
QosMedium = "medium"
QosHigh = "high"
QosCritical = "critical"
QosLevels = (QosLow, QosMedium, QosHigh, QosCritical)

# ------------------------------------------------------------- isolation
IsolationShared = "shared"
IsolationSoft = "soft"  # ERL token bucket via LD_PRELOAD limiter
IsolationHard = "hard"  # CU mask + VRAM hard cap
IsolationPartitioned = "partitioned"  # SPX/CPX x NPS compute partitions
IsolationModes = (IsolationShared, IsolationSoft, IsolationHard, IsolationPartitioned)

# ----------------------------------------------------------------- paths
DataRoot = "/run/tensor-fusion"
ShmRoot = f"{DataRoot}/shm"  # /run/tensor-fusion/shm/<ns>/<pod>/shm
PreloadFile = "/etc/ld.so.preload"
LimiterLibName = "libtfhip_limiter.so"
ClientLibName = "libtfhip_client.so"
AcceleratorLibName = "libaccelerator_amd.so"


def shm_path(namespace: str, pod: str) -> str:
    return f"{ShmRoot}/{namespace}/{pod}/shm"


# ------------------------------------------------------------------- env
# Contract consumed by the limiter / client stub inside workload containers
# (reference env.go:100-216).
EnvIsolationMode = "TF_ISOLATION_MODE"
EnvShmPath = "TF_SHM_PATH"
EnvPodNamespace = "POD_NAMESPACE"
EnvPodName = "POD_NAME"
EnvContainerName = "CONTAINER_NAME"
EnvHypervisorIP = "HYPERVISOR_IP"
EnvHypervisorPort = "HYPERVISOR_PORT"
EnvConnectionName = "TENSOR_FUSION_CONNECTION_NAME"
EnvConnectionNamespace = "TENSOR_FUSION_CONNECTION_NAMESPACE"
EnvOperatorEndpoint = "TF_OPERATOR_ENDPOINT"
EnvVisibleDevices = "HIP_VISIBLE_DEVICES"
EnvCuMask = "HSA_CU_MASK"  # hard isolation: per-queue CU mask honoured by ROCr
EnvVramLimit = "TF_VRAM_LIMIT_BYTES"
EnvUpLimitPercent = "TF_UP_LIMIT_PERCENT"
EnvRemoteWorkerURL = "TF_WORKER_URL"  # native+<ip>+<port>+<worker>-<rev>
EnvRemoteSocket = "TF_WORKER_SOCKET"  # same-node shm-ring endpoint

# ----------------------------------------------------------------- ports
OperatorHTTPPort = 8080  # client /connection lookup
HypervisorHTTPPort = 8001  # worker/limiter facing API
WorkerPort = 8000  # GPU-over-IP endpoint
NodePortRangeStart = 40000
NodePortRangeEnd = 42000
ClusterPortRangeStart = 42000
ClusterPortRangeEnd = 62000

SchedulerName = "tensor-fusion-scheduler"
IndexResourcePrefix = f"{Domain}/index-"  # index-1 .. index-32
MaxWorkersPerNode = 32

# ------------------------------------------------------------- GPU model
# MI355X hardware facts used across the stack (see docs/DESIGN.md §hardware).
MI355X_MODEL = "MI355X"
MI355X_VRAM_BYTES = 288 * 1024**3  # HBM3E
MI355X_CUS = 256
MI355X_XCDS = 8
MI355X_BF16_TFLOPS = 2500.0  # dense MFMA peak
MI355X_FP32_TFLOPS = 157.3
MI355X_XGMI_LINKS = 7  # point-to-point, ~153 GB/s each
MI355X_XGMI_LINK_GBPS = 153.0

# xGMI topology tiers (reference gpu_types.go:140-157 normalises NVLink tiers;
# on MI355X every intra-node pair has a dedicated xGMI link => tier 0).
TopoTierXGMI = 0  # direct xGMI link (all intra-node pairs on MI355X)
TopoTierNUMA = 1  # same NUMA via host
TopoTierXNUMA = 2  # cross NUMA
TopoTierUnknown = 3
