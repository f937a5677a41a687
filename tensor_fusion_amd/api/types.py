"""Resource model — the 12 CRD-equivalents of the reference stack.

Reference: api/v1/*.go. Same object names and field
semantics, re-modelled as plain dataclasses served by an embedded object
store (tensor_fusion_amd.api.store) instead of a kube-apiserver; on a real
cluster these serialize 1:1 into CRDs.

Resources (amounts) are plain numbers: tflops (float, bf16 dense TFLOPS),
vram (int bytes), compute_percent (float 0-100; 1 CU on MI355X ≈ 0.39%).
"""
from __future__ import annotations

import copy
import time
import uuid as _uuid
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

from .. import constants as C

# ------------------------------------------------------------------ base


@dataclass
class ObjectMeta:
    name: str = ""
    namespace: str = ""
    uid: str = field(default_factory=lambda: str(_uuid.uuid4()))
    labels: Dict[str, str] = field(default_factory=dict)
    annotations: Dict[str, str] = field(default_factory=dict)
    creation_ts: float = field(default_factory=time.time)
    resource_version: int = 0
    deletion_ts: Optional[float] = None
    owner: Optional[str] = None  # "<Kind>/<namespace>/<name>"
    finalizers: List[str] = field(default_factory=list)

    @property
    def key(self) -> str:
        return f"{self.namespace}/{self.name}" if self.namespace else self.name


_SCALARS = (str, int, float, bool, type(None))


def fast_deepcopy(x):
    """Structure-aware deepcopy for the object model: every stored type
    is a tree of dataclasses / dicts / lists / scalars (no cycles, no
    sets, no Any), so a direct recursive copy beats copy.deepcopy's
    generic memo machinery ~5x. The store copies on every read, write
    and event — this is its hottest helper. Unknown types fall back to
    copy.deepcopy."""

    t = type(x)
    if t in _SCALARS:
        return x
    if t is dict:
        return {k: fast_deepcopy(v) for k, v in x.items()}
    if t is list:
        return [fast_deepcopy(v) for v in x]
    if hasattr(t, "__dataclass_fields__") and hasattr(x, "__dict__"):
        new = t.__new__(t)
        nd = new.__dict__
        for k, v in x.__dict__.items():
            tv = type(v)
            nd[k] = v if tv in _SCALARS else fast_deepcopy(v)
        return new
    return copy.deepcopy(x)


@dataclass
class TFObject:
    meta: ObjectMeta = field(default_factory=ObjectMeta)

    kind: str = ""

    def deepcopy(self):
        return fast_deepcopy(self)

    def __deepcopy__(self, memo=None):
        return fast_deepcopy(self)


# ------------------------------------------------------------- resources


@dataclass
class Resource:
    """One amount of the three first-class vGPU resources."""

    tflops: float = 0.0
    vram: int = 0
    compute_percent: float = 0.0

    def add(self, o: "Resource") -> "Resource":
        return Resource(self.tflops + o.tflops, self.vram + o.vram,
                        self.compute_percent + o.compute_percent)

    def sub(self, o: "Resource") -> "Resource":
        return Resource(self.tflops - o.tflops, self.vram - o.vram,
                        self.compute_percent - o.compute_percent)

    def fits_in(self, o: "Resource") -> bool:
        return (self.tflops <= o.tflops + 1e-9 and self.vram <= o.vram
                and self.compute_percent <= o.compute_percent + 1e-9)

    def any_negative(self) -> bool:
        return self.tflops < -1e-9 or self.vram < 0 or self.compute_percent < -1e-9


@dataclass
class Requirements:
    requests: Resource = field(default_factory=Resource)
    limits: Resource = field(default_factory=Resource)


# --------------------------------------------------------------- GPU (CR)


@dataclass
class GPUPartition:
    """A bound compute partition on a device (MI355X: CPX slot / CU-mask slab)."""

    partition_id: str = ""
    template_id: str = ""
    workload: str = ""
    pod: str = ""
    resource: Resource = field(default_factory=Resource)
    xcds: List[int] = field(default_factory=list)  # XCDs backing a CPX slot


@dataclass
class GPUStatus:
    capacity: Resource = field(default_factory=Resource)
    available: Resource = field(default_factory=Resource)
    uuid: str = ""
    index: int = 0
    numa_node: int = 0
    model: str = C.MI355X_MODEL
    vendor: str = "AMD"
    phase: str = "Ready"  # Pending|Ready|Unknown|Migrating|Destroying
    used_by: str = "tensor-fusion"  # or "external-device-plugin"
    node: str = ""
    pool: str = ""
    running_apps: List[str] = field(default_factory=list)  # workload names
    # xGMI peers: device uuid -> topology tier (0 = direct xGMI link).
    topology: Dict[str, int] = field(default_factory=dict)
    isolation_mode: str = C.IsolationSoft
    allocated_partitions: List[GPUPartition] = field(default_factory=list)


@dataclass
class GPU(TFObject):
    kind: str = "GPU"
    status: GPUStatus = field(default_factory=GPUStatus)


# ---------------------------------------------------------------- GPUNode


@dataclass
class GPUNodeStatus:
    phase: str = "Pending"  # Pending|Running|Unknown|Destroying
    total: Resource = field(default_factory=Resource)
    available: Resource = field(default_factory=Resource)
    gpu_count: int = 0
    gpus: List[str] = field(default_factory=list)  # GPU object names
    hypervisor_ready: bool = False
    node_ip: str = ""
    kernel: str = ""
    rocm_version: str = ""


@dataclass
class GPUNode(TFObject):
    kind: str = "GPUNode"
    manage_mode: str = "AutoSelect"  # AutoSelect|Manual|Provisioned
    pool: str = ""
    status: GPUNodeStatus = field(default_factory=GPUNodeStatus)


# ---------------------------------------------------------------- GPUPool


@dataclass
class Oversubscription:
    """Reference gpupool_types.go:64-86."""

    vram_expand_to_host_mem_percent: int = 50
    vram_expand_to_host_disk_percent: int = 70
    tflops_oversell_ratio: int = 500  # percent


@dataclass
class CapacityConfig:
    min_resources: Resource = field(default_factory=Resource)
    max_resources: Resource = field(default_factory=Resource)
    oversubscription: Oversubscription = field(default_factory=Oversubscription)


@dataclass
class NodeManagerConfig:
    provisioning_mode: str = "AutoSelect"  # AutoSelect|Provisioned|Manual
    node_selector: Dict[str, str] = field(default_factory=dict)
    isolation_default: str = C.IsolationSoft
    defrag_enabled: bool = False
    defrag_schedule: str = ""  # cron
    rolling_update_batch_percent: int = 25
    rolling_update_interval_s: int = 60


@dataclass
class QosPricing:
    qos: str = C.QosMedium
    tflops_per_hour: float = 0.0
    vram_gb_per_hour: float = 0.0


@dataclass
class ComponentConfig:
    """Pod templates for hypervisor/worker/client (reference :398-473)."""

    hypervisor_image: str = "tensor-fusion/hypervisor:latest"
    worker_image: str = "tensor-fusion/worker:latest"
    client_image: str = "tensor-fusion/client:latest"
    hypervisor_template: Dict[str, Any] = field(default_factory=dict)
    worker_template: Dict[str, Any] = field(default_factory=dict)
    client_template: Dict[str, Any] = field(default_factory=dict)


@dataclass
class GPUPoolStatus:
    phase: str = "Pending"
    node_count: int = 0
    gpu_count: int = 0
    total: Resource = field(default_factory=Resource)
    available: Resource = field(default_factory=Resource)
    virtual_total: Resource = field(default_factory=Resource)  # after oversell
    virtual_available: Resource = field(default_factory=Resource)


@dataclass
class GPUPool(TFObject):
    kind: str = "GPUPool"
    cluster: str = ""
    capacity: CapacityConfig = field(default_factory=CapacityConfig)
    node_manager: NodeManagerConfig = field(default_factory=NodeManagerConfig)
    qos_pricing: List[QosPricing] = field(default_factory=list)
    components: ComponentConfig = field(default_factory=ComponentConfig)
    scheduling_template: str = ""  # SchedulingConfigTemplate name
    status: GPUPoolStatus = field(default_factory=GPUPoolStatus)


# ------------------------------------------------------- TensorFusionCluster


@dataclass
class ComputingVendor:
    name: str = ""  # aws | alibaba | karpenter | mock
    auth: Dict[str, str] = field(default_factory=dict)
    enabled: bool = False


@dataclass
class TensorFusionClusterStatus:
    phase: str = "Pending"
    pool_count: int = 0


@dataclass
class TensorFusionCluster(TFObject):
    kind: str = "TensorFusionCluster"
    pools: List[GPUPool] = field(default_factory=list)  # templates; controller owns real pools
    vendors: List[ComputingVendor] = field(default_factory=list)
    status: TensorFusionClusterStatus = field(default_factory=TensorFusionClusterStatus)


# --------------------------------------------------------- WorkloadProfile


@dataclass
class AutoScalingConfig:
    enabled: bool = False
    recommender: str = "percentile"  # percentile | cron | external
    target_percentile: float = 0.9
    margin: float = 0.15
    cron_rules: List[Dict[str, Any]] = field(default_factory=list)
    external_url: str = ""


@dataclass
class GangSchedulingConfig:
    enabled: bool = False
    min_members: int = 0
    timeout_s: float = 60.0
    group_key: str = ""


@dataclass
class WorkloadProfile(TFObject):
    """The vGPU request model (reference workloadprofile_types.go:37-174)."""

    kind: str = "WorkloadProfile"
    resources: Requirements = field(default_factory=Requirements)
    qos: str = C.QosMedium
    isolation_mode: str = C.IsolationSoft
    is_local_gpu: bool = False
    sidecar_worker: bool = False
    gpu_count: int = 1  # 1..128
    gpu_model: str = ""
    gpu_vendor: str = ""
    gpu_indices: List[int] = field(default_factory=list)
    pool: str = ""
    auto_scaling: AutoScalingConfig = field(default_factory=AutoScalingConfig)
    gang: GangSchedulingConfig = field(default_factory=GangSchedulingConfig)


# ----------------------------------------------------- TensorFusionWorkload


@dataclass
class GangSchedulingStatus:
    group: str = ""
    phase: str = ""  # Pending|Scheduled|Failed
    members_total: int = 0
    members_scheduled: int = 0


@dataclass
class Recommendation:
    resources: Requirements = field(default_factory=Requirements)
    reason: str = ""
    ts: float = 0.0


@dataclass
class TensorFusionWorkloadStatus:
    phase: str = "Pending"
    replicas: int = 0
    ready_replicas: int = 0
    worker_pods: List[str] = field(default_factory=list)
    gang: GangSchedulingStatus = field(default_factory=GangSchedulingStatus)
    recommendation: Optional[Recommendation] = None


@dataclass
class TensorFusionWorkload(TFObject):
    kind: str = "TensorFusionWorkload"
    profile: WorkloadProfile = field(default_factory=WorkloadProfile)
    replicas: int = 1
    pool: str = ""
    status: TensorFusionWorkloadStatus = field(default_factory=TensorFusionWorkloadStatus)


# --------------------------------------------------- TensorFusionConnection


@dataclass
class TensorFusionConnectionStatus:
    phase: str = "Pending"
    connection_url: str = ""  # native+<ip>+<port>+<worker>-<rev>
    worker: str = ""


@dataclass
class TensorFusionConnection(TFObject):
    kind: str = "TensorFusionConnection"
    workload: str = ""
    client_pod: str = ""
    status: TensorFusionConnectionStatus = field(default_factory=TensorFusionConnectionStatus)


def format_connection_url(ip: str, port: int, worker: str, rev: int) -> str:
    """Reference tensorfusionconnection_controller.go:136-137."""

    return f"native+{ip}+{port}+{worker}-{rev}"


# ------------------------------------------------------- GPUResourceQuota


@dataclass
class GPUResourceQuotaSpec:
    total: Resource = field(default_factory=Resource)
    single_max: Resource = field(default_factory=Resource)
    max_workers: int = 0  # 0 = unlimited
    alert_threshold_percent: int = 95


@dataclass
class GPUResourceQuotaStatus:
    used: Resource = field(default_factory=Resource)
    worker_count: int = 0


@dataclass
class GPUResourceQuota(TFObject):
    kind: str = "GPUResourceQuota"
    spec: GPUResourceQuotaSpec = field(default_factory=GPUResourceQuotaSpec)
    status: GPUResourceQuotaStatus = field(default_factory=GPUResourceQuotaStatus)


# --------------------------------------------- AllocRequest (allocator currency)


@dataclass
class AllocRequest:
    """What the scheduler asks the allocator for (reference
    gpuresourcequota_types.go:168-215)."""

    workload: str = ""
    pod_name: str = ""
    namespace: str = "default"
    pool: str = ""
    request: Resource = field(default_factory=Resource)
    limit: Resource = field(default_factory=Resource)
    gpu_count: int = 1
    gpu_model: str = ""
    gpu_vendor: str = ""
    gpu_indices: List[int] = field(default_factory=list)
    qos: str = C.QosMedium
    isolation_mode: str = C.IsolationSoft
    partitioned: bool = False
    node_affinity: Dict[str, str] = field(default_factory=dict)
    gang_group: str = ""

    @property
    def pod_key(self) -> str:
        return f"{self.namespace}/{self.pod_name}"


# ------------------------------------------- GPUNodeClass / GPUNodeClaim


@dataclass
class GPUNodeClass(TFObject):
    kind: str = "GPUNodeClass"
    vendor: str = "mock"
    instance_types: List[str] = field(default_factory=list)
    region: str = ""
    gpu_model: str = C.MI355X_MODEL
    gpus_per_node: int = 8
    launch_template: Dict[str, Any] = field(default_factory=dict)


@dataclass
class GPUNodeClaimStatus:
    phase: str = "Pending"  # Pending|Creating|Bound|Failed
    instance_id: str = ""
    node_name: str = ""


@dataclass
class GPUNodeClaim(TFObject):
    kind: str = "GPUNodeClaim"
    node_class: str = ""
    pool: str = ""
    instance_type: str = ""
    status: GPUNodeClaimStatus = field(default_factory=GPUNodeClaimStatus)


# --------------------------------------------- SchedulingConfigTemplate


@dataclass
class ElasticRateLimitParams:
    """PID gains of the hypervisor ERL controller (reference :252-266)."""

    kp: float = 0.9
    ki: float = 0.35
    kd: float = 0.10
    ema_alpha: float = 0.25
    deadband_percent: float = 3.0
    slew_up_percent: float = 35.0
    slew_down_percent: float = 25.0
    loop_interval_s: float = 0.5
    min_rate: float = 10.0
    max_rate: float = 200_000.0


@dataclass
class AutoFreezeRule:
    qos: str = C.QosLow
    freeze_to_mem_ttl_s: int = 0
    freeze_to_disk_ttl_s: int = 0
    enable: bool = False


@dataclass
class SchedulingConfigTemplate(TFObject):
    kind: str = "SchedulingConfigTemplate"
    placement_mode: str = "CompactFirst"  # CompactFirst | LowLoadFirst
    vram_weight: float = 0.7
    tflops_weight: float = 0.3
    topo_mode: str = "soft"  # soft | hard
    auto_freeze: List[AutoFreezeRule] = field(default_factory=list)
    rebalance_interval_s: int = 0
    erl: ElasticRateLimitParams = field(default_factory=ElasticRateLimitParams)


# ----------------------------------------------------------- ProviderConfig


@dataclass
class PartitionTemplate:
    """One partition size (MI355X: a CU/XCD slab; analogous to a MIG profile).

    On CDNA4 a partition is expressed as N of the 8 XCDs (CPX-style) or a
    CU-mask slab; placement bitmask is over the 8 XCD slots.
    """

    id: str = ""
    name: str = ""  # e.g. "1xcd.36gb"
    xcds: int = 1
    compute_percent: float = 12.5
    tflops: float = C.MI355X_BF16_TFLOPS / 8
    vram: int = C.MI355X_VRAM_BYTES // 8
    placements: List[int] = field(default_factory=list)  # allowed start XCDs
    # AMD compute-partition pairing: the device-global mode this template
    # belongs to (SPX/DPX/QPX/CPX) and the memory interleave it expects
    # (NPS1/NPS2/NPS4). Templates of different modes cannot coexist on
    # one device (partitioning.py enforces it).
    mode: str = ""
    memory_mode: str = ""


@dataclass
class HardwareModel:
    model: str = C.MI355X_MODEL
    vendor: str = "AMD"
    fp16_tflops: float = C.MI355X_BF16_TFLOPS
    vram: int = C.MI355X_VRAM_BYTES
    compute_units: int = C.MI355X_CUS
    cost_per_hour: float = 0.0


@dataclass
class ProviderConfig(TFObject):
    kind: str = "ProviderConfig"
    vendor: str = "AMD"
    accelerator_lib: str = C.AcceleratorLibName
    limiter_lib: str = C.LimiterLibName
    device_nodes: List[str] = field(default_factory=lambda: ["/dev/kfd", "/dev/dri"])
    mount_libs: List[str] = field(default_factory=list)
    models: List[HardwareModel] = field(default_factory=list)
    partition_templates: List[PartitionTemplate] = field(default_factory=list)
    device_plugin_prefixes: List[str] = field(default_factory=lambda: ["amd.com/gpu"])


def default_mi355x_partition_templates() -> List[PartitionTemplate]:
    """Partition templates for one MI355X, one per AMD compute-partition
    mode (the device-global SPX/DPX/QPX/CPX split over the 8 XCDs) with
    its paired NPS memory interleave. The mode is enforced exclusively
    per device by allocator/partitioning.py."""

    mode_by_xcds = {1: ("CPX", "NPS4"), 2: ("QPX", "NPS4"),
                    4: ("DPX", "NPS2"), 8: ("SPX", "NPS1")}
    out = []
    for xcds in (1, 2, 4, 8):
        frac = xcds / C.MI355X_XCDS
        mode, nps = mode_by_xcds[xcds]
        out.append(PartitionTemplate(
            id=f"xcd{xcds}",
            name=f"{xcds}xcd.{int(288 * frac)}gb",
            xcds=xcds,
            compute_percent=100.0 * frac,
            tflops=C.MI355X_BF16_TFLOPS * frac,
            vram=int(C.MI355X_VRAM_BYTES * frac),
            placements=[s for s in range(0, C.MI355X_XCDS, xcds)],
            mode=mode,
            memory_mode=nps,
        ))
    return out


# ------------------------------------------------------------- Pod (node obj)
# Minimal pod model: what the webhook mutates and the scheduler schedules.
# On a real cluster this is corev1.Pod; the embedded store serves the same
# shape for tests and the single-node backend.


@dataclass
class Container:
    name: str = "main"
    image: str = ""
    command: List[str] = field(default_factory=list)
    env: Dict[str, str] = field(default_factory=dict)
    resources: Dict[str, str] = field(default_factory=dict)
    volume_mounts: List[Dict[str, str]] = field(default_factory=list)


@dataclass
class PodStatus:
    phase: str = "Pending"  # Pending|Scheduled|Running|Succeeded|Failed
    node: str = ""
    pod_ip: str = ""
    host_ip: str = ""
    pid: int = 0  # single-node backend: host pid of main process


@dataclass
class Pod(TFObject):
    kind: str = "Pod"
    scheduler_name: str = "default"
    containers: List[Container] = field(default_factory=list)
    node_selector: Dict[str, str] = field(default_factory=dict)
    status: PodStatus = field(default_factory=PodStatus)


@dataclass
class Node(TFObject):
    """A compute node (corev1.Node equivalent)."""

    kind: str = "Node"
    labels_: Dict[str, str] = field(default_factory=dict)
    capacity: Dict[str, float] = field(default_factory=dict)
    status_phase: str = "Ready"
    address: str = "127.0.0.1"
    taints: List[Dict[str, str]] = field(default_factory=list)


ALL_KINDS = [
    "TensorFusionCluster", "GPUPool", "GPUNode", "GPU", "TensorFusionWorkload",
    "WorkloadProfile", "TensorFusionConnection", "GPUResourceQuota",
    "GPUNodeClass", "GPUNodeClaim", "SchedulingConfigTemplate", "ProviderConfig",
    "Pod", "Node",
]
