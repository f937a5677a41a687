"""Annotation parsing → AllocRequest (reference utils/resource.go:132-210
ComposeAllocationRequest + tf_parser.go ParseTensorFusionInfo)."""
from __future__ import annotations

import re
from typing import Optional

from .. import constants as C
from ..api.types import AllocRequest, Pod, WorkloadProfile

_QUANT_RE = re.compile(r"^\s*([0-9.]+)\s*([KMGTPE]i?)?\s*$")
_SUFFIX = {
    None: 1, "K": 10**3, "M": 10**6, "G": 10**9, "T": 10**12, "P": 10**15,
    "E": 10**18, "Ki": 2**10, "Mi": 2**20, "Gi": 2**30, "Ti": 2**40,
    "Pi": 2**50, "Ei": 2**60,
}


def parse_quantity(s: str) -> float:
    """k8s-style quantity: '8Gi', '100', '1.5T'."""

    if isinstance(s, (int, float)):
        return float(s)
    m = _QUANT_RE.match(s)
    if not m:
        raise ValueError(f"bad quantity {s!r}")
    return float(m.group(1)) * _SUFFIX[m.group(2)]


def parse_tflops(s: str) -> float:
    # tflops annotations are plain numbers ("620") or suffixed ("1.2T" = 1.2
    # peta-flops worth of tera units is nonsense — treat suffix as SI count).
    return parse_quantity(s)


def profile_from_annotations(pod: Pod,
                             base: Optional[WorkloadProfile] = None
                             ) -> WorkloadProfile:
    """Merge pod annotations over a WorkloadProfile (annotations win)."""

    import copy
    p = copy.deepcopy(base) if base else WorkloadProfile()
    a = pod.meta.annotations

    def q(key, default=None):
        return a.get(key, default)

    res = p.resources
    if q(C.AnnoTflopsRequest):
        res.requests.tflops = parse_tflops(a[C.AnnoTflopsRequest])
    if q(C.AnnoTflopsLimit):
        res.limits.tflops = parse_tflops(a[C.AnnoTflopsLimit])
    if q(C.AnnoVramRequest):
        res.requests.vram = int(parse_quantity(a[C.AnnoVramRequest]))
    if q(C.AnnoVramLimit):
        res.limits.vram = int(parse_quantity(a[C.AnnoVramLimit]))
    if q(C.AnnoComputePercentRequest):
        res.requests.compute_percent = float(a[C.AnnoComputePercentRequest])
    if q(C.AnnoComputePercentLimit):
        res.limits.compute_percent = float(a[C.AnnoComputePercentLimit])
    if q(C.AnnoGpuCount):
        p.gpu_count = int(a[C.AnnoGpuCount])
    if q(C.AnnoGpuModel):
        p.gpu_model = a[C.AnnoGpuModel]
    if q(C.AnnoGpuVendor):
        p.gpu_vendor = a[C.AnnoGpuVendor]
    if q(C.AnnoGpuIndices):
        p.gpu_indices = [int(x) for x in a[C.AnnoGpuIndices].split(",") if x]
    if q(C.AnnoQos):
        p.qos = a[C.AnnoQos]
    if q(C.AnnoIsolation):
        p.isolation_mode = a[C.AnnoIsolation]
    if q(C.AnnoIsLocalGpu):
        p.is_local_gpu = a[C.AnnoIsLocalGpu].lower() == "true"
    if q(C.AnnoSidecarWorker):
        p.sidecar_worker = a[C.AnnoSidecarWorker].lower() == "true"
    if q(C.AnnoGangEnabled):
        p.gang.enabled = a[C.AnnoGangEnabled].lower() == "true"
    if q(C.AnnoGangMinMembers):
        p.gang.min_members = int(a[C.AnnoGangMinMembers])
    if q(C.AnnoGangTimeout):
        p.gang.timeout_s = float(a[C.AnnoGangTimeout])
    if q(C.AnnoGangGroupKey):
        p.gang.group_key = a[C.AnnoGangGroupKey]
    if q(C.AnnoAutoscale):
        p.auto_scaling.enabled = a[C.AnnoAutoscale].lower() == "true"
    if q(C.AnnoPartition):
        p.isolation_mode = C.IsolationPartitioned
    # defaults: limits >= requests
    if res.limits.tflops < res.requests.tflops:
        res.limits.tflops = res.requests.tflops
    if res.limits.vram < res.requests.vram:
        res.limits.vram = res.requests.vram
    if res.limits.compute_percent < res.requests.compute_percent:
        res.limits.compute_percent = res.requests.compute_percent
    return p


def compose_allocation_request(pod: Pod, profile: WorkloadProfile,
                               pool: str = "") -> AllocRequest:
    gang_group = ""
    if profile.gang.enabled:
        gang_group = profile.gang.group_key or \
            pod.meta.labels.get(C.LabelWorkload, "") or pod.meta.name

    return AllocRequest(
        workload=pod.meta.labels.get(C.LabelWorkload, pod.meta.name),
        pod_name=pod.meta.name,
        namespace=pod.meta.namespace or "default",
        pool=pool or profile.pool,
        request=profile.resources.requests,
        limit=profile.resources.limits,
        gpu_count=profile.gpu_count,
        gpu_model=profile.gpu_model,
        gpu_vendor=profile.gpu_vendor,
        gpu_indices=profile.gpu_indices,
        qos=profile.qos,
        isolation_mode=profile.isolation_mode,
        partitioned=profile.isolation_mode == C.IsolationPartitioned,
        node_affinity=dict(pod.node_selector),
        gang_group=gang_group,
    )
