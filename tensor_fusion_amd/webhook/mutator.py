"""Pod admission mutator — the stack's front door.

Reference: internal/webhook/v1/pod_webhook.go:84 (Handle), tf_parser.go
(ParseTensorFusionInfo), utils/compose.go:328-512 (client injection),
auto_migration.go (adopting plain GPU pods). A pod opts in with the
`tensor-fusion.ai/enabled=true` label (or by requesting a plain
`amd.com/gpu`, which auto-migration converts); the mutator

  1. parses annotations over an optional WorkloadProfile base,
  2. derives the QoS level (pod_webhook.go:787 calculateQoSLevel),
  3. creates/updates a TensorFusionWorkload CR (remote mode) so the
     workload controller can spawn worker pods,
  4. patches the client pod: scheduler name (local mode), limiter/client
     env, LD_PRELOAD preload mount, index placeholder resource, strips
     plain-GPU resource requests,
and is idempotent: re-admitting a mutated pod is a no-op.
"""
from __future__ import annotations

from typing import Optional

from .. import constants as C
from ..api.store import AlreadyExists, Store
from ..api.types import (Pod, TensorFusionWorkload, WorkloadProfile)
from ..utils.resource import profile_from_annotations

PLAIN_GPU_RESOURCES = ("amd.com/gpu", "nvidia.com/gpu")

# QoS inferred from requests when unset (reference calculateQoSLevel):
# bigger fractions of a device → higher default QoS.
_QOS_BY_FRACTION = ((0.75, C.QosHigh), (0.25, C.QosMedium), (0.0, C.QosLow))


class PodMutator:
    def __init__(self, store: Store, index_allocator=None, port_allocator=None,
                 adoption_percent: int = 100):
        self.store = store
        self.index_allocator = index_allocator
        self.port_allocator = port_allocator
        # grey release (reference pod_counter.go): only N % of eligible
        # plain-GPU pods are auto-migrated; explicit opt-ins always pass
        self.adoption_percent = adoption_percent
        self.counters = {"seen": 0, "handled": 0, "skipped_grey": 0}

    # ------------------------------------------------------------ gating

    def should_handle(self, pod: Pod) -> bool:
        self.counters["seen"] += 1
        # never re-mutate the stack's own pods: on a real cluster worker/
        # hypervisor pods pass through admission too (the controllers
        # create them via the apiserver) and a worker inheriting the
        # enabled label must not be turned into a client pod
        # (reference pod_webhook.go skips component-labelled pods)
        if pod.meta.labels.get(C.LabelComponent) in (
                C.ComponentWorker, C.ComponentHypervisor,
                C.ComponentOperator):
            return False
        if pod.meta.labels.get(C.LabelEnabled) == "true":
            return True
        if not self.should_auto_migrate(pod):
            return False
        if self.adoption_percent < 100:
            import zlib
            bucket = zlib.crc32(pod.meta.key.encode()) % 100
            if bucket >= self.adoption_percent:
                self.counters["skipped_grey"] += 1
                return False
        return True

    def should_auto_migrate(self, pod: Pod) -> bool:
        """Adopt pods that request a plain GPU extended resource
        (reference auto_migration.go ShouldAutoMigrateGPUPod)."""

        for c in pod.containers:
            for r in PLAIN_GPU_RESOURCES:
                if r in c.resources:
                    return True
        return False

    # ----------------------------------------------------------- parsing

    def parse(self, pod: Pod) -> WorkloadProfile:
        base: Optional[WorkloadProfile] = None
        prof_name = pod.meta.annotations.get(C.AnnoWorkloadProfile)
        if prof_name:
            obj = self.store.try_get("WorkloadProfile", prof_name,
                                     pod.meta.namespace)
            if obj is None:
                raise ValueError(f"WorkloadProfile {prof_name} not found")
            base = obj
        profile = profile_from_annotations(pod, base)
        if self.should_auto_migrate(pod) and profile.gpu_count == 1:
            n = 0
            for c in pod.containers:
                for r in PLAIN_GPU_RESOURCES:
                    if r in c.resources:
                        n += int(float(c.resources[r]))
            profile.gpu_count = max(profile.gpu_count, n or 1)
            # plain-GPU pods get whole devices
            if profile.resources.requests.compute_percent == 0:
                profile.resources.requests.compute_percent = 100.0
                profile.resources.limits.compute_percent = 100.0
            if profile.resources.requests.vram == 0:
                profile.resources.requests.vram = C.MI355X_VRAM_BYTES
                profile.resources.limits.vram = C.MI355X_VRAM_BYTES
            profile.is_local_gpu = True
        if not pod.meta.annotations.get(C.AnnoQos):
            profile.qos = self.calculate_qos(profile)
        self._apply_recommendation(pod, profile)
        return profile

    def _apply_recommendation(self, pod: Pod, profile: WorkloadProfile):
        """Autoscaler feedback at admission (reference webhook :349):
        when the pod's workload carries a status.recommendation and the
        user did NOT pin resources explicitly via annotations, admit
        with the recommended requests/limits so new replicas start at
        the learned size."""

        wl_name = pod.meta.labels.get(C.LabelWorkload)
        if not wl_name and pod.meta.annotations.get(C.AnnoAutoscale):
            wl_name = f"{pod.meta.name}-wl"
        if not wl_name:
            return
        wl = self.store.try_get("TensorFusionWorkload", wl_name,
                                pod.meta.namespace)
        rec = getattr(getattr(wl, "status", None), "recommendation", None)
        if not rec or not rec.resources.requests.vram and \
                not rec.resources.requests.tflops:
            return
        a = pod.meta.annotations
        req, lim = rec.resources.requests, rec.resources.limits
        if C.AnnoTflopsRequest not in a and req.tflops:
            profile.resources.requests.tflops = req.tflops
            profile.resources.limits.tflops = lim.tflops or req.tflops
        if C.AnnoVramRequest not in a and req.vram:
            profile.resources.requests.vram = req.vram
            profile.resources.limits.vram = lim.vram or req.vram

    @staticmethod
    def calculate_qos(profile: WorkloadProfile) -> str:
        req = profile.resources.requests
        frac = max(req.compute_percent / 100.0,
                   req.tflops / C.MI355X_BF16_TFLOPS if C.MI355X_BF16_TFLOPS else 0,
                   req.vram / C.MI355X_VRAM_BYTES)
        for lo, qos in _QOS_BY_FRACTION:
            if frac >= lo:
                return qos
        return C.QosLow

    # ---------------------------------------------------------- mutation

    def handle(self, pod: Pod) -> Pod:
        """Mutate in place and return the pod (the admission response)."""

        if not self.should_handle(pod):
            return pod
        self.counters["handled"] += 1
        profile = self.parse(pod)
        # persist parsed values so re-admission (and controllers) see them
        # even after plain-GPU resources are stripped below
        a = pod.meta.annotations
        a[C.AnnoGpuCount] = str(profile.gpu_count)
        a.setdefault(C.AnnoComputePercentRequest,
                     str(profile.resources.requests.compute_percent))
        a.setdefault(C.AnnoComputePercentLimit,
                     str(profile.resources.limits.compute_percent))
        a.setdefault(C.AnnoVramRequest, str(int(profile.resources.requests.vram)))
        a.setdefault(C.AnnoVramLimit, str(int(profile.resources.limits.vram)))
        a.setdefault(C.AnnoQos, profile.qos)
        if profile.is_local_gpu:
            a.setdefault(C.AnnoIsLocalGpu, "true")
        pod.meta.labels[C.LabelEnabled] = "true"
        pod.meta.labels[C.LabelManaged] = "tensor-fusion"

        workload_name = pod.meta.labels.get(
            C.LabelWorkload) or f"{pod.meta.name}-wl"
        pod.meta.labels[C.LabelWorkload] = workload_name

        if profile.is_local_gpu:
            self._patch_local(pod, profile)
        else:
            self._ensure_workload(pod, workload_name, profile)
            self._patch_client(pod, workload_name, profile)
        self._assign_host_port(pod)
        return pod

    def _ensure_workload(self, pod: Pod, name: str, profile: WorkloadProfile):
        """Create/update the TensorFusionWorkload backing this client pod
        (reference pod_webhook.go:284 createOrUpdateWorkload)."""

        ns = pod.meta.namespace
        existing = self.store.try_get("TensorFusionWorkload", name, ns)
        if existing is None:
            wl = TensorFusionWorkload()
            wl.meta.name = name
            wl.meta.namespace = ns
            wl.meta.labels[C.LabelComponent] = C.ComponentWorker
            wl.profile = profile
            wl.pool = profile.pool
            wl.replicas = max(1, profile.gpu_count
                              if profile.gang.enabled else 1)
            try:
                self.store.create(wl)
            except AlreadyExists:
                pass
        else:
            # skip the no-op patch entirely when the profile is unchanged:
            # re-admissions of replica pods hit this path per pod, and the
            # store's no-op detection still costs two full asdict() walks
            # — comparing the profiles here is ~6x cheaper (webhook QPS)
            if existing.profile == profile:
                return

            def _p(obj):
                obj.profile = profile
            self.store.patch("TensorFusionWorkload", name, ns, _p)

    def _patch_local(self, pod: Pod, profile: WorkloadProfile):
        """Local mode: the pod itself runs on the GPU under the limiter."""

        pod.scheduler_name = C.SchedulerName
        res = profile.resources
        for c in self._target_containers(pod):
            for r in PLAIN_GPU_RESOURCES:
                c.resources.pop(r, None)
            c.env.update({
                C.EnvIsolationMode: profile.isolation_mode,
                C.EnvShmPath: f"{C.ShmRoot}/shm",
                C.EnvPodNamespace: pod.meta.namespace,
                C.EnvPodName: pod.meta.name,
                C.EnvContainerName: c.name,
                C.EnvHypervisorPort: str(C.HypervisorHTTPPort),
                C.EnvVramLimit: str(int(res.limits.vram) or C.MI355X_VRAM_BYTES),
                C.EnvUpLimitPercent: str(int(
                    res.limits.compute_percent or 100)),
                "LD_PRELOAD": f"/tensor-fusion/{C.LimiterLibName}",
            })
            c.volume_mounts.append(
                {"name": "tf-libs", "mountPath": "/tensor-fusion"})
        self._attach_index_resource(pod)
        # annotations the scheduler consumes
        a = pod.meta.annotations
        a.setdefault(C.AnnoQos, profile.qos)
        a.setdefault(C.AnnoIsolation, profile.isolation_mode)

    def _patch_client(self, pod: Pod, workload: str, profile: WorkloadProfile):
        """Remote mode: GPU-less client; stub dials the worker
        (reference compose.go:328 AddTFDefaultClientConfBeforePatch)."""

        conn_name = f"{pod.meta.name}-conn"
        for c in self._target_containers(pod):
            for r in PLAIN_GPU_RESOURCES:
                c.resources.pop(r, None)
            c.env.update({
                C.EnvConnectionName: conn_name,
                C.EnvConnectionNamespace: pod.meta.namespace,
                C.EnvOperatorEndpoint:
                    f"http://tensor-fusion-operator:{C.OperatorHTTPPort}",
                C.EnvPodNamespace: pod.meta.namespace,
                C.EnvPodName: pod.meta.name,
                "TF_PREPEND_PATH": "/tensor-fusion",
                "TF_LD_LIBRARY_PATH": "/tensor-fusion",
                "LD_PRELOAD": f"/tensor-fusion/{C.ClientLibName}",
            })
            c.volume_mounts.append(
                {"name": "tf-libs", "mountPath": "/tensor-fusion"})
        pod.meta.labels[C.LabelComponent] = C.ComponentClient

    def _target_containers(self, pod: Pod):
        only = pod.meta.annotations.get(C.AnnoInjectContainer)
        if only:
            names = {x.strip() for x in only.split(",")}
            return [c for c in pod.containers if c.name in names]
        return pod.containers

    def _attach_index_resource(self, pod: Pod):
        """Placeholder extended resource tensor-fusion.ai/index-N so the
        device plugin can correlate container→pod (reference
        indexallocator.go:29, compose.go:499)."""

        if self.index_allocator is None:
            return
        if any(k.startswith(C.IndexResourcePrefix)
               for c in pod.containers for k in c.resources):
            return  # idempotent
        idx = self.index_allocator.assign(pod.meta.key)
        if pod.containers:
            pod.containers[0].resources[f"{C.IndexResourcePrefix}{idx}"] = "1"
        pod.meta.annotations[C.AnnoPodIndex] = str(idx)

    def _assign_host_port(self, pod: Pod):
        if self.port_allocator is None:
            return
        if pod.meta.annotations.get(C.AnnoHostPort) == "auto" and \
                not pod.meta.annotations.get(C.AnnoHostPortAssigned):
            port = self.port_allocator.assign_cluster_port(pod.meta.key)
            pod.meta.annotations[C.AnnoHostPortAssigned] = str(port)
