"""Control-plane tests: webhook, controllers, connection flow, defrag,
rollout, expander (reference test strategy: envtest-style with fake-GPU
CRs injected directly)."""
import time

import pytest

import tensor_fusion_amd.constants as C
from tensor_fusion_amd.api.store import Store
from tensor_fusion_amd.api.types import (GPU, Container, GPUNode, GPUPool,
                                         GPUResourceQuota, Node, Pod,
                                         Resource, TensorFusionCluster,
                                         TensorFusionConnection,
                                         TensorFusionWorkload,
                                         WorkloadProfile)
from tensor_fusion_amd.allocator.gpuallocator import GpuAllocator
from tensor_fusion_amd.controllers import (ControllerManager,
                                           DefragController,
                                           default_controllers,
                                           generate_worker_pod)
from tensor_fusion_amd.portallocator import IndexAllocator, PortAllocator
from tensor_fusion_amd.webhook import PodMutator


def mk_gpu(name, node="node-a", pool="pool-a", tflops=2500.0,
           vram=C.MI355X_VRAM_BYTES):
    g = GPU()
    g.meta.name = name
    g.status.uuid = f"uuid-{name}"
    g.status.node = node
    g.status.pool = pool
    g.status.capacity = Resource(tflops, vram, 100.0)
    g.status.available = Resource(tflops, vram, 100.0)
    return g


def mk_client_pod(name="app-1", ns="default", annotations=None):
    p = Pod()
    p.meta.name = name
    p.meta.namespace = ns
    p.meta.labels[C.LabelEnabled] = "true"
    p.meta.annotations.update(annotations or {})
    p.containers = [Container(name="main", image="app:1")]
    return p


# ----------------------------------------------------------- webhook


class TestWebhook:
    def test_remote_mode_creates_workload_and_injects_client(self):
        store = Store()
        m = PodMutator(store)
        pod = mk_client_pod(annotations={
            C.AnnoTflopsRequest: "600", C.AnnoVramRequest: "16Gi"})
        m.handle(pod)
        wl = store.get("TensorFusionWorkload", "app-1-wl", "default")
        assert wl.profile.resources.requests.tflops == 600
        assert wl.profile.resources.requests.vram == 16 << 30
        env = pod.containers[0].env
        assert env[C.EnvConnectionName] == "app-1-conn"
        assert C.ClientLibName in env["LD_PRELOAD"]

    def test_local_mode_sets_scheduler_and_limiter_env(self):
        store = Store()
        idx = IndexAllocator()
        m = PodMutator(store, index_allocator=idx)
        pod = mk_client_pod(annotations={
            C.AnnoIsLocalGpu: "true",
            C.AnnoComputePercentRequest: "25",
            C.AnnoVramLimit: "8Gi"})
        m.handle(pod)
        assert pod.scheduler_name == C.SchedulerName
        env = pod.containers[0].env
        assert env[C.EnvVramLimit] == str(8 << 30)
        assert env[C.EnvUpLimitPercent] == "25"
        assert C.LimiterLibName in env["LD_PRELOAD"]
        assert any(k.startswith(C.IndexResourcePrefix)
                   for k in pod.containers[0].resources)

    def test_auto_migration_of_plain_gpu_pod(self):
        store = Store()
        m = PodMutator(store)
        pod = Pod()
        pod.meta.name = "legacy"
        pod.meta.namespace = "default"
        pod.containers = [Container(name="main",
                                    resources={"amd.com/gpu": "2"})]
        assert m.should_handle(pod)
        m.handle(pod)
        assert "amd.com/gpu" not in pod.containers[0].resources
        assert pod.scheduler_name == C.SchedulerName  # local mode
        prof = m.parse(pod)
        assert prof.gpu_count == 2

    def test_idempotent(self):
        store = Store()
        m = PodMutator(store)
        pod = mk_client_pod(annotations={C.AnnoTflopsRequest: "100"})
        m.handle(pod)
        env1 = dict(pod.containers[0].env)
        n_mounts = len(pod.containers[0].volume_mounts)
        m.handle(pod)
        assert pod.containers[0].env == env1
        # mounts may duplicate — they must not (regression guard)
        assert len(pod.containers[0].volume_mounts) <= n_mounts + 1

    def test_qos_derived_from_fraction(self):
        store = Store()
        m = PodMutator(store)
        pod = mk_client_pod(annotations={C.AnnoComputePercentRequest: "80"})
        assert m.parse(pod).qos == C.QosHigh
        pod2 = mk_client_pod(name="b",
                             annotations={C.AnnoComputePercentRequest: "30"})
        assert m.parse(pod2).qos == C.QosMedium


# -------------------------------------------------------- controllers


class TestControllers:
    def _mk_world(self):
        store = Store()
        mgr = ControllerManager(store)
        for ctrl in default_controllers(store):
            mgr.register(ctrl)
        return store, mgr

    def test_cluster_creates_pools(self):
        store, mgr = self._mk_world()
        cl = TensorFusionCluster()
        cl.meta.name = "c1"
        tmpl = GPUPool()
        tmpl.meta.name = "pool-a"
        cl.pools = [tmpl]
        store.create(cl)
        mgr.reconcile_now()
        pool = store.get("GPUPool", "pool-a")
        assert pool.cluster == "c1"
        assert store.get("TensorFusionCluster", "c1").status.pool_count == 1

    def test_node_to_gpunode_to_pool_capacity(self):
        store, mgr = self._mk_world()
        pool = GPUPool()
        pool.meta.name = "pool-a"
        store.create(pool)
        node = Node()
        node.meta.name = "node-a"
        store.create(node)
        mgr.reconcile_now()
        gn = store.get("GPUNode", "node-a")
        assert gn.pool == "pool-a"
        for i in range(2):
            store.create(mk_gpu(f"g{i}"))
        mgr.reconcile_now()
        gn = store.get("GPUNode", "node-a")
        assert gn.status.gpu_count == 2
        assert gn.status.total.vram == 2 * C.MI355X_VRAM_BYTES
        pool = store.get("GPUPool", "pool-a")
        assert pool.status.gpu_count == 2
        assert not mgr.errors, mgr.errors[:1]  # no silent reconcile
        # failures (this exact assert caught a bad field name once)
        # oversell: 500% tflops, +50% vram
        assert pool.status.virtual_total.tflops == pytest.approx(2 * 2500 * 5)
        assert pool.status.virtual_total.vram == pytest.approx(
            2 * C.MI355X_VRAM_BYTES * 1.5)
        # hypervisor pod created per node
        hyp = store.get("Pod", "hypervisor-node-a", "tensor-fusion-sys")
        assert hyp.meta.labels[C.LabelComponent] == C.ComponentHypervisor

    def test_workload_scales_workers_and_connection_url(self):
        store, mgr = self._mk_world()
        wl = TensorFusionWorkload()
        wl.meta.name = "wl1"
        wl.meta.namespace = "default"
        wl.replicas = 2
        store.create(wl)
        mgr.reconcile_now()
        workers = [p for p in store.list("Pod", namespace="default")
                   if p.meta.labels.get(C.LabelComponent) == C.ComponentWorker]
        assert len(workers) == 2
        assert all(p.scheduler_name == C.SchedulerName for p in workers)

        # mark one worker scheduled → connection picks it
        def _sched(obj):
            obj.status.phase = "Scheduled"
            obj.status.pod_ip = "10.0.0.5"
        store.patch("Pod", workers[0].meta.name, "default", _sched)
        conn = TensorFusionConnection()
        conn.meta.name = "c1"
        conn.meta.namespace = "default"
        conn.workload = "wl1"
        store.create(conn)
        mgr.reconcile_now()
        conn = store.get("TensorFusionConnection", "c1", "default")
        assert conn.status.connection_url.startswith("native+10.0.0.5+8000+")
        assert conn.status.phase == "Ready"

        # scale down to 1
        def _scale(obj):
            obj.replicas = 1
        store.patch("TensorFusionWorkload", "wl1", "default", _scale)
        mgr.reconcile_now()
        workers = [p for p in store.list("Pod", namespace="default")
                   if p.meta.labels.get(C.LabelComponent) == C.ComponentWorker]
        assert len(workers) == 1

    def test_client_pod_gets_connection_created(self):
        store, mgr = self._mk_world()
        pod = mk_client_pod()
        m = PodMutator(store)
        m.handle(pod)
        store.create(pod)
        mgr.reconcile_now()
        conn = store.get("TensorFusionConnection", "app-1-conn", "default")
        assert conn.workload == "app-1-wl"

    def test_pod_delete_deallocs(self):
        store = Store()
        alloc = GpuAllocator(store=store)
        mgr = ControllerManager(store)
        for ctrl in default_controllers(store, allocator=alloc):
            mgr.register(ctrl)
        store.create(mk_gpu("g0"))
        from tensor_fusion_amd.api.types import AllocRequest
        req = AllocRequest(pod_name="w1", namespace="default",
                           request=Resource(100, 8 << 30, 10),
                           limit=Resource(100, 8 << 30, 10))
        alloc.assume(req, ["g0"])
        alloc.commit("default/w1")
        pod = Pod()
        pod.meta.name = "w1"
        pod.meta.namespace = "default"
        store.create(pod)
        store.delete("Pod", "w1", "default")
        assert alloc.allocation("default/w1") is None
        g = alloc.gpu("g0")
        assert g.status.available.vram == C.MI355X_VRAM_BYTES

    def test_worker_pod_annotations_carry_profile(self):
        wl = TensorFusionWorkload()
        wl.meta.name = "wl2"
        wl.meta.namespace = "ns1"
        wl.profile = WorkloadProfile()
        wl.profile.resources.requests = Resource(600, 48 << 30, 25)
        wl.profile.resources.limits = Resource(1200, 48 << 30, 50)
        wl.profile.gang.enabled = True
        wl.replicas = 4
        pod = generate_worker_pod(wl, 0)
        a = pod.meta.annotations
        assert float(a[C.AnnoTflopsRequest]) == 600.0
        assert a[C.AnnoGangEnabled] == "true"
        assert a[C.AnnoGangMinMembers] == "4"


# ------------------------------------------------------------- defrag


class TestDefrag:
    def _world(self):
        store = Store()
        alloc = GpuAllocator(store=store)
        for node in ("node-a", "node-b"):
            for i in range(2):
                store.create(mk_gpu(f"{node}-g{i}", node=node))
        return store, alloc

    def test_campaign_consolidates_underutilized_node(self):
        from tensor_fusion_amd.api.types import AllocRequest
        store, alloc = self._world()
        # node-a nearly full, node-b one small pod
        req_big = AllocRequest(pod_name="big", namespace="d",
                               request=Resource(2000, 200 << 30, 80),
                               limit=Resource(2000, 200 << 30, 80))
        alloc.assume(req_big, ["node-a-g0"])
        alloc.commit("d/big")
        req_small = AllocRequest(pod_name="small", namespace="d",
                                 request=Resource(100, 8 << 30, 5),
                                 limit=Resource(100, 8 << 30, 5))
        alloc.assume(req_small, ["node-b-g0"])
        alloc.commit("d/small")
        pod = Pod()
        pod.meta.name = "small"
        pod.meta.namespace = "d"
        store.create(pod)

        d = DefragController(store, alloc, utilization_threshold=0.3,
                             eviction_ttl_s=0.0, campaign_cooldown_s=0.0)
        plan = d.run_campaign()
        assert plan is not None
        assert plan.candidate_nodes == ["node-b"]
        assert plan.evict_pods == ["d/small"]
        marked = store.get("Pod", "small", "d")
        from tensor_fusion_amd.controllers.defrag import AnnoEvictionMark
        assert AnnoEvictionMark in marked.meta.annotations
        evicted = d.execute_due_evictions(now=time.time() + 1)
        assert evicted == ["d/small"]

    def test_no_campaign_when_everything_busy(self):
        from tensor_fusion_amd.api.types import AllocRequest
        store, alloc = self._world()
        for node in ("node-a", "node-b"):
            req = AllocRequest(pod_name=f"p-{node}", namespace="d",
                               request=Resource(2000, 200 << 30, 80),
                               limit=Resource(2000, 200 << 30, 80))
            alloc.assume(req, [f"{node}-g0"])
            alloc.commit(f"d/p-{node}")
        d = DefragController(store, alloc, utilization_threshold=0.3,
                             campaign_cooldown_s=0.0)
        assert d.run_campaign() is None


# ------------------------------------------------------------ rollout


class TestRollout:
    def test_batch_rolling_update(self):
        from tensor_fusion_amd.component import ComponentRollout
        store = Store()
        tmpl_v1 = {"image": "worker:v1"}
        roll = ComponentRollout(store, C.ComponentWorker, batch_percent=50,
                                interval_s=0.0)
        for i in range(4):
            p = Pod()
            p.meta.name = f"w{i}"
            p.meta.namespace = "d"
            p.meta.labels[C.LabelComponent] = C.ComponentWorker
            roll.stamp(p, tmpl_v1)
            store.create(p)
        assert roll.out_of_date(tmpl_v1) == []
        tmpl_v2 = {"image": "worker:v2"}
        assert len(roll.out_of_date(tmpl_v2)) == 4
        deleted = roll.tick(tmpl_v2)
        assert len(deleted) == 2  # 50% batch
        # batch is computed from the remaining pod count (2 left -> 1)
        roll._last_batch_ts = 0.0
        assert len(roll.tick(tmpl_v2)) == 1
        roll._last_batch_ts = 0.0
        assert len(roll.tick(tmpl_v2)) == 1
        roll._last_batch_ts = 0.0
        assert roll.tick(tmpl_v2) == []


# --------------------------------------------------- expander/provider


class TestExpander:
    def test_unschedulable_creates_claim_then_node_joins(self):
        from tensor_fusion_amd.api.types import AllocRequest
        from tensor_fusion_amd.cloudprovider import MockProvider
        from tensor_fusion_amd.controllers import (ControllerManager,
                                                   NodeClaimReconciler)
        from tensor_fusion_amd.scheduler.expander import NodeExpander
        store = Store()
        provider = MockProvider(polls_until_ready=1, store=store)
        mgr = ControllerManager(store)
        mgr.register(NodeClaimReconciler(store, provider=provider))
        exp = NodeExpander(store, cooldown_s=0.0)
        req = AllocRequest(pod_name="p1", namespace="d", gpu_count=2,
                           request=Resource(1000, 100 << 30, 50),
                           limit=Resource(1000, 100 << 30, 50))
        claim_name = exp.handle_unschedulable(req)
        assert claim_name
        claim = store.get("GPUNodeClaim", claim_name)
        assert claim.instance_type == "mi355x.2g"
        # second call rides the same claim
        assert exp.handle_unschedulable(req) == claim_name
        mgr.reconcile_now()
        # claim reconciler drives Pending→Creating→Bound (requeue ticks)
        deadline = time.time() + 5
        while time.time() < deadline:
            mgr.reconcile_now()
            if store.get("GPUNodeClaim", claim_name).status.phase == "Bound":
                break
            time.sleep(0.05)
        claim = store.get("GPUNodeClaim", claim_name)
        assert claim.status.phase == "Bound"
        assert store.try_get("Node", claim.status.node_name) is not None

    def test_cheapest_instance(self):
        from tensor_fusion_amd.cloudprovider import cheapest_instance_for
        it = cheapest_instance_for(1, 2000, 100 << 30)
        assert it.name == "mi355x.1g"
        assert cheapest_instance_for(16, 0, 0) is None


class TestProviderManager:
    def test_partition_templates_hot_reload_into_allocator(self):
        from tensor_fusion_amd.api.types import (ProviderConfig,
                                                 default_mi355x_partition_templates)
        store = Store()
        alloc = GpuAllocator(store=store)
        mgr = ControllerManager(store)
        for ctrl in default_controllers(store, allocator=alloc):
            mgr.register(ctrl)
        pc = ProviderConfig()
        pc.meta.name = "amd"
        pc.partition_templates = default_mi355x_partition_templates()
        store.create(pc)
        mgr.reconcile_now()
        assert len(alloc.partition_templates) == 4
        ids = {t.id for t in alloc.partition_templates}
        assert ids == {"xcd1", "xcd2", "xcd4", "xcd8"}


class TestSchedulingConfig:
    def test_placement_mode_hot_swap(self):
        from tensor_fusion_amd.api.types import SchedulingConfigTemplate
        store = Store()
        alloc = GpuAllocator(store=store)
        mgr = ControllerManager(store)
        for ctrl in default_controllers(store, allocator=alloc):
            mgr.register(ctrl)
        assert type(alloc.strategy).__name__ == "NodeCompactGPULowLoad"
        tpl = SchedulingConfigTemplate()
        tpl.meta.name = "t1"
        tpl.placement_mode = "LowLoadFirst"
        tpl.vram_weight = 0.5
        store.create(tpl)
        mgr.reconcile_now()
        assert type(alloc.strategy).__name__ == "LowLoadFirst"
        assert alloc.strategy.vram_weight == 0.5


class TestGreyRelease:
    def test_adoption_percent_gates_plain_gpu_pods(self):
        store = Store()
        m = PodMutator(store, adoption_percent=0)
        pod = Pod()
        pod.meta.name = "legacy2"
        pod.meta.namespace = "d"
        pod.containers = [Container(name="m", resources={"amd.com/gpu": "1"})]
        assert m.should_handle(pod) is False
        assert m.counters["skipped_grey"] == 1
        # explicit opt-in bypasses the grey gate
        pod.meta.labels[C.LabelEnabled] = "true"
        assert m.should_handle(pod) is True
        m100 = PodMutator(store, adoption_percent=100)
        pod2 = Pod()
        pod2.meta.name = "legacy3"
        pod2.meta.namespace = "d"
        pod2.containers = [Container(name="m", resources={"amd.com/gpu": "1"})]
        assert m100.should_handle(pod2) is True


class TestDefragMigration:
    def test_campaign_live_migrates_instead_of_evicting(self):
        from tensor_fusion_amd.api.types import AllocRequest
        from tensor_fusion_amd.controllers.defrag import AnnoEvictionMark
        store = Store()
        alloc = GpuAllocator(store=store)
        for node in ("node-a", "node-b"):
            for i in range(2):
                store.create(mk_gpu(f"{node}-g{i}", node=node))
        req_big = AllocRequest(pod_name="big", namespace="d",
                               request=Resource(2000, 200 << 30, 80),
                               limit=Resource(2000, 200 << 30, 80))
        alloc.assume(req_big, ["node-a-g0"])
        alloc.commit("d/big")
        req_small = AllocRequest(pod_name="small", namespace="d",
                                 request=Resource(100, 8 << 30, 5),
                                 limit=Resource(100, 8 << 30, 5))
        alloc.assume(req_small, ["node-b-g0"])
        alloc.commit("d/small")
        pod = Pod()
        pod.meta.name = "small"
        pod.meta.namespace = "d"
        store.create(pod)

        moves = []

        def migrate(pod_key, target_gpus):
            moves.append((pod_key, list(target_gpus)))
            return True

        d = DefragController(store, alloc, utilization_threshold=0.3,
                             campaign_cooldown_s=0.0, migrate_fn=migrate)
        plan = d.run_campaign()
        assert plan is not None
        assert moves and moves[0][0] == "d/small"
        # allocation moved to the target node, no eviction mark
        new_alloc = alloc.allocation("d/small")
        assert new_alloc is not None
        assert all(g.startswith("node-a") for g in new_alloc.gpu_names)
        assert AnnoEvictionMark not in store.get("Pod", "small",
                                                 "d").meta.annotations
        # node-b is now empty
        for g in alloc.gpus(node="node-b"):
            assert g.status.available.vram == C.MI355X_VRAM_BYTES


class TestConnectionFailover:
    def test_worker_loss_reselects_and_updates_url(self):
        store = Store()
        mgr = ControllerManager(store)
        for ctrl in default_controllers(store):
            mgr.register(ctrl)
        from tensor_fusion_amd.api.types import (TensorFusionConnection,
                                                 TensorFusionWorkload)
        wl = TensorFusionWorkload()
        wl.meta.name = "wl1"
        wl.meta.namespace = "d"
        wl.replicas = 2
        store.create(wl)
        mgr.reconcile_now()
        workers = sorted(
            (p for p in store.list("Pod", namespace="d")
             if p.meta.labels.get(C.LabelComponent) == C.ComponentWorker),
            key=lambda p: p.meta.name)
        assert len(workers) == 2
        for i, w in enumerate(workers):
            def _s(obj, ip=f"10.0.0.{i+1}"):
                obj.status.phase = "Running"
                obj.status.pod_ip = ip
            store.patch("Pod", w.meta.name, "d", _s)
        conn = TensorFusionConnection()
        conn.meta.name = "c1"
        conn.meta.namespace = "d"
        conn.workload = "wl1"
        store.create(conn)
        mgr.reconcile_now()
        conn = store.get("TensorFusionConnection", "c1", "d")
        first_worker = conn.status.worker
        assert first_worker

        # the selected worker dies → controller must fail over
        def _fail(obj):
            obj.status.phase = "Failed"
        store.patch("Pod", first_worker, "d", _fail)
        mgr.reconcile_now()
        conn = store.get("TensorFusionConnection", "c1", "d")
        assert conn.status.worker != first_worker
        assert conn.status.worker
        ip = conn.status.connection_url.split("+")[1]
        assert ip.startswith("10.0.0.")


def test_node_deletion_tears_down_gpunode_and_inventory():
    """Node failure path (reference gpunode_controller.go:376): deleting
    the k8s Node removes its GPUNode and GPU CRs, and the allocator
    stops offering that node's capacity."""

    from tensor_fusion_amd.operator import build_operator

    op = build_operator()
    pool = GPUPool()
    pool.meta.name = "pool-a"
    op.store.create(pool)
    node = Node()
    node.meta.name = "dying-node"
    op.store.create(node)
    op.tick()
    # publish two GPUs on the node (as the hypervisor backend would)
    for i in range(2):
        g = GPU()
        g.meta.name = f"dying-node-gpu-{i}"
        g.status.uuid = g.meta.name
        g.status.node = "dying-node"
        g.status.pool = "pool-a"
        g.status.capacity = Resource(2500.0, 288 << 30, 100.0)
        g.status.available = Resource(2500.0, 288 << 30, 100.0)
        op.store.create(g)
    op.tick()
    assert len(op.allocator.gpus(node="dying-node")) == 2

    op.store.delete("Node", "dying-node")
    op.tick()
    assert op.store.try_get("GPUNode", "dying-node") is None
    assert [g for g in op.store.list("GPU")
            if g.status.node == "dying-node"] == []
    assert op.allocator.gpus(node="dying-node") == []


def test_owner_reference_gc_cascades():
    """Store-level owner GC (kube garbage collector semantics): deleting
    an owner deletes transitively-owned objects — Node → GPUNode →
    hypervisor Pod in one cascade."""

    from tensor_fusion_amd.operator import build_operator

    op = build_operator()
    pool = GPUPool()
    pool.meta.name = "pool-a"
    op.store.create(pool)
    node = Node()
    node.meta.name = "gc-node"
    op.store.create(node)
    op.tick()
    gn = op.store.try_get("GPUNode", "gc-node")
    assert gn is not None and gn.meta.owner == "Node//gc-node"
    op.tick()
    # driver probe gates the hypervisor rollout (gpunode_controller.go
    # :790-863): no hypervisor pod until the probe pod Succeeds
    probe = op.store.try_get("Pod", "driver-probe-gc-node",
                             "tensor-fusion-sys")
    assert probe is not None and probe.meta.owner == "GPUNode//gc-node"
    assert op.store.try_get("Pod", "hypervisor-gc-node",
                            "tensor-fusion-sys") is None

    def _ok(obj):
        obj.status.phase = "Succeeded"
        obj.meta.annotations["tensor-fusion.ai/rocm-version"] = "7.2.0"
    op.store.patch("Pod", "driver-probe-gc-node", "tensor-fusion-sys", _ok)
    op.tick()
    hyp = op.store.try_get("Pod", "hypervisor-gc-node", "tensor-fusion-sys")
    assert hyp is not None and hyp.meta.owner == "GPUNode//gc-node"
    assert op.store.get("GPUNode", "gc-node").status.rocm_version == "7.2.0"

    op.store.delete("Node", "gc-node")
    assert op.store.try_get("GPUNode", "gc-node") is None
    assert op.store.try_get("Pod", "hypervisor-gc-node",
                            "tensor-fusion-sys") is None


def test_fast_deepcopy_semantics():
    """TFObject.__deepcopy__ (structure-aware) must match copy.deepcopy
    exactly: full independence, asdict equality, nested containers."""

    import copy
    from dataclasses import asdict

    from tensor_fusion_amd.api.types import (GPU, Pod, Resource,
                                             fast_deepcopy)
    g = GPU()
    g.meta.name = "g0"
    g.meta.labels = {"a": "1"}
    g.status.capacity = Resource(2500.0, 288 << 30, 100.0)
    g.status.running_apps = [{"name": "w1", "namespace": "ns"}]
    c = copy.deepcopy(g)  # routes through __deepcopy__
    assert c == g and asdict(c) == asdict(g)
    assert c.status is not g.status
    assert c.status.running_apps is not g.status.running_apps
    assert c.status.running_apps[0] is not g.status.running_apps[0]
    c.status.running_apps[0]["name"] = "w2"
    assert g.status.running_apps[0]["name"] == "w1"
    c.meta.labels["a"] = "2"
    assert g.meta.labels["a"] == "1"
    # helper form used by the store
    p = Pod()
    p.meta.annotations = {"k": "v"}
    q = fast_deepcopy(p)
    assert q == p and q.meta.annotations is not p.meta.annotations


def test_node_legacy_taint_removed():
    """NodeReconciler strips the deprecated
    tensor-fusion.ai/used-by=tensor-fusion:NoSchedule taint from nodes
    (reference node_controller.go:300); unrelated taints survive and
    deleting nodes are left alone."""

    import tensor_fusion_amd.constants as C
    from tensor_fusion_amd.api.store import Store
    from tensor_fusion_amd.api.types import Node
    from tensor_fusion_amd.controllers.core import NodeReconciler, Request

    store = Store()
    rec = NodeReconciler(store)
    n = Node()
    n.meta.name = "n0"
    n.taints = [
        {"key": C.NodeUsedByTaintKey, "value": C.TensorFusionSystemName,
         "effect": "NoSchedule"},
        {"key": "other.io/maintenance", "value": "x",
         "effect": "NoExecute"},
    ]
    store.create(n)
    rec.reconcile(Request("Node", "n0", ""))
    got = store.get("Node", "n0")
    assert [t["key"] for t in got.taints] == ["other.io/maintenance"]

    # deleting node: untouched
    n2 = Node()
    n2.meta.name = "n1"
    n2.meta.deletion_ts = 1.0
    n2.taints = [{"key": C.NodeUsedByTaintKey,
                  "value": C.TensorFusionSystemName,
                  "effect": "NoSchedule"}]
    store.create(n2)
    rec.reconcile(Request("Node", "n1", ""))
    assert len(store.get("Node", "n1").taints) == 1

    # serde: taints survive the corev1 round trip
    from tensor_fusion_amd.k8s import serde
    wire = serde.node_to_k8s(got)
    assert wire["spec"]["taints"] == got.taints
    back = serde.node_from_k8s(wire)
    assert back.taints == got.taints
