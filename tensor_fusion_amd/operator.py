"""Operator entry — wires the whole control plane in one process.

Reference: cmd/main.go:131-297 (manager wiring: provider manager, metrics
recorder, index+GPU+port allocators, autoscaler, webhook, scheduler, 16
controllers, client HTTP server, TSDB+alerts). `build_operator()` returns
the assembled Operator for embedding (tests, single-node runtime);
`python -m tensor_fusion_amd.operator` runs it standalone against the
embedded store with optional persistence.
"""
from __future__ import annotations

import argparse
import threading
import time
from dataclasses import dataclass, field
from typing import List, Optional

from . import constants as C
from .alert import AlertEvaluator
from .allocator.gpuallocator import GpuAllocator
from .api.store import Store
from .autoscaler import Autoscaler
from .cloudprovider import MockProvider
from .config import ConfigWatcher
from .controllers import ControllerManager, default_controllers
from .controllers.defrag import DefragController
from .gang.manager import GangManager
from .metrics import MetricsRecorder, PoolMetrics, TSDB
from .portallocator import IndexAllocator, PortAllocator
from .quota.quota_store import QuotaStore
from .scheduler.expander import NodeExpander
from .scheduler.framework import Scheduler
from .scheduler.gpuresources import GPUResourcesFit
from .scheduler.gputopo import GPUNetworkTopologyAware
from .webhook import PodMutator


@dataclass
class Operator:
    store: Store
    allocator: GpuAllocator
    quota: QuotaStore
    gang: GangManager
    scheduler: Scheduler
    mutator: PodMutator
    controllers: ControllerManager
    autoscaler: Autoscaler
    metrics: MetricsRecorder
    tsdb: TSDB
    alerts: AlertEvaluator
    port_allocator: PortAllocator
    index_allocator: IndexAllocator
    expander: NodeExpander
    defrag: DefragController
    config: Optional[ConfigWatcher] = None
    _threads: List[threading.Thread] = field(default_factory=list)
    _stop: threading.Event = field(default_factory=threading.Event)
    _last_rebalance: float = 0.0

    # ------------------------------------------------------ admission

    def admit(self, pod):
        """Webhook path: mutate + persist the pod (the kube-apiserver's
        admission hook, embedded)."""

        self.mutator.handle(pod)
        existing = self.store.try_get("Pod", pod.meta.name,
                                      pod.meta.namespace)
        if existing is None:
            self.store.create(pod)
        else:
            pod.meta.resource_version = existing.meta.resource_version
            self.store.update(pod)
        return pod

    def recover(self) -> int:
        """Operator-restart recovery: rebuild the allocator's committed
        state from scheduled worker pods' gpu-ids annotations (reference
        reconcileAllocationState gpuallocator.go:2906)."""

        from .utils.resource import (compose_allocation_request,
                                     profile_from_annotations)
        records = []
        for pod in self.store.list("Pod"):
            gpus = pod.meta.annotations.get(C.AnnoContainerGpus)
            if not gpus or not pod.status.node:
                continue
            if pod.meta.labels.get(C.LabelComponent) not in (
                    C.ComponentWorker, None) and \
                    not pod.meta.annotations.get(C.AnnoIsLocalGpu):
                continue
            try:
                profile = profile_from_annotations(pod)
                req = compose_allocation_request(pod, profile)
            except (ValueError, KeyError):
                continue
            records.append((req, gpus.split(",")))
        self.allocator.reconcile_from_allocations(records)
        return len(records)

    # -------------------------------------------------------- lifecycle

    def tick(self):
        """One deterministic pass of every loop (tests + single-node)."""

        self.controllers.reconcile_now()
        self.scheduler.schedule_pending()
        self.controllers.reconcile_now()
        self.allocator.sync_dirty()
        self.quota.sync_dirty()
        self.autoscaler.tick()
        self._maybe_rebalance()
        self._record_pool_metrics()
        self.metrics.flush()
        self.alerts.evaluate()

    def _maybe_rebalance(self, now: Optional[float] = None):
        """Periodic re-balancer (reference SchedulingConfigTemplate
        :241): when a pool enables defrag, or a scheduling template
        sets rebalanceIntervalS, run a defrag campaign on that cadence
        (DefragController applies its own campaign cooldown on top)."""

        import time as _t
        now = now if now is not None else _t.time()
        interval = 0
        for t in self.store.list("SchedulingConfigTemplate"):
            if t.rebalance_interval_s > 0:
                interval = (t.rebalance_interval_s if not interval
                            else min(interval, t.rebalance_interval_s))
        enabled = interval > 0 or any(
            p.node_manager.defrag_enabled
            for p in self.store.list("GPUPool"))
        if not enabled:
            return
        if interval and now - self._last_rebalance < interval:
            return
        self._last_rebalance = now
        try:
            self.defrag.run_campaign(now)
        except Exception:
            pass

    def _record_pool_metrics(self):
        for pool in self.store.list("GPUPool"):
            s = pool.status
            self.metrics.set_pool(PoolMetrics(
                pool=pool.meta.name, node_count=s.node_count,
                gpu_count=s.gpu_count, total_tflops=s.total.tflops,
                total_vram=s.total.vram,
                allocated_tflops=s.total.tflops - s.available.tflops,
                allocated_vram=s.total.vram - s.available.vram))

    def start(self, scheduler_interval_s: float = 0.5):
        self.controllers.start()
        self.autoscaler.start()
        self.metrics.start()
        self.alerts.start()
        if self.config:
            self.config.start()

        def sched_loop():
            while not self._stop.wait(scheduler_interval_s):
                try:
                    self._maybe_rebalance()
                    self.scheduler.schedule_pending()
                    self.allocator.sync_dirty()
                    self.quota.sync_dirty()
                    self.gang.sweep_timeouts()
                    self.allocator.sweep_stale_assumed(
                        gang_active=self.gang.active_groups())
                except Exception:
                    pass
        t = threading.Thread(target=sched_loop, daemon=True,
                             name="scheduler-loop")
        t.start()
        self._threads.append(t)

    def stop(self):
        self._stop.set()
        self.controllers.stop()
        self.autoscaler.stop()
        self.metrics.stop()
        self.alerts.stop()
        if self.config:
            self.config.stop()


def build_operator(persist_dir: Optional[str] = None,
                   metrics_dir: str = "",
                   config_path: Optional[str] = None,
                   provider=None, store: Optional[Store] = None) -> Operator:
    """Assemble the control plane. `store` swaps the backing state plane:
    the embedded Store (default) or a k8s.bridge.K8sStore, in which case
    the whole operator runs against a real kube-apiserver (reference
    cmd/main.go:131-297 manager wiring)."""

    store = store or Store(persist_dir=persist_dir)
    quota = QuotaStore(store)
    allocator = GpuAllocator(store=store, quota=quota)
    gang = GangManager(store)
    tsdb = TSDB()
    metrics = MetricsRecorder(out_dir=metrics_dir, tsdb=tsdb)
    alerts = AlertEvaluator(tsdb)
    ports = PortAllocator(store)
    indexes = IndexAllocator(store)
    expander = NodeExpander(store)
    mutator = PodMutator(store, index_allocator=indexes,
                         port_allocator=ports)
    fit = GPUResourcesFit(store, allocator, gang=gang,
                          index_allocator=indexes, expander=expander,
                          metrics=metrics)
    topo = GPUNetworkTopologyAware(allocator)
    scheduler = Scheduler(store, [fit, topo])
    provider = provider or MockProvider(store=store)
    mgr = ControllerManager(store)
    for ctrl in default_controllers(store, allocator=allocator,
                                    provider=provider):
        mgr.register(ctrl)
    autoscaler = Autoscaler(store, tsdb=tsdb, allocator=allocator)
    defrag = DefragController(store, allocator)
    cfg = ConfigWatcher(config_path) if config_path else None
    if cfg is not None:
        from .alert import default_rules, rules_from_config

        def _apply_alert_rules(conf):
            extra = rules_from_config(conf.alert_rules)
            alerts.rules = default_rules() + extra
        _apply_alert_rules(cfg.config)
        cfg.on_change(_apply_alert_rules)
    return Operator(
        store=store, allocator=allocator, quota=quota, gang=gang,
        scheduler=scheduler, mutator=mutator, controllers=mgr,
        autoscaler=autoscaler, metrics=metrics, tsdb=tsdb, alerts=alerts,
        port_allocator=ports, index_allocator=indexes, expander=expander,
        defrag=defrag, config=cfg)


def run_leader_election(cli, lease_name: str, namespace: str,
                        duration_s: int = 15, identity: str = "",
                        on_lost=None, _max_wait: float = 0.0) -> str:
    """Block until this replica holds the operator Lease, then keep
    renewing it from a daemon thread (reference cmd/main.go manager
    leader election). Losing the lease calls `on_lost` (default:
    hard-exit so kubernetes restarts us into a follower)."""

    import os
    import socket
    import time as _t
    identity = identity or f"{socket.gethostname()}-{os.getpid()}"
    waited = 0.0
    while not cli.acquire_lease(lease_name, namespace, identity,
                                duration_s=duration_s):
        _t.sleep(duration_s / 5)
        waited += duration_s / 5
        if _max_wait and waited >= _max_wait:
            return ""  # tests: give up instead of blocking forever

    def renew():
        while True:
            _t.sleep(duration_s / 3)
            try:
                ok = cli.acquire_lease(lease_name, namespace, identity,
                                       duration_s=duration_s)
            except Exception:
                ok = True  # transient apiserver error: keep trying
            if not ok:
                if on_lost:
                    on_lost()
                    return
                os._exit(1)

    threading.Thread(target=renew, daemon=True,
                     name="leader-lease-renew").start()
    return identity


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--persist-dir", default="")
    ap.add_argument("--metrics-dir", default="")
    ap.add_argument("--config", default="")
    ap.add_argument("--http-port", type=int, default=C.OperatorHTTPPort)
    # --- Kubernetes mode (reference cmd/main.go): the operator runs
    # against a real apiserver instead of the embedded store ---
    ap.add_argument("--k8s", action="store_true",
                    help="back the control plane with a kube-apiserver "
                         "(kubeconfig / in-cluster / TF_K8S_URL)")
    ap.add_argument("--kubeconfig", default="")
    ap.add_argument("--namespace", default="default",
                    help="namespace watched for namespaced CRDs/pods")
    ap.add_argument("--install-crds", action="store_true",
                    help="apply the generated CRD manifests on startup")
    ap.add_argument("--webhook-port", type=int, default=9443)
    ap.add_argument("--webhook-certs", default="",
                    help="dir with tls.crt/tls.key for the webhook server "
                         "(generated with tensor_fusion_amd.k8s.certs)")
    ap.add_argument("--webhook-workers", type=int, default=1,
                    help=">1: serve admission from that many forked "
                         "processes on one shared socket (~1.6k adm/s "
                         "per worker, near-linear to core count)")
    ap.add_argument("--leader-elect", action="store_true",
                    help="Lease-based leader election: block until this "
                         "replica holds the lease, exit if it loses it "
                         "(k8s restarts us; followers keep serving HTTP "
                         "reads and proxy writes to the leader)")
    ap.add_argument("--lease-name", default="tensor-fusion-operator")
    args = ap.parse_args()

    store = None
    if args.k8s:
        from .k8s.bridge import K8sStore
        from .k8s.client import K8sClient
        cli = (K8sClient.from_kubeconfig(args.kubeconfig)
               if args.kubeconfig else K8sClient.auto())
        if args.install_crds:
            from .k8s.client import ApiError
            from .k8s.crdgen import all_crds
            for crd in all_crds().values():
                try:
                    cli.create(crd)
                except ApiError as e:
                    if not e.conflict:
                        raise
        store = K8sStore(cli, namespace=args.namespace).start()
        if args.leader_elect:
            run_leader_election(cli, args.lease_name, args.namespace)

    op = build_operator(persist_dir=args.persist_dir or None,
                        metrics_dir=args.metrics_dir,
                        config_path=args.config or None,
                        store=store)
    op.start()

    import uvicorn

    from .server import create_operator_app
    app = create_operator_app(op.store, allocator=op.allocator,
                              port_allocator=op.port_allocator,
                              index_allocator=op.index_allocator,
                              expander=getattr(op, "expander", None))
    if args.k8s:
        # serve the AdmissionReview webhook the apiserver calls (TLS when
        # certs are provided; MutatingWebhookConfiguration in deploy/)
        from .server.webhook_server import (create_webhook_app,
                                            serve_multiprocess)
        whapp = create_webhook_app(op.mutator)
        crt = key = ""
        if args.webhook_certs:
            import os
            crt = os.path.join(args.webhook_certs, "tls.crt")
            key = os.path.join(args.webhook_certs, "tls.key")
        if args.webhook_workers > 1:
            # each forked worker rebuilds its own informer-backed store
            # (watch threads do not survive fork)
            def wh_factory():
                from .k8s.bridge import K8sStore
                from .k8s.client import K8sClient
                from .webhook import PodMutator
                # fresh client: the parent's HTTP session fds must not
                # be shared across forked processes
                child_cli = (K8sClient.from_kubeconfig(args.kubeconfig)
                             if args.kubeconfig else K8sClient.auto())
                st = K8sStore(child_cli, namespace=args.namespace).start()
                return create_webhook_app(PodMutator(st))

            serve_multiprocess(whapp, args.webhook_port,
                               workers=args.webhook_workers,
                               ssl_certfile=crt, ssl_keyfile=key,
                               app_factory=wh_factory)
        else:
            kw = {}
            if crt:
                kw = {"ssl_certfile": crt, "ssl_keyfile": key}
            wh = uvicorn.Server(uvicorn.Config(
                whapp, host="0.0.0.0", port=args.webhook_port,
                log_level="warning", **kw))
            threading.Thread(target=wh.run, daemon=True,
                             name="webhook-server").start()
    uvicorn.run(app, host="0.0.0.0", port=args.http_port, log_level="warning")


if __name__ == "__main__":
    main()
