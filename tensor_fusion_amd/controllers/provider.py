"""Provider manager — ProviderConfig hot-reload into allocator/config.

Reference: internal/provider/manager.go:37-212 + providerconfig
controller — ProviderConfig CRDs carry per-vendor hardware tables
(model → tflops/vram), partition templates, device mounts and the
limiter/accelerator lib names; changes hot-reload into the GpuAllocator
(partition templates) and the GpuInfo cache consumed by pricing and the
node expander.
"""
from __future__ import annotations

from typing import Dict, Optional

from ..api.store import Store
from ..api.types import HardwareModel, ProviderConfig
from .base import Reconciler, Request


class ProviderManager:
    """Caches merged provider data; reconciler pushes updates here."""

    def __init__(self, allocator=None):
        self.allocator = allocator
        self.models: Dict[str, HardwareModel] = {}
        self.partition_templates: Dict[str, list] = {}  # vendor → templates
        self.device_nodes: Dict[str, list] = {}
        self.mount_libs: Dict[str, list] = {}

    def apply(self, pc: ProviderConfig):
        for m in pc.models:
            self.models[m.model] = m
        self.partition_templates[pc.vendor] = list(pc.partition_templates)
        self.device_nodes[pc.vendor] = list(pc.device_nodes)
        self.mount_libs[pc.vendor] = list(pc.mount_libs)
        if self.allocator is not None and pc.partition_templates:
            self.allocator.partition_templates = list(pc.partition_templates)

    def drop(self, pc: ProviderConfig):
        self.partition_templates.pop(pc.vendor, None)
        self.device_nodes.pop(pc.vendor, None)
        self.mount_libs.pop(pc.vendor, None)

    def tflops_of(self, model: str, default: float = 0.0) -> float:
        m = self.models.get(model)
        return m.fp16_tflops if m else default


class ProviderConfigReconciler(Reconciler):
    kind = "ProviderConfig"

    def __init__(self, store: Store, manager: Optional[ProviderManager] = None):
        super().__init__(store)
        self.manager = manager or ProviderManager()
        store.on_change("ProviderConfig", self._on_event)

    def _on_event(self, event: str, obj):
        if event == "DELETED":
            self.manager.drop(obj)

    def reconcile(self, req: Request):
        pc = self.store.get(self.kind, req.name, req.namespace)
        self.manager.apply(pc)
        return 0.0


class SchedulingConfigReconciler(Reconciler):
    """SchedulingConfigTemplate → live scheduler/allocator behavior:
    placement mode (CompactFirst/LowLoadFirst/default), score weights and
    ERL gains hot-swap without restarts (reference
    schedulingconfigtemplate controller + scheduler_config.go)."""

    kind = "SchedulingConfigTemplate"

    def __init__(self, store: Store, allocator=None, erl_params=None):
        super().__init__(store)
        self.allocator = allocator
        self.erl_params = erl_params  # mutated in place for the ERL loop

    def reconcile(self, req: Request):
        tpl = self.store.get(self.kind, req.name, req.namespace)
        if self.allocator is not None:
            from ..allocator.strategy import make_strategy
            self.allocator.strategy = make_strategy(
                tpl.placement_mode,
                vram_weight=tpl.vram_weight,
                tflops_weight=tpl.tflops_weight)
        if self.erl_params is not None:
            src = tpl.erl
            for f in ("kp", "ki", "kd", "ema_alpha", "deadband_percent",
                      "slew_up_percent", "slew_down_percent",
                      "loop_interval_s", "min_rate", "max_rate"):
                setattr(self.erl_params, f, getattr(src, f))
        return 0.0
