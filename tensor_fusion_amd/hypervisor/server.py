"""Hypervisor HTTP API (:8001) — consumed by limiters, workers, the
operator and the TUI.

Reference: pkg/hypervisor/server/server.go:95-123 + handlers/ — devices,
workers, pod info (limits/qos/auto-freeze), process registration (container
pid → host pid → shm), limiter list, VRAM-pressure /trap with low-QoS
victim selection, snapshot/resume.
"""
from __future__ import annotations

import os
from typing import Optional

from fastapi import FastAPI, Query
from fastapi.responses import JSONResponse
from pydantic import BaseModel

from .. import constants as C
from .device import DeviceController
from .worker import WorkerController


class ProcessRegistration(BaseModel):
    container_pid: int
    container_name: str = ""
    namespace: str = ""
    pod_name: str = ""


class TrapRequest(BaseModel):
    device_uuid: str = ""
    bytes_needed: int = 0


def resolve_host_pid(container_pid: int, namespace: str, pod: str) -> int:
    """Container-pid → host-pid via /proc scan (reference legacy.go:448-481
    walks /proc matching the pod's cgroup). Single-node mode: identity."""

    target = f"{namespace}/{pod}"
    try:
        for entry in os.listdir("/proc"):
            if not entry.isdigit():
                continue
            try:
                with open(f"/proc/{entry}/environ", "rb") as f:
                    env = f.read().split(b"\x00")
                envmap = dict(e.split(b"=", 1) for e in env if b"=" in e)
                if (envmap.get(b"POD_NAMESPACE", b"").decode() == namespace and
                        envmap.get(b"POD_NAME", b"").decode() == pod):
                    # NSpid match: the inner pid of this host process
                    with open(f"/proc/{entry}/status") as f:
                        for line in f:
                            if line.startswith("NSpid:"):
                                inner = line.split()[-1]
                                if int(inner) == container_pid:
                                    return int(entry)
            except (OSError, ValueError):
                continue
    except OSError:
        pass
    return container_pid  # same pid namespace


def build_app(devices: DeviceController, workers: WorkerController,
              auto_freeze_rules=None, snapshot_fn=None) -> FastAPI:
    app = FastAPI(title="tensor-fusion-amd hypervisor")
    auto_freeze_rules = auto_freeze_rules or {}

    @app.get("/healthz")
    def healthz():
        return {"ok": True}

    @app.get("/api/v1/devices")
    def get_devices():
        out = []
        for d in devices.devices():
            m = None
            try:
                m = devices.metrics(d.index)
            except Exception:
                pass
            nworkers = sum(
                1 for st in workers.list()
                if any(dev.uuid == d.uuid for dev in st.allocation.devices))
            part = None
            try:
                part = devices.accel.compute_partition(d.index)
            except Exception:
                pass
            out.append({
                "uuid": d.uuid, "index": d.index, "name": d.name,
                "numa_node": d.numa_node, "vram_total": d.vram_total,
                "compute_units": d.compute_units, "xcd_count": d.xcd_count,
                "fp16_tflops": d.fp16_tflops, "is_mock": d.is_mock,
                "vram_used": m.vram_used if m else 0,
                "busy_percent": m.gfx_activity if m else 0,
                "compute_partition": part or "SPX",
                "worker_count": nworkers,
                "metrics": None if m is None else {
                    "gfx_activity": m.gfx_activity,
                    "umc_activity": m.umc_activity,
                    "vram_used": m.vram_used,
                },
            })
        return {"success": True, "data": out}

    @app.get("/api/v1/workers")
    def get_workers():
        """Full per-worker records (the TUI's worker view + shm
        inspector read this shape)."""

        import time as _t
        out = []
        for st in workers.list():
            a = st.allocation
            entry = None
            try:
                entry = st.shm.device(0)
            except (OSError, ValueError):
                pass
            ns, name = st.key.split("/", 1)
            out.append({
                "namespace": ns, "pod": name,
                "workload": a.spec.workload,
                "qos": a.spec.qos, "isolation": a.spec.isolation,
                "shm_path": st.shm.path,
                "device_uuid": a.devices[0].uuid if a.devices else "",
                # shm heartbeat is CLOCK_MONOTONIC ns → wall-clock ts
                "heartbeat_ts": (
                    _t.time() - max(0.0, _t.monotonic()
                                    - st.shm.heartbeat() / 1e9)
                    if st.shm and st.shm.heartbeat() else 0),
                "pids": st.pids or [],
                "flags": st.shm.flags() if st.shm else 0,
                "limits": {
                    "vram": entry.mem_limit_bytes if entry else
                    a.spec.vram_limit,
                    "compute_percent": a.up_limit_percent,
                },
                "usage": {
                    "vram": entry.pod_memory_used if entry else 0,
                    "vmm_bytes": entry.vmm_bytes if entry else 0,
                    "erl_rate": entry.erl_refill_rate if entry else 0,
                    "erl_capacity": entry.erl_capacity if entry else 0,
                    "erl_tokens": entry.erl_tokens if entry else 0,
                    "launches": entry.launch_count if entry else 0,
                    "block_ns": entry.block_ns_total if entry else 0,
                },
            })
        return {"success": True, "data": out}

    @app.get("/api/v1/workers/metrics")
    def get_worker_metrics():
        """Flat join rows (the recorder's input shape, reference
        worker metrics join :212-290)."""

        return {"success": True, "data": workers.worker_metrics()}

    @app.get("/api/v1/metrics")
    def get_metrics():
        """Node metric rows for the TUI metrics view."""

        rows = []
        devs = devices.devices()
        used = total = 0
        for d in devs:
            try:
                m = devices.metrics(d.index)
                used += m.vram_used
            except Exception:
                pass
            total += d.vram_total
        wl = workers.list()
        launches = blocked = 0
        detail = []
        for st in wl:
            try:
                e = st.shm.device(0)
                launches += e.launch_count
                blocked += e.block_ns_total
                detail.append(f"{st.key}:{e.launch_count}")
            except (OSError, ValueError):
                pass
        rows = [
            {"name": "devices", "value": len(devs), "detail": ""},
            {"name": "workers", "value": len(wl), "detail": ""},
            {"name": "vram used GiB",
             "value": f"{used / (1 << 30):.1f}/{total / (1 << 30):.0f}",
             "detail": ""},
            {"name": "launches", "value": launches,
             "detail": ", ".join(detail[:8])},
            {"name": "throttled ms", "value": f"{blocked / 1e6:.0f}",
             "detail": ""},
        ]
        return {"success": True, "data": rows}

    @app.get("/api/v1/pod")
    def get_pod(namespace: str = Query("default"), pod: str = Query(...)):
        st = workers.get(f"{namespace}/{pod}")
        if st is None:
            return JSONResponse(status_code=404, content={
                "success": False, "message": "worker not found"})
        spec = st.allocation.spec
        af = auto_freeze_rules.get(spec.qos)
        return {"success": True, "data": {
            "pod_name": spec.name, "namespace": spec.namespace,
            "gpu_uuids": spec.gpu_uuids,
            "tflops_limit": spec.tflops_limit,
            "vram_limit": spec.vram_limit,
            "qos_level": spec.qos,
            "compute_shard": st.allocation.up_limit_percent,
            "isolation": spec.isolation,
            "shm_path": st.allocation.shm_path,
            "auto_freeze": {
                "enable": bool(af and af.enable),
                "freeze_to_mem_ttl": af.freeze_to_mem_ttl_s if af else 0,
                "freeze_to_disk_ttl": af.freeze_to_disk_ttl_s if af else 0,
            },
        }}

    @app.post("/api/v1/process")
    def register_process(reg: ProcessRegistration):
        host_pid = resolve_host_pid(reg.container_pid, reg.namespace,
                                    reg.pod_name)
        ok = workers.register_pid(f"{reg.namespace}/{reg.pod_name}", host_pid)
        if not ok:
            return JSONResponse(status_code=404, content={
                "success": False, "message": "worker not found"})
        return {"success": True, "data": {
            "host_pid": host_pid, "container_pid": reg.container_pid}}

    @app.get("/api/v1/limiter")
    def limiter_list():
        out = []
        for st in workers.list():
            spec = st.allocation.spec
            out.append({
                "worker": st.key, "qos": spec.qos,
                "vram_limit": spec.vram_limit,
                "up_limit_percent": st.allocation.up_limit_percent,
                "devices": [d.uuid for d in st.allocation.devices],
            })
        return {"success": True, "data": out}

    @app.post("/api/v1/trap")
    def trap(req: TrapRequest):
        """VRAM pressure: pick low-QoS victims whose working set should be
        tiered/frozen (reference legacy.go:124-150)."""

        order = {q: i for i, q in enumerate(C.QosLevels)}
        victims = sorted(
            (st for st in workers.list()
             if order.get(st.allocation.spec.qos, 1) <= order[C.QosMedium]
             and (not req.device_uuid or
                  req.device_uuid in st.allocation.spec.gpu_uuids)),
            key=lambda st: (order.get(st.allocation.spec.qos, 1), -st.started))
        picked = []
        freed = 0
        for st in victims:
            entry = st.shm.device(0)
            picked.append({"worker": st.key,
                           "vram_used": entry.pod_memory_used,
                           "qos": st.allocation.spec.qos})
            st.shm.set_flag(2, True)  # FLAG_VRAM_PRESSURE
            freed += entry.pod_memory_used
            if req.bytes_needed and freed >= req.bytes_needed:
                break
        return {"success": True, "data": {"victims": picked}}

    @app.post("/api/v1/workers/{namespace}/{pod}/freeze")
    def freeze(namespace: str, pod: str):
        st = workers.get(f"{namespace}/{pod}")
        if st is None:
            return JSONResponse(status_code=404, content={"success": False})
        st.shm.freeze(True)
        return {"success": True}

    @app.post("/api/v1/workers/{namespace}/{pod}/resume")
    def resume(namespace: str, pod: str):
        st = workers.get(f"{namespace}/{pod}")
        if st is None:
            return JSONResponse(status_code=404, content={"success": False})
        st.shm.freeze(False)
        return {"success": True}

    @app.post("/api/v1/workers/{namespace}/{pod}/snapshot")
    def snapshot(namespace: str, pod: str):
        st = workers.get(f"{namespace}/{pod}")
        if st is None:
            return JSONResponse(status_code=404, content={"success": False})
        if snapshot_fn is None:
            return JSONResponse(status_code=501, content={
                "success": False,
                "message": "no vGPU worker manager attached"})
        res = snapshot_fn(st)
        return {"success": True, "data": res}

    return app


def attach_vgpu_manager(app: FastAPI, manager) -> None:
    """Mount vGPU worker snapshot/resume/migration routes (the endpoints
    the reference returns 501 for — here backed by the VA-stable worker
    snapshot machinery, hypervisor/vgpu_manager.py)."""

    @app.get("/api/v1/vgpu")
    def vgpu_status():
        return {"success": True, "data": manager.status()}

    @app.post("/api/v1/vgpu/{namespace}/{pod}/snapshot")
    def vgpu_snapshot(namespace: str, pod: str):
        try:
            path = manager.snapshot(f"{namespace}/{pod}")
        except KeyError:
            return JSONResponse(status_code=404, content={"success": False})
        except Exception as e:
            return JSONResponse(status_code=500,
                                content={"success": False, "message": str(e)})
        return {"success": True, "data": {"snapshot": path}}

    @app.post("/api/v1/vgpu/{namespace}/{pod}/resume")
    def vgpu_resume(namespace: str, pod: str,
                    device: Optional[int] = Query(None)):
        try:
            w = manager.resume(f"{namespace}/{pod}", device_index=device)
        except KeyError:
            return JSONResponse(status_code=404, content={"success": False})
        except Exception as e:
            return JSONResponse(status_code=500,
                                content={"success": False, "message": str(e)})
        return {"success": True, "data": {"device": w.device_index}}

    @app.post("/api/v1/vgpu/{namespace}/{pod}/migrate")
    def vgpu_migrate(namespace: str, pod: str, device: int = Query(...)):
        try:
            w = manager.migrate(f"{namespace}/{pod}", device)
        except KeyError:
            return JSONResponse(status_code=404, content={"success": False})
        except Exception as e:
            return JSONResponse(status_code=500,
                                content={"success": False, "message": str(e)})
        return {"success": True,
                "data": {"device": w.device_index,
                         "migrations": w.migrations}}
