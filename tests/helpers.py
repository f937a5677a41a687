"""Shared test fixtures: fake MI355X fleets (the envtest-with-fake-GPUs
pattern of the reference suite)."""
from tensor_fusion_amd import constants as C
from tensor_fusion_amd.api.types import (GPU, AllocRequest, GPUStatus,
                                         ObjectMeta, Resource)

TFLOPS = C.MI355X_BF16_TFLOPS
VRAM = C.MI355X_VRAM_BYTES


def make_gpu(name, node="node-0", pool="pool-a", index=0, numa=0,
             tflops=TFLOPS, vram=VRAM, phase="Ready"):
    cap = Resource(tflops=tflops, vram=vram, compute_percent=100.0)
    return GPU(
        meta=ObjectMeta(name=name),
        status=GPUStatus(
            capacity=cap,
            available=Resource(tflops=tflops, vram=vram, compute_percent=100.0),
            uuid=f"uuid-{name}", index=index, numa_node=numa,
            node=node, pool=pool, phase=phase,
        ),
    )


def make_node_gpus(node, pool="pool-a", count=8):
    """One MI355X node: 8 GPUs, xGMI full mesh (tier 0), NUMA split 4+4."""

    gpus = [make_gpu(f"{node}-gpu-{i}", node=node, pool=pool, index=i,
                     numa=0 if i < count // 2 else 1) for i in range(count)]
    for g in gpus:
        for o in gpus:
            if o is not g:
                g.status.topology[o.status.uuid] = (
                    C.TopoTierXGMI)  # full mesh intra-node
    return gpus


def make_request(pod="p1", ns="default", tflops=100.0, vram=16 << 30,
                 count=1, pool="pool-a", **kw):
    return AllocRequest(
        workload=kw.pop("workload", "wl-1"),
        pod_name=pod, namespace=ns, pool=pool,
        request=Resource(tflops=tflops, vram=vram,
                         compute_percent=kw.pop("compute_percent", 0.0)),
        limit=Resource(tflops=tflops * 2, vram=vram),
        gpu_count=count, **kw)
