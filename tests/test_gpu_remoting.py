"""GPU-over-IP remoting on a real MI355X.

The client process runs with HIP_VISIBLE_DEVICES="" — it cannot touch the
GPU except through the worker, so passing results prove genuine remoting.
"""
import json
import os
import subprocess
import sys
import time

import pytest

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
NATIVE = os.path.join(REPO, "tensor_fusion_amd", "_native")

sys.path.insert(0, REPO)
from tensor_fusion_amd.client.runtime import client_env, start_worker  # noqa


@pytest.fixture()
def worker(tmp_path):
    h = start_worker(str(tmp_path / "vgpu.sock"), device_index=0)
    yield h
    h.stop()


def test_testapp_native_baseline():
    out = subprocess.run([os.path.join(NATIVE, "tf_remote_testapp")],
                         capture_output=True, text=True, timeout=120)
    assert "TESTAPP_OK" in out.stdout, out.stdout + out.stderr


def test_testapp_remoted(worker):
    env = client_env(worker.socket_path)
    out = subprocess.run([os.path.join(NATIVE, "tf_remote_testapp")],
                         capture_output=True, text=True, timeout=180, env=env)
    assert "TESTAPP_OK" in out.stdout, out.stdout + out.stderr
    assert "devices=1" in out.stdout
    # the vGPU is a full MI355X: 288 GB visible through the remote path
    assert "vram_total_gb=" in out.stdout


def test_torch_tiny_decode_remoted(worker, tmp_path):
    """Full PyTorch through the remoting path: tiny Llama decode."""

    env = client_env(worker.socket_path)
    env["PYTORCH_NO_HIP_MEMORY_CACHING"] = env.get(
        "PYTORCH_NO_HIP_MEMORY_CACHING", "0")
    out = subprocess.run(
        [sys.executable, "-m", "tensor_fusion_amd.models.llama", "--model",
         "tiny", "--batch", "2", "--ctx", "16", "--steps", "4", "--warmup",
         "1"],
        capture_output=True, text=True, timeout=600, env=env, cwd=REPO)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-4000:]
    r = json.loads(out.stdout.strip().splitlines()[-1])
    assert r["tok_s"] > 0


MIGRATE_CLIENT = r"""
import os, sys, time
# phase 1: allocate + compute through the remote vGPU, report, then wait
# for the marker file (while the hypervisor migrates the worker under us),
# then keep computing on the SAME device pointers.
import ctypes
hip = ctypes.CDLL(None)  # libtfhip_client.so is LD_PRELOADed

def ck(rc, what):
    if rc != 0:
        print(f"FAIL {what} rc={rc}", flush=True)
        sys.exit(1)

N = 1 << 20
ptr = ctypes.c_void_p()
ck(hip.hipMalloc(ctypes.byref(ptr), N * 4), "malloc")
host = (ctypes.c_float * N)(*( [1.5] * N ))
ck(hip.hipMemcpy(ptr, host, N * 4, 1), "h2d")
print("PHASE1 ready", flush=True)
marker = sys.argv[1]
while not os.path.exists(marker):
    time.sleep(0.2)
# phase 2: the worker behind the socket is now a DIFFERENT process that
# restored the snapshot; the pointer must still hold our data.
back = (ctypes.c_float * N)()
ck(hip.hipMemcpy(back, ptr, N * 4, 2), "d2h-after-migrate")
ok = all(abs(back[i] - 1.5) < 1e-9 for i in range(0, N, 65536))
# and the device must still execute new work
ck(hip.hipMemset(ptr, 0, N * 4), "memset-after-migrate")
ck(hip.hipDeviceSynchronize(), "sync")
ck(hip.hipMemcpy(back, ptr, 4, 2), "d2h2")
ok2 = back[0] == 0.0
print(f"PHASE2 ok={ok} ok2={ok2}", flush=True)
sys.exit(0 if (ok and ok2) else 1)
"""


def test_live_migration_snapshot_restore(tmp_path):
    """Snapshot worker A, restore into worker B on the same socket while
    the client stays alive: device pointers (VA-stable VMM heap) and data
    survive. The reference's snapshot/resume endpoints return 501
    (handlers/worker.go:103-137) — this is the native implementation."""

    from tensor_fusion_amd.client.runtime import (client_env, migrate_worker,
                                                  start_worker)
    sock = str(tmp_path / "vgpu.sock")
    snap = str(tmp_path / "snap.bin")
    marker = str(tmp_path / "go2")
    w = start_worker(sock, device_index=0, snapshot_path=snap)
    env = client_env(sock)
    cli = subprocess.Popen([sys.executable, "-c", MIGRATE_CLIENT, marker],
                           env=env, stdout=subprocess.PIPE,
                           stderr=subprocess.PIPE, text=True, cwd=REPO)
    try:
        # wait for phase 1
        line = cli.stdout.readline()
        assert "PHASE1" in line, line + cli.stderr.read()
        w2 = migrate_worker(w, snap, new_device_index=0)
        try:
            with open(marker, "w") as f:
                f.write("go")
            out, err = cli.communicate(timeout=180)
            assert cli.returncode == 0, out + err
            assert "PHASE2 ok=True ok2=True" in out, out + err
        finally:
            w2.stop()
    finally:
        if cli.poll() is None:
            cli.kill()
        w.stop()


def test_graph_decode_remoted(worker):
    """hipGraph capture/replay forwarded through GPU-over-IP: the tiny
    Llama decode in graph mode must run and produce tokens (vLLM-class
    interception completeness, SURVEY §7 hard part #1)."""

    env = client_env(worker.socket_path)
    out = subprocess.run(
        [sys.executable, "-m", "tensor_fusion_amd.models.llama", "--model",
         "tiny", "--batch", "2", "--ctx", "16", "--steps", "8", "--warmup",
         "2", "--graphs"],
        capture_output=True, text=True, timeout=600, env=env, cwd=REPO)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-4000:]
    r = json.loads(out.stdout.strip().splitlines()[-1])
    assert r["tok_s"] > 0


def test_graph_fused_decode_remoted(worker):
    """The bench.py headline path through GPU-over-IP: hipGraph-captured
    decode with the fused gfx950 rmsnorm/add_rmsnorm kernels (libtfops
    launched via ctypes → hipLaunchKernel interposed and its code object
    forwarded by the client stub). Proves the strongest native
    configuration also runs remoted — the same-mode comparison bench.py
    now defaults to."""

    env = client_env(worker.socket_path)
    env["TF_FUSED_OPS"] = "1"
    out = subprocess.run(
        [sys.executable, "-m", "tensor_fusion_amd.models.llama", "--model",
         "tiny", "--batch", "2", "--ctx", "16", "--steps", "8", "--warmup",
         "2", "--graphs"],
        capture_output=True, text=True, timeout=600, env=env, cwd=REPO)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-4000:]
    r = json.loads(out.stdout.strip().splitlines()[-1])
    assert r["tok_s"] > 0


def test_resnet50_inference_remoted(worker):
    """Conv workloads through GPU-over-IP: MIOpen loads its kernels via a
    different path than rocBLAS/Tensile — this catches interception gaps
    beyond the Llama surface."""

    env = client_env(worker.socket_path)
    out = subprocess.run(
        [sys.executable, "-m", "tensor_fusion_amd.models.resnet",
         "--batch", "16", "--steps", "8", "--warmup", "2"],
        capture_output=True, text=True, timeout=900, env=env, cwd=REPO)
    assert out.returncode == 0, out.stdout[-1500:] + out.stderr[-4000:]
    r = json.loads(out.stdout.strip().splitlines()[-1])
    assert r["img_s"] > 0


def test_tcp_transport_cross_node_shape(tmp_path):
    """GPU-over-IP over actual IP: the TCP transport carries the same
    command stream (the reference's cross-node mode, README 'Ethernet/
    InfiniBand'). Same-host loopback here; the wire format is the
    cross-node one."""

    from tensor_fusion_amd.client.runtime import client_env, start_worker
    h = start_worker("", device_index=0, tcp_port=47821)
    try:
        env = client_env("", tcp="127.0.0.1:47821")
        out = subprocess.run([os.path.join(NATIVE, "tf_remote_testapp")],
                             capture_output=True, text=True, timeout=300,
                             env=env)
        assert "TESTAPP_OK" in out.stdout, out.stdout + out.stderr
        # full PyTorch over TCP
        out2 = subprocess.run(
            [sys.executable, "-m", "tensor_fusion_amd.models.llama",
             "--model", "tiny", "--batch", "2", "--ctx", "16", "--steps",
             "4", "--warmup", "1"],
            capture_output=True, text=True, timeout=600, env=env, cwd=REPO)
        assert out2.returncode == 0, out2.stdout[-1500:] + out2.stderr[-3000:]
        r = json.loads(out2.stdout.strip().splitlines()[-1])
        assert r["tok_s"] > 0
    finally:
        h.stop()


def test_two_workers_one_gpu_concurrent(tmp_path):
    """Multi-tenant GPU-over-IP: two vGPU workers share one physical
    MI355X, each serving its own GPU-less client concurrently (the
    pooling deployment shape — oversubscribed device, isolated command
    streams)."""

    import threading
    w1 = start_worker(str(tmp_path / "a.sock"), device_index=0)
    w2 = start_worker(str(tmp_path / "b.sock"), device_index=0)
    results = {}

    def client(name, h):
        env = client_env(h.socket_path)
        out = subprocess.run([os.path.join(NATIVE, "tf_remote_testapp")],
                             capture_output=True, text=True, timeout=240,
                             env=env)
        results[name] = out
    try:
        t1 = threading.Thread(target=client, args=("a", w1))
        t2 = threading.Thread(target=client, args=("b", w2))
        t1.start(); t2.start()
        t1.join(timeout=260); t2.join(timeout=260)
    finally:
        w1.stop()
        w2.stop()
    for name in ("a", "b"):
        out = results.get(name)
        assert out is not None, f"client {name} did not finish"
        assert "TESTAPP_OK" in out.stdout, (name, out.stdout, out.stderr)


def test_remote_worker_under_erl_quota(tmp_path):
    """Composition of the two isolation layers: the vGPU worker process
    itself runs under the LD_PRELOAD ERL limiter (the worker-pod shape —
    hypervisor/allocation.py attaches LD_PRELOAD for soft isolation), so
    a GPU-less remote client is throttled by its vGPU's quota."""

    import time

    def run(env_extra, tag):
        h = start_worker(str(tmp_path / f"{tag}.sock"), device_index=0,
                         env=env_extra)
        try:
            env = client_env(h.socket_path)
            t0 = time.perf_counter()
            out = subprocess.run(
                [sys.executable, "-m", "tensor_fusion_amd.models.llama",
                 "--model", "tiny", "--batch", "2", "--ctx", "16",
                 "--steps", "12", "--warmup", "2"],
                capture_output=True, text=True, timeout=600, env=env,
                cwd=REPO)
            dt = time.perf_counter() - t0
            assert out.returncode == 0, out.stdout[-1000:] + out.stderr[-3000:]
            return json.loads(out.stdout.strip().splitlines()[-1]), dt
        finally:
            h.stop()

    limiter = os.path.join(NATIVE, "libtfhip_limiter.so")
    free, _ = run({}, "free")
    capped, _ = run({"LD_PRELOAD": limiter,
                     "TF_UP_LIMIT_PERCENT": "25",
                     "TF_ERL_RATE": "300", "TF_ERL_CAPACITY": "30"}, "cap")
    # a decode step is dozens of launches; at 300 launches/s the capped
    # worker must be far below the free worker's token rate
    assert capped["tok_s"] < 0.5 * free["tok_s"], (free, capped)


SYNC_RTT_CHILD = r"""
import json, time
import torch
torch.cuda.init()
x = torch.ones(8, device="cuda")
torch.cuda.synchronize()
# sync-op RTT: each synchronize is one OP_DEVICE_SYNC round trip
N = 100
t0 = time.perf_counter()
for _ in range(N):
    torch.cuda.synchronize()
rtt_ms = (time.perf_counter() - t0) / N * 1e3
# async burst: launches are fire-and-forget; only the final sync waits
t0 = time.perf_counter()
for _ in range(200):
    x = x * 1.0001
torch.cuda.synchronize()
burst_ms = (time.perf_counter() - t0) * 1e3
print(json.dumps({"sync_rtt_ms": round(rtt_ms, 3),
                  "burst200_ms": round(burst_ms, 2)}))
"""


def test_tcp_latency_sensitivity(tmp_path):
    """Cross-node latency sensitivity of the GPU-over-IP wire: the same
    TCP command stream through a relay injecting 0 / 1 / 3 ms one-way
    delay (no iproute2 in this image ⇒ no netns/netem; the relay is the
    substitute — two extra processes on the same byte stream). Expect:
    sync-op RTT tracks the injected delay (~2x one-way + base), while
    async launch bursts stay delay-insensitive (fire-and-forget ring)."""

    from tensor_fusion_amd.client.runtime import client_env, start_worker
    base_port = 48500
    w = start_worker("", device_index=0, tcp_port=base_port)
    results = {}
    try:
        for delay_ms in (0.0, 1.0, 3.0):
            proxy_port = base_port + 1 + int(delay_ms * 10) % 97
            proxy = subprocess.Popen(
                [sys.executable,
                 os.path.join(REPO, "tools", "tcp_latency_proxy.py"),
                 "--listen", str(proxy_port),
                 "--connect", f"127.0.0.1:{base_port}",
                 "--delay-ms", str(delay_ms)],
                stdout=subprocess.PIPE, text=True)
            try:
                assert "PROXY_READY" in proxy.stdout.readline()
                env = client_env("", tcp=f"127.0.0.1:{proxy_port}")
                out = subprocess.run(
                    [sys.executable, "-c", SYNC_RTT_CHILD], env=env,
                    capture_output=True, text=True, timeout=600, cwd=REPO)
                assert out.returncode == 0, \
                    out.stdout[-800:] + out.stderr[-3000:]
                results[delay_ms] = json.loads(
                    out.stdout.strip().splitlines()[-1])
            finally:
                proxy.kill()
    finally:
        w.stop()
    r0, r1, r3 = results[0.0], results[1.0], results[3.0]
    print(f"\nsync RTT ms @0/1/3ms delay: {r0['sync_rtt_ms']} / "
          f"{r1['sync_rtt_ms']} / {r3['sync_rtt_ms']}; "
          f"200-launch async burst: {r0['burst200_ms']} / "
          f"{r1['burst200_ms']} / {r3['burst200_ms']}")
    # sync ops pay ~2x the injected one-way delay
    assert r1["sync_rtt_ms"] >= r0["sync_rtt_ms"] + 1.0
    assert r3["sync_rtt_ms"] >= r0["sync_rtt_ms"] + 4.0
    # async bursts must NOT pay per-launch latency (<< 200 * delay)
    assert r3["burst200_ms"] < 200 * 3.0


EXPANDABLE_CHILD = r"""
import json, os
import torch
# expandable_segments drives the VMM surface: hipMemAddressReserve /
# hipMemCreate / hipMemMap / hipMemSetAccess / hipMemUnmap / hipMemRelease
assert "expandable_segments" in os.environ.get("PYTORCH_ALLOC_CONF", "") \
    or "expandable_segments" in os.environ.get("PYTORCH_HIP_ALLOC_CONF", "")
a = torch.randn(4 << 20, device="cuda")
b = torch.randn(4 << 20, device="cuda")
c = (a * 2 + b).sum()
del a
torch.cuda.empty_cache()  # unmap path
d = torch.randn(8 << 20, device="cuda")  # grow the segment (remap)
torch.cuda.synchronize()
print(json.dumps({"ok": bool(torch.isfinite(c).item()),
                  "sum": float(c.item()), "d0": float(d[0].item())}))
"""


def test_expandable_segments_allocator_remoted(worker):
    """PyTorch's expandable_segments caching-allocator backend through
    GPU-over-IP: the client-side VMM surface (reserve/create/map/
    set-access/unmap/release) is forwarded to the worker — the allocator
    machinery vLLM-class servers sit on (vLLM itself is not installed in
    this offline image; this is the same HIP surface)."""

    env = client_env(worker.socket_path)
    env["PYTORCH_ALLOC_CONF"] = "expandable_segments:True"
    env["PYTORCH_HIP_ALLOC_CONF"] = "expandable_segments:True"
    out = subprocess.run([sys.executable, "-c", EXPANDABLE_CHILD], env=env,
                         capture_output=True, text=True, timeout=600,
                         cwd=REPO)
    assert out.returncode == 0, out.stdout[-1000:] + out.stderr[-4000:]
    r = json.loads(out.stdout.strip().splitlines()[-1])
    assert r["ok"] is True


TRANSFORMERS_CHILD = r"""
import json
import torch
from transformers import LlamaConfig, LlamaForCausalLM
cfg = LlamaConfig(vocab_size=256, hidden_size=128, num_hidden_layers=2,
                  num_attention_heads=4, num_key_value_heads=2,
                  intermediate_size=256, max_position_embeddings=512)
torch.manual_seed(0)
model = LlamaForCausalLM(cfg).to("cuda", torch.bfloat16).eval()
ids = torch.randint(0, 256, (2, 12), device="cuda")
with torch.no_grad():
    out = model.generate(ids, max_new_tokens=16, do_sample=False,
                         pad_token_id=0)
assert out.shape == (2, 28), out.shape
print(json.dumps({"ok": True, "tokens": out.shape[1]}))
"""


def test_transformers_generate_remoted(worker):
    """HuggingFace transformers LlamaForCausalLM.generate() through the
    remoting path (random-init tiny config — no network for weights).
    Exercises the HF serving stack's op surface (SDPA attention, KV
    cache, sampling machinery) as the closest available stand-in for the
    vLLM config-3 workload."""

    env = client_env(worker.socket_path)
    out = subprocess.run([sys.executable, "-c", TRANSFORMERS_CHILD],
                         env=env, capture_output=True, text=True,
                         timeout=900, cwd=REPO)
    assert out.returncode == 0, out.stdout[-1000:] + out.stderr[-5000:]
    r = json.loads(out.stdout.strip().splitlines()[-1])
    assert r["ok"] is True and r["tokens"] == 28


FREEZE_CLIENT = r"""
import ctypes, os, sys, time
hip = ctypes.CDLL(None)  # libtfhip_client.so is LD_PRELOADed

def ck(rc, what):
    if rc != 0:
        print(f"FAIL {what} rc={rc}", flush=True)
        sys.exit(1)

N = 1 << 20
ptr = ctypes.c_void_p()
ck(hip.hipMalloc(ctypes.byref(ptr), N * 4), "malloc")
host = (ctypes.c_float * N)(*([2.5] * N))
ck(hip.hipMemcpy(ptr, host, N * 4, 1), "h2d")
ck(hip.hipDeviceSynchronize(), "sync")
print("PHASE1 ready", flush=True)
marker = sys.argv[1]
while not os.path.exists(marker):   # stay IDLE: the controller freezes us
    time.sleep(0.2)
# worker is now frozen to disk; this op must dial the parked socket,
# trigger auto-resume, and then read back the pre-freeze data
back = (ctypes.c_float * N)()
ck(hip.hipMemcpy(back, ptr, N * 4, 2), "d2h-after-freeze")
ok = all(abs(back[i] - 2.5) < 1e-9 for i in range(0, N, 65536))
print(f"PHASE2 ok={ok}", flush=True)
sys.exit(0 if ok else 1)
"""


def test_auto_freeze_idle_worker_and_resume_on_dial(tmp_path):
    """Auto-freeze end to end on hardware: an idle remote worker is
    snapshot to disk and STOPPED (its HBM freed); the client's next HIP
    call dials the parked socket, the controller resumes the worker from
    the snapshot, and the pre-freeze device data reads back intact
    (reference surface AutoFreezeConfig http_types.go:85-91; execution
    was closed-source there)."""

    from tensor_fusion_amd.hypervisor.vgpu_manager import (
        AutoFreezeController, VgpuWorkerManager)
    mgr = VgpuWorkerManager(run_dir=str(tmp_path))
    w = mgr.start("default/afw", 0)
    ctl = AutoFreezeController(
        mgr, rules={"low": {"enable": True, "freeze_to_disk_ttl_s": 3}},
        interval_s=0.5)
    ctl.register("default/afw", "low")
    marker = str(tmp_path / "go")
    env = client_env(w.socket_path)
    cli = subprocess.Popen([sys.executable, "-c", FREEZE_CLIENT, marker],
                           env=env, stdout=subprocess.PIPE,
                           stderr=subprocess.PIPE, text=True, cwd=REPO)
    try:
        line = cli.stdout.readline()
        assert "PHASE1" in line, line + cli.stderr.read()
        ctl.start()
        # wait for the idle TTL to freeze the worker
        deadline = time.time() + 60
        st = ctl.states["default/afw"]
        while time.time() < deadline and st.phase != "frozen_disk":
            time.sleep(0.5)
        assert st.phase == "frozen_disk", st
        # worker process actually exited (HBM released)
        assert mgr.workers["default/afw"].handle.proc.poll() is not None
        # release the client: its next op triggers auto-resume
        with open(marker, "w") as f:
            f.write("go")
        out, err = cli.communicate(timeout=180)
        assert cli.returncode == 0, out + err
        assert "PHASE2 ok=True" in out, out + err
        assert st.resumes == 1 and st.freezes == 1
    finally:
        ctl.stop()
        if cli.poll() is None:
            cli.kill()
        mgr.stop("default/afw")


def test_tcp_wire_robustness_garbage_inputs(tmp_path):
    """A worker's TCP port must survive hostile/buggy peers: a wrong
    handshake magic is rejected, a valid handshake followed by garbage
    frames ends that session cleanly, and a LEGITIMATE client afterwards
    still gets a full working session (the accept loop recovers)."""

    import socket as _s
    import struct

    from tensor_fusion_amd.client.runtime import client_env, start_worker
    port = 48777
    w = start_worker("", device_index=0, tcp_port=port)
    try:
        # 1) wrong magic → closed without echo
        c = _s.create_connection(("127.0.0.1", port), timeout=10)
        c.sendall(b"NOPE")
        c.settimeout(3)
        got = b""
        try:
            got = c.recv(4)
        except _s.timeout:
            pass
        assert got == b""  # no magic echo, connection dropped
        c.close()

        # 2) good magic, then a hostile frame header (absurd rec_len)
        c = _s.create_connection(("127.0.0.1", port), timeout=10)
        c.sendall(struct.pack("<I", 0x54465443))
        assert c.recv(4) == struct.pack("<I", 0x54465443)
        c.sendall(struct.pack("<IIQII", 0, 0xFFFFFFFF, 0, 0, 0))
        c.sendall(os.urandom(4096))
        c.close()

        # 3) a real client afterwards works end to end
        time.sleep(0.5)
        env = client_env("", tcp=f"127.0.0.1:{port}")
        out = subprocess.run([os.path.join(NATIVE, "tf_remote_testapp")],
                             capture_output=True, text=True, timeout=180,
                             env=env)
        assert "TESTAPP_OK" in out.stdout, out.stdout + out.stderr
    finally:
        w.stop()


HOT_MIGRATE_CLIENT = r"""
import ctypes, os, sys, time
hip = ctypes.CDLL(None)

def ck(rc, what):
    if rc != 0:
        print(f"FAIL {what} rc={rc}", flush=True)
        sys.exit(1)

N = 1 << 18
a = ctypes.c_void_p(); b = ctypes.c_void_p()
ck(hip.hipMalloc(ctypes.byref(a), N * 4), "malloc-a")
ck(hip.hipMalloc(ctypes.byref(b), N * 4), "malloc-b")
host = (ctypes.c_float * N)(*([1.0] * N))
ck(hip.hipMemcpy(a, host, N * 4, 1), "h2d")
print("LOOPING", flush=True)
stop_marker = sys.argv[1]
iters = 0
# continuous device work: copy a->b and read one element back, no pauses
while not os.path.exists(stop_marker):
    ck(hip.hipMemcpy(b, a, N * 4, 3), f"d2d-{iters}")
    probe = (ctypes.c_float * 1)()
    ck(hip.hipMemcpy(probe, b, 4, 2), f"d2h-{iters}")
    if abs(probe[0] - 1.0) > 1e-9:
        print(f"FAIL corrupt probe {probe[0]} at {iters}", flush=True)
        sys.exit(1)
    iters += 1
back = (ctypes.c_float * N)()
ck(hip.hipMemcpy(back, a, N * 4, 2), "final-d2h")
ok = all(abs(back[i] - 1.0) < 1e-9 for i in range(0, N, 16384))
print(f"DONE iters={iters} ok={ok}", flush=True)
sys.exit(0 if (ok and iters > 0) else 1)
"""


def test_live_migration_under_continuous_load(tmp_path):
    """Migrate the worker while the client is MID-LOOP issuing device
    work with no cooperation — ops stall during the snapshot/restore
    window, then complete; no op ever fails and the data survives
    (config-5's live-migrate-one-vGPU story under load)."""

    from tensor_fusion_amd.client.runtime import (client_env,
                                                  migrate_worker,
                                                  start_worker)
    sock = str(tmp_path / "vgpu.sock")
    snap = str(tmp_path / "snap.bin")
    stop = str(tmp_path / "stop")
    w = start_worker(sock, device_index=0, snapshot_path=snap)
    env = client_env(sock)
    cli = subprocess.Popen([sys.executable, "-c", HOT_MIGRATE_CLIENT,
                            stop],
                           env=env, stdout=subprocess.PIPE,
                           stderr=subprocess.PIPE, text=True, cwd=REPO)
    w2 = None
    try:
        assert "LOOPING" in cli.stdout.readline()
        time.sleep(1.0)  # client is deep in its loop
        w2 = migrate_worker(w, snap, new_device_index=0)
        time.sleep(1.0)  # client keeps hammering the restored worker
        with open(stop, "w") as f:
            f.write("x")
        out, err = cli.communicate(timeout=180)
        assert cli.returncode == 0, out + err
        assert "DONE" in out and "ok=True" in out, out + err
        iters = int(out.split("iters=")[1].split()[0])
        assert iters > 10, f"client barely ran: {iters}"
    finally:
        if cli.poll() is None:
            cli.kill()
        if w2 is not None:
            w2.stop()
        w.stop()
