"""Expert parallelism (parallel/ep.py) — 2-rank gloo CPU test: the
EP-sharded switch-MoE reproduces the single-process dense reference
exactly (token routing + two all_to_all_single exchanges + return
scatter)."""
import os
import socket
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_ep_matches_reference_2rank():
    with socket.socket() as sk:
        sk.bind(("127.0.0.1", 0))
        port = sk.getsockname()[1]
    procs = []
    for rank in range(2):
        env = dict(os.environ, RANK=str(rank), WORLD_SIZE="2",
                   MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                   TF_REPO=REPO)
        procs.append(subprocess.Popen(
            [sys.executable, "-m",
             "tensor_fusion_amd.parallel._ep_test_worker"],
            env=env, cwd=REPO, stdout=subprocess.PIPE,
            stderr=subprocess.PIPE, text=True))
    outs = [p.communicate(timeout=180) for p in procs]
    assert all(p.returncode == 0 for p in procs), \
        "\n".join(o + e for o, e in outs)
    assert "EP_OK" in outs[0][0]
    assert "EP_EMPTY_OK" in outs[0][0]  # zero-token expert path
