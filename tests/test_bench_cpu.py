"""bench.py multi-rank aggregation path on CPU — 2 gloo ranks with
stubbed GPU children. Validates the exact flow the driver's 8-GPU
SCALE run uses: init_process_group(gloo) from torchrun-style env,
all_gather aggregation, rank-0 single JSON line with whole-job values."""
import json
import multiprocessing as mp
import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _rank_main(rank: int, world: int, port: int, q):
    os.environ.update({
        "RANK": str(rank), "LOCAL_RANK": str(rank),
        "WORLD_SIZE": str(world), "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": str(port),
    })
    sys.path.insert(0, REPO)
    sys.argv = ["bench.py", "--gpus", str(world), "--steps", "4",
                "--warmup", "1"]
    import bench

    # stub the GPU children: rank-dependent throughputs so aggregation
    # (sum over ranks, max ms_per_step) is actually checked
    def fake_child(mode, args, local_rank, graphs=True, fused=True):
        # graph-mode rows are the headline; the eager row is slower on
        # both sides (disclosure only)
        base = 100.0 if mode == "native" else 98.0
        if not graphs:
            base *= 0.5
        return {"tok_s": base + local_rank, "ms_per_step": 10.0 + local_rank}

    bench.run_child = fake_child
    import io
    from contextlib import redirect_stdout
    buf = io.StringIO()
    with redirect_stdout(buf):
        bench.main()
    q.put((rank, buf.getvalue()))


def test_bench_two_rank_gloo_aggregation():
    import socket
    world = 2
    with socket.socket() as sk:
        sk.bind(("127.0.0.1", 0))
        port = sk.getsockname()[1]
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_rank_main, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    outs = {}
    for _ in procs:
        rank, out = q.get(timeout=120)
        outs[rank] = out
    for p in procs:
        p.join(timeout=30)
        assert p.exitcode == 0
    # only rank 0 prints; exactly one JSON line
    assert outs[1].strip() == ""
    line = outs[0].strip().splitlines()[-1]
    rec = json.loads(line)
    assert rec["n_gpus"] == 2
    assert rec["scaling"] == "weak"
    assert rec["higher_is_better"] is False
    # native = 100+101, vgpu = 98+99 → overhead = 100*(1-197/201)
    assert abs(rec["config"]["native_tok_s"] - 201.0) < 1e-6
    assert abs(rec["config"]["vgpu_tok_s"] - 197.0) < 1e-6
    assert abs(rec["value"] - 100.0 * (1 - 197.0 / 201.0)) < 1e-3
    assert rec["ms_per_step"] == 11.0  # max over ranks
    assert rec["config"]["parallelism"] == "dp2"
    assert rec["config"]["global_batch"] == 16
